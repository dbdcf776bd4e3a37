"""Platform crawler implementations (one module per platform).

Each module exposes `register(registry)` — called by
crawler_amd.registry.register_all_crawlers (the registrar pattern of
crawler/common/registrar.go:11).
"""
