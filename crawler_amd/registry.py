"""Platform crawler plugin registry.

Mirrors the reference's plugin system (crawler/crawler.go:49-126:
`Crawler` interface + `DefaultCrawlerFactory`, and
crawler/common/registrar.go:11 `RegisterAllCrawlers`): platforms
register a factory under their platform type; the CLI resolves the
configured platform with one lookup. Adding a platform means adding a
module under `crawler_amd/platforms/` and one `register` call — no CLI
changes.
"""
from __future__ import annotations

import abc
import dataclasses
from typing import Callable, Dict, List, Optional


@dataclasses.dataclass
class CrawlTarget:
    """crawler/crawler.go CrawlTarget: what to crawl on which platform."""

    platform: str
    id: str
    metadata: Optional[dict] = None


@dataclasses.dataclass
class CrawlContext:
    """Everything a platform crawler needs from the CLI layer."""

    cfg: object
    args: object
    urls: List[str]
    feed: object = None


class PlatformCrawler(abc.ABC):
    """crawler/crawler.go:49-67 Crawler interface, MI355X-shaped: the
    run loop owns its state manager / GPU engine internally."""

    @abc.abstractmethod
    def platform_type(self) -> str:
        ...

    def initialize(self, ctx: CrawlContext) -> None:
        """Set up resources (pools, state managers)."""

    def validate_target(self, target: CrawlTarget) -> None:
        """Raise ValueError if the target is not valid for this
        platform (crawler.go ValidateTarget)."""
        if target.platform != self.platform_type():
            raise ValueError(
                f"target platform {target.platform!r} does not match "
                f"crawler {self.platform_type()!r}")
        if not target.id:
            raise ValueError("empty target id")

    @abc.abstractmethod
    def run(self, ctx: CrawlContext) -> dict:
        """Execute the configured crawl; returns stats."""

    def close(self) -> None:
        """Release resources (crawler.go Close)."""


class CrawlerRegistry:
    """DefaultCrawlerFactory (crawler/crawler.go:79-106): register once
    per platform, duplicate registration and unknown lookups raise."""

    def __init__(self):
        self._creators: Dict[str, Callable[[], PlatformCrawler]] = {}

    def register(self, platform: str,
                 creator: Callable[[], PlatformCrawler]) -> None:
        if platform in self._creators:
            raise ValueError(
                f"crawler for platform {platform!r} already registered")
        self._creators[platform] = creator

    def get(self, platform: str) -> PlatformCrawler:
        creator = self._creators.get(platform)
        if creator is None:
            raise ValueError(
                f"no crawler registered for platform {platform!r} "
                f"(registered: {sorted(self._creators)})")
        return creator()

    def platforms(self) -> List[str]:
        return sorted(self._creators)


default_registry = CrawlerRegistry()


def register_all_crawlers(registry: Optional[CrawlerRegistry] = None
                          ) -> CrawlerRegistry:
    """crawler/common/registrar.go:11 — import every platform module
    and let each register itself. Idempotent on the default registry."""
    reg = registry if registry is not None else default_registry
    from .platforms import telegram, youtube  # noqa: PLC0415

    for mod in (telegram, youtube):
        try:
            mod.register(reg)
        except ValueError:
            pass  # already registered (idempotent re-entry)
    return reg


def get_crawler(platform: str) -> PlatformCrawler:
    return default_registry.get(platform)
