"""Crawler plugin registry tests (ref crawler/crawler.go:49-126 +
crawler/common/registrar.go)."""
import pytest

from crawler_amd.registry import (CrawlContext, CrawlerRegistry,
                                  CrawlTarget, PlatformCrawler,
                                  register_all_crawlers)


class FakePlatform(PlatformCrawler):
    def __init__(self):
        self.ran = False

    def platform_type(self):
        return "fake"

    def run(self, ctx):
        self.ran = True
        return {"ok": 1}


def test_register_and_get():
    reg = CrawlerRegistry()
    reg.register("fake", FakePlatform)
    c = reg.get("fake")
    assert isinstance(c, FakePlatform)
    assert c.run(None) == {"ok": 1}


def test_duplicate_registration_rejected():
    reg = CrawlerRegistry()
    reg.register("fake", FakePlatform)
    with pytest.raises(ValueError, match="already registered"):
        reg.register("fake", FakePlatform)


def test_unknown_platform_rejected():
    reg = CrawlerRegistry()
    with pytest.raises(ValueError, match="no crawler registered"):
        reg.get("myspace")


def test_register_all_crawlers_idempotent():
    reg = register_all_crawlers()
    reg2 = register_all_crawlers()
    assert reg is reg2
    assert "telegram" in reg.platforms()
    assert "youtube" in reg.platforms()


def test_third_platform_needs_no_cli_change():
    """Adding a platform = registering a creator; the CLI's lookup
    resolves it with zero dispatch-code changes."""
    reg = register_all_crawlers()
    try:
        reg.register("mastodon", FakePlatform)
        assert isinstance(reg.get("mastodon"), FakePlatform)
        assert "mastodon" in reg.platforms()
    finally:
        reg._creators.pop("mastodon", None)


def test_validate_target():
    c = FakePlatform()
    c.validate_target(CrawlTarget(platform="fake", id="chan1"))
    with pytest.raises(ValueError, match="does not match"):
        c.validate_target(CrawlTarget(platform="telegram", id="x"))
    with pytest.raises(ValueError, match="empty target id"):
        c.validate_target(CrawlTarget(platform="fake", id=""))


def test_cli_uses_registry_for_platform_dispatch():
    """cli.py contains no platform if/else ladder anymore."""
    import inspect

    from crawler_amd import cli

    src = inspect.getsource(cli)
    assert "get_crawler(cfg.platform)" in src
    assert 'cfg.platform == "youtube"' not in src


def test_context_fields():
    ctx = CrawlContext(cfg=1, args=2, urls=["a"], feed=3)
    assert ctx.urls == ["a"]
