"""Panic-containment parity: unexpected exceptions are recovered at
message, page and video granularity (reference: tdutils.go:395-405
recover() per message parse, crawl runner recoverFromPanic per channel,
crawler/youtube/panic_test.go per video)."""
import random

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page, RandomWalkStore
from crawler_amd.engine.pipeline import run_for_channel
from crawler_amd.engine.runner import RandomWalkRunner, StandaloneRunner
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.client import ConnectionPool


def mk_env(tmp_path, **cfg_kw):
    cfg_kw.setdefault("min_users", 1)
    cfg = CrawlerConfig(crawl_id="pr1", storage_root=str(tmp_path),
                        disable_rate_limits=True, **cfg_kw)
    feed = SyntheticFeed(FeedConfig(seed=5, universe=50,
                                    posts_per_channel=10))
    pool = ConnectionPool(feed, 2, cfg.rate_limit, posts_per_channel=10,
                          disable_rate_limits=True)
    return cfg, feed, pool, LocalStateManager(cfg)


def test_message_parse_exception_recovered(tmp_path, monkeypatch):
    """One message that blows up in parse is skipped; the rest of the
    channel still stores."""
    cfg, feed, pool, sm = mk_env(tmp_path)
    import crawler_amd.engine.pipeline as P

    real = P.G.parse_message
    calls = {"n": 0}

    def bomb(m, **kw):
        calls["n"] += 1
        if calls["n"] == 3:
            raise ValueError("malformed message")
        return real(m, **kw)

    monkeypatch.setattr(P.G, "parse_message", bomb)
    client = pool.get_connection()
    try:
        res = run_for_channel(client, Page(id="p", url="c0000000001"),
                              sm, cfg, rng=random.Random(0))
    finally:
        pool.release_connection(client)
    assert res.status == "fetched"
    assert res.parse_errors == 1
    assert res.posts_stored == 9  # 10 messages, one recovered
    sm.close()


def test_standalone_page_exception_marks_error_not_crash(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="channel")

    def exploding(pool_, page, sm_, cfg_, **kw):
        if page.url == "c0000000002":
            raise RuntimeError("unexpected bug in channel processing")
        from crawler_amd.engine.pipeline import run_for_channel_with_pool
        return run_for_channel_with_pool(pool_, page, sm_, cfg_, **kw)

    runner = StandaloneRunner(cfg, sm, pool, run_for_channel_fn=exploding)
    stats = runner.run(["c0000000001", "c0000000002", "c0000000003"])
    assert stats["pages"] == 2          # the other two completed
    assert stats["errors"] == 1
    sm2 = LocalStateManager(cfg)
    assert sm2.load_state()
    bad = [p for p in sm2.pages.values() if p.url == "c0000000002"]
    assert bad and bad[0].status == "error"
    assert "recovered" in bad[0].error


def test_randomwalk_page_exception_drops_page_and_continues(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="random-walk",
                                 walkback_rate=0)
    rw = RandomWalkStore()
    n = {"calls": 0}

    def exploding(pool_, page, sm_, cfg_, **kw):
        n["calls"] += 1
        raise RuntimeError("boom")

    runner = RandomWalkRunner(cfg, sm, rw, pool,
                              run_for_channel_fn=exploding)
    runner.seed(["c0000000001"])
    stats = runner.run(max_pages=5, max_seconds=5)
    assert stats["errors"] == 1
    assert n["calls"] == 1
    assert rw.get_pages(10) == []       # page deleted, loop exited cleanly


def test_youtube_video_conversion_exception_skips_video(tmp_path,
                                                        monkeypatch):
    from crawler_amd.youtube import runner as yr

    cfg = CrawlerConfig(crawl_id="yt1", storage_root=str(tmp_path),
                        sampling_method="random", max_posts=20)
    real = yr.convert_video_to_post
    seen = {"n": 0}

    def bomb(v, ch, crawl_label=""):
        seen["n"] += 1
        if seen["n"] % 5 == 0:
            raise KeyError("missing field")
        return real(v, ch, crawl_label=crawl_label)

    monkeypatch.setattr(yr, "convert_video_to_post", bomb)
    stats = yr.run_youtube(cfg, [])
    assert stats["videos"] == 20
    assert stats["posts"] == 16         # 4 of 20 recovered-and-skipped
