"""Crawl engine tests: per-channel pipeline, runners, random-walk, faults.

Mirrors the reference's crawl/ test coverage areas (channel
pipeline, FLOOD_WAIT behavior, 400 replacement matrix, tandem batching,
layer iteration, layerless loop)."""
import json
import random

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page, RandomWalkStore
from crawler_amd.engine import errors as E
from crawler_amd.engine.pipeline import run_for_channel_with_pool
from crawler_amd.engine.runner import RandomWalkRunner, StandaloneRunner
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.client import ConnectionPool, FaultConfig


def mk_env(tmp_path, universe=200, posts=40, pool_size=2, faults=None,
           **cfg_kw):
    cfg_kw.setdefault("min_users", 1)
    cfg = CrawlerConfig(
        crawl_id="t1", storage_root=str(tmp_path),
        disable_rate_limits=True, **cfg_kw,
    )
    feed = SyntheticFeed(FeedConfig(seed=11, universe=universe,
                                    posts_per_channel=posts))
    pool = ConnectionPool(feed, pool_size, cfg.rate_limit, faults=faults,
                          posts_per_channel=posts, disable_rate_limits=True)
    sm = LocalStateManager(cfg)
    return cfg, feed, pool, sm


def test_run_for_channel_stores_posts(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, posts=30)
    page = Page(id="p1", url="c0000000003", depth=0)
    res = run_for_channel_with_pool(pool, page, sm, cfg)
    assert res.status == "fetched"
    assert res.posts_stored == 30
    sm.close()
    path = tmp_path / "t1" / "c0000000003" / "posts" / "posts.jsonl"
    lines = path.read_bytes().splitlines()
    assert len(lines) == 30
    obj = json.loads(lines[0])
    assert obj["platform_name"] == "Telegram"
    assert obj["channel_data"]["channel_engagement_data"]["post_count"] == 30


def test_channel_mode_processes_seeds_only(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="channel",
                                 max_depth=3)
    runner = StandaloneRunner(cfg, sm, pool)
    stats = runner.run(["c0000000001", "c0000000002"])
    assert stats["pages"] == 2
    sm2 = LocalStateManager(cfg)
    assert sm2.load_state()
    assert sm2.get_max_depth() == 0  # no expansion in channel mode


def test_snowball_expands_layers(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="snowball",
                                 max_depth=1, posts=60)
    runner = StandaloneRunner(cfg, sm, pool)
    stats = runner.run(["c0000000001"])
    sm2 = LocalStateManager(cfg)
    assert sm2.load_state()
    assert sm2.get_max_depth() >= 1
    layer1 = sm2.get_layer_by_depth(1)
    assert layer1, "snowball must discover outlinked channels"
    # crawl completed: layer-1 pages processed too
    assert all(p.status in ("fetched", "deadend", "error")
               for p in layer1)


def test_snowball_max_depth_respected(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="snowball",
                                 max_depth=0, posts=60)
    runner = StandaloneRunner(cfg, sm, pool)
    runner.run(["c0000000001"])
    sm2 = LocalStateManager(cfg)
    sm2.load_state()
    # discovered pages may exist at depth 1 but must be unprocessed
    for p in sm2.get_layer_by_depth(1):
        assert p.status == "unfetched"


def test_resume_skips_fetched_pages(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="channel")
    runner = StandaloneRunner(cfg, sm, pool)

    calls = []
    orig = runner.run_for_channel_fn

    def spy(pool_, page, sm_, cfg_, **kw):
        calls.append(page.url)
        return orig(pool_, page, sm_, cfg_, **kw)

    runner.run_for_channel_fn = spy
    runner.run(["c0000000001", "c0000000002"])
    assert len(calls) == 2

    # a completed crawl is not resumed; a fresh incomplete one re-runs only
    # unfetched pages
    cfg2, _, pool2, sm2 = mk_env(tmp_path, sampling_method="channel")
    sm2.metadata.crawl_id = "t1"
    exec_id, ok = sm2.find_incomplete_crawl("t1")
    assert not ok  # crawl completed above


def test_flood_wait_retire_aborts_when_pool_empty(tmp_path):
    faults = FaultConfig(long_flood_permille=1000)
    cfg, feed, pool, sm = mk_env(tmp_path, pool_size=1, faults=faults,
                                 sampling_method="channel")
    runner = StandaloneRunner(cfg, sm, pool)
    with pytest.raises(E.FloodWaitRetire):
        runner.run(["c0000000001"])
    assert pool.stats()["retired"] == 1


def test_tdlib_400_marks_error_page(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="channel")
    runner = StandaloneRunner(cfg, sm, pool)
    runner.run(["c0000009999"])  # outside universe=200 -> USERNAME_NOT_OCCUPIED
    sm2 = LocalStateManager(cfg)
    sm2.load_state()
    page = sm2.get_layer_by_depth(0)[0]
    assert page.status == "error"
    assert "USERNAME_NOT_OCCUPIED" in page.error


def test_deadend_on_min_users(tmp_path):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="channel",
                                 min_users=10_000_000)
    runner = StandaloneRunner(cfg, sm, pool)
    stats = runner.run(["c0000000001"])
    assert stats["deadends"] == 1


# ---------- random walk ----------

def rw_env(tmp_path, **kw):
    cfg, feed, pool, sm = mk_env(tmp_path, sampling_method="random-walk",
                                 posts=50, universe=300, **kw)
    rw = RandomWalkStore()
    return cfg, feed, pool, sm, rw


def test_random_walk_loop_walks_and_records_edges(tmp_path):
    cfg, feed, pool, sm, rw = rw_env(tmp_path)
    runner = RandomWalkRunner(cfg, sm, rw, pool,
                              rng=random.Random(7))
    runner.seed(["c0000000001"])
    stats = runner.run(max_pages=10)
    assert stats["pages"] == 10
    assert rw.edge_records, "walk must record edges"
    followed = [e for e in rw.edge_records if not e.skipped]
    assert followed
    # forward edges propagate sequence ids; pages remain in buffer = frontier
    assert rw.buffer_size() >= 1


def test_random_walk_walkback_decision_sequences(tmp_path):
    cfg, feed, pool, sm, rw = rw_env(tmp_path, walkback_rate=100)
    runner = RandomWalkRunner(cfg, sm, rw, pool, rng=random.Random(3))
    runner.seed(["c0000000001", "c0000000002"])
    runner.run(max_pages=6)
    wb = [e for e in rw.edge_records if e.walkback]
    assert wb, "walkback_rate=100 must produce walkback edges"


def test_random_walk_400_replacement(tmp_path):
    cfg, feed, pool, sm, rw = rw_env(tmp_path)
    runner = RandomWalkRunner(cfg, sm, rw, pool, rng=random.Random(5))
    runner.seed(["c0000000001", "c0000000002"])
    # inject a page pointing outside the universe -> 400 path
    bad = Page(id="bad1", url="c0000009999", depth=1, sequence_id="seqX",
               status="unfetched")
    rw.add_page(bad)
    runner.run(max_pages=8)
    assert runner.stats["invalid_400"] >= 1
    assert rw.is_invalid_channel("c0000009999")
    assert "bad1" not in rw.page_buffer  # deleted after replacement


def test_400_seed_replacement_picks_valid_seed(tmp_path):
    """handle400SeedReplacement (runner.go:263-284): a 400'd SEED
    channel (no incoming edge) is replaced by a random VALID seed —
    fresh sequence chain, NO edge record, and never an invalidated
    seed."""
    import datetime as dt

    from crawler_amd.engine import randomwalk as RW

    cfg, feed, pool, sm, rw = rw_env(tmp_path)
    for u in ("c0000000001", "c0000000002", "c0000000003"):
        sm.add_discovered_channel(u)
        rw.upsert_seed_channel(u)
        sm._seed_channels[u] = True  # is_seed_channel
    rw.mark_seed_channel_invalid("c0000000002")
    bad = Page(id="seedbad", url="c0000000001", depth=0,
               sequence_id="sq0", status="unfetched")
    rw.add_page(bad)
    n_edges = rw.edge_count()
    RW.handle_400_replacement(sm, rw, bad, cfg, random.Random(3))
    # replacement page exists, on a fresh chain, from the seed table
    repl = [p for p in rw.page_buffer.values() if p.id != "seedbad"]
    assert len(repl) == 1
    # the failed seed was just invalidated, so only c...3 is eligible
    assert repl[0].url == "c0000000003"
    assert repl[0].sequence_id != "sq0"
    # invalidated seeds are never picked
    for _ in range(20):
        pick = rw.get_random_seed_channel(random.Random(_))
        assert pick != "c0000000002"
    # NO edge record is written for seed replacement
    assert rw.edge_count() == n_edges


def test_tandem_mode_writes_pending_edges(tmp_path):
    cfg, feed, pool, sm, rw = rw_env(tmp_path, tandem_crawl=True)
    runner = RandomWalkRunner(cfg, sm, rw, pool, rng=random.Random(9))
    runner.seed(["c0000000001"])
    runner.run(max_pages=1)
    # either edges were streamed into an open->closed batch, or a forced
    # walkback page was produced
    if rw.pending_batches:
        b = next(iter(rw.pending_batches.values()))
        assert b.status == "closed"
        assert rw.edges_of_batch(b.batch_id)
    else:
        assert rw.buffer_size() >= 1


def test_walk_stats_and_seed_marking(tmp_path):
    cfg, feed, pool, sm, rw = rw_env(tmp_path)
    runner = RandomWalkRunner(cfg, sm, rw, pool, rng=random.Random(1))
    runner.seed(["c0000000005"])
    runner.run(max_pages=3)
    assert rw.get_channel_last_crawled("c0000000005") is not None


def test_resample_marker_skips_fetched_and_marks_deleted(tmp_path):
    """resampleMarker semantics (crawl/runner.go:1572-1633): re-crawling a
    page skips already-fetched messages (no duplicate posts) and marks
    tracked-but-vanished messages deleted."""
    from crawler_amd.engine.state import PageMessage

    cfg, feed, pool, sm = mk_env(tmp_path, posts=20)
    page = Page(id="pr1", url="c0000000003", depth=0)
    sm.add_layer([page])
    res1 = run_for_channel_with_pool(pool, page, sm, cfg)
    assert res1.posts_stored == 20
    page = sm.get_page("pr1")
    assert len(page.messages) == 20
    assert all(m.status == "fetched" for m in page.messages)

    # track a phantom message that the next fetch won't contain
    page.messages.append(PageMessage(chat_id=page.messages[0].chat_id,
                                     message_id=999 << 20, status="unfetched",
                                     page_id="pr1"))
    res2 = run_for_channel_with_pool(pool, page, sm, cfg)
    assert res2.posts_stored == 0          # nothing re-stored
    assert res2.skipped_fetched == 20
    phantom = [m for m in page.messages if m.message_id == 999 << 20][0]
    assert phantom.status == "deleted"
    sm.close()
    path = tmp_path / "t1" / "c0000000003" / "posts" / "posts.jsonl"
    assert path.read_bytes().count(b"\n") == 20  # no duplicates on disk
