"""GPU random-walk tests: batched hops, walk semantics, frontier checkpoint."""
import datetime as dt
import random

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page, RandomWalkStore
from crawler_amd.feed import FeedConfig, SyntheticFeed

pytestmark = pytest.mark.gpu

NOW = dt.datetime(2026, 1, 1, tzinfo=dt.timezone.utc)


def mk(tmp_path, **kw):
    from crawler_amd.engine.gpu_randomwalk import GpuRandomWalk

    kw.setdefault("walkback_rate", 15)
    cfg = CrawlerConfig(crawl_id="grw1", storage_root=str(tmp_path),
                        sampling_method="random-walk", min_users=1, **kw)
    feed = SyntheticFeed(FeedConfig(seed=44, universe=500,
                                    posts_per_channel=64))
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    eng = GpuRandomWalk(cfg, sm, rw, feed, posts_per_hop=64, walkers=16,
                        rng=random.Random(3))
    return cfg, sm, rw, eng


def test_gpu_walk_progresses_and_records_edges(tmp_path):
    cfg, sm, rw, eng = mk(tmp_path)
    eng.seed(["c0000000001", "c0000000002"])
    stats = eng.run(max_pages=12, now=NOW)
    assert stats["pages"] == 12
    assert stats["posts"] == 12 * 64
    followed = [e for e in rw.edge_records if not e.skipped]
    assert followed
    # frontier persists (the checkpoint)
    assert rw.buffer_size() >= 1
    # JSONL landed for the seed channel and matches the CPU oracle bytes
    import numpy as np

    from crawler_amd.ops.golden_batch import encode_batch

    path = tmp_path / "grw1" / "c0000000001" / "posts" / "posts.jsonl"
    got = path.read_bytes()
    golden_batch = eng.feed.build_batch(np.array([1]), posts_per_channel=64)
    lines, _ = encode_batch(golden_batch, now=NOW)
    assert got == b"".join(lines)


def test_gpu_walk_400_replacement(tmp_path):
    cfg, sm, rw, eng = mk(tmp_path)
    eng.seed(["c0000000001", "c0000000002"])
    rw.add_page(Page(id="badX", url="c0000999999", depth=1,
                     sequence_id="sq", status="unfetched"))
    eng.run(max_pages=6, now=NOW)
    assert eng.stats["invalid_400"] >= 1
    assert rw.is_invalid_channel("c0000999999")
    assert "badX" not in rw.page_buffer


def test_gpu_walkback_rate_100(tmp_path):
    cfg, sm, rw, eng = mk(tmp_path, walkback_rate=100)
    eng.seed(["c0000000001", "c0000000002", "c0000000003"])
    eng.run(max_pages=6, now=NOW)
    wb = [e for e in rw.edge_records if e.walkback]
    assert wb


def test_gpu_walk_pipelined(tmp_path):
    """Pipelined two-pool mode: same per-page semantics (pages, edges,
    JSONL lines = posts stored), device stage overlapped on a worker
    thread."""
    cfg, sm, rw, eng = mk(tmp_path)
    eng.seed(["c%010d" % i for i in range(1, 9)])
    stats = eng.run(max_pages=24, now=NOW, pipelined=True)
    assert stats["pages"] == 24
    assert stats["posts"] == 24 * 64
    followed = [e for e in rw.edge_records if not e.skipped]
    assert followed
    assert rw.buffer_size() >= 1
    # every stored JSONL line is intact (count == posts, all parse)
    import json as _json

    n_lines = 0
    for f in (tmp_path / "grw1").rglob("posts.jsonl"):
        for line in f.read_bytes().splitlines():
            _json.loads(line)
            n_lines += 1
    assert n_lines == stats["posts"]


def test_cli_gpu_random_walk(tmp_path):
    """--gpu --sampling random-walk through the CLI dispatch runs the
    batched GPU engine (pipelined) end to end."""
    from crawler_amd.cli import main

    rc = main([
        "--mode", "standalone", "--gpu", "--sampling", "random-walk",
        "--seed-size", "8", "--max-pages", "24",
        "--storage-root", str(tmp_path), "--crawl-id", "gcliwalk1",
        "--synthetic-universe", "500", "--synthetic-posts", "32",
        "--concurrency", "16", "--min-users", "1", "--skip-media",
    ])
    assert rc == 0
    posts = list((tmp_path / "gcliwalk1").rglob("posts.jsonl"))
    assert posts
    assert sum(p.stat().st_size for p in posts) > 0
