#!/usr/bin/env python3
"""End-to-end GPU crawl ENGINE measurement (not the bench.py microbench):
a real snowball crawl through GpuCrawlEngine — device feed generation,
HIP parse+encode, seen-set claims, per-channel JSONL files on disk,
state checkpoints — reporting whole-crawl posts/sec including the host
filesystem writes. This is the config #3 shape on one GPU.

Usage: python scripts/gpu_engine_bench.py [--seeds 200 --posts 2000
       --max-depth 2 --storage /tmp/gpu-crawl]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from crawler_amd.config import CrawlerConfig  # noqa: E402
from crawler_amd.engine import LocalStateManager  # noqa: E402
from crawler_amd.engine.gpu_runner import GpuCrawlEngine  # noqa: E402
from crawler_amd.feed import FeedConfig, SyntheticFeed  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seeds", type=int, default=200)
    ap.add_argument("--posts", type=int, default=2000)
    ap.add_argument("--max-depth", type=int, default=2)
    ap.add_argument("--universe", type=int, default=1_000_000)
    ap.add_argument("--storage", default="/tmp/gpu-crawl")
    ap.add_argument("--max-pages", type=int, default=1500)
    ap.add_argument("--mode", choices=["snowball", "randomwalk"],
                    default="snowball")
    ap.add_argument("--walkers", type=int, default=512,
                    help="randomwalk: concurrent walker chains per hop")
    ap.add_argument("--walkback-rate", type=int, default=15)
    args = ap.parse_args()

    if args.mode == "randomwalk":
        return run_randomwalk(args)

    cfg = CrawlerConfig(
        crawl_id="engine-bench", storage_root=args.storage,
        sampling_method="snowball", max_depth=args.max_depth, min_users=1,
        max_pages=args.max_pages,
        skip_media_download=True,
    )
    feed = SyntheticFeed(FeedConfig(seed=2026, universe=args.universe,
                                    posts_per_channel=args.posts))
    sm = LocalStateManager(cfg)
    eng = GpuCrawlEngine(cfg, sm, feed, posts_per_channel=args.posts,
                         chunk_channels=512)
    seeds = [feed.username_of(i) for i in range(args.seeds)]
    t0 = time.perf_counter()
    stats = eng.run(seeds)
    elapsed = time.perf_counter() - t0
    out = {
        "metric": "engine posts/sec (full crawl incl. disk JSONL)",
        "value": round(stats["posts"] / elapsed, 1),
        "elapsed_s": round(elapsed, 2),
        "pages": stats["pages"],
        "posts": stats["posts"],
        "jsonl_gb": round(stats["jsonl_bytes"] / 1e9, 2),
        "discovered": stats["discovered"],
        "deadends": stats["deadends"],
        "max_depth": args.max_depth,
    }
    if getattr(eng, "timings", None):
        out["phase_s"] = {k: round(v, 3) for k, v in eng.timings.items()}
    print(json.dumps(out))


def run_randomwalk(args):
    """Batched random-walk measurement (BASELINE random-walk shape;
    VERDICT r01 item 5: >=2M posts/s with identical walk semantics)."""
    import random

    from crawler_amd.engine import RandomWalkStore
    from crawler_amd.engine.gpu_randomwalk import GpuRandomWalk

    cfg = CrawlerConfig(
        crawl_id="rw-bench", storage_root=args.storage,
        sampling_method="random-walk", min_users=1,
        walkback_rate=args.walkback_rate, skip_media_download=True,
    )
    feed = SyntheticFeed(FeedConfig(seed=2027, universe=args.universe,
                                    posts_per_channel=args.posts))
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    eng = GpuRandomWalk(cfg, sm, rw, feed, posts_per_hop=args.posts,
                        walkers=args.walkers, rng=random.Random(11))
    eng.seed([feed.username_of(i) for i in range(args.walkers)])
    t0 = time.perf_counter()
    stats = eng.run(max_pages=args.max_pages)
    elapsed = time.perf_counter() - t0
    print(json.dumps({
        "metric": "random-walk posts/sec (batched hops incl. disk)",
        "value": round(stats["posts"] / elapsed, 1),
        "elapsed_s": round(elapsed, 2),
        "walkers": args.walkers,
        "pages": stats["pages"],
        "posts": stats["posts"],
        "edges": stats["edges"],
        "walkback_exhausted": stats["walkback_exhausted"],
        "phase_s": {k: round(v, 3) for k, v in eng.timings.items()},
    }))


if __name__ == "__main__":
    main()
