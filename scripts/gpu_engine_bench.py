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
    ap.add_argument("--mode", choices=["snowball", "randomwalk", "media"],
                    default="snowball")
    ap.add_argument("--blobs", type=int, default=2000,
                    help="media mode: number of blobs to fetch+upload")
    ap.add_argument("--now", default=None,
                    help="pin the capture timestamp (ISO, UTC) so "
                         "tools/verify_jsonl.py can re-derive the bytes")
    ap.add_argument("--walkers", type=int, default=512,
                    help="randomwalk: concurrent walker chains per hop")
    ap.add_argument("--walkback-rate", type=int, default=15)
    ap.add_argument("--pipelined", action="store_true",
                    help="two half-pools: device stage overlaps host tail")
    args = ap.parse_args()

    if args.mode == "randomwalk":
        return run_randomwalk(args)
    if args.mode == "media":
        return run_media(args)

    cfg = CrawlerConfig(
        crawl_id="engine-bench", storage_root=args.storage,
        sampling_method="snowball", max_depth=args.max_depth, min_users=1,
        max_pages=args.max_pages,
        skip_media_download=True,
    )
    feed = SyntheticFeed(FeedConfig(seed=2026, universe=args.universe,
                                    posts_per_channel=args.posts))
    sm = LocalStateManager(cfg)
    fixed_now = None
    if args.now:
        import datetime as dt
        fixed_now = dt.datetime.fromisoformat(args.now).replace(
            tzinfo=dt.timezone.utc)
    eng = GpuCrawlEngine(cfg, sm, feed, posts_per_channel=args.posts,
                         chunk_channels=512, fixed_now=fixed_now)
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
    stats = eng.run(max_pages=args.max_pages,
                    pipelined=bool(args.pipelined))
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
        "per_hop": [
            {"pages": h[0], "posts": h[1],
             "posts_per_s": round(h[1] / h[2], 1) if h[2] else None,
             "s": h[2], "phases": (h[3] if len(h) > 3 else {})}
            for h in eng.hop_log
        ],
    }))


def run_media(args):
    """BASELINE config #5 measurement: synthetic
    blobs staged in HBM, spilled through the pinned bounce ring to the
    local blob store (Azure-Blob mock). Reports blobs/s and spill GB/s
    incl. the host write; media-cache dedup + 150 MB cap semantics are
    live (engine/media.py)."""
    from crawler_amd.engine.media import MediaEngine, synth_blob_size

    cfg = CrawlerConfig(crawl_id="media-bench", storage_root=args.storage,
                        sampling_method="channel", min_users=1)
    sm = LocalStateManager(cfg)
    eng = MediaEngine(sm, device="cuda:0")
    # pre-plan ids; sizes are deterministic (mostly 10KB-2MB, some MBs)
    ids = [f"m{100000 + i}" for i in range(args.blobs)]
    planned = sum(synth_blob_size(i) for i in ids
                  if synth_blob_size(i) < 150 * 1024 * 1024)
    import torch
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i, mid in enumerate(ids):
        eng.fetch_and_upload(f"chan{i % 32}", mid)
    eng.flush()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    eng.close()
    print(json.dumps({
        "metric": "media blobs/sec (HBM gen -> pinned spill -> disk)",
        "value": round(eng.stats["stored"] / elapsed, 1),
        "elapsed_s": round(elapsed, 2),
        "blobs_stored": eng.stats["stored"],
        "deduped": eng.stats["deduped"],
        "over_cap": eng.stats["over_cap"],
        "bytes_gb": round(eng.stats["bytes"] / 1e9, 3),
        "spill_gb_s": round(eng.stats["bytes"] / 1e9 / elapsed, 3),
        "planned_bytes_gb": round(planned / 1e9, 3),
    }))


if __name__ == "__main__":
    main()
