#!/usr/bin/env python3
"""Multi-GPU orchestrated snowball crawl (BASELINE config #3 shape).

Launch on an 8-GPU node:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
      --master-addr 127.0.0.1 scripts/gpu_dist_crawl.py \\
      --seeds 400 --posts 1000 --max-depth 3 --max-pages 4000

Rank 0 hosts the TCPStore work queue and the persisted checkpoints; all
ranks claim channel chunks, run the HIP hot path, write their own JSONL
shards, and exchange discoveries collectively (RCCL over xGMI).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402
from torch.distributed import TCPStore  # noqa: E402

from crawler_amd.config import CrawlerConfig  # noqa: E402
from crawler_amd.engine import LocalStateManager  # noqa: E402
from crawler_amd.engine.gpu_runner import GpuCrawlEngine  # noqa: E402
from crawler_amd.feed import FeedConfig, SyntheticFeed  # noqa: E402
from crawler_amd.parallel.orchestrated import OrchestratedCrawl  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seeds", type=int, default=400)
    ap.add_argument("--posts", type=int, default=1000)
    ap.add_argument("--max-depth", type=int, default=3)
    ap.add_argument("--max-pages", type=int, default=4000)
    ap.add_argument("--universe", type=int, default=1_000_000)
    ap.add_argument("--storage", default="/tmp/gpu-dist-crawl")
    ap.add_argument("--chunk-channels", type=int, default=64)
    ap.add_argument("--fake-engine", action="store_true",
                    help="CPU dry run: gloo backend + a stub hot path "
                         "(exercises the exact launcher wiring at "
                         "world>1 without a GPU)")
    ap.add_argument("--store-port", type=int, default=29761)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if not args.fake_engine:
        # wrap onto visible devices so a 1-GPU box can dry-run world>1
        # (two ranks share cuda:0; use CRAWL_DIST_BACKEND=gloo then)
        local_rank = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)
    if world > 1:
        backend = os.environ.get(
            "CRAWL_DIST_BACKEND",
            "gloo" if args.fake_engine else "nccl")
        dist.init_process_group(backend)
    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    store = TCPStore(host, args.store_port, is_master=(rank == 0),
                     wait_for_workers=False)

    cfg = CrawlerConfig(
        crawl_id="dist-crawl", storage_root=f"{args.storage}/r{rank}",
        sampling_method="snowball", max_depth=args.max_depth,
        max_pages=args.max_pages, min_users=1, skip_media_download=True,
    )
    feed = SyntheticFeed(FeedConfig(seed=2026, universe=args.universe,
                                    posts_per_channel=args.posts))
    sm = LocalStateManager(cfg)
    if args.fake_engine:
        eng = _FakeEngine(feed, args.posts)
        device = None
    else:
        eng = GpuCrawlEngine(cfg, sm, feed, device=f"cuda:{local_rank}",
                             posts_per_channel=args.posts,
                             chunk_channels=256)
        device = torch.device("cuda", local_rank)
    crawl = OrchestratedCrawl(
        cfg, sm, store, rank, world,
        process_fn=(
            (lambda names: eng.process_channels(names))
            if (args.fake_engine
                or os.environ.get("CRAWL_DIST_STR_NAMES")) else
            (lambda names: eng.process_channels(names, as_arrays=True,
                                                drain=False))
        ),
        flush_fn=(None if args.fake_engine else eng.drain_spills),
        chunk_channels=args.chunk_channels,
        dist=dist if world > 1 else _SoloDist(),
        device=device,
        deadends_fn=(None if args.fake_engine
                     else (lambda: eng.last_deadends)),
    )
    seeds = [feed.username_of(i) for i in range(args.seeds)]
    t0 = time.perf_counter()
    stats = crawl.run(seeds)
    elapsed = time.perf_counter() - t0
    stats.update(eng.stats)
    if rank == 0:
        print(json.dumps({
            "metric": "distributed engine posts/sec (rank 0 share)",
            "elapsed_s": round(elapsed, 2),
            "world": world,
            **stats,
            "engine_phase_s": {k: round(v, 3) for k, v in
                               getattr(eng, "timings", {}).items()},
        }))
    if world > 1:
        dist.destroy_process_group()


class _SoloDist:
    """world=1 stand-in for torch.distributed collectives."""

    @staticmethod
    def barrier():
        pass

    @staticmethod
    def all_gather(out_list, t):
        out_list[0].copy_(t)


class _FakeEngine:
    """--fake-engine stub hot path: deterministic discovery fan-out so
    a CPU dry run exercises the full launcher wiring (TCPStore queue,
    chunk claiming, gloo collectives, checkpoints) at world>1."""

    def __init__(self, feed, posts):
        self.feed = feed
        self.posts = posts
        self.stats = {"fake_pages": 0}

    def process_channels(self, names):
        discovered = []
        for n in names:
            cid = int(n[1:]) if n[1:].isdigit() else 0
            for k in range(2):
                discovered.append(self.feed.username_of(
                    (cid * 31 + 7 * (k + 1)) % self.feed.cfg.universe))
        self.stats["fake_pages"] += len(names)
        return discovered, len(names) * self.posts


if __name__ == "__main__":
    main()
