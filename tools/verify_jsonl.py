#!/usr/bin/env python3
"""Verify a synthetic-feed crawl's JSONL output against the CPU oracle.

Operator tool: re-derives the expected byte stream for each channel in
a crawl directory (crawlID/channel/posts/posts.jsonl) from the
deterministic feed + ops/golden_batch and byte-compares. Catches any
divergence between what a (GPU or CPU) crawl wrote and the reference
semantics — without needing a GPU.

Usage:
  python tools/verify_jsonl.py --crawl-dir /path/storage/crawlID \
      --seed 1234 --universe 1000000 --posts 2000 --now 2026-01-01T00:00:00

The `--now` must match the crawl's capture time (the GPU engine stamps
one `now` per process_channels call; crawls driven through bench or the
engine tests use fixed stamps). Channels whose posts count differs from
--posts (date filters, resume truncation) are reported, not compared.
"""
import argparse
import datetime as dt
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from crawler_amd.feed import FeedConfig, SyntheticFeed  # noqa: E402
from crawler_amd.ops.golden_batch import encode_batch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--crawl-dir", required=True)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--universe", type=int, default=1_000_000)
    ap.add_argument("--posts", type=int, required=True)
    ap.add_argument("--now", required=True,
                    help="capture time, ISO (UTC assumed)")
    ap.add_argument("--limit", type=int, default=0,
                    help="verify at most N channels (0 = all)")
    args = ap.parse_args()

    now = dt.datetime.fromisoformat(args.now)
    if now.tzinfo is None:
        now = now.replace(tzinfo=dt.timezone.utc)
    feed = SyntheticFeed(FeedConfig(seed=args.seed,
                                    universe=args.universe))

    channels = sorted(
        d for d in os.listdir(args.crawl_dir)
        if os.path.isfile(os.path.join(args.crawl_dir, d, "posts",
                                       "posts.jsonl"))
        and d.startswith("c") and d[1:].isdigit()
    )
    if args.limit:
        channels = channels[: args.limit]
    ok = bad = skipped = 0
    for ch in channels:
        path = os.path.join(args.crawl_dir, ch, "posts", "posts.jsonl")
        got = open(path, "rb").read()
        cid = int(ch[1:])
        batch = feed.build_batch(np.array([cid]),
                                 posts_per_channel=args.posts)
        lines, _ = encode_batch(batch, now=now)
        expect = b"".join(lines)
        n_got = got.count(b"\n")
        if n_got != len(lines):
            skipped += 1
            print(f"SKIP {ch}: {n_got} lines vs {len(lines)} expected "
                  "(filtered/partial crawl?)")
            continue
        if got == expect:
            ok += 1
        else:
            bad += 1
            # locate first divergence
            i = next((k for k in range(min(len(got), len(expect)))
                      if got[k] != expect[k]), min(len(got), len(expect)))
            print(f"FAIL {ch}: first divergence at byte {i}")
    print(f"verified={ok} failed={bad} skipped={skipped} "
          f"of {len(channels)} channels")
    return 1 if bad else 0


if __name__ == "__main__":
    sys.exit(main())
