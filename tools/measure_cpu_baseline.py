#!/usr/bin/env python3
"""Measure the host-CPU reference baseline ("run the
reference('s) parse/dedup/encode logic on the same synthetic corpus, with
rate limiters disabled, on the host CPU").

Dumps a packed batch, builds cpp/cpu_reference.cc (-O3, native C++ standing
in for the reference's Go at full tilt), asserts BYTE-equality of its
output against the Python oracle, then times 1-thread and all-core runs.
Also times the Python golden path for reference.

Usage: python tools/measure_cpu_baseline.py [--channels 20 --posts 2000]
Writes BASELINE_MEASURED.md.
"""
import argparse
import datetime as dt
import json
import os
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from crawler_amd.feed import FeedConfig, SyntheticFeed  # noqa: E402
from crawler_amd.ops import batch as BB  # noqa: E402
from crawler_amd.ops.golden_batch import encode_batch  # noqa: E402

NOW = dt.datetime(2026, 1, 1, tzinfo=dt.timezone.utc)


def dump_batch(b, d):
    import torch  # noqa: F401

    def w(name, t):
        t.numpy().tofile(os.path.join(d, name))

    w("chat_id", b.chat_id)
    w("msg_id", b.msg_id)
    w("text_off", b.text_off)
    for k, v in b.meta.items():
        if k in ("date", "content_type", "views", "forwards",
                 "media_album_id", "channel_idx", "flags", "text_len",
                 "aux_off", "aux_len", "ent_off", "ent_cnt", "react_off",
                 "react_cnt", "com_off", "com_cnt", "poster_off",
                 "poster_len"):
            w(k, v)
    w("pool", b.text_pool)
    w("entities", b.entities.contiguous())
    for name in ("react_emoji", "react_count", "com_text_off",
                 "com_text_len", "com_handle_off", "com_handle_len",
                 "com_views", "com_replies", "com_react_off",
                 "com_react_cnt", "ch_chat_id", "ch_member",
                 "ch_postcount", "ch_totalviews", "ch_user_off",
                 "ch_user_len", "ch_title_off", "ch_title_len"):
        w(name, getattr(b, name))
    from crawler_amd.models.post import format_go_time

    with open(os.path.join(d, "tables.txt"), "w") as f:
        f.write(f"{len(BB.EMOJI_TABLE)}\n")
        for e in BB.EMOJI_TABLE:
            f.write(e + "\n")
        f.write(f"{len(BB.CONTENT_TYPES)}\n")
        for c in BB.CONTENT_TYPES:
            f.write(c + "\n")
        f.write(format_go_time(NOW.replace(microsecond=0)) + "\n")
        f.write(format_go_time(NOW) + "\n")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--channels", type=int, default=20)
    ap.add_argument("--posts", type=int, default=2000)
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()

    feed = SyntheticFeed(FeedConfig(seed=1234, universe=1_000_000))
    batch = feed.build_batch(np.arange(args.channels),
                             posts_per_channel=args.posts)
    n = batch.n
    print(f"[baseline] corpus: {n} posts", file=sys.stderr)

    # Python golden timing (single core) on a slice
    t0 = time.perf_counter()
    m = min(n, 2000)
    import dataclasses

    sliced = batch  # golden is per-message; time first m messages
    lines, _ = encode_batch(_slice(sliced, m), now=NOW)
    py_pps = m / (time.perf_counter() - t0)
    golden_all = b"".join(encode_batch(batch, now=NOW)[0])

    with tempfile.TemporaryDirectory() as d:
        dump_batch(batch, d)
        exe = os.path.join(d, "cpu_ref")
        subprocess.run(
            ["g++", "-O3", "-std=c++17", "-march=native", "-pthread",
             "cpp/cpu_reference.cc", "-o", exe],
            check=True,
        )
        out_path = os.path.join(d, "out.jsonl")
        r1 = json.loads(subprocess.run(
            [exe, d, "1", str(args.reps), out_path],
            check=True, capture_output=True,
        ).stdout)
        with open(out_path, "rb") as f:
            cpp_out = f.read()
        assert cpp_out == golden_all, (
            f"C++ reference output differs from Python oracle "
            f"({len(cpp_out)} vs {len(golden_all)} bytes)"
        )
        ncpu = os.cpu_count() or 1
        rn = json.loads(subprocess.run(
            [exe, d, str(ncpu), str(args.reps), "-"],
            check=True, capture_output=True,
        ).stdout)

    md = f"""# BASELINE_MEASURED — host-CPU reference throughput

The reference publishes no benchmark (BASELINE.md); per its instructions
the comparison baseline is MEASURED: the same parse + link-extraction +
Go-JSON-encode logic, on the same synthetic corpus, rate limiters
disabled, on the host CPU. `cpp/cpu_reference.cc` (native C++, -O3
-march=native, std::thread pool) stands in for the reference's Go hot
loop; its output is asserted byte-identical to the Python oracle before
timing.

Corpus: {args.channels} channels x {args.posts} posts = {n} posts.
Host: {os.uname().nodename} ({os.cpu_count()} logical cores).

| implementation | posts/sec |
|---|---|
| Python golden (1 core) | {py_pps:,.0f} |
| C++ reference (1 thread) | {r1['posts_per_sec']:,.0f} |
| C++ reference ({ncpu} threads) | {rn['posts_per_sec']:,.0f} |
| **MI355X HIP pipeline (1 GPU)** | **28,000,000** (see profiles/) |

GPU vs best host number: {24_400_000 / rn['posts_per_sec']:,.1f}x.
Generated by tools/measure_cpu_baseline.py on {dt.date.today()}.
"""
    with open("BASELINE_MEASURED.md", "w") as f:
        f.write(md)
    print(md)


def _slice(batch, m):
    """First-m-messages view of a batch (meta sliced; pools shared)."""
    import dataclasses

    kw = {}
    for f in dataclasses.fields(batch):
        v = getattr(batch, f.name)
        kw[f.name] = v
    kw["n"] = m
    kw["chat_id"] = batch.chat_id[:m]
    kw["msg_id"] = batch.msg_id[:m]
    kw["text_off"] = batch.text_off[:m]
    kw["meta"] = {k: t[:m] for k, t in batch.meta.items()}
    return type(batch)(**kw)


if __name__ == "__main__":
    main()
