"""GPU aux kernels: segmented channel stats + batched HTML classification."""
import os
import random

import numpy as np
import pytest
import torch

from crawler_amd.engine.htmlvalidator import parse_channel_html
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.tme import MockTMe

pytestmark = pytest.mark.gpu

FIX = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "fixtures", "telegram-html")


def test_channel_stats_matches_numpy():
    from crawler_amd.ops import gpu

    feed = SyntheticFeed(FeedConfig(seed=9, universe=1000))
    batch = feed.build_batch_device(np.arange(16), torch.device("cuda:0"),
                                    posts_per_channel=500)
    out = gpu.channel_stats(batch)
    torch.cuda.synchronize()
    views = batch.meta["views"].cpu().numpy().reshape(16, 500)
    fwd = batch.meta["forwards"].cpu().numpy().reshape(16, 500)
    rep = batch.meta["reply_count"].cpu().numpy().reshape(16, 500)
    assert (out["views"].cpu().numpy()
            == views.sum(axis=1, dtype=np.int64)).all()
    assert (out["forwards"].cpu().numpy()
            == fwd.sum(axis=1, dtype=np.int64)).all()
    assert (out["replies"].cpu().numpy()
            == rep.sum(axis=1, dtype=np.int64)).all()
    assert (out["posts"].cpu().numpy() == 500).all()
    totals = out["totals"].cpu().numpy()
    assert totals[0] == views.sum(dtype=np.int64)
    assert totals[3] == 16 * 500


def test_html_classify_fixtures_and_mock():
    from crawler_amd.ops import gpu

    docs = []
    for name in ("valid-channel.html", "not-a-supergroup.html",
                 "username-not-occupied.html", "invalid-channel.html"):
        with open(os.path.join(FIX, name), "rb") as f:
            docs.append(f.read())
    # plus a spread of mock-universe pages and junk
    tme = MockTMe(universe=500)
    rng = random.Random(7)
    for i in range(120):
        _status, body = tme("c%010d" % rng.randrange(600))
        docs.append(body)
    docs.append(b"<html><head></head><body>no title</body></html>")
    docs.append(b"")

    got = gpu.html_classify(docs)
    torch.cuda.synchronize()
    for i, d in enumerate(docs):
        oracle = parse_channel_html(d)
        assert got[i] == (oracle.status, oracle.reason), (
            f"doc {i}: GPU {got[i]} != oracle "
            f"{(oracle.status, oracle.reason)}"
        )


def test_reservoir_sample_matches_oracle_and_encodes():
    from crawler_amd.ops import batch as B
    from crawler_amd.ops import gpu
    from crawler_amd.ops.golden_batch import encode_batch
    import datetime as dt

    NOW = dt.datetime(2026, 1, 1, tzinfo=dt.timezone.utc)
    feed = SyntheticFeed(FeedConfig(seed=10, universe=400))
    K, P, k = 6, 200, 25
    dev_batch = feed.build_batch_device(np.arange(K),
                                        torch.device("cuda:0"),
                                        posts_per_channel=P)
    idx = gpu.reservoir_sample(dev_batch, k, seed=123)
    torch.cuda.synchronize()
    got = idx.cpu().numpy().tolist()
    expect = gpu.reservoir_sample_oracle(P, K, k, seed=123)
    assert got == expect
    # uniform-without-replacement invariants
    for c in range(K):
        rows = got[c]
        assert len(set(rows)) == k
        assert all(c * P <= r < (c + 1) * P for r in rows)

    # the sampled subset flows through the full encode path byte-exactly
    sub_idx = idx.flatten()
    sub = B.select_rows(dev_batch, sub_idx)
    res = gpu.parse_encode(sub, now=NOW)
    torch.cuda.synchronize()
    cpu_batch = feed.build_batch(np.arange(K), posts_per_channel=P)
    cpu_sub = B.select_rows(cpu_batch, sub_idx.cpu())
    golden_lines, _ = encode_batch(cpu_sub, now=NOW)
    assert bytes(res.out.cpu().numpy()) == b"".join(golden_lines)
