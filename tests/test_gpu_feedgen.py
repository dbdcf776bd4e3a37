"""Device feed generator (csrc/feedgen.hip) vs the numpy oracle:
every tensor of the packed batch must be bit-identical."""
import dataclasses

import numpy as np
import pytest
import torch

from crawler_amd.feed import FeedConfig, SyntheticFeed

pytestmark = pytest.mark.gpu


def _assert_batches_equal(cpu_b, dev_b):
    for f in dataclasses.fields(cpu_b):
        a = getattr(cpu_b, f.name)
        b = getattr(dev_b, f.name)
        if isinstance(a, dict):
            for k in a:
                bb = b[k].cpu()
                assert torch.equal(a[k], bb), (
                    f"meta[{k}] differs: "
                    f"{(a[k] != bb).nonzero()[:5].flatten().tolist()}"
                )
        elif isinstance(a, torch.Tensor):
            bb = b.cpu()
            assert a.shape == bb.shape, f"{f.name}: {a.shape} vs {bb.shape}"
            assert torch.equal(a, bb), (
                f"{f.name} differs at "
                f"{(a != bb).nonzero()[:5].flatten().tolist()}"
            )
        else:
            assert a == b, f"{f.name}: {a} != {b}"


def test_device_generation_bit_identical():
    feed = SyntheticFeed(FeedConfig(seed=77, universe=100_000))
    ids = np.arange(10, 22)
    cpu_b = feed.build_batch(ids, posts_per_channel=500)
    dev_b = feed.build_batch_device(ids, torch.device("cuda:0"),
                                    posts_per_channel=500)
    torch.cuda.synchronize()
    _assert_batches_equal(cpu_b, dev_b)


def test_device_generation_then_encode_matches_golden():
    import datetime as dt

    from crawler_amd.ops import gpu
    from crawler_amd.ops.golden_batch import encode_batch

    now = dt.datetime(2026, 1, 1, tzinfo=dt.timezone.utc)
    feed = SyntheticFeed(FeedConfig(seed=78, universe=5000))
    ids = np.arange(6)
    cpu_b = feed.build_batch(ids, posts_per_channel=200)
    dev_b = feed.build_batch_device(ids, torch.device("cuda:0"),
                                    posts_per_channel=200)
    golden_lines, _ = encode_batch(cpu_b, now=now)
    res = gpu.parse_encode(dev_b, now=now)
    torch.cuda.synchronize()
    assert bytes(res.out.cpu().numpy()) == b"".join(golden_lines)
