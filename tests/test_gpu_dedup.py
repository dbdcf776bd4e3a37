"""GPU seen-set (hash claim + bloom) tests vs Python set semantics."""
import datetime as dt

import numpy as np
import pytest
import torch

from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.ops.golden_batch import encode_batch

pytestmark = pytest.mark.gpu

NOW = dt.datetime(2026, 1, 1, tzinfo=dt.timezone.utc)


def test_claim_exactly_once_within_and_across_batches():
    from crawler_amd.ops import gpu

    feed = SyntheticFeed(FeedConfig(seed=5, universe=300))
    seen = gpu.SeenSet(torch.device("cuda:0"), slots_log2=16)
    py_seen = set()
    total_gpu_new = 0
    total_py_new = 0
    for trial in range(3):
        batch = feed.build_batch(np.arange(trial * 5, trial * 5 + 5),
                                 posts_per_channel=100)
        _, golden_links = encode_batch(batch, now=NOW)
        res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
        new_mask = seen.claim(res)
        torch.cuda.synchronize()
        total_gpu_new += seen.new_count()
        for row in golden_links:
            for (name, _src) in row:
                if name not in py_seen:
                    py_seen.add(name)
                    total_py_new += 1
    assert total_gpu_new == total_py_new
    assert total_gpu_new > 0


def test_new_mask_marks_first_claim_only():
    from crawler_amd.ops import gpu

    feed = SyntheticFeed(FeedConfig(seed=6, universe=50))
    batch = feed.build_batch(np.arange(3), posts_per_channel=200)
    _, golden_links = encode_batch(batch, now=NOW)
    seen = gpu.SeenSet(torch.device("cuda:0"), slots_log2=14)
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    new_mask = seen.claim(res).cpu().numpy()
    torch.cuda.synchronize()
    # Count of set bits == number of distinct names in the batch
    distinct = {name for row in golden_links for (name, _s) in row}
    assert new_mask.sum() == len(distinct)
    # claiming the same batch again yields zero new
    res2 = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    mask2 = seen.claim(res2)
    torch.cuda.synchronize()
    assert int(mask2.sum().item()) == 0
    assert seen.new_count() == 0


def test_insert_hashes_prevents_future_claims():
    from crawler_amd.ops import gpu

    feed = SyntheticFeed(FeedConfig(seed=7, universe=40))
    batch = feed.build_batch(np.arange(2), posts_per_channel=150)
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    torch.cuda.synchronize()

    seen_a = gpu.SeenSet(torch.device("cuda:0"), slots_log2=14)
    mask_a = seen_a.claim(res)
    torch.cuda.synchronize()
    flat = mask_a.flatten().bool()
    new_hashes = res.link_hash.flatten()[flat]

    # simulate a remote rank merging these and then claiming the same batch
    seen_b = gpu.SeenSet(torch.device("cuda:0"), slots_log2=14)
    seen_b.insert_hashes(new_hashes)
    mask_b = seen_b.claim(res)
    torch.cuda.synchronize()
    assert int(mask_b.sum().item()) == 0


def test_fnv_hash_matches_python_oracle():
    from crawler_amd.ops import gpu

    feed = SyntheticFeed(FeedConfig(seed=8, universe=60))
    batch = feed.build_batch(np.arange(2), posts_per_channel=100)
    res = gpu.parse_encode(batch.to("cuda:0"), now=NOW)
    torch.cuda.synchronize()
    dev_links = gpu.links_to_python(res)
    hashes = res.link_hash.cpu().numpy().astype(np.uint64)
    checked = 0
    for i, row in enumerate(dev_links):
        for k, (name, _src) in enumerate(row):
            assert int(hashes[i, k]) == gpu.fnv1a64(name.encode()), name
            checked += 1
    assert checked > 0


def test_compact_claimed_matches_mask_gather():
    """claim_compact_kernel densification == host nonzero+gather over the
    same new_mask (set equality; compact order is atomic)."""
    from crawler_amd.ops import gpu as gpu_mod

    feed = SyntheticFeed(FeedConfig(seed=5, universe=1000))
    batch = feed.build_batch(np.arange(300, 340), posts_per_channel=100)
    res = gpu_mod.parse_encode(batch.to("cuda:0"), now=NOW)
    seen = gpu_mod.SeenSet(torch.device("cuda:0"))
    new_mask = seen.claim(res)
    torch.cuda.synchronize()
    names_c, hashes_c = seen.compact_claimed(res, new_mask)
    # host reference gather
    nz = new_mask.nonzero()
    rows, cols = nz[:, 0], nz[:, 1]
    names_h = res.link_name[rows, cols].cpu().numpy()
    lens_h = res.link_len[rows, cols].cpu().numpy()
    ref = {bytes(names_h[i][: lens_h[i]]) for i in range(len(lens_h))}
    got_rows = names_c.cpu().numpy()
    got = {bytes(r).rstrip(b"\0") for r in got_rows}
    assert got == ref
    assert names_c.shape[0] == len(lens_h)
    ref_h = set(res.link_hash[rows, cols].cpu().tolist())
    assert set(hashes_c.cpu().tolist()) == ref_h
    # second claim of the same batch claims nothing
    mask2 = seen.claim(res)
    n2, _ = seen.compact_claimed(res, mask2)
    assert n2.shape[0] == 0


def test_merge_remote_exchange_on_device():
    """SeenSet.merge_remote end-to-end on the GPU with a fake 2-rank
    dist: remote hashes insert into the table (blocking later claims)
    and the bloom union runs without device/backend crashes (the
    world>1 path the driver's multi-GPU run exercises for real)."""
    from crawler_amd.ops import gpu

    gpu.require_lib()
    feed = SyntheticFeed(FeedConfig(seed=77, universe=20_000))
    batch = feed.build_batch(np.arange(4), posts_per_channel=64)
    res = gpu.parse_encode(batch.to("cuda:0"),
                           now=dt.datetime(2026, 1, 1,
                                           tzinfo=dt.timezone.utc))
    seen = gpu.SeenSet(torch.device("cuda:0"))
    new_mask = seen.claim(res)
    torch.cuda.synchronize()
    _names, my_hashes = seen.compact_claimed(res, new_mask)

    remote = torch.tensor([987654321987, 123456789123],
                          dtype=torch.int64)

    class Fake2Dist:
        """rank 0 of 2; 'rank 1' contributes `remote`."""

        @staticmethod
        def get_backend(group=None):
            return "gloo"

        @staticmethod
        def get_rank(group=None):
            return 0

        @staticmethod
        def all_gather(out_list, t):
            out_list[0].copy_(t)
            if t.dtype == torch.int64 and t.numel() >= remote.numel():
                out_list[1].zero_()
                out_list[1][:remote.numel()] = remote
            else:
                out_list[1].copy_(t)

        @staticmethod
        def all_reduce(t, op=None):
            pass

        class ReduceOp:
            BOR = "bor"

    bloom_before = seen.bloom.clone()
    inserted = seen.merge_remote(my_hashes, Fake2Dist, 2)
    torch.cuda.synchronize()
    assert inserted == remote.numel()
    # the remote hashes now occupy the table: inserting them again via
    # the claim-side bulk insert is a no-op, and our own claims remain
    seen.insert_hashes(remote.to("cuda:0"))
    mask2 = seen.claim(res)
    n2, _ = seen.compact_claimed(res, mask2)
    assert n2.shape[0] == 0  # nothing re-claims
    # bloom unchanged by the (identity) gloo union but still valid
    assert torch.equal(seen.bloom, bloom_before)
