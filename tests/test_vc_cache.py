"""Unit tests for the random-walk cross-hop validation cache
(GpuRandomWalk._vc_*): a pure-numpy open-addressing hash table
(vectorized linear probe, amortized growth). CPU-only — the methods
never touch the device, so the object is built via __new__."""
import numpy as np
import pytest

from crawler_amd.engine.gpu_randomwalk import GpuRandomWalk


def make_cache(slots=1 << 10):
    g = GpuRandomWalk.__new__(GpuRandomWalk)
    g._vc_init(slots)
    return g


def ref_lookup(ref, hashes):
    adm = np.zeros(len(hashes), dtype=bool)
    miss = np.ones(len(hashes), dtype=bool)
    for i, h in enumerate(hashes):
        if h in ref:
            adm[i] = ref[h]
            miss[i] = False
    return adm, miss


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_vc_cache_matches_dict_reference(seed):
    rng = np.random.default_rng(seed)
    g = make_cache()
    ref = {}
    for hop in range(12):
        q = rng.integers(0, 5000, size=400).astype(np.int64)
        adm = np.zeros(len(q), dtype=bool)
        miss = g._vc_lookup(q, adm)
        ra, rmiss = ref_lookup(ref, q)
        assert (miss == rmiss).all(), f"hop {hop} miss mask"
        hit = ~miss
        assert (adm[hit] == ra[hit]).all()
        # insert the unique misses exactly like _hop does
        if miss.any():
            mh, first = np.unique(q[miss], return_index=True)
            ma = (mh % 3 == 0)
            g._vc_insert(mh, ma)
            for h, a in zip(mh, ma):
                ref[h] = a
    # table bookkeeping: entry count matches the reference dict
    assert g._vc_n == len(ref)
    assert (g._vc_keys != g._VC_EMPTY).sum() == len(ref)


def test_vc_collision_chains():
    """Keys engineered to land on the same initial slot still resolve
    (linear probing), both within one insert batch and across
    batches."""
    g = make_cache(slots=1 << 4)
    # craft keys with identical mixed slot: brute-force search
    base_slot = None
    ks = []
    k = 0
    while len(ks) < 6:
        s = g._vc_slots_of(np.array([k], dtype=np.int64))[0]
        if base_slot is None:
            base_slot = s
        if s == base_slot:
            ks.append(k)
        k += 1
    ks = np.array(ks, dtype=np.int64)
    adm = (ks % 2 == 0)
    g._vc_insert(ks[:3], adm[:3])   # same-slot batch
    g._vc_insert(ks[3:], adm[3:])   # cross-batch chain
    out_a = np.zeros(len(ks), dtype=bool)
    miss = g._vc_lookup(ks, out_a)
    assert not miss.any()
    assert (out_a == adm).all()
    # absent key that hashes into the chain is still reported missing
    probe = k
    while g._vc_slots_of(np.array([probe], dtype=np.int64))[0] != base_slot:
        probe += 1
    miss2 = g._vc_lookup(np.array([probe], dtype=np.int64),
                         np.zeros(1, dtype=bool))
    assert miss2.all()


def test_vc_growth_preserves_entries():
    g = make_cache(slots=1 << 10)
    # force several table doublings
    total = {}
    rng = np.random.default_rng(7)
    for _ in range(3):
        mh = np.unique(rng.integers(0, 2**40, size=150_000))
        ma = (mh % 2 == 0)
        # drop keys already cached (insert contract: misses only)
        fresh = g._vc_lookup(mh, np.zeros(len(mh), dtype=bool))
        g._vc_insert(mh[fresh], ma[fresh])
        for h, a in zip(mh[fresh], ma[fresh]):
            total[h] = a
    q = np.array(sorted(total)[:5000], dtype=np.int64)
    adm = np.zeros(len(q), dtype=bool)
    miss = g._vc_lookup(q, adm)
    assert not miss.any()
    assert all(adm[i] == total[int(h)] for i, h in enumerate(q))


def test_inv_snapshot_versioning():
    """The invalid-name snapshot rebuilds only on store-version moves
    (or TTL expiry) and answers membership exactly."""
    import datetime as dt

    from crawler_amd.engine.state import RandomWalkStore

    g = GpuRandomWalk.__new__(GpuRandomWalk)
    g.rw = RandomWalkStore()
    now = dt.datetime.now(dt.timezone.utc)
    a0 = g._inv_snapshot(now, 32)
    assert a0.size == 0
    g.rw.mark_invalid_channel("badchan")
    g.rw.mark_invalid_channel("worse_chan")
    a1 = g._inv_snapshot(now, 32)
    assert sorted(a1.tolist()) == [b"badchan", b"worse_chan"]
    # same version -> cached object reused
    assert g._inv_snapshot(now, 32) is a1
    # membership helper
    q = np.array([b"badchan", b"goodchan", b"worse_chan"], dtype="S32")
    assert GpuRandomWalk._in_sorted(a1, q).tolist() == [True, False, True]
    # expired entries drop on rebuild after the TTL boundary
    g.rw.invalid_channels["badchan"] = now - dt.timedelta(days=31)
    g.rw.mark_invalid_channel("third")  # bump version
    a2 = g._inv_snapshot(now, 32)
    assert b"badchan" not in a2.tolist()
    assert {b"worse_chan", b"third"} <= set(a2.tolist())
    # version unchanged + TTL not crossed -> still cached even though
    # the dict was edited behind the store's back (documented contract)
    g.rw.invalid_channels["sneaky"] = now
    assert g._inv_snapshot(now, 32) is a2


def test_walk_tail_rows_matches_fast():
    """_walk_tail_rows (S32 byte rows, lazy-decoded blocks) makes the
    SAME decisions with the SAME rng consumption as
    randomwalk.walk_tail_fast (str lists) — pages, followed edges and
    materialized skipped edges are identical."""
    import datetime as dt
    import random

    from crawler_amd.config import CrawlerConfig
    from crawler_amd.engine import randomwalk
    from crawler_amd.engine.state import (
        LocalStateManager,
        Page,
        RandomWalkStore,
    )

    now = dt.datetime(2026, 2, 3, tzinfo=dt.timezone.utc)

    def setup(tmp):
        cfg = CrawlerConfig(crawl_id="wt1", storage_root=tmp,
                            walkback_rate=40)
        sm = LocalStateManager(cfg)
        for i in range(30):
            sm.add_discovered_channel("c%010d" % (100 + i))
        rw = RandomWalkStore()
        return cfg, sm, rw

    import tempfile

    name_sets = [
        sorted("c%010d" % v for v in s)
        for s in ([1, 2, 3], [], [7], list(range(20, 40)), [], [5, 9])
    ]
    with tempfile.TemporaryDirectory() as ta, \
            tempfile.TemporaryDirectory() as tb:
        cfg_a, sm_a, rw_a = setup(ta)
        cfg_b, sm_b, rw_b = setup(tb)
        g = GpuRandomWalk.__new__(GpuRandomWalk)
        g.cfg, g.sm, g.rw = cfg_b, sm_b, rw_b
        rng_a = random.Random(77)
        g.rng = random.Random(77)
        pages_a, pages_b = [], []
        for i, names in enumerate(name_sets):
            owner_a = Page(id=f"p{i}", url="c%010d" % i, depth=i,
                           sequence_id=f"sq{i}", status="unfetched")
            owner_b = Page(id=f"p{i}", url="c%010d" % i, depth=i,
                           sequence_id=f"sq{i}", status="unfetched")
            pa = randomwalk.walk_tail_fast(owner_a, list(names), sm_a,
                                           rw_a, cfg_a, rng_a, now)
            rows = np.array(names, dtype="S32")
            pb = g._walk_tail_rows(owner_b, rows, now)
            assert pb.url == pa.url, f"set {i}"
            assert pb.depth == pa.depth
            assert (pb.sequence_id == owner_b.sequence_id) == (
                pa.sequence_id == owner_a.sequence_id)
            pages_a.append(pa)
            pages_b.append(pb)
        ea = [(e.destination_channel, e.source_channel, e.walkback,
               e.skipped, e.sequence_id) for e in rw_a.edge_records]
        eb = [(e.destination_channel, e.source_channel, e.walkback,
               e.skipped, e.sequence_id) for e in rw_b.edge_records]
        assert ea == eb
        assert rw_a.edge_count() == rw_b.edge_count()
