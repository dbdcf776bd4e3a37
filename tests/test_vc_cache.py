"""Unit tests for the random-walk cross-hop validation cache
(GpuRandomWalk._vc_*): pure-numpy two-level sorted cache with
sorted-query probing and O(n) disjoint-key merges. CPU-only — the
methods never touch the device, so the object is built via __new__."""
import numpy as np
import pytest

from crawler_amd.engine.gpu_randomwalk import GpuRandomWalk


def make_cache():
    g = GpuRandomWalk.__new__(GpuRandomWalk)
    g._vc = [[np.zeros(0, dtype=np.int64),
              np.zeros(0, dtype=object),
              np.zeros(0, dtype=bool)] for _ in range(2)]
    return g


def ref_lookup(ref, hashes):
    names = np.empty(len(hashes), dtype=object)
    adm = np.zeros(len(hashes), dtype=bool)
    miss = np.ones(len(hashes), dtype=bool)
    for i, h in enumerate(hashes):
        if h in ref:
            names[i], adm[i] = ref[h]
            miss[i] = False
    return names, adm, miss


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_vc_cache_matches_dict_reference(seed):
    rng = np.random.default_rng(seed)
    g = make_cache()
    ref = {}
    for hop in range(12):
        q = rng.integers(0, 5000, size=400).astype(np.int64)
        names = np.empty(len(q), dtype=object)
        adm = np.zeros(len(q), dtype=bool)
        miss = g._vc_lookup(q, names, adm)
        rn, ra, rmiss = ref_lookup(ref, q)
        assert (miss == rmiss).all(), f"hop {hop} miss mask"
        hit = ~miss
        assert all(names[i] == rn[i] for i in np.flatnonzero(hit))
        assert (adm[hit] == ra[hit]).all()
        # insert the unique misses exactly like _hop does
        if miss.any():
            mh, first = np.unique(q[miss], return_index=True)
            mn = np.array([f"name{h}" for h in mh], dtype=object)
            ma = (mh % 3 == 0)
            g._vc_insert(mh, mn, ma)
            for h, n, a in zip(mh, mn, ma):
                ref[h] = (n, a)
    # cache levels stay sorted and disjoint
    for lvl in g._vc:
        assert (np.diff(lvl[0]) > 0).all()
    inter = np.intersect1d(g._vc[0][0], g._vc[1][0])
    assert len(inter) == 0


def test_merge_sorted_interleave():
    a = [np.array([1, 4, 9], dtype=np.int64),
         np.array(["a", "d", "i"], dtype=object),
         np.array([True, False, True])]
    b = [np.array([2, 3, 10], dtype=np.int64),
         np.array(["b", "c", "j"], dtype=object),
         np.array([False, True, False])]
    m = GpuRandomWalk._merge_sorted(a, b)
    assert m[0].tolist() == [1, 2, 3, 4, 9, 10]
    assert m[1].tolist() == ["a", "b", "c", "d", "i", "j"]
    assert m[2].tolist() == [True, False, True, False, True, False]


def test_merge_sorted_empty_sides():
    e = [np.zeros(0, dtype=np.int64), np.zeros(0, dtype=object),
         np.zeros(0, dtype=bool)]
    b = [np.array([5], dtype=np.int64), np.array(["x"], dtype=object),
         np.array([True])]
    assert GpuRandomWalk._merge_sorted(e, b)[0].tolist() == [5]
    assert GpuRandomWalk._merge_sorted(b, e)[0].tolist() == [5]
    assert GpuRandomWalk._merge_sorted(e, e)[0].size == 0


def test_vc_main_spill_preserves_entries():
    g = make_cache()
    # force several pending->main spills past the 1<<18 floor by
    # shrinking the threshold: insert enough to trigger len>262144
    total = {}
    rng = np.random.default_rng(7)
    for _ in range(3):
        mh = np.unique(rng.integers(0, 2**40, size=150_000))
        mn = np.array([str(h) for h in mh], dtype=object)
        ma = (mh % 2 == 0)
        # drop keys already cached (insert contract: misses only)
        seen = np.concatenate([g._vc[0][0], g._vc[1][0]])
        fresh = ~np.isin(mh, seen)
        g._vc_insert(mh[fresh], mn[fresh], ma[fresh])
        for h, n, a in zip(mh[fresh], mn[fresh], ma[fresh]):
            total[h] = (n, a)
    q = np.array(sorted(total)[:5000], dtype=np.int64)
    names = np.empty(len(q), dtype=object)
    adm = np.zeros(len(q), dtype=bool)
    miss = g._vc_lookup(q, names, adm)
    assert not miss.any()
    assert all(names[i] == total[int(h)][0] for i, h in enumerate(q))


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
