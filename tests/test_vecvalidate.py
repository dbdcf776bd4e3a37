"""Vectorized name validation == scalar oracle (engine/vecvalidate.py
vs ops/golden.filter_username + the universe id rule)."""
import numpy as np

from crawler_amd.engine.vecvalidate import decode_names, validate_names
from crawler_amd.ops.golden import filter_username


def _cid_of(username: str, universe: int):
    if username.startswith("c") and username[1:].isdigit():
        cid = int(username[1:])
        if cid < universe:
            return cid
    return None


CASES = [
    # the transcribed reference matrix (username_filter_test.go)
    "testchannel", "test_channel", "channel123", "abcde",
    "abcdefghijklmnopqrstuvwxyz123456",
    "abcd", "a", "", "abcdefghijklmnopqrstuvwxyz1234567",
    "1channel", "_channel", "channel_",
    "test channel", "test-channel", "test.channel",
    "some_bot", "SomeBot", "SomeBOT", "Test_Bot",
    "usr/local", "home~user", "file.name",
    # universe id shapes
    "c0000000001", "c0000999999", "c1000000000", "c123", "c12345",
    "cabcde", "c00000x0001", "c999999999999", "c9999999999999",
    # edge shapes
    "abbot", "robot", "botts", "bot", "xbot5", "a_b_c", "A2345",
]


def test_vector_matches_scalar_oracle_on_cases():
    universe = 1_000_000
    # S40: wider than the 32-byte kernel rows so the too_long case
    # is representable (the hop path regex caps names at 32 anyway)
    arr = np.array(CASES, dtype="S40")
    ok_v, cid_ok_v, cids = validate_names(arr, universe)
    for i, nm in enumerate(CASES):
        ok_s, _ = filter_username(nm)
        cid_s = _cid_of(nm, universe)
        assert bool(ok_v[i]) == ok_s, nm
        assert bool(cid_ok_v[i]) == (cid_s is not None), nm
        if cid_s is not None:
            assert int(cids[i]) == cid_s, nm


def test_vector_matches_scalar_on_random_ascii():
    rng = np.random.default_rng(7)
    alphabet = list("abcXYZ019_/.~- ")
    names = []
    for _ in range(3000):
        n = rng.integers(0, 33)
        names.append("".join(rng.choice(alphabet, size=n)))
    # regex-extracted names never contain NUL; vec path is NUL-padded
    names = [n.replace("\x00", "") for n in names]
    universe = 50_000
    arr = np.array(names, dtype="S32")
    ok_v, cid_ok_v, cids = validate_names(arr, universe)
    for i, nm in enumerate(names):
        ok_s, _ = filter_username(nm)
        assert bool(ok_v[i]) == ok_s, repr(nm)
        cid_s = _cid_of(nm, universe)
        assert bool(cid_ok_v[i]) == (cid_s is not None), repr(nm)


def test_decode_names_roundtrip():
    names = ["alpha", "beta_gamma", "c0000000123", ""]
    arr = np.array(names, dtype="S32")
    assert decode_names(arr) == names
    assert decode_names(arr[:0]) == []


def test_fnv1a64_rows_matches_scalar_oracle():
    from crawler_amd.engine.vecvalidate import fnv1a64_rows
    from crawler_amd.ops.gpu import fnv1a64

    names = ["chan1", "c0000000042", "x", "", "a" * 32]
    arr = np.array(names, dtype="S32")
    got = fnv1a64_rows(arr)
    for i, nm in enumerate(names):
        want = fnv1a64(nm.encode())
        # device/oracle hash is uint64; rows variant returns int64 view
        assert int(got.view(np.uint64)[i]) == want, nm


def test_torch_matches_numpy():
    """validate_names_torch (the device path the GPU random-walk uses)
    decides identically to the numpy oracle, on the curated cases and
    on random ASCII rows (CPU torch here; same code runs on cuda)."""
    import torch

    from crawler_amd.engine.vecvalidate import validate_names_torch

    rng = np.random.default_rng(5)
    rand = []
    for _ in range(4000):
        n = rng.integers(0, 33)
        rand.append(bytes(rng.integers(32, 127, n).astype(np.uint8)))
    for universe in (1, 1000, 10**6, 10**12):
        for pool in (CASES, rand):
            arr = np.array(pool, dtype="S40")
            ok_n, cid_n, val_n = validate_names(arr, universe)
            rows = torch.from_numpy(
                np.ascontiguousarray(arr).view(np.uint8).reshape(
                    len(arr), 40))
            ok_t, cid_t, val_t = validate_names_torch(rows, universe)
            assert (ok_t.numpy() == ok_n).all()
            assert (cid_t.numpy() == cid_n).all()
            assert (val_t.numpy()[cid_n] == val_n[cid_n]).all()
