"""Property-based tests (hypothesis) for the byte-exactness invariants.

These complement the fixed-vector suites: hypothesis explores the
input space and shrinks failures. Bounded example counts keep the CPU
suite fast.
"""
import json
import math

import numpy as np
import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from crawler_amd.models.post import _enc_float  # noqa: E402
from crawler_amd.ops.golden import filter_username  # noqa: E402
from crawler_amd.ops.gpu import fnv1a64  # noqa: E402
from crawler_amd.engine import vecvalidate  # noqa: E402

FINITE = st.floats(allow_nan=False, allow_infinity=False)


@settings(max_examples=300, deadline=None)
@given(FINITE)
def test_enc_float_round_trips(v):
    """Go floatEncoder output must parse back to the same float
    (shortest round-trip digits) and follow the format rules."""
    s = _enc_float(v)
    assert float(s) == v or (v == 0 and float(s) == 0)
    av = abs(v)
    if v != int(v) if av < 1e18 else True:
        pass
    if "e" in s:
        # exponent form only outside [1e-6, 1e21); no e-0X form
        assert not (1e-6 <= av < 1e21)
        mant, _, es = s.partition("e")
        # Go's only exponent cleanup is e-0X -> e-X; positive
        # exponents keep their '+' (strconv 'e' format)
        assert not es.startswith("-0"), s
    elif "." in s:
        # fixed form only inside the Go 'f' window
        assert 1e-6 <= av < 1e21
    # json must accept it verbatim
    assert json.loads(s) == pytest.approx(v, abs=0.0)


@settings(max_examples=200, deadline=None)
@given(st.integers(-9007199254740993, 9007199254740993).map(float))
def test_enc_float_integral_prints_bare(v):
    s = _enc_float(v)
    assert "." not in s and "e" not in s
    assert float(s) == v


@settings(max_examples=300, deadline=None)
@given(st.text(
    alphabet=st.characters(min_codepoint=32, max_codepoint=126),
    min_size=0, max_size=40,
))
def test_vectorized_filter_matches_scalar(name):
    """vecvalidate.validate_names == golden.filter_username on
    arbitrary printable-ASCII candidates (the t.me extractor only
    emits [A-Za-z0-9_], but the validator must agree on anything)."""
    b = name.encode("ascii")
    if b"\x00" in b:
        return
    arr = np.array([b], dtype="S48")
    ok_v, _cid_ok, _ = vecvalidate.validate_names(arr, 10)
    ok_s, _reason = filter_username(name)
    assert bool(ok_v[0]) == ok_s, name


@settings(max_examples=200, deadline=None)
@given(st.binary(min_size=0, max_size=32).filter(lambda b: b"\x00" not in b))
def test_fnv_rows_matches_scalar(data):
    """Vectorized fnv1a64_rows == scalar oracle for any NUL-free key
    up to the row width."""
    rows = np.zeros((1, 32), dtype=np.uint8)
    rows[0, :len(data)] = np.frombuffer(data, dtype=np.uint8)
    h = vecvalidate.fnv1a64_rows(rows)[0]
    assert int(np.uint64(h)) == fnv1a64(data)


@settings(max_examples=100, deadline=None)
@given(st.lists(st.integers(0, 2**62), min_size=0, max_size=200,
                unique=True))
def test_vc_table_total_recall(keys):
    """The open-addressing cache returns exactly the inserted adm bits
    for every inserted key and misses for absent ones."""
    from crawler_amd.engine.gpu_randomwalk import GpuRandomWalk

    g = GpuRandomWalk.__new__(GpuRandomWalk)
    g._vc_init(1 << 6)  # tiny: forces growth + collisions
    ks = np.array(keys, dtype=np.int64)
    adm = ks % 3 == 0
    g._vc_insert(ks, adm)
    out = np.zeros(len(ks), dtype=bool)
    miss = g._vc_lookup(ks, out)
    assert not miss.any()
    assert (out == adm).all()
    absent = np.array([k + 2**62 + 1 for k in range(5)], dtype=np.int64)
    absent = absent[~np.isin(absent, ks)]
    if len(absent):
        m2 = g._vc_lookup(absent, np.zeros(len(absent), dtype=bool))
        assert m2.all()
