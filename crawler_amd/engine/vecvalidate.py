"""Vectorized channel-name validation (numpy, fixed-width byte rows).

Semantics are EXACTLY FilterUsername (username_filter.go:26-81, mirrored
scalar in ops/golden.py:filter_username) plus the synthetic-universe
SearchPublicChat check ("c" + digits < universe). The GPU random-walk
validates ~150k distinct names per hop; the scalar python path costs
~1.5 s/hop — these run in ~10 ms. Equality with the scalar oracle is
pinned by tests/test_vecvalidate.py.
"""
from __future__ import annotations

import numpy as np

# bytes legal inside a username: [0-9A-Za-z_]
_VALID = np.zeros(256, dtype=bool)
_VALID[48:58] = True   # 0-9
_VALID[65:91] = True   # A-Z
_VALID[97:123] = True  # a-z
_VALID[95] = True      # _


def _lower(c: np.ndarray) -> np.ndarray:
    return np.where((c >= 65) & (c <= 90), c + 32, c)


def validate_names(unames: np.ndarray, universe: int):
    """unames: S<w> array of NUL-padded names (ASCII — the t.me link
    regex admits only [A-Za-z0-9_]). Returns (ok, cid_ok, cid):
    ok       bool[N] — FilterUsername passes;
    cid_ok   bool[N] — name decodes to a universe channel id;
    cid      int64[N] — the id (valid where cid_ok).
    """
    n = len(unames)
    if n == 0:
        z = np.zeros(0, dtype=bool)
        return z, z, np.zeros(0, dtype=np.int64)
    w = unames.dtype.itemsize
    U = np.ascontiguousarray(unames).view(np.uint8).reshape(n, w)
    L = (U != 0).sum(1)
    rows = np.arange(n)
    first = U[:, 0]
    is_letter = (((first >= 65) & (first <= 90))
                 | ((first >= 97) & (first <= 122)))
    colm = np.arange(w)[None, :] < L[:, None]
    chars_ok = (_VALID[U] | ~colm).all(1)
    lastpos = np.maximum(L - 1, 0)
    no_trail_us = U[rows, lastpos] != 95
    has3 = L >= 3
    b1 = _lower(U[rows, np.maximum(L - 3, 0)])
    b2 = _lower(U[rows, np.maximum(L - 2, 0)])
    b3 = _lower(U[rows, lastpos])
    bot = has3 & (b1 == 98) & (b2 == 111) & (b3 == 116)  # "bot"
    ok = ((L >= 5) & (L <= 32) & is_letter & chars_ok
          & no_trail_us & ~bot)

    # synthetic SearchPublicChat: "c" + all-digits, value < universe
    dig = (U[:, 1:] >= 48) & (U[:, 1:] <= 57)
    digcols = np.arange(1, w)[None, :] < L[:, None]
    all_dig = (dig | ~digcols).all(1)
    nd = L - 1
    cand = (first == 99) & all_dig & (nd >= 1) & (nd <= 12)
    val = np.zeros(n, dtype=np.int64)
    for j in range(1, min(13, w)):
        has = j < L
        d = U[:, j].astype(np.int64) - 48
        val = np.where(has & cand, val * 10 + d, val)
    cid_ok = cand & (val < universe)
    return ok, cid_ok, val


def validate_names_torch(rows, universe: int):
    """validate_names on a torch uint8 tensor [N, w] (any device).

    Same decisions as validate_names (pinned by
    tests/test_vecvalidate.py::test_torch_matches_numpy); used by the
    GPU random-walk to validate the deduped name rows ON DEVICE, so
    the host miss path only indexes precomputed results instead of
    re-scanning ~150k rows of bytes per hop."""
    import torch

    n, w = rows.shape
    dev = rows.device
    if n == 0:
        z = torch.zeros(0, dtype=torch.bool, device=dev)
        return z, z, torch.zeros(0, dtype=torch.int64, device=dev)
    U = rows.to(torch.int64)
    L = (U != 0).sum(1)
    first = U[:, 0]
    is_letter = (((first >= 65) & (first <= 90))
                 | ((first >= 97) & (first <= 122)))
    valid_lut = torch.from_numpy(_VALID.copy()).to(dev)
    colm = torch.arange(w, device=dev)[None, :] < L[:, None]
    chars_ok = (valid_lut[U] | ~colm).all(1)
    rows_i = torch.arange(n, device=dev)
    lastpos = (L - 1).clamp(min=0)
    no_trail_us = U[rows_i, lastpos] != 95

    def low(c):
        return torch.where((c >= 65) & (c <= 90), c + 32, c)

    has3 = L >= 3
    b1 = low(U[rows_i, (L - 3).clamp(min=0)])
    b2 = low(U[rows_i, (L - 2).clamp(min=0)])
    b3 = low(U[rows_i, lastpos])
    bot = has3 & (b1 == 98) & (b2 == 111) & (b3 == 116)
    ok = ((L >= 5) & (L <= 32) & is_letter & chars_ok
          & no_trail_us & ~bot)

    dig = (U[:, 1:] >= 48) & (U[:, 1:] <= 57)
    digcols = torch.arange(1, w, device=dev)[None, :] < L[:, None]
    all_dig = (dig | ~digcols).all(1)
    nd = L - 1
    cand = (first == 99) & all_dig & (nd >= 1) & (nd <= 12)
    val = torch.zeros(n, dtype=torch.int64, device=dev)
    for j in range(1, min(13, w)):
        has = j < L
        d = U[:, j] - 48
        val = torch.where(has & cand, val * 10 + d, val)
    cid_ok = cand & (val < universe)
    return ok, cid_ok, val


def fnv1a64_rows(rows: np.ndarray) -> np.ndarray:
    """Vectorized FNV-1a 64 over NUL-padded uint8[N, W] name rows —
    matches the device hash (csrc/common.h fnv1a64) and the scalar
    oracle (ops/golden.fnv1a64): hash of the unpadded bytes."""
    if rows.ndim != 2:
        rows = np.ascontiguousarray(rows).view(np.uint8).reshape(
            len(rows), rows.dtype.itemsize)
    n, w = rows.shape
    # NOTE: the repo-wide basis (csrc/common.h, ops/golden.py) is
    # 1469598103934665603 — not the standard FNV offset basis
    h = np.full(n, 1469598103934665603, dtype=np.uint64)
    prime = np.uint64(1099511628211)
    lens = (rows != 0).sum(axis=1)
    for j in range(w):
        live = j < lens
        hj = (h ^ rows[:, j].astype(np.uint64)) * prime
        h = np.where(live, hj, h)
    return h.view(np.int64)


def decode_names(unames: np.ndarray) -> list:
    """One bulk ascii decode of NUL-padded S<w> rows -> python strings."""
    n = len(unames)
    if n == 0:
        return []
    w = unames.dtype.itemsize
    U = np.ascontiguousarray(unames).view(np.uint8).reshape(n, w)
    lens = (U != 0).sum(1)
    blob = unames.tobytes().decode("ascii", "replace")
    return [blob[i * w:i * w + int(l)] for i, l in enumerate(lens)]
