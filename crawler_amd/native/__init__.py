"""Native runtime pieces (C++, built in-tree).

fanout_native: thread-pool per-channel JSONL spill writer used by the
GPU crawl engine's storage fan-out (the runtime-native counterpart of
the reference's Go writer path, state/daprstate.go:1106-1248). Built by
`python -m crawler_amd.ops.build` alongside the HIP kernels; plain C++,
no GPU dependency, so it also serves the CPU engine.
"""
from __future__ import annotations

_sink_mod = None
_err = None


def load():
    """Import the built extension; raises ImportError if not built."""
    global _sink_mod, _err
    if _sink_mod is not None:
        return _sink_mod
    try:
        from . import fanout_native as m  # built .so next to this file
        _sink_mod = m
        return m
    except ImportError as e:  # pragma: no cover - build missing
        _err = e
        raise


def available() -> bool:
    try:
        load()
        return True
    except ImportError:
        return False
