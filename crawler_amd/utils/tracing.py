"""Tracing hooks (reference §5.1).

- trace_range: roctx/nvtx range context for rocprofv3 --marker-trace
  correlation (replaces the reference's pprof-on-:6060 and per-message
  trace IDs as the kernel-level tracing story);
- latency_class: the reference's cache-vs-server probe
  (DetectCacheOrServer, telegramutils.go:855-879: <5ms local cache,
  >=15ms server) kept for the synthetic client facade.
"""
from __future__ import annotations

import contextlib

try:
    import torch

    _HAVE_NVTX = torch.cuda.is_available()
except Exception:  # pragma: no cover
    _HAVE_NVTX = False


@contextlib.contextmanager
def trace_range(name: str):
    """roctx range when on GPU; no-op otherwise."""
    if _HAVE_NVTX:
        import torch

        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def latency_class(seconds: float) -> str:
    """telegramutils.go:855-879 thresholds."""
    ms = seconds * 1000.0
    if ms < 5.0:
        return "cache"
    if ms >= 15.0:
        return "server"
    return "ambiguous"
