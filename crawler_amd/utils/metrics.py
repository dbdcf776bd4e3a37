"""Counter-log metrics (reference §5.5: counter logs as metrics —
SearchPublicChat hit/miss every 100 lookups crawl/runner.go:1040-1079,
layer statistics standalone/runner.go:862-882, orchestrator progress
every 30s orchestrator.go:561-593). Emits the BASELINE-native metrics:
posts/sec and p50 channel latency."""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional


class Counter:
    def __init__(self, name: str, log_every: int = 0, logger=None):
        self.name = name
        self.value = 0
        self.log_every = log_every
        self.logger = logger
        self._lock = threading.Lock()

    def inc(self, n: int = 1) -> int:
        with self._lock:
            self.value += n
            v = self.value
        if self.log_every and self.logger and v % self.log_every == 0:
            self.logger.info(f"{self.name}={v}", log_tag="metrics")
        return v


class LatencyTracker:
    """Windowed latency percentiles (p50 channel latency is a BASELINE
    headline metric)."""

    def __init__(self, window: int = 1024):
        self.window = window
        self.samples: List[float] = []
        self._lock = threading.Lock()

    def observe(self, seconds: float):
        with self._lock:
            self.samples.append(seconds)
            if len(self.samples) > self.window:
                self.samples = self.samples[-self.window:]

    def percentile(self, q: float) -> Optional[float]:
        with self._lock:
            if not self.samples:
                return None
            s = sorted(self.samples)
            idx = min(len(s) - 1, int(q / 100.0 * len(s)))
            return s[idx]

    def p50_ms(self) -> Optional[float]:
        p = self.percentile(50)
        return None if p is None else p * 1000.0

    def time(self):
        tracker = self

        class _Ctx:
            def __enter__(self):
                self.t0 = time.perf_counter()
                return self

            def __exit__(self, *a):
                tracker.observe(time.perf_counter() - self.t0)

        return _Ctx()


class MetricsRegistry:
    """posts/sec + latency + arbitrary counters, periodically loggable."""

    def __init__(self, logger=None):
        self.t0 = time.monotonic()
        self.posts = Counter("posts")
        self.pages = Counter("pages")
        self.errors = Counter("errors")
        self.channel_latency = LatencyTracker()
        self.counters: Dict[str, Counter] = {}
        self.logger = logger

    def counter(self, name: str, log_every: int = 0) -> Counter:
        if name not in self.counters:
            self.counters[name] = Counter(name, log_every, self.logger)
        return self.counters[name]

    def posts_per_sec(self) -> float:
        dt = time.monotonic() - self.t0
        return self.posts.value / dt if dt > 0 else 0.0

    def snapshot(self) -> dict:
        out = {
            "posts": self.posts.value,
            "pages": self.pages.value,
            "errors": self.errors.value,
            "posts_per_sec": round(self.posts_per_sec(), 1),
            "p50_channel_latency_ms": self.channel_latency.p50_ms(),
        }
        out.update({k: c.value for k, c in self.counters.items()})
        return out
