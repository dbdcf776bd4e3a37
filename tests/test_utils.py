"""Metrics / logging / tracing utility tests."""
import io

from crawler_amd.utils import (
    Counter,
    LatencyTracker,
    MetricsRegistry,
    latency_class,
    trace_range,
)
from crawler_amd.utils.logging import TaggedLogger


def test_counter_and_registry():
    m = MetricsRegistry()
    m.posts.inc(100)
    m.pages.inc()
    m.counter("search_hits", log_every=0).inc(5)
    snap = m.snapshot()
    assert snap["posts"] == 100
    assert snap["search_hits"] == 5
    assert snap["posts_per_sec"] > 0


def test_latency_percentiles():
    t = LatencyTracker()
    for v in [0.001, 0.002, 0.003, 0.100]:
        t.observe(v)
    assert 1.9 < t.p50_ms() < 3.1
    with t.time():
        pass
    assert len(t.samples) == 5


def test_latency_class_thresholds():
    # telegramutils.go:855-879: <5ms cache, >=15ms server
    assert latency_class(0.001) == "cache"
    assert latency_class(0.020) == "server"
    assert latency_class(0.010) == "ambiguous"


def test_trace_range_noop_on_cpu():
    with trace_range("step"):
        x = 1
    assert x == 1


def test_tagged_logger_json_fields():
    buf = io.StringIO()
    lg = TaggedLogger("test", level="debug", stream=buf)
    lg.info("hello", log_tag="rw_edge", n=3)
    import json

    rec = json.loads(buf.getvalue())
    assert rec["msg"] == "hello" and rec["log_tag"] == "rw_edge"
    # level filtering
    buf2 = io.StringIO()
    lg2 = TaggedLogger("t2", level="error", stream=buf2)
    lg2.info("hidden")
    assert buf2.getvalue() == ""
