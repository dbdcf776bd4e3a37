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


def test_debug_server_endpoints():
    """pprof-:6060 analog (reference main.go pprof server)."""
    import json
    import urllib.request

    from crawler_amd.utils.debugserver import DebugServer
    from crawler_amd.utils.metrics import MetricsRegistry

    reg = MetricsRegistry()
    reg.posts.inc(7)
    srv = DebugServer(0, metrics=reg).start()
    try:
        base = f"http://127.0.0.1:{srv.port}"
        assert urllib.request.urlopen(base + "/healthz").read() == b"ok\n"
        snap = json.loads(urllib.request.urlopen(base + "/metrics").read())
        assert snap["posts"] == 7
        stacks = urllib.request.urlopen(base + "/debug/stacks").read()
        assert b"thread" in stacks and b"MainThread" in stacks
        v = json.loads(
            urllib.request.urlopen(base + "/debug/vars").read())
        assert v["pid"] > 0 and "uptime_s" in v
        import urllib.error
        try:
            urllib.request.urlopen(base + "/nope")
            assert False, "expected 404"
        except urllib.error.HTTPError as e:
            assert e.code == 404
    finally:
        srv.stop()


def test_debug_server_maybe_start_disabled():
    from crawler_amd.utils.debugserver import maybe_start

    assert maybe_start(0) is None
    assert maybe_start(None) is None


def test_latency_tracker_window_trims():
    from crawler_amd.utils.metrics import LatencyTracker

    lt = LatencyTracker(window=8)
    for i in range(20):
        lt.observe(float(i))
    assert len(lt.samples) == 8
    assert lt.samples == [float(i) for i in range(12, 20)]
    # p50 over the surviving window only
    assert lt.percentile(50) in (15.0, 16.0)


def test_float_encoding_go_rules():
    """Go encoding/json: NaN/Inf are unsupported values; integral floats
    print bare (model/post.py _enc_float)."""
    from crawler_amd.models.post import _enc_float

    assert _enc_float(2.0) == "2"
    assert _enc_float(0.5) == "0.5"
    assert _enc_float(-3.0) == "-3"
    import pytest

    for bad in (float("nan"), float("inf"), float("-inf")):
        with pytest.raises(ValueError):
            _enc_float(bad)
