"""Native fan-out sink (crawler_amd/native/fanout_sink.cc): byte-parity
with the pure-Python spill path, LRU fd cap, append semantics, and the
state-manager batch API that the GPU engine calls."""
import os

import numpy as np
import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine.state import LocalStateManager

native = pytest.importorskip("crawler_amd.native.fanout_native")


def mk_buffer(n_channels=20, lines_per=30):
    parts = []
    bounds = []
    pos = 0
    for c in range(n_channels):
        blob = b"".join(
            b'{"ch":%d,"line":%d}\n' % (c, i) for i in range(lines_per)
        )
        parts.append(blob)
        bounds.append((pos, pos + len(blob)))
        pos += len(blob)
    return np.frombuffer(b"".join(parts), dtype=np.uint8), bounds


def test_sink_matches_python_path(tmp_path):
    buf, bounds = mk_buffer()
    channels = [f"chan{c:03d}" for c in range(len(bounds))]

    cfg_n = CrawlerConfig(crawl_id="n1", storage_root=str(tmp_path / "n"))
    sm_n = LocalStateManager(cfg_n)
    sm_n.store_post_lines_batch(
        [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)],
        memoryview(buf),
    )
    sm_n.close()

    cfg_p = CrawlerConfig(crawl_id="n1", storage_root=str(tmp_path / "p"))
    sm_p = LocalStateManager(cfg_p)
    os.environ["CRAWL_NO_NATIVE_SINK"] = "1"
    try:
        sm_p.store_post_lines_batch(
            [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)],
            memoryview(buf),
        )
    finally:
        del os.environ["CRAWL_NO_NATIVE_SINK"]
    sm_p.close()

    for ch in channels:
        a = (tmp_path / "n" / "n1" / ch / "posts" /
             "posts.jsonl").read_bytes()
        b = (tmp_path / "p" / "n1" / ch / "posts" /
             "posts.jsonl").read_bytes()
        assert a == b and a


def test_sink_appends_across_batches(tmp_path):
    buf, bounds = mk_buffer(n_channels=4)
    channels = [f"chan{c}" for c in range(4)]
    cfg = CrawlerConfig(crawl_id="n2", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    items = [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)]
    sm.store_post_lines_batch(items, memoryview(buf))
    sm.store_post_lines_batch(items, memoryview(buf))
    sm.close()
    for ch, (lo, hi) in zip(channels, bounds):
        data = (tmp_path / "n2" / ch / "posts" / "posts.jsonl").read_bytes()
        assert data == bytes(buf[lo:hi]) * 2


def test_sink_respects_truncate(tmp_path):
    buf, bounds = mk_buffer(n_channels=2)
    channels = ["chanA", "chanB"]
    cfg = CrawlerConfig(crawl_id="n3", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    items = [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)]
    sm.store_post_lines_batch(items, memoryview(buf))
    for ch in channels:
        sm.truncate_posts(ch)
    sm.store_post_lines_batch(items, memoryview(buf))
    sm.close()
    for ch, (lo, hi) in zip(channels, bounds):
        data = (tmp_path / "n3" / ch / "posts" / "posts.jsonl").read_bytes()
        assert data == bytes(buf[lo:hi])  # one copy after truncate


def test_sink_mixed_with_python_handles_no_interleave(tmp_path):
    """A channel first written through the buffered Python path then via
    the native sink must keep byte order (handle flushed+dropped)."""
    cfg = CrawlerConfig(crawl_id="n4", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    sm.store_post_lines("chanX", b"first-python\n")
    buf = np.frombuffer(b"then-native\n", dtype=np.uint8)
    sm.store_post_lines_batch([("chanX", 0, len(buf))], memoryview(buf))
    sm.close()
    data = (tmp_path / "n4" / "chanX" / "posts" /
            "posts.jsonl").read_bytes()
    assert data == b"first-python\nthen-native\n"


def test_fd_cap_eviction(tmp_path):
    sink = native.FanoutSink(4, 8)
    buf = np.frombuffer(b"z" * 100, dtype=np.uint8)
    paths = [str(tmp_path / f"f{i}.jsonl") for i in range(30)]
    sink.write_batch(paths, memoryview(buf),
                     [0] * 30, [100] * 30)
    assert sink.open_files <= 8
    assert sink.bytes_written == 3000
    sink.write_batch(paths[:5], memoryview(buf), [0] * 5, [50] * 5)
    sink.close()
    assert os.path.getsize(paths[0]) == 150
    assert os.path.getsize(paths[29]) == 100


def test_bad_slice_raises(tmp_path):
    sink = native.FanoutSink(2, 4)
    buf = np.frombuffer(b"abc", dtype=np.uint8)
    with pytest.raises(Exception):
        sink.write_batch([str(tmp_path / "x")], memoryview(buf),
                         [0], [99])
    sink.close()


def test_nowait_then_drain(tmp_path):
    buf, bounds = mk_buffer(n_channels=6)
    channels = [f"chan{c}" for c in range(6)]
    cfg = CrawlerConfig(crawl_id="n5", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    items = [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)]
    sm.store_post_lines_batch(items, memoryview(buf), nowait=True)
    sm.drain_post_writes()
    sm.store_post_lines_batch(items, memoryview(buf), nowait=True)
    sm.drain_post_writes()
    sm.close()
    for ch, (lo, hi) in zip(channels, bounds):
        data = (tmp_path / "n5" / ch / "posts" / "posts.jsonl").read_bytes()
        assert data == bytes(buf[lo:hi]) * 2


def test_open_failure_surfaces_at_barrier(tmp_path):
    """Opens happen in the WORKER pool (parallel metadata ops), so a
    path that cannot be opened (here: a directory) surfaces as a write
    failure at the drain/wait barrier, not at enqueue."""
    sink = native.FanoutSink(2, 4)
    buf = np.frombuffer(b"x" * 10, dtype=np.uint8)
    ro_dir = tmp_path / "ro"
    ro_dir.mkdir()
    sink.write_batch_nowait([str(ro_dir)], memoryview(buf), [0], [10])
    with pytest.raises(Exception):
        sink.drain()
    sink.close()


def test_ticket_open_failure_surfaces_at_wait(tmp_path):
    sink = native.FanoutSink(2, 4)
    buf = np.frombuffer(b"x" * 10, dtype=np.uint8)
    ro_dir = tmp_path / "ro2"
    ro_dir.mkdir()
    t = sink.write_batch_ticket([str(ro_dir)], memoryview(buf), [0], [10])
    with pytest.raises(Exception):
        sink.wait_ticket(t)
    sink.close()


def test_ticketed_batches_independent_waits(tmp_path):
    sink = native.FanoutSink(3, 16)
    bufs = [np.frombuffer(bytes([65 + k]) * 1000, dtype=np.uint8)
            for k in range(4)]
    tickets = []
    for k in range(4):
        tickets.append(sink.write_batch_ticket(
            [str(tmp_path / f"t{k}.jsonl")], memoryview(bufs[k]),
            [0], [1000]))
    # waits may happen out of order
    sink.wait_ticket(tickets[2])
    sink.wait_ticket(tickets[0])
    sink.wait_ticket(tickets[3])
    sink.wait_ticket(tickets[1])
    sink.close()
    for k in range(4):
        assert (tmp_path / f"t{k}.jsonl").read_bytes() == \
            bytes([65 + k]) * 1000


def test_state_ticket_api_roundtrip(tmp_path):
    buf, bounds = mk_buffer(n_channels=3)
    channels = ["chanA", "chanB", "chanC"]
    cfg = CrawlerConfig(crawl_id="n6", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    items = [(ch, lo, hi) for ch, (lo, hi) in zip(channels, bounds)]
    t1 = sm.store_post_lines_batch(items, memoryview(buf), ticket=True)
    t2 = sm.store_post_lines_batch(items, memoryview(buf), ticket=True)
    sm.wait_post_write(t1)
    sm.wait_post_write(t2)
    sm.wait_post_write(None)  # no-op
    sm.close()
    for ch, (lo, hi) in zip(channels, bounds):
        data = (tmp_path / "n6" / ch / "posts" / "posts.jsonl").read_bytes()
        assert data == bytes(buf[lo:hi]) * 2


def test_tsan_stress_harness(tmp_path):
    """Compile FanoutCore's stress harness under ThreadSanitizer and run
    it (the sanitizer gate the reference lacks). Any data
    race fails the run (halt_on_error)."""
    import shutil
    import subprocess

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    src = os.path.join(root, "tools", "tsan_sink_stress.cc")
    exe = str(tmp_path / "tsan_sink")
    if shutil.which("g++") is None:
        pytest.skip("no g++")
    build = subprocess.run(
        ["g++", "-fsanitize=thread", "-O1", "-g", "-std=c++17",
         "-pthread", src, "-o", exe],
        capture_output=True, text=True, timeout=120,
    )
    if build.returncode != 0 and "tsan" in build.stderr.lower():
        pytest.skip("libtsan unavailable")
    assert build.returncode == 0, build.stderr
    run = subprocess.run(
        [exe, str(tmp_path / "out")],
        env={**os.environ, "TSAN_OPTIONS": "halt_on_error=1"},
        capture_output=True, text=True, timeout=120,
    )
    assert run.returncode == 0, run.stdout + run.stderr
    assert "tsan-stress OK" in run.stdout
    assert "ThreadSanitizer" not in run.stderr


def test_ticketed_pipeline_evicts_fds(tmp_path):
    """A long ticketed pipeline must not accumulate open fds past the
    cap (EMFILE regression: the GPU random-walk opens ~1k new channel
    files per hop and only drains at crawl end)."""
    from crawler_amd.native import load

    sink = load().FanoutSink(4, 32)  # cap at 32 open files
    data = b"x" * 64
    last = None
    for hop in range(8):
        paths = [str(tmp_path / f"h{hop}" / f"c{k}" / "posts.jsonl")
                 for k in range(64)]
        t = sink.write_batch_ticket(paths, data,
                                    [0] * len(paths),
                                    [len(data)] * len(paths))
        if last is not None:
            sink.wait_ticket(last)
            assert sink.open_files <= 32 + 64  # cap + one in-flight batch
        last = t
    sink.wait_ticket(last)
    sink.drain()
    assert sink.open_files <= 32


def test_write_failures_surface_with_errno(tmp_path):
    """Unwritable targets raise with the captured errno instead of
    silently dropping data (the EMFILE incident made errno part of the
    contract)."""
    import os

    import pytest

    from crawler_amd.native import load

    sink = load().FanoutSink(2, 8)
    # a FILE in the parent position defeats even root (ENOTDIR)
    blocker = tmp_path / "blocker"
    blocker.write_text("file, not a directory")
    data = b"x" * 16
    with pytest.raises(RuntimeError, match=r"errno=\d+"):
        sink.write_batch([str(blocker / "sub" / "posts.jsonl")],
                         data, [0], [len(data)])
    sink.drain()
    assert os.path.exists(blocker)
