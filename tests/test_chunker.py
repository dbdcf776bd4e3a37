"""Chunker pipeline tests (reference chunk/main_test.go coverage areas:
temp->watch protocol, trigger/hard-cap batching, rotation gating,
timeout flush, VerifyCleanup crash recovery)."""
import os

import pytest

from crawler_amd.engine.chunker import Chunker


class FakeClock:
    def __init__(self):
        self.t = 0.0

    def __call__(self):
        return self.t


@pytest.fixture
def env(tmp_path):
    uploads = []
    clock = FakeClock()
    ch = Chunker(
        str(tmp_path / "temp"), str(tmp_path / "watch"),
        str(tmp_path / "write"),
        upload=lambda p: uploads.append(open(p, "rb").read()),
        trigger_bytes=100, hard_cap_bytes=150, batch_timeout_s=60,
        rotation_interval_s=900, clock=clock,
    )
    return ch, uploads, clock


def test_temp_then_watch_protocol(env, tmp_path):
    ch, uploads, clock = env
    path = ch.write_temp_then_watch("a.jsonl", b"x" * 10)
    assert os.path.dirname(path).endswith("watch")
    assert not os.listdir(tmp_path / "temp")


def test_trigger_size_combines(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 60)
    ch.write_temp_then_watch("b.jsonl", b"b" * 60)
    ch.scan_once()
    assert len(uploads) == 1
    assert uploads[0] == b"a" * 60 + b"b" * 60
    assert ch.stats["input_files"] == 2


def test_hard_cap_never_exceeded(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 90)
    ch.write_temp_then_watch("b.jsonl", b"b" * 90)
    ch.write_temp_then_watch("c.jsonl", b"c" * 90)
    ch.scan_once()
    ch.flush()
    assert all(len(u) <= 150 for u in uploads)
    assert sum(len(u) for u in uploads) == 270


def test_below_trigger_waits_then_timeout_flush(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 10)
    ch.scan_once()
    assert uploads == []
    clock.t += 61
    ch.scan_once()
    assert len(uploads) == 1


def test_files_not_recombined_after_scan(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 120)
    ch.scan_once()
    assert len(uploads) == 1
    # file deleted after combine; but even a lingering name is gated
    ch.write_temp_then_watch("a.jsonl", b"a" * 120)
    ch.scan_once()
    assert len(uploads) == 1  # seen-map gates the same name
    # after two rotations the name ages out of both maps
    clock.t += 901
    ch.scan_once()
    clock.t += 901
    ch.scan_once()
    assert len(uploads) == 2


def test_verify_cleanup_reuploads_and_rebatches(env, tmp_path):
    ch, uploads, clock = env
    # leftover combined file in write dir (crash before upload)
    with open(tmp_path / "write" / "combined-left.jsonl", "wb") as f:
        f.write(b"leftover")
    # unprocessed watch file
    ch.write_temp_then_watch("w.jsonl", b"w" * 20)
    # temp leftover (writer crashed mid-write)
    with open(tmp_path / "temp" / "t.jsonl", "wb") as f:
        f.write(b"partial")
    report = ch.verify_cleanup()
    assert report["reuploaded"] == 1
    assert report["rebatched"] == 1
    assert report["temp_leftovers"] == 1
    assert b"leftover" in uploads
    assert b"w" * 20 in uploads


def test_threaded_start_stop(tmp_path):
    uploads = []
    ch = Chunker(
        str(tmp_path / "t"), str(tmp_path / "w"), str(tmp_path / "o"),
        upload=lambda p: uploads.append(os.path.basename(p)),
        trigger_bytes=10, hard_cap_bytes=100, batch_timeout_s=60,
    )
    ch.start(poll_interval=0.01)
    ch.write_temp_then_watch("x.jsonl", b"y" * 20)
    import time

    for _ in range(100):
        if uploads:
            break
        time.sleep(0.01)
    ch.stop()
    assert uploads
