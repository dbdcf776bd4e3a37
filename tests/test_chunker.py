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


def test_rotation_double_buffer_remembers_one_interval(env):
    """A NAME stays 'seen' across ONE rotation (prev buffer) and is
    forgotten after TWO (rotateMap/shouldRotate, chunk/main.go:454-480):
    a same-named file re-appearing within the window is skipped, after
    two rotations it is processed again."""
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"x" * 120)  # >= trigger
    assert ch.scan_once() == 1
    assert len(uploads) == 1                          # combined + deleted
    # same name re-appears: still in the seen map -> skipped
    ch.write_temp_then_watch("a.jsonl", b"y" * 120)
    assert ch.scan_once() == 0
    clock.t += 900
    assert ch.scan_once() == 0          # rotation 1: remembered via prev
    assert ch.stats["rotations"] == 1
    clock.t += 900
    got = ch.scan_once()                # rotation 2: name forgotten
    assert ch.stats["rotations"] == 2
    assert got == 1
    assert len(uploads) == 2 and uploads[1] == b"y" * 120


def test_rotation_does_not_fire_early(env):
    ch, uploads, clock = env
    clock.t += 899.9
    ch.scan_once()
    assert ch.stats["rotations"] == 0


def test_batch_timeout_flushes_partial(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 30)   # below trigger 100
    ch.scan_once()
    assert uploads == []
    clock.t += 59
    ch.scan_once()
    assert uploads == []                 # not yet
    clock.t += 1.5
    ch.scan_once()
    assert len(uploads) == 1 and uploads[0] == b"a" * 30


def test_hard_cap_preflushes_batch(env):
    """Adding a file that would push the batch past the hard cap flushes
    the current batch FIRST; the new file starts the next batch."""
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 90)
    ch.scan_once()                       # 90 < trigger, pending
    ch.write_temp_then_watch("b.jsonl", b"b" * 90)
    ch.scan_once()                       # 90+90 > cap 150 -> flush a first
    assert uploads == [b"a" * 90]
    # b (90 < trigger 100) starts the NEXT batch and stays pending
    ch.flush()
    assert uploads == [b"a" * 90, b"b" * 90]


def test_single_file_over_hard_cap_combined_alone(env):
    ch, uploads, clock = env
    big = b"z" * 400                     # > hard cap 150
    ch.write_temp_then_watch("big.jsonl", big)
    ch.scan_once()
    assert len(uploads) == 1 and uploads[0] == big


def test_trigger_exact_boundary_flushes(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 50)
    ch.write_temp_then_watch("b.jsonl", b"b" * 50)
    ch.scan_once()                       # 50+50 == trigger 100 -> flush
    assert len(uploads) == 1
    assert uploads[0] == b"a" * 50 + b"b" * 50   # sorted name order


def test_stats_accounting(env):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 60)
    ch.write_temp_then_watch("b.jsonl", b"b" * 60)
    ch.scan_once()
    assert ch.stats["combined_files"] == 1
    assert ch.stats["input_files"] == 2
    assert ch.stats["bytes_combined"] == 120


def test_inputs_deleted_after_combine(env, tmp_path):
    ch, uploads, clock = env
    ch.write_temp_then_watch("a.jsonl", b"a" * 120)
    ch.scan_once()
    assert os.listdir(tmp_path / "watch") == []
    assert os.listdir(tmp_path / "write") == []  # combined file removed
