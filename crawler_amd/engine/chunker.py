"""Chunker: JSONL file combiner pipeline (combine-files mode).

Parity (reference chunk/main.go):
- temp -> watch dir rename protocol (writers write to temp_dir then move
  into watch_dir so the watcher only sees complete files; chunk/main.go:22-51)
- watcher -> batcher -> consumer pipeline (5 goroutine stages,
  chunk/main.go:105-150; here: scan-based watcher thread + worker thread)
- trigger-size batching with a hard cap no combined file may exceed
  (processBatches, chunk/main.go:292-347; sizes in MiB via config
  combine_trigger_size / combine_hard_cap, main.go:800-801)
- batch timeout flushes partial batches (chunk/main.go:93)
- double-buffered seen-map with rotation so re-scans don't re-combine
  files (rotateMap/shouldRotate, chunk/main.go:454-480)
- VerifyCleanup crash recovery: leftover write-dir files re-uploaded,
  watch-dir files re-batched, temp files reported (chunk/main.go:523-680)
"""
from __future__ import annotations

import os
import threading
import time
import uuid
from typing import Callable, List, Optional


class Chunker:
    def __init__(self, temp_dir: str, watch_dir: str, write_dir: str,
                 upload: Callable[[str], None],
                 trigger_bytes: int, hard_cap_bytes: int,
                 batch_timeout_s: float = 300.0,
                 rotation_interval_s: float = 900.0,
                 clock=time.monotonic):
        assert trigger_bytes <= hard_cap_bytes
        self.temp_dir = temp_dir
        self.watch_dir = watch_dir
        self.write_dir = write_dir
        self.upload = upload
        self.trigger = trigger_bytes
        self.hard_cap = hard_cap_bytes
        self.batch_timeout = batch_timeout_s
        self.rotation_interval = rotation_interval_s
        self.clock = clock
        for d in (temp_dir, watch_dir, write_dir):
            os.makedirs(d, exist_ok=True)
        self._seen_cur = set()
        self._seen_prev = set()
        self._last_rotation = clock()
        self._pending: List[str] = []   # files accumulating toward trigger
        self._pending_bytes = 0
        self._batch_started = None
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"combined_files": 0, "input_files": 0,
                      "bytes_combined": 0, "rotations": 0}

    # ---- writer-side API (state manager writes via the chunker) ----

    def write_temp_then_watch(self, name: str, data: bytes) -> str:
        """The temp->watch rename protocol (chunk/main.go:22-51)."""
        tmp = os.path.join(self.temp_dir, name)
        with open(tmp, "wb") as f:
            f.write(data)
        dst = os.path.join(self.watch_dir, name)
        os.replace(tmp, dst)
        return dst

    # ---- seen-map rotation (chunk/main.go:454-480) ----

    def _seen(self, name: str) -> bool:
        return name in self._seen_cur or name in self._seen_prev

    def _mark_seen(self, name: str):
        self._seen_cur.add(name)

    def _maybe_rotate(self):
        if self.clock() - self._last_rotation >= self.rotation_interval:
            self._seen_prev = self._seen_cur
            self._seen_cur = set()
            self._last_rotation = self.clock()
            self.stats["rotations"] += 1

    # ---- batching (chunk/main.go:292-347) ----

    def scan_once(self) -> int:
        """One watcher pass: pick up unseen watch-dir files into the
        pending batch; flush on trigger/timeout. Returns new files seen."""
        with self._lock:
            self._maybe_rotate()
            new = 0
            for name in sorted(os.listdir(self.watch_dir)):
                path = os.path.join(self.watch_dir, name)
                if not os.path.isfile(path) or self._seen(name):
                    continue
                self._mark_seen(name)
                size = os.path.getsize(path)
                if self._pending_bytes + size > self.hard_cap and self._pending:
                    self._flush_locked()
                self._pending.append(path)
                self._pending_bytes += size
                if self._batch_started is None:
                    self._batch_started = self.clock()
                new += 1
                if self._pending_bytes >= self.trigger:
                    self._flush_locked()
            if (self._pending and self._batch_started is not None
                    and self.clock() - self._batch_started
                    >= self.batch_timeout):
                self._flush_locked()
            return new

    def flush(self):
        with self._lock:
            if self._pending:
                self._flush_locked()

    def _flush_locked(self):
        paths = self._pending
        self._pending = []
        self._pending_bytes = 0
        self._batch_started = None
        if not paths:
            return
        out_name = f"combined-{int(time.time())}-{uuid.uuid4().hex[:8]}.jsonl"
        out_path = os.path.join(self.write_dir, out_name)
        total = 0
        with open(out_path, "wb") as out:
            for p in paths:
                try:
                    with open(p, "rb") as f:
                        data = f.read()
                except OSError:
                    continue
                out.write(data)
                total += len(data)
        for p in paths:
            try:
                os.remove(p)
            except OSError:
                pass
        self.upload(out_path)
        try:
            os.remove(out_path)
        except OSError:
            pass
        self.stats["combined_files"] += 1
        self.stats["input_files"] += len(paths)
        self.stats["bytes_combined"] += total

    # ---- lifecycle ----

    def start(self, poll_interval: float = 0.5):
        def loop():
            while not self._stop.is_set():
                self.scan_once()
                self._stop.wait(poll_interval)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=10)
        # final sweep: pick up files the poll loop had not seen yet, then
        # flush the partial batch
        self.scan_once()
        self.flush()

    def verify_cleanup(self) -> dict:
        """Crash recovery (chunk/main.go:523-680): re-upload leftover
        combined files, re-batch watch-dir files, report temp leftovers."""
        report = {"reuploaded": 0, "rebatched": 0, "temp_leftovers": 0}
        for name in sorted(os.listdir(self.write_dir)):
            path = os.path.join(self.write_dir, name)
            if os.path.isfile(path):
                self.upload(path)
                os.remove(path)
                report["reuploaded"] += 1
        with self._lock:
            self._seen_cur.clear()
            self._seen_prev.clear()
        report["rebatched"] = self.scan_once()
        self.flush()
        report["temp_leftovers"] = sum(
            1 for n in os.listdir(self.temp_dir)
            if os.path.isfile(os.path.join(self.temp_dir, n))
        )
        return report
