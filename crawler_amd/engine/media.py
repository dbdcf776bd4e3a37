"""Media pipeline: HBM-staged synthetic blobs with pinned host spill.

BASELINE config #5 and reference parity:
- fetchAndUploadMedia semantics (tdutils.go:226-358): skip when
  skip_media; dedup via the media cache (HasProcessedMedia /
  MarkMediaAsProcessed, tdutils.go:736-776); 150 MB size cap
  (tdutils.go:293) — over-cap files are fetched-but-not-stored;
- StoreFile storage layout crawlID/media/channel/<name>
  (storageproviders.go StoreFile);
- the Azure-Blob binding becomes a local-FS mock written by a host
  writer thread fed from a pinned bounce ring (the hipMemcpyAsync spill).

Blob bytes are generated deterministically ON DEVICE (a cheap mixing
expression over torch ops — no extra kernel needed), staged in an HBM
arena, and spilled D2H through pinned buffers.
"""
from __future__ import annotations

import os
import queue
import threading
import zlib
from typing import Optional

import torch


def _stable_hash(s: str) -> int:
    return zlib.crc32(s.encode()) & 0xFFFFFFFF

MEDIA_SIZE_CAP_MB = 150.0  # tdutils.go:293


def synth_blob_size(media_id: str, seed: int = 0) -> int:
    """Deterministic blob size: mostly 10KB-2MB thumbnails, occasional
    multi-MB videos, and a rare >150MB to exercise the cap."""
    h = _stable_hash(f"{media_id}:{seed}")
    if h % 997 == 0:
        return 151 * 1024 * 1024 + (h % 4096)  # over-cap
    if h % 11 == 0:
        return 2 * 1024 * 1024 + (h % (6 * 1024 * 1024))  # video-ish
    return 10 * 1024 + (h % (2 * 1024 * 1024))  # thumbnail-ish


_IDX_CACHE = {}  # device -> cached arange (grown on demand)


def device_blob(media_id: str, size: int, device,
                seed: int = 0) -> torch.Tensor:
    """Deterministic device-resident blob (uint8[size])."""
    base = (_stable_hash(f"{media_id}:{seed}:blob") & 0x7FFFFFFF) or 1
    key = str(device)
    idx = _IDX_CACHE.get(key)
    if idx is None or idx.numel() < size:
        idx = torch.arange(max(size, 8 << 20), device=device,
                           dtype=torch.int64)
        _IDX_CACHE[key] = idx
    return ((idx[:size] * 2654435761 + base) >> 7).to(torch.uint8)


class MediaEngine:
    """HBM arena -> pinned ring -> host writer thread -> local blob store."""

    def __init__(self, sm, out_root: Optional[str] = None,
                 device="cuda:0", ring_slots: int = 4,
                 slot_bytes: int = 32 * 1024 * 1024, seed: int = 0,
                 use_gpu: bool = True):
        self.sm = sm
        self.device = torch.device(device) if use_gpu else None
        self.seed = seed
        self.use_gpu = use_gpu
        self.out_root = out_root
        self.stats = {"stored": 0, "deduped": 0, "over_cap": 0,
                      "bytes": 0}
        self._q: "queue.Queue" = queue.Queue(maxsize=ring_slots * 2)
        self._writer = threading.Thread(target=self._write_loop, daemon=True)
        self._writer.start()
        if use_gpu:
            self._ring = [
                torch.empty(slot_bytes, dtype=torch.uint8, pin_memory=True)
                for _ in range(ring_slots)
            ]
            self._ring_free: "queue.Queue" = queue.Queue()
            for r in self._ring:
                self._ring_free.put(r)
            self._slot_bytes = slot_bytes
            self._stream = torch.cuda.Stream()

    # ---- host writer ----

    def _write_loop(self):
        while True:
            item = self._q.get()
            if item is None:
                return
            path, data, ring_buf = item
            os.makedirs(os.path.dirname(path), exist_ok=True)
            with open(path, "wb") as f:
                f.write(data)
            if ring_buf is not None:
                self._ring_free.put(ring_buf)

    def _store_path(self, channel: str, file_name: str) -> str:
        root = self.out_root or self.sm._crawl_dir()
        return os.path.join(root, "media", channel, file_name)

    # ---- the fetch+upload entry (fetchAndUploadMedia semantics) ----

    def fetch_and_upload(self, channel: str, media_id: str,
                         skip_media: bool = False) -> str:
        """Returns the stored path ("" when skipped/deduped/over-cap)."""
        if not media_id:
            return ""
        if skip_media:
            return ""
        if self.sm.has_processed_media(media_id):
            self.stats["deduped"] += 1
            return ""
        size = synth_blob_size(media_id, self.seed)
        if size / (1024.0 * 1024.0) > MEDIA_SIZE_CAP_MB:
            self.stats["over_cap"] += 1
            self.sm.mark_media_as_processed(media_id)
            return ""
        path = self._store_path(channel, media_id + ".bin")
        if self.use_gpu:
            self._spill_gpu(media_id, size, path)
        else:
            # CPU mode: generate host-side with the same expression
            idx = torch.arange(size, dtype=torch.int64)
            base = (_stable_hash(f"{media_id}:{self.seed}:blob")
                    & 0x7FFFFFFF) or 1
            data = (((idx * 2654435761 + base) >> 7)
                    .to(torch.uint8).numpy().tobytes())
            self._q.put((path, data, None))
        self.sm.mark_media_as_processed(media_id)
        self.stats["stored"] += 1
        self.stats["bytes"] += size
        return path

    def _spill_gpu(self, media_id: str, size: int, path: str):
        """Stage in HBM, spill via the pinned ring in slot-sized pieces.

        Issue and drain are PIPELINED: when the ring is exhausted the
        oldest in-flight piece is drained before issuing the next (a
        blocking get() with all slots in flight would deadlock)."""
        import collections

        import numpy as np

        blob = device_blob(media_id, size, self.device, self.seed)
        pending = collections.deque()  # (buf, n, ev, dst_off)
        data = bytearray(size)
        view = np.frombuffer(data, dtype=np.uint8)

        def drain_one():
            buf, n, ev, dst = pending.popleft()
            ev.synchronize()
            # single copy pinned->bytearray (bytes() cost a second one)
            view[dst:dst + n] = buf[:n].numpy()
            self._ring_free.put(buf)

        for off in range(0, size, self._slot_bytes):
            n = min(self._slot_bytes, size - off)
            while self._ring_free.empty() and pending:
                drain_one()
            buf = self._ring_free.get()
            with torch.cuda.stream(self._stream):
                buf[:n].copy_(blob[off:off + n], non_blocking=True)
                ev = torch.cuda.Event()
                ev.record(self._stream)
            pending.append((buf, n, ev, off))
        while pending:
            drain_one()
        # hand the bytearray itself to the writer (no bytes() copy; we
        # never touch it again)
        self._q.put((path, data, None))

    def flush(self):
        import time

        while not self._q.empty():
            time.sleep(0.01)

    def close(self):
        self.flush()
        self._q.put(None)
        self._writer.join(timeout=10)
