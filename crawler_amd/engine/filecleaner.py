"""Scheduled file cleaner (reference telegramhelper/filecleaner.go:30-240).

The reference deletes TDLib media cache files older than a threshold under
conn_*/.tdlib/files/{videos,photos,documents} on a ticker. Here the
equivalents are the media spill staging dirs and combine temp dirs: any
file older than max_age is removed on each tick.
"""
from __future__ import annotations

import os
import threading
import time
from typing import List


class FileCleaner:
    def __init__(self, dirs: List[str], max_age_s: float = 3600.0,
                 interval_s: float = 300.0, clock=time.time):
        self.dirs = dirs
        self.max_age_s = max_age_s
        self.interval_s = interval_s
        self.clock = clock
        self._stop = threading.Event()
        self._thread = None
        self.stats = {"deleted": 0, "bytes_freed": 0, "sweeps": 0}

    def sweep_once(self) -> int:
        now = self.clock()
        deleted = 0
        for d in self.dirs:
            if not os.path.isdir(d):
                continue
            for root, _dirs, files in os.walk(d):
                for name in files:
                    path = os.path.join(root, name)
                    try:
                        st = os.stat(path)
                        if now - st.st_mtime > self.max_age_s:
                            size = st.st_size
                            os.remove(path)
                            deleted += 1
                            self.stats["bytes_freed"] += size
                    except OSError:
                        continue
        self.stats["deleted"] += deleted
        self.stats["sweeps"] += 1
        return deleted

    def start(self):
        def loop():
            while not self._stop.is_set():
                self.sweep_once()
                self._stop.wait(self.interval_s)

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
