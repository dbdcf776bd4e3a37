"""Worker: claim work items, run the per-channel pipeline, publish results.

Parity (reference worker/worker.go):
- subscribe + one-item-at-a-time processing (handleWorkMessage, :164-231)
- ProcessWorkItem -> RunForChannelWithPool (:302-381)
- 30s heartbeats (:234-252)
- retryability classification by error kind (shouldRetryError, :436):
  FLOOD_WAIT / transient -> retry, TDLib 400 -> permanent error
"""
from __future__ import annotations

import threading
import time
from typing import Optional

from ..engine import errors as E
from ..engine.pipeline import run_for_channel_with_pool
from ..engine.state import LocalStateManager, Page
from . import messages as M
from .queue import Heartbeats, StoreQueue

HEARTBEAT_INTERVAL_S = 30.0  # worker/worker.go:237


class Worker:
    def __init__(self, worker_id: str, cfg, pool, store,
                 sm: Optional[LocalStateManager] = None):
        self.worker_id = worker_id
        self.cfg = cfg
        self.pool = pool
        self.sm = sm or LocalStateManager(cfg)
        self.work_q = StoreQueue(store, M.TOPIC_WORK_QUEUE)
        self.result_q = StoreQueue(store, M.TOPIC_RESULTS)
        self.status_q = StoreQueue(store, M.TOPIC_WORKER_STATUS)
        self.heartbeats = Heartbeats(store)
        self.processed = 0
        self._stop = threading.Event()
        self._hb_thread = None

    def start_heartbeats(self, interval_s: float = HEARTBEAT_INTERVAL_S):
        self.heartbeats.register(self.worker_id)
        self.heartbeats.beat(self.worker_id, M.WORKER_ACTIVE)

        def loop():
            while not self._stop.is_set():
                try:
                    self.heartbeats.beat(self.worker_id, M.WORKER_ACTIVE)
                except Exception:
                    return  # store gone: main loop is exiting too
                self._stop.wait(interval_s)

        self._hb_thread = threading.Thread(target=loop, daemon=True)
        self._hb_thread.start()

    @staticmethod
    def should_retry_error(err: Exception) -> bool:
        """worker/worker.go:436-456 classification, exactly: permanent
        for "not found"/"access denied"/"forbidden" (and our typed
        TDLib 400), retry for connection/timeout/temporary, and RETRY
        BY DEFAULT for unknown errors (the reference's fall-through)."""
        if isinstance(err, E.TDLib400):
            return False
        msg = str(err).lower()
        if ("not found" in msg or "access denied" in msg
                or "forbidden" in msg):
            return False
        return True

    def _merged_cfg(self, item: M.WorkItem):
        """WorkItemConfig overrides the worker's base config
        (worker.go:302-381 applies the orchestrator-sent config)."""
        import dataclasses as _dc

        cfg = self.cfg
        overrides = {k: v for k, v in (item.config or {}).items()
                     if hasattr(cfg, k) and v not in (None, "", -1)}
        return _dc.replace(cfg, **overrides) if overrides else cfg

    def process_item(self, item: M.WorkItem) -> M.WorkResult:
        """ProcessWorkItem (worker.go:302-381)."""
        t0 = time.perf_counter()
        page = Page(id=item.id, url=item.url, depth=item.depth,
                    parent_id=item.parent_id)
        try:
            res = run_for_channel_with_pool(
                self.pool, page, self.sm, self._merged_cfg(item)
            )
        except Exception as err:  # classified below
            status = (M.STATUS_RETRY if self.should_retry_error(err)
                      else M.STATUS_ERROR)
            return M.WorkResult(
                work_item_id=item.id, worker_id=self.worker_id,
                status=status, error=str(err), page_status="error",
                duration_ms=(time.perf_counter() - t0) * 1000,
                trace_id=item.trace_id,
            )
        self.processed += 1
        return M.WorkResult(
            work_item_id=item.id, worker_id=self.worker_id,
            status=(M.STATUS_SUCCESS if res.status != "error"
                    else M.STATUS_ERROR),
            error=res.error, page_status=res.status,
            posts_stored=res.posts_stored,
            discovered=list(res.discovered),
            duration_ms=(time.perf_counter() - t0) * 1000,
            trace_id=item.trace_id,
        )

    def run_once(self, timeout_s: float = 0.5) -> bool:
        """Claim and process one item. Returns False on poison pill/empty."""
        try:
            raw = self.work_q.claim(timeout_s=timeout_s)
        except Exception:
            # The store master (orchestrator) is gone: the crawl is over
            # — exit cleanly instead of crashing mid-poll (the reference
            # worker likewise dies with its sidecar; we prefer a clean
            # stop so state files close properly).
            return False
        if raw is None:
            return True  # idle, keep polling
        item = M.WorkItem.from_json(raw)
        if item.platform == "poison_pill":
            return False
        self.heartbeats.beat(self.worker_id, M.WORKER_BUSY)
        result = self.process_item(item)
        self.result_q.publish(result.to_json())
        self.heartbeats.beat(self.worker_id, M.WORKER_IDLE)
        return True

    def run(self, max_items: int = 1_000_000):
        self.start_heartbeats()
        self.status_q.publish(M.StatusMessage(
            message_type=M.MSG_WORKER_STARTED, worker_id=self.worker_id,
        ).to_json())
        try:
            for _ in range(max_items):
                if not self.run_once(timeout_s=1.0):
                    break
        finally:
            self._stop.set()
            try:
                self.status_q.publish(M.StatusMessage(
                    message_type=M.MSG_WORKER_STOPPING,
                    worker_id=self.worker_id, processed=self.processed,
                ).to_json())
            except Exception:
                pass  # store may already be gone
            self.sm.close()


def worker_main(cfg, feed) -> int:
    """CLI entry (--mode worker)."""
    import os

    from torch.distributed import TCPStore

    from ..feed.client import ConnectionPool

    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29571"))
    worker_id = (getattr(cfg, "_cli", None) and cfg._cli.worker_id) or \
        f"worker-{os.getpid()}"
    store = TCPStore(host, port, is_master=False)
    pool = ConnectionPool(feed, 2, cfg.rate_limit,
                          disable_rate_limits=cfg.disable_rate_limits)
    w = Worker(worker_id, cfg, pool, store)
    w.run()
    return 0
