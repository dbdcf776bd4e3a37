"""Orchestrator: depth-synchronous work distribution + health monitoring.

Parity (reference orchestrator/orchestrator.go):
- BFS depth cursor + work distributor (distributeWork, :182-277): publish
  unfetched pages of the current depth as WorkItems, mark them
  "processing", track assignments
- result handler (handleResultMessage, :315-383): page status updates,
  discovered pages -> AddLayer(depth+1), depth advance when the layer
  drains, completion metadata (:561-593 progress logging)
- worker registry via heartbeats; offline after `worker_timeout`
  (default 5 min) -> that worker's in-flight items republished at high
  priority with retry_count+1 (checkWorkerHealth/reassignWork, :472-559)
"""
from __future__ import annotations

import datetime as _dt
import time
import uuid
from typing import Dict

from ..engine.state import LocalStateManager, Page
from . import messages as M
from .queue import Heartbeats, StoreQueue

DEFAULT_WORKER_TIMEOUT_S = 300.0   # orchestrator.go:498 (5 min)
DISTRIBUTE_INTERVAL_S = 5.0        # orchestrator.go:163
HEALTH_INTERVAL_S = 30.0           # orchestrator.go:475


class Orchestrator:
    def __init__(self, cfg, sm: LocalStateManager, store,
                 worker_timeout_s: float = DEFAULT_WORKER_TIMEOUT_S,
                 clock=time.monotonic):
        self.cfg = cfg
        self.sm = sm
        self.work_q = StoreQueue(store, M.TOPIC_WORK_QUEUE)
        self.result_q = StoreQueue(store, M.TOPIC_RESULTS)
        self.status_q = StoreQueue(store, M.TOPIC_WORKER_STATUS)
        self.control_q = StoreQueue(store, M.TOPIC_ORCHESTRATOR)
        self.heartbeats = Heartbeats(store)
        self.worker_timeout_s = worker_timeout_s
        self.clock = clock
        self.current_depth = 0
        # work_item_id -> (WorkItem, assigned_monotonic)
        self.in_flight: Dict[str, tuple] = {}
        self.page_of_item: Dict[str, str] = {}
        self.done = False
        self.stats = {"distributed": 0, "results": 0, "reassigned": 0,
                      "errors": 0, "retried": 0}

    MAX_RETRIES = 3  # matches RandomWalkStore.MAX_ATTEMPTS poison cap

    # ---- distribution (distributeWork, :182-277) ----

    def distribute(self) -> int:
        layer = self.sm.get_layer_by_depth(self.current_depth)
        n = 0
        for page in layer:
            if page.status != "unfetched":
                continue
            page.status = "processing"
            self.sm.update_page(page)
            item = M.WorkItem(
                id=str(uuid.uuid4()), url=page.url, depth=page.depth,
                crawl_id=self.sm.metadata.crawl_id,
                platform=self.cfg.platform, parent_id=page.parent_id,
                created_at=_dt.datetime.now(_dt.timezone.utc),
                trace_id=M.new_trace_id(),
                config={
                    "max_posts": self.cfg.max_posts,
                    "max_comments": self.cfg.max_comments,
                    "min_users": self.cfg.min_users,
                    "skip_media_download": self.cfg.skip_media_download,
                    "sampling_method": self.cfg.sampling_method,
                },
            )
            item.validate()
            self.work_q.publish(item.to_json())
            self.in_flight[item.id] = (item, self.clock())
            self.page_of_item[item.id] = page.id
            n += 1
        self.stats["distributed"] += n
        return n

    # ---- results (handleResultMessage, :315-383) ----

    def pump_results(self) -> int:
        n = 0
        while True:
            raw = self.result_q.claim(timeout_s=0.0)
            if raw is None:
                break
            res = M.WorkResult.from_json(raw)
            n += 1
            self.stats["results"] += 1
            entry = self.in_flight.pop(res.work_item_id, None)
            if res.status == M.STATUS_RETRY and entry is not None:
                # transient failure (FLOOD_WAIT / timeout): republish at
                # high priority up to MAX_RETRIES, page stays 'processing'
                # (the reference gets this via pub/sub NACK redelivery,
                # worker/worker.go:436 + orchestrator reassignment)
                item = entry[0]
                if item.retry_count < self.MAX_RETRIES:
                    item.retry_count += 1
                    item.priority = M.PRIORITY_HIGH
                    self.work_q.publish(item.to_json())
                    self.in_flight[item.id] = (item, self.clock())
                    self.stats["retried"] += 1
                    continue
            page_id = self.page_of_item.pop(res.work_item_id, None)
            if page_id is not None:
                page = self.sm.get_page(page_id)
                page.status = res.page_status
                page.error = res.error
                self.sm.update_page(page)
            if res.status in (M.STATUS_ERROR, M.STATUS_RETRY):
                self.stats["errors"] += 1
            if (res.discovered
                    and self.cfg.sampling_method == "snowball"
                    and (self.cfg.max_depth < 0
                         or self.current_depth < self.cfg.max_depth)):
                depth = (entry[0].depth if entry else self.current_depth) + 1
                pages = []
                for name in res.discovered:
                    self.sm.add_discovered_channel(name)
                    pages.append(Page(url=name, depth=depth,
                                      status="unfetched"))
                self.sm.add_layer(pages)
        self._maybe_advance()
        return n

    def _maybe_advance(self):
        layer = self.sm.get_layer_by_depth(self.current_depth)
        if not layer:
            self._complete()
            return
        if any(p.status in ("unfetched", "processing") for p in layer):
            return
        nxt = self.sm.get_layer_by_depth(self.current_depth + 1)
        if nxt and (self.cfg.max_depth < 0
                    or self.current_depth < self.cfg.max_depth):
            self.current_depth += 1
            return
        self._complete()

    def _complete(self):
        if not self.done:
            self.done = True
            self.sm.update_crawl_metadata(self.sm.metadata.crawl_id, {
                "status": "completed",
            })
            self.sm.save_state()

    # ---- health (checkWorkerHealth/reassignWork, :472-559) ----

    def check_worker_health(self) -> int:
        # offline workers are tracked for observability; reassignment is
        # driven by in-flight age, which covers both a dead worker and a
        # lost message (the reference republishes a failed worker's items)
        self.offline = set(
            self.heartbeats.offline_workers(self.worker_timeout_s)
        )
        n = 0
        for item_id, (item, t0) in list(self.in_flight.items()):
            if self.clock() - t0 > self.worker_timeout_s:
                item.retry_count += 1
                item.priority = M.PRIORITY_HIGH
                self.work_q.publish(item.to_json())
                self.in_flight[item_id] = (item, self.clock())
                n += 1
        self.stats["reassigned"] += n
        return n

    def broadcast_stop(self, n_workers: int):
        for _ in range(n_workers):
            self.work_q.publish(
                M.WorkItem(id="poison", url="-", crawl_id="-",
                           platform="poison_pill").to_json()
            )

    # ---- drive loop ----

    def run(self, seed_urls, n_workers: int, max_rounds: int = 10_000,
            idle_sleep: float = 0.02) -> dict:
        self.sm.initialize(seed_urls)
        self.sm.save_state()
        rounds = 0
        while not self.done and rounds < max_rounds:
            self.distribute()
            got = self.pump_results()
            self.check_worker_health()
            if not got:
                time.sleep(idle_sleep)
            rounds += 1
        self.broadcast_stop(n_workers)
        # Grace period: hold the store (this process hosts the TCPStore
        # master) until every worker acked with WORKER_STOPPING or the
        # timeout passes — otherwise workers crash on a dead store
        # mid-poll instead of consuming their poison pill.
        deadline = self.clock() + 15.0
        stopping = 0
        while stopping < n_workers and self.clock() < deadline:
            raw = self.status_q.claim(timeout_s=0.2)
            if raw is None:
                continue
            msg = M.StatusMessage.from_json(raw)
            if msg.message_type == M.MSG_WORKER_STOPPING:
                stopping += 1
        return dict(self.stats)


def orchestrator_main(cfg, urls) -> int:
    """CLI entry (--mode orchestrator): hosts the TCPStore and drives the
    crawl; workers join via --mode worker on the same MASTER_ADDR."""
    import os

    from torch.distributed import TCPStore

    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29571"))
    n_workers = int(os.environ.get("CRAWLER_NUM_WORKERS", "1"))
    store = TCPStore(host, port, is_master=True, wait_for_workers=False)
    sm = LocalStateManager(cfg)
    orch = Orchestrator(cfg, sm, store)
    stats = orch.run(urls, n_workers=n_workers)
    print(f"orchestrator complete: {stats}")
    return 0
