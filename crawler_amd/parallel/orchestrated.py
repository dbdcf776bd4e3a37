"""Orchestrated multi-rank crawl: work-queue fan-out + collective discovery.

BASELINE config #3 ("orchestrator + 8 workers on RCCL, snowball, seen-set
all-reduce over xGMI") in its MI355X-native shape:

- control plane: rank 0 publishes each BFS layer's channels as chunked
  WorkItems on a TCPStore queue (the Dapr work-queue equivalent,
  orchestrator.go:182-277); ALL ranks — including rank 0 — claim chunks
  dynamically, so load balances even when channels vary in cost;
- data plane: every rank runs the per-chunk hot path (on the GPU via
  GpuCrawlEngine.process_channels) and the newly-claimed discovery names
  are all-gathered per layer (RCCL over xGMI on GPU groups; gloo in CPU
  tests) — the seen-set union of SURVEY §2.6;
- results flow back on a results queue for rank 0's bookkeeping
  (orchestrator.go:315-383); rank 0 owns the persisted state/checkpoints,
  every rank writes its own JSONL shard;
- layer synchronization is depth-barriered exactly like the reference's
  depth-synchronous distribution; every rank derives the next layer
  deterministically from the same gathered name set.

Heartbeats remain for observability; a rank that dies mid-layer parts the
collective (RCCL semantics), which the launcher surfaces — reassignment
at the work-queue level only helps before the layer barrier, matching the
reference's at-least-once republish model.
"""
from __future__ import annotations

import uuid
from typing import Callable, List, Tuple

import numpy as np

from ..engine.state import LocalStateManager, Page
from . import collectives as C
from . import messages as M
from .queue import Heartbeats, StoreQueue


class OrchestratedCrawl:
    def __init__(self, cfg, sm: LocalStateManager, store, rank: int,
                 world: int,
                 process_fn: Callable[[List[str]], Tuple[List[str], int]],
                 chunk_channels: int = 64, dist=None, device=None,
                 deadends_fn: Callable[[], set] = None,
                 flush_fn: Callable[[], None] = None):
        """process_fn(names) -> (discovered_names, posts_stored).
        `dist` is torch.distributed (injected so CPU tests can pass gloo
        and unit tests can fake it); `device` is the compute device used
        for collective buffers when the backend is NCCL/RCCL;
        `deadends_fn` (optional) reports the channels the LAST
        process_fn call classified as deadends, so page statuses and
        the MaxPages deadend-replacement budget (state/base.go:284)
        stay correct — and identical across ranks via the exchange.
        `flush_fn` (optional) is called once per layer after all
        chunks complete, BEFORE statuses/checkpoint — engines that
        pipeline disk writes across chunks (process_channels
        drain=False) drain them here."""
        self.cfg = cfg
        self.sm = sm
        self.rank = rank
        self.world = world
        self.process_fn = process_fn
        self.chunk = chunk_channels
        if dist is None:
            import torch.distributed as dist  # noqa: PLC0415

        self.dist = dist
        self.device = device
        self.deadends_fn = deadends_fn
        self.flush_fn = flush_fn
        self.work_q = StoreQueue(store, M.TOPIC_WORK_QUEUE)
        self.result_q = StoreQueue(store, M.TOPIC_RESULTS)
        self.heartbeats = Heartbeats(store)
        self.stats = {"chunks": 0, "pages": 0, "posts": 0,
                      "discovered": 0, "layers": 0}

    # ---- name exchange (count-sized fixed-width all-gather, placed on
    # the backend's device: RCCL wants device tensors, gloo wants CPU) ----

    def _allgather_names(self, names: List[str], width: int = 32
                         ) -> List[str]:
        rows = C.names_to_rows(names, width)
        out_rows = C.allgather_rows(rows, self.dist, self.world,
                                    device=self.device)
        return C.rows_to_names(out_rows)

    # ---- one layer ----

    def _run_layer(self, names: List[str], depth: int) -> List[str]:
        dist = self.dist
        self.heartbeats.register(f"rank{self.rank}")
        self.heartbeats.beat(f"rank{self.rank}", M.WORKER_ACTIVE)
        if self.rank == 0:
            for i in range(0, len(names), self.chunk):
                item = M.WorkItem(
                    id=str(uuid.uuid4()), url=f"chunk-{depth}-{i}",
                    depth=depth, crawl_id=self.sm.metadata.crawl_id,
                    trace_id=M.new_trace_id(),
                    config={"channels": names[i:i + self.chunk]},
                )
                item.validate()
                self.work_q.publish(item.to_json())
        dist.barrier()  # queue fully published before claims start

        my_discovered: List = []   # str names, or uint8[N,32] arrays
        rows_mode = False
        my_dead: set = set()
        while True:
            raw = self.work_q.claim(timeout_s=0.0)
            if raw is None:
                break
            item = M.WorkItem.from_json(raw)
            chans = item.config["channels"]
            discovered, posts = self.process_fn(chans)
            if self.deadends_fn is not None:
                my_dead |= set(self.deadends_fn())
            if isinstance(discovered, np.ndarray):
                # rows mode (engine as_arrays=True): discovery names
                # stay zero-padded uint8[N, 32] rows end to end — the
                # exchange, dedup and budget below never build ~1M
                # python strings (only the admitted next layer decodes)
                rows_mode = True
                my_discovered.append(discovered)
            else:
                my_discovered.extend(discovered)
            self.stats["chunks"] += 1
            self.stats["pages"] += len(chans)
            self.stats["posts"] += posts
            # bookkeeping only: the next layer comes from the all-gather,
            # so publish the COUNT, not ~1M names as JSON
            self.result_q.publish(M.WorkResult(
                work_item_id=item.id, worker_id=f"rank{self.rank}",
                posts_stored=posts, discovered_count=len(discovered),
                trace_id=item.trace_id,
            ).to_json())
        self.heartbeats.beat(f"rank{self.rank}", M.WORKER_IDLE)
        if self.flush_fn is not None:
            self.flush_fn()   # drain pipelined disk writes (layer safe point)
        all_dead = (set(self._allgather_names(sorted(my_dead)))
                    if self.deadends_fn is not None else set())
        if rows_mode:
            import torch

            rows_np = (np.concatenate(my_discovered)
                       if my_discovered
                       else np.zeros((0, 32), dtype=np.uint8))
            out = C.allgather_rows(
                torch.from_numpy(np.ascontiguousarray(rows_np)),
                self.dist, self.world, device=self.device)
            return out.cpu().numpy(), all_dead
        return self._allgather_names(my_discovered), all_dead

    # ---- the crawl ----

    def run(self, seed_urls: List[str]) -> dict:
        sm = self.sm
        sm.initialize(seed_urls)
        depth = 0
        while True:
            layer = [p for p in sm.get_layer_by_depth(depth)
                     if p.status == "unfetched"]
            if not layer:
                break
            names = [p.url for p in layer]
            all_discovered, all_dead = self._run_layer(names, depth)
            self.stats["layers"] += 1
            self.stats["discovered"] += len(all_discovered)
            for p in layer:
                p.status = "deadend" if p.url in all_dead else "fetched"
                sm.update_page(p)
            if self.rank == 0:
                # drain results for bookkeeping (page counts / errors)
                self.result_q.drain()
            if (self.cfg.sampling_method == "snowball"
                    and (self.cfg.max_depth < 0
                         or depth < self.cfg.max_depth)
                    and len(all_discovered)):
                # every rank derives the SAME next layer (sorted set ->
                # deterministic admission order). Pre-apply add_layer's
                # URL-dedup + MaxPages budget before building Page
                # objects (same rule as gpu_runner.run) — deterministic,
                # so all ranks still agree.
                max_pages = getattr(self.cfg, "max_pages", 0) or 0
                budget = None
                if max_pages > 0:
                    total = len(sm.pages)
                    deadends = sum(1 for p in sm.pages.values()
                                   if p.status == "deadend")
                    budget = max(0, max_pages - total) + deadends
                existing = {p.url for p in sm.pages.values()}
                url_dedup = getattr(sm, "url_dedup", {})
                if isinstance(all_discovered, np.ndarray):
                    # rows mode: sorted-unique on S32 views (ASCII byte
                    # order == str sort order), vectorized membership,
                    # decode ONLY the admitted budget-capped slice
                    un = np.unique(np.ascontiguousarray(all_discovered)
                                   .view("S32").ravel())
                    known = {u.encode() for u in existing}
                    known.update(u.encode() for u in url_dedup)
                    if known:
                        karr = np.array(sorted(known), dtype="S32")
                        pos = np.searchsorted(karr, un)
                        pos_c = np.clip(pos, 0, len(karr) - 1)
                        un = un[karr[pos_c] != un]
                    if budget is not None:
                        un = un[:budget]
                    cand = [b.decode() for b in un.tolist()]
                else:
                    cand = []
                    for n in sorted(set(all_discovered)):
                        if budget is not None and len(cand) >= budget:
                            break
                        if n in existing or n in url_dedup:
                            continue
                        existing.add(n)
                        cand.append(n)
                pages = [Page(url=n, depth=depth + 1, status="unfetched")
                         for n in cand]
                sm.add_layer(pages)
            if self.rank == 0:
                sm.save_state()
            if self.cfg.sampling_method == "channel":
                break
            if self.cfg.max_depth >= 0 and depth >= self.cfg.max_depth:
                break
            depth += 1
        sm.update_crawl_metadata(sm.metadata.crawl_id,
                                 {"status": "completed"})
        if self.rank == 0:
            sm.save_state()
        sm.close()
        return dict(self.stats)
