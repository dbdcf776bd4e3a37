"""Tandem validator: pending-edge HTTP validation + walkback processing.

Parity (reference crawl/validator.go):
- edge claim loop: RunValidationLoop goroutine A (validator.go:79-183),
  ClaimPendingEdges batches, per-edge outcome classification
  (validateSingleEdge, validator.go:194-310): cached-invalid / duplicate
  fast paths, rate-limited HTTP validate, valid -> ClaimDiscoveredChannel
  exactly-once + seed upsert, not_channel/invalid -> invalid cache,
  blocked/transient -> edge left pending;
- IP-block state machine (validator.go:34-38, 112-169): 5 consecutive
  blocked outcomes -> blocked state + access_event("ip_blocked"); canary
  probe of t.me/telegram on an interval until access returns;
- walkback batch processor (validator.go:314-487): claim closed+drained
  batch, decide walkback vs forward over valid first-claimed channels,
  write page_buffer + primary/skipped edge records, complete batch, flush
  per-source-type stats; stale/orphan recovery at startup.
"""
from __future__ import annotations

import random
import time
import uuid
from typing import Callable, Optional

from ..feed.tme import MockTMe
from . import errors as E
from .htmlvalidator import (
    ERR_BLOCKED,
    ValidationHTTPError,
    validate_channel_http,
)
from .randomwalk import pick_walkback_channel
from .state import EdgeRecord, Page, RandomWalkStore

BLOCKED_THRESHOLD = 5          # validator.go:36
PROBE_INTERVAL_S = 300.0       # validator.go:37 (5 min)


class TandemValidator:
    def __init__(self, cfg, sm, rw: RandomWalkStore,
                 fetcher: Optional[Callable] = None,
                 validate_fn: Optional[Callable] = None,
                 rng=None, clock=time.monotonic,
                 probe_interval: float = PROBE_INTERVAL_S,
                 rate_limiter=None):
        self.cfg = cfg
        self.sm = sm
        self.rw = rw
        self.fetcher = fetcher
        self.validate_fn = validate_fn or (
            lambda username: validate_channel_http(username, self.fetcher)
        )
        self.rng = rng or random.Random()
        self.clock = clock
        self.probe_interval = probe_interval
        self.rate_limiter = rate_limiter  # TokenBucket or None
        self.consecutive_blocked = 0
        self.blocked = False
        self.last_probe = 0.0
        self.stats = {"validated": 0, "valid": 0, "invalid": 0,
                      "duplicate": 0, "not_channel": 0, "batches": 0,
                      "blocked_events": 0}

    # ---- startup recovery (dapr/standalone.go:288-306) ----

    def recover(self):
        self.rw.recover_stale_claims()
        self.rw.delete_orphan_edges()

    # ---- blocked-state machine ----

    def _record_blocked(self):
        self.consecutive_blocked += 1
        if (not self.blocked
                and self.consecutive_blocked >= BLOCKED_THRESHOLD):
            self.blocked = True
            self.stats["blocked_events"] += 1
            self.rw.insert_access_event("ip_blocked", "validator paused")
            self.last_probe = self.clock()

    def _maybe_probe(self) -> bool:
        """In blocked state: canary-probe t.me/telegram on the interval.
        Returns True when access has returned."""
        if not self.blocked:
            return True
        if self.clock() - self.last_probe < self.probe_interval:
            return False
        self.last_probe = self.clock()
        try:
            res = self.validate_fn("telegram")
        except ValidationHTTPError:
            return False
        if res.status == "valid":
            self.blocked = False
            self.consecutive_blocked = 0
            self.rw.insert_access_event("ip_unblocked", "validator resumed")
            return True
        return False

    # ---- edge validation (goroutine A) ----

    def _validate_one(self, edge):
        """validateSingleEdge (validator.go:194-310).
        Returns (status, reason, outcome) where outcome in
        definitive|transient|blocked."""
        ch = edge.destination_channel
        if self.rw.is_invalid_channel(ch):
            return "invalid", "cached_invalid", "definitive"
        if ch in self.rw.discovered_channels:
            return "duplicate", "", "definitive"
        if self.rate_limiter is not None:
            self.rate_limiter.acquire()
        try:
            res = self.validate_fn(ch)
        except ValidationHTTPError as err:
            if err.kind == ERR_BLOCKED:
                return "pending", "", "blocked"
            return "pending", "", "transient"
        if res.status == "valid":
            claimed = self.rw.claim_discovered_channel(ch, edge.crawl_id)
            if not claimed:
                return "duplicate", "", "definitive"
            self.sm.add_discovered_channel(ch)
            self.sm.upsert_seed_channel_chat_id(ch, 0)
            self.rw.upsert_seed_channel(ch, 0)
            return "valid", "", "definitive"
        if res.status == "not_channel":
            self.rw.mark_invalid_channel(ch)
            return "not_channel", res.reason, "definitive"
        self.rw.mark_invalid_channel(ch)
        return "invalid", res.reason, "definitive"

    def pump_edges(self) -> int:
        """One claim+validate round. Returns edges processed."""
        if not self._maybe_probe():
            return 0
        edges = self.rw.claim_pending_edges(
            self.cfg.validator_claim_batch_size
        )
        if not edges:
            return 0
        updates = []
        for e in edges:
            status, reason, outcome = self._validate_one(e)
            updates.append((e.pending_id, status, reason))
            if outcome == "blocked":
                self._record_blocked()
            else:
                self.consecutive_blocked = 0
            if outcome == "definitive":
                self.stats["validated"] += 1
                if status in self.stats:
                    self.stats[status] += 1
        self.rw.update_pending_edges(updates)
        return len(edges)

    # ---- walkback processing (goroutine B) ----

    def pump_walkback(self) -> bool:
        """Claim one closed+drained batch and process it
        (processWalkbackBatch, validator.go:360-487)."""
        batch = self.rw.claim_walkback_batch()
        if batch is None:
            return False
        all_edges = self.rw.edges_of_batch(batch.batch_id)
        valid_first = [e.destination_channel for e in all_edges
                       if e.validation_status == "valid"]

        walkback = False
        if not valid_first:
            walkback = True
        else:
            rnd = self.rng.randint(1, 100)
            if self.cfg.walkback_rate >= rnd:
                walkback = True

        if walkback:
            exclude = set(valid_first)
            try:
                next_url = pick_walkback_channel(
                    self.sm, batch.source_channel, exclude, self.rng
                )
            except E.WalkbackExhausted:
                # leave batch processing; stale recovery will retry
                return False
            sequence_id = batch.sequence_id
            page_seq = str(uuid.uuid4())
        else:
            idx = self.rng.randrange(len(valid_first))
            next_url = valid_first.pop(idx)
            sequence_id = batch.sequence_id
            page_seq = batch.sequence_id

        page = Page(
            id=str(uuid.uuid4()), parent_id=batch.source_page_id,
            depth=batch.source_depth + 1, url=next_url,
            sequence_id=page_seq, status="unfetched",
            crawl_id=batch.crawl_id,
        )
        self.rw.add_page(page)
        records = [EdgeRecord(
            destination_channel=next_url,
            source_channel=batch.source_channel, walkback=walkback,
            skipped=False, sequence_id=sequence_id,
            crawl_id=batch.crawl_id,
        )]
        for ch in valid_first:
            records.append(EdgeRecord(
                destination_channel=ch,
                source_channel=batch.source_channel, walkback=False,
                skipped=True, sequence_id=batch.sequence_id,
                crawl_id=batch.crawl_id,
            ))
        self.rw.save_edge_records(records)
        self.rw.complete_batch(batch.batch_id)
        # flush per-source-type stats then drop the batch's edges
        counts = {}
        for e in all_edges:
            if e.validation_status == "valid":
                counts[e.source_type] = counts.get(e.source_type, 0) + 1
        self.rw.flush_batch_stats(batch.crawl_id, counts)
        with self.rw._lock:
            for e in list(self.rw.pending_edges.values()):
                if e.batch_id == batch.batch_id:
                    del self.rw.pending_edges[e.pending_id]
        self.stats["batches"] += 1
        return True

    def run(self, max_rounds: int = 1000) -> dict:
        """Drive both pumps until idle (tests/CLI; production uses threads
        the same way the reference uses an errgroup)."""
        self.recover()
        idle = 0
        for _ in range(max_rounds):
            n = self.pump_edges()
            b = self.pump_walkback()
            if n == 0 and not b:
                idle += 1
                if idle >= 2:
                    break
            else:
                idle = 0
        return dict(self.stats)


def run_validation_loop(cfg, sm, rw, fetcher=None, **kw):
    """CLI entry (crawl/validator.go:53 RunValidationLoop)."""
    fetcher = fetcher or MockTMe(universe=1_000_000)
    v = TandemValidator(cfg, sm, rw, fetcher=fetcher, **kw)
    return v.run()
