"""Execution-mode runners: standalone layer BFS + layerless random walk.

Parity map:
- StandaloneRunner: standalone/runner.go:37-873 (resume via
  FindIncompleteCrawl, seed layer 0, depth loop, per-page pipeline,
  AddLayer of discovered pages, SaveState after pages) with the
  layer-parallel channel pool of dapr/standalone.go:417-689 (P1: semaphore
  of `concurrency` workers per layer, barrier between layers, URL dedup
  within a layer via AddLayer).
- RandomWalkRunner: RunRandomWalkLayerless (dapr/standalone.go:792-946):
  poll page_buffer (limit = max workers), in-flight set, per-page worker,
  delete page after processing; WalkbackExhausted leaves the page in the
  buffer; FloodWaitRetire leaves it and aborts when the pool is empty;
  TDLib400 -> Handle400Replacement + delete page. The frontier IS the
  checkpoint (restart resumes from the buffer).
"""
from __future__ import annotations

import concurrent.futures as cf
import random
import time
import uuid
from typing import Callable, List, Optional

from . import errors as E
from . import pipeline, randomwalk
from .state import LocalStateManager, Page, RandomWalkStore


class StandaloneRunner:
    """channel / snowball sampling over a layer BFS."""

    def __init__(self, cfg, sm: LocalStateManager, pool, rng=None,
                 run_for_channel_fn: Optional[Callable] = None):
        self.cfg = cfg
        self.sm = sm
        self.pool = pool
        self.rng = rng or random.Random()
        # injectable for fault tests (crawl/runner.go:290-294 pattern)
        self.run_for_channel_fn = (
            run_for_channel_fn or pipeline.run_for_channel_with_pool
        )
        self.stats = {"pages": 0, "posts": 0, "errors": 0, "deadends": 0}

    def run(self, seed_urls: List[str], resume: bool = True) -> dict:
        sm = self.sm
        resumed = False
        if resume:
            exec_id, ok = sm.find_incomplete_crawl(sm.metadata.crawl_id)
            if ok and sm.load_state():
                resumed = True
        if not resumed:
            sm.initialize(seed_urls)
            sm.save_state()
        # cross-crawl URL dedup (daprstate.go:550-623); random-walk mode
        # deliberately skips it (duplicate visits allowed)
        if self.cfg.sampling_method != "random-walk":
            sm.load_url_dedup_cache()

        depth = 0
        max_depth = self.cfg.max_depth
        while True:
            layer = [p for p in sm.get_layer_by_depth(depth)
                     if p.status == "unfetched"]
            all_layer = sm.get_layer_by_depth(depth)
            if not all_layer:
                break
            if layer:
                self._process_layer(layer, depth)
                sm.save_state()
            if max_depth >= 0 and depth >= max_depth:
                break
            if self.cfg.sampling_method == "channel":
                break  # channel mode: seeds only, no expansion
            depth += 1
        sm.update_crawl_metadata(sm.metadata.crawl_id, {
            "status": "completed",
            "messagesCount": self.stats["posts"],
            "errorsCount": self.stats["errors"],
        })
        sm.save_state()
        sm.close()
        return dict(self.stats)

    def _process_layer(self, layer: List[Page], depth: int):
        """P1: worker pool with a barrier at layer end
        (dapr/standalone.go:417-689)."""
        sm = self.sm
        discovered: List[Page] = []
        lock = __import__("threading").Lock()

        def work(page: Page):
            page.status = "processing"
            sm.update_page(page)
            try:
                res = self.run_for_channel_fn(
                    self.pool, page, sm, self.cfg, rng=self.rng
                )
            except E.TDLib400 as err:
                page.status = "error"
                page.error = str(err)
                sm.update_page(page)
                self.stats["errors"] += 1
                return
            except E.FloodWaitRetire as err:
                page.status = "unfetched"  # retryable by a later run
                sm.update_page(page)
                if self.pool.empty():
                    raise
                return
            except Exception as err:  # panic containment per page
                # (recoverFromPanic, crawl/runner.go: one bad channel
                # never kills the layer)
                page.status = "error"
                page.error = f"recovered: {err}"
                sm.update_page(page)
                self.stats["errors"] += 1
                return
            page.status = res.status
            page.error = res.error
            sm.update_page(page)
            self.stats["pages"] += 1
            self.stats["posts"] += res.posts_stored
            if res.status == "deadend":
                self.stats["deadends"] += 1
            if (self.cfg.sampling_method == "snowball"
                    and res.status == "fetched"):
                with lock:
                    for name in res.discovered:
                        sm.add_discovered_channel(name)
                        discovered.append(Page(
                            url=name, depth=depth + 1, status="unfetched",
                            parent_id=page.id,
                        ))

        n_workers = max(1, self.cfg.concurrency)
        if n_workers == 1:
            for p in layer:
                work(p)
        else:
            with cf.ThreadPoolExecutor(max_workers=n_workers) as ex:
                list(ex.map(work, layer))
        if discovered:
            sm.add_layer(discovered)


class RandomWalkRunner:
    """Layerless random-walk loop (dapr/standalone.go:792-946)."""

    def __init__(self, cfg, sm: LocalStateManager, rw: RandomWalkStore,
                 pool, rng=None, poll_interval: float = 0.0,
                 run_for_channel_fn: Optional[Callable] = None):
        self.cfg = cfg
        self.sm = sm
        self.rw = rw
        self.pool = pool
        self.rng = rng or random.Random()
        self.poll_interval = poll_interval  # 5s in the reference; 0 in tests
        self.run_for_channel_fn = (
            run_for_channel_fn or pipeline.run_for_channel_with_pool
        )
        self.stats = {"pages": 0, "posts": 0, "errors": 0, "invalid_400": 0,
                      "walkback_exhausted": 0}

    def seed(self, seed_urls: List[str]):
        for u in seed_urls:
            self.sm.add_discovered_channel(u)
            self.rw.upsert_seed_channel(u)
            self.rw.add_page(Page(
                id=str(uuid.uuid4()), url=u, depth=0,
                sequence_id=str(uuid.uuid4()), status="unfetched",
            ))

    def _process_page(self, page: Page) -> bool:
        """Returns True if the page should be deleted from the buffer."""
        cfg = self.cfg

        def rw_outlinks(owner, result, client):
            if cfg.tandem_crawl:
                randomwalk.tandem_tail(
                    owner, result.discovered_links, self.sm, self.rw, cfg,
                    self.rng,
                )
            else:
                new_channels = randomwalk.validate_outlinks(
                    result.discovered_links, owner.url, self.sm, self.rw,
                    client, cfg,
                )
                randomwalk.walk_tail(owner, new_channels, self.sm, self.rw,
                                     cfg, self.rng)

        try:
            res = self.run_for_channel_fn(
                self.pool, page, self.sm, cfg, rw=self.rw,
                mode_hooks={"outlinks": rw_outlinks}, rng=self.rng,
            )
        except E.WalkbackExhausted:
            self.stats["walkback_exhausted"] += 1
            return False  # leave in buffer (standalone.go:902)
        except E.FloodWaitRetire:
            if self.pool.empty():
                raise
            return False  # leave in buffer (standalone.go:905-911)
        except E.TDLib400:
            self.stats["invalid_400"] += 1
            try:
                randomwalk.handle_400_replacement(
                    self.sm, self.rw, page, cfg, self.rng
                )
            except E.WalkbackExhausted:
                self.stats["walkback_exhausted"] += 1
            return True  # delete page (standalone.go:912-921)
        except Exception:  # panic containment: drop the page, keep walking
            self.stats["errors"] += 1
            return True
        self.stats["pages"] += 1
        self.stats["posts"] += res.posts_stored
        if res.status == "error":
            self.stats["errors"] += 1
        self.rw.mark_channel_crawled(
            page.url, 0
        )
        return True

    def run(self, max_pages: Optional[int] = None,
            max_seconds: Optional[float] = None) -> dict:
        deadline = time.monotonic() + max_seconds if max_seconds else None
        cap = max_pages or self.cfg.max_pages
        max_workers = max(1, self.cfg.concurrency)
        in_flight = set()
        blocked_since = None  # tandem circuit breaker state
        while self.stats["pages"] < cap:
            if deadline and time.monotonic() > deadline:
                break
            pages = [p for p in self.rw.get_pages(max_workers)
                     if p.id not in in_flight]
            if not pages:
                if not self.cfg.tandem_crawl:
                    break
                # Tandem: an empty buffer with incomplete batches means we
                # are waiting on the validator; a circuit breaker aborts
                # after validator_timeout of zero progress
                # (dapr/standalone.go:837-867).
                incomplete = self.rw.count_incomplete_batches(
                    self.cfg.crawl_id
                )
                if incomplete == 0:
                    break  # crawl complete
                now_t = time.monotonic()
                if blocked_since is None:
                    blocked_since = now_t
                timeout = self.cfg.validator_timeout_s
                if timeout and now_t - blocked_since > timeout:
                    self.stats["circuit_breaker"] = 1
                    raise E.PoolExhausted(
                        f"validator made no progress for {timeout}s with "
                        f"{incomplete} incomplete batches — aborting "
                        "(circuit breaker)"
                    )
                time.sleep(self.poll_interval or 0.01)
                continue
            blocked_since = None
            for page in pages:
                if self.stats["pages"] >= cap:
                    break
                in_flight.add(page.id)
                try:
                    delete = self._process_page(page)
                finally:
                    in_flight.discard(page.id)
                if delete:
                    self.rw.delete_pages([page.id])
            if self.poll_interval:
                time.sleep(self.poll_interval)
        self.sm.save_state()
        self.sm.close()
        return dict(self.stats)
