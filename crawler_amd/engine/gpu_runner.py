"""GPU crawl engine: whole-layer batch execution of the per-post hot path.

This is the MI355X-native execution mode of the crawl engine (BASELINE
configs #2/#3): instead of per-message host parsing, a BFS layer's
channels are generated/fetched as one packed device batch and pushed
through the HIP kernels:

    feed (device feedgen or host batch)  ->  parse+encode (JSONL bytes)
      ->  SeenSet.claim (exactly-once discovery, atomicCAS)
      ->  host spill: per-channel JSONL slices appended to the same
          posts.jsonl layout the CPU path writes
      ->  newly-claimed link names (small D2H) become the next layer

Equivalence contract: for the same channels the JSONL bytes are identical
to the CPU pipeline's Post.to_jsonl() output (enforced by the byte-exact
kernel tests) and the discovered-channel set matches golden extraction.

Cross-rank: ranks shard the layer's channels; newly-claimed hashes are
all-gathered and merged into each rank's SeenSet (bench.py does the same
per step) so a channel is crawled by exactly one rank per crawl.
"""
from __future__ import annotations

import datetime as _dt
from typing import List, Optional

import numpy as np
import torch

from ..feed.synth import SyntheticFeed
from ..parallel import collectives as C
from .state import LocalStateManager, Page


class GpuCrawlEngine:
    def __init__(self, cfg, sm: LocalStateManager, feed: SyntheticFeed,
                 device="cuda:0", posts_per_channel: Optional[int] = None,
                 chunk_channels: int = 256, use_device_gen: bool = True,
                 fixed_now: Optional[_dt.datetime] = None):
        from ..ops import gpu as gpu_mod

        self.gpu = gpu_mod
        gpu_mod.require_lib()
        self.cfg = cfg
        self.sm = sm
        self.feed = feed
        self.device = torch.device(device)
        self.ppc = posts_per_channel or feed.cfg.posts_per_channel
        self.chunk_channels = chunk_channels
        self.seen = gpu_mod.SeenSet(self.device)
        # pin the capture timestamp for byte-reproducible crawls
        # (tools/verify_jsonl.py re-derives expected bytes from it)
        self.fixed_now = fixed_now
        self._pin_ring = [None, None]      # reusable pinned host slots
        self._spill_tickets = [None, None]  # sink ticket per ring slot
        import collections
        self.timings = collections.defaultdict(float)  # phase seconds
        self.stats = {"pages": 0, "posts": 0, "jsonl_bytes": 0,
                      "discovered": 0, "deadends": 0}
        self.last_deadends = set()

    # ---- helpers ----

    def _cid_of(self, username: str) -> Optional[int]:
        if username.startswith("c") and username[1:].isdigit():
            cid = int(username[1:])
            if cid < self.feed.cfg.universe:
                return cid
        return None

    def process_channels(self, usernames: List[str],
                         now: Optional[_dt.datetime] = None,
                         as_arrays: bool = False,
                         drain: bool = True):
        """Process a list of channels; returns (newly discovered names,
        posts stored). as_arrays=True returns the names as ONE
        zero-padded uint8[M, 32] array instead of python strings — the
        BFS loop sorts/dedups those at numpy speed (857k string objects
        per layer cost ~0.5s to build and sort)."""
        now = now or self.fixed_now or _dt.datetime.now(_dt.timezone.utc)
        discovered: List[str] = []
        discovered_arrays = []
        posts_total = 0
        t = self.timings
        import time as _time
        valid = [(u, self._cid_of(u)) for u in usernames]
        bad = [u for u, c in valid if c is None]
        ok = [(u, c) for u, c in valid if c is not None]
        # deadend channels of THIS call (invalid username or zero posts
        # after filters) — run() uses this to set page status so the
        # add_layer deadend-replacement budget activates (base.go:284)
        self.last_deadends = set(bad)
        for u in bad:
            self.stats["deadends"] += 1
        for ci, i in enumerate(range(0, len(ok), self.chunk_channels)):
            chunk = ok[i:i + self.chunk_channels]
            cids = np.array([c for _u, c in chunk], dtype=np.int64)
            t0 = _time.perf_counter()
            batch = self.feed.build_batch_device(
                cids, self.device, posts_per_channel=self.ppc
            )
            res = self.gpu.parse_encode(
                batch, now=now, min_post_date=self.cfg.min_post_date
            )
            new_mask = self.seen.claim(res)
            torch.cuda.synchronize()
            t["gen+kernels"] += _time.perf_counter() - t0; t0 = _time.perf_counter()

            # host spill: per-channel JSONL slices. Messages are grouped
            # by channel (K x P layout), so channel k owns lines
            # [k*P, (k+1)*P) -> bytes [line_off[kP], line_off[(k+1)P-1]+len).
            # The pinned HOST RING (2 slots, allocated once) replaces a
            # per-chunk 1GB pin_memory allocation (which cost ~0.1s/chunk
            # — the d2h phase was allocation-bound, not copy-bound); the
            # slot's previous SINK TICKET is awaited before the D2H may
            # overwrite it, making the spill pipeline 2 chunks deep.
            slot = ci % 2
            self.sm.wait_post_write(self._spill_tickets[slot])
            self._spill_tickets[slot] = None
            need = int(res.out.numel())
            ring = self._pin_ring
            if ring[slot] is None or ring[slot].numel() < need:
                ring[slot] = torch.empty(need + need // 8,
                                         dtype=torch.uint8,
                                         pin_memory=True)
            out_host = ring[slot][:need]
            out_host.copy_(res.out, non_blocking=True)
            line_off = res.line_off.cpu().numpy()
            line_len = res.line_len.cpu().numpy()
            torch.cuda.synchronize()
            t["d2h"] += _time.perf_counter() - t0; t0 = _time.perf_counter()
            buf = memoryview(out_host.numpy())
            P = self.ppc
            K = len(chunk)
            # vectorized per-channel ranges/counts (a python loop over
            # numpy scalars costs ~0.3s per 1500 channels)
            off2 = line_off.reshape(K, P)
            len2 = line_len.reshape(K, P)
            lo_k = off2[:, 0]
            hi_k = off2[:, -1] + len2[:, -1]
            n_lines_k = (len2 > 0).sum(axis=1)
            posts_total += int(n_lines_k.sum())
            self.stats["pages"] += K
            items = []
            for k, (uname, _cid) in enumerate(chunk):
                lo = int(lo_k[k])
                hi = int(hi_k[k])
                if hi > lo:
                    # truncate-then-write = exactly-once per channel even
                    # when a crash forces the layer to re-process
                    self.sm.truncate_posts(uname)
                    items.append((uname, lo, hi))
                # an all-filtered window is NOT a deadend: the
                # reference only deadends on inactivity/zero-message/
                # min-users (runner.go:635) — synthetic channels always
                # have messages, so only invalid usernames deadend here
            # one fan-out call for the whole chunk: the native sink
            # (crawler_amd/native) appends all channels in parallel with
            # the GIL released; the ticket is awaited when this ring slot
            # comes around again (2-deep: chunk i's disk writes run under
            # chunks i+1 AND i+2's GPU work).
            self._spill_tickets[slot] = self.sm.store_post_lines_batch(
                items, buf, ticket=True
            )
            self.stats["jsonl_bytes"] += int(buf.shape[0])
            t["spill"] += _time.perf_counter() - t0; t0 = _time.perf_counter()

            # newly-claimed names (first-discovery rows only). One bulk
            # D2H + ONE ascii decode of the packed name block, then
            # string slices — a per-name bytes().decode() loop costs
            # seconds at ~1M discoveries per crawl.
            if as_arrays:
                # device-side densification (claim_compact_kernel):
                # ships exactly M zero-padded rows, no host gather
                names_g, _hashes_g = self.seen.compact_claimed(res,
                                                               new_mask)
                if names_g.shape[0]:
                    discovered_arrays.append(names_g.cpu().numpy())
            else:
                nz = new_mask.nonzero()
                if nz.numel():
                    rows = nz[:, 0]
                    cols = nz[:, 1]
                    names_g = res.link_name[rows, cols]
                    lens_g = res.link_len[rows, cols]
                    w = names_g.shape[1]
                    # zero-pad ON THE GPU so the D2H ships clean
                    # fixed-width byte strings
                    col_idx = torch.arange(w, device=names_g.device,
                                           dtype=lens_g.dtype)
                    padded_g = torch.where(
                        col_idx[None, :] < lens_g[:, None], names_g,
                        torch.zeros_like(names_g))
                    padded = padded_g.cpu().numpy()
                    lens = lens_g.cpu().numpy()
                    blob = padded.tobytes().decode("ascii", "replace")
                    discovered.extend(
                        blob[i * w:i * w + ln]
                        for i, ln in enumerate(lens.tolist())
                    )
            t["names"] += _time.perf_counter() - t0
        if as_arrays:
            discovered = (np.concatenate(discovered_arrays)
                          if discovered_arrays
                          else np.zeros((0, 32), dtype=np.uint8))
        self.stats["posts"] += posts_total
        self.stats["discovered"] += len(discovered)
        if drain:
            # barrier: all spill writes down before the layer's
            # save_state. Callers that invoke process_channels once per
            # work CHUNK (the orchestrated crawl) pass drain=False and
            # call drain_spills() once per LAYER instead — otherwise
            # every chunk waits out its own disk writes and the 2-deep
            # spill pipeline never overlaps the next chunk's GPU work.
            self.drain_spills()
        return discovered, posts_total

    def drain_spills(self) -> None:
        """Wait out all in-flight ticketed sink writes (safe point for
        save_state; truncate-then-write keeps re-processing
        exactly-once if a crash lands before this)."""
        for slot in (0, 1):
            self.sm.wait_post_write(self._spill_tickets[slot])
            self._spill_tickets[slot] = None
        self.sm.drain_post_writes()

    # ---- BFS crawl (snowball / channel) ----

    def run(self, seed_urls: List[str], comm=None,
            resume: bool = True) -> dict:
        """comm: optional torch.distributed process group for multi-rank
        discovery exchange (rank-sharded layers). Resume follows the
        standalone rules (never resume completed; state.json required)."""
        import torch.distributed as dist

        sm = self.sm
        resumed = False
        if resume:
            _exec, ok = sm.find_incomplete_crawl(sm.metadata.crawl_id)
            if ok and sm.load_state():
                resumed = True
        if not resumed:
            sm.initialize(seed_urls)
        depth = 0
        while True:
            all_layer = sm.get_layer_by_depth(depth)
            if not all_layer:
                break
            # Resume fix: a depth whose pages are all
            # fetched (crash happened after its save_state) must NOT end
            # the crawl — skip processing and advance to the next depth,
            # like the CPU runner's all_layer check.
            layer = [p for p in all_layer if p.status == "unfetched"]
            import time as _time
            if layer:
                names = [p.url for p in layer]
                if comm is not None:
                    rank = dist.get_rank()
                    world = dist.get_world_size()
                    mine = names[rank::world]
                else:
                    mine = names
                discovered, _ = self.process_channels(mine,
                                                      as_arrays=True)
                dead = set(self.last_deadends)
                t0 = _time.perf_counter()
                if comm is not None:
                    # exchange discoveries as packed uint8 rows on the
                    # backend's device (RCCL: cuda; gloo: cpu) —
                    # count-sized, no cap
                    rows = torch.from_numpy(
                        np.ascontiguousarray(discovered))
                    out_rows = C.allgather_rows(
                        rows, dist, world, device=self.device)
                    discovered = out_rows.numpy()
                    # seen-set union: fold every rank's
                    # claims into the local table + OR-union the bloom
                    # so later layers dedup identically on all ranks
                    if discovered.shape[0]:
                        from . import vecvalidate as _vv
                        hs = torch.from_numpy(
                            _vv.fnv1a64_rows(discovered))
                        self.seen.insert_hashes(hs.to(self.device))
                    C.bloom_union(self.seen.bloom, dist, world)
                    # deadend statuses must agree across ranks: the
                    # next layer's deadend-replacement budget is
                    # computed independently by every rank
                    dead = set(C.rows_to_names(C.allgather_rows(
                        C.names_to_rows(sorted(dead)), dist, world,
                        device=self.device)))
                for p in layer:
                    p.status = ("deadend" if p.url in dead
                                else "fetched")
                    sm.update_page(p)
            else:
                discovered = np.zeros((0, 32), dtype=np.uint8)
                t0 = _time.perf_counter()
            if (self.cfg.sampling_method == "snowball"
                    and (self.cfg.max_depth < 0
                         or depth < self.cfg.max_depth)
                    and len(discovered)):
                # Pre-apply add_layer's own skip rules (URL dedup +
                # MaxPages/deadend budget) BEFORE constructing Page
                # objects: a dense crawl discovers ~1M names per layer
                # and the cap keeps ~1k — building 1M dataclasses to
                # throw them away costs seconds.
                max_pages = getattr(self.cfg, "max_pages", 0) or 0
                budget = None
                if max_pages > 0:
                    total = len(sm.pages)
                    deadends = sum(1 for p in sm.pages.values()
                                   if p.status == "deadend")
                    budget = max(0, max_pages - total) + deadends
                existing = {p.url for p in sm.pages.values()}
                url_dedup = getattr(sm, "url_dedup", {})
                # numpy sorted-unique over fixed-width byte rows (NUL
                # padding sorts exactly like the shorter string), then
                # decode ONLY as many rows as the budget admits — the
                # same lexicographic admission order as sorted(set(...))
                # without 857k python string objects
                uniq = np.unique(
                    np.ascontiguousarray(discovered).view("S32").ravel()
                )
                cand = []
                for b in uniq:
                    if budget is not None and len(cand) >= budget:
                        break
                    n = b.decode("ascii", "replace")
                    if n in existing or n in url_dedup:
                        continue
                    existing.add(n)
                    cand.append(n)
                pages = [Page(url=n, depth=depth + 1, status="unfetched")
                         for n in cand]
                sm.add_layer(pages)
            self.timings["layer-build"] += _time.perf_counter() - t0
            t0 = _time.perf_counter()
            sm.save_state()
            self.timings["save_state"] += _time.perf_counter() - t0
            if self.cfg.sampling_method == "channel":
                break
            if self.cfg.max_depth >= 0 and depth >= self.cfg.max_depth:
                break
            depth += 1
        sm.update_crawl_metadata(sm.metadata.crawl_id,
                                 {"status": "completed"})
        sm.save_state()
        sm.close()
        return dict(self.stats)
