"""GPU random-walk: W independent walker chains batched per hop.

The reference's layerless random-walk loop (dapr/standalone.go:792-946)
processes one channel per goroutine; here every hop of every walker is ONE
device batch: the frontier (page_buffer = the checkpoint)
yields up to W channels, the HIP pipeline parses/encodes/claims them in a
single launch, and the walk decisions (walkback %, skipped edges,
sequence-id chains — engine/randomwalk.py, identical code to the CPU
path) run host-side per walker as SURVEY §2.6 prescribes ("tiny; keep
host-side").

Outlink validation in the synthetic world: a name is a valid channel iff
it decodes to a universe id (the SearchPublicChat equivalent — the live
check the reference does at crawl/runner.go:1310-1383) and passes
FilterUsername and the invalid-channel cache.
"""
from __future__ import annotations

import datetime as _dt
import random
from typing import List, Optional

import numpy as np
import torch

from ..feed.synth import SyntheticFeed
from . import randomwalk, vecvalidate
from .state import LocalStateManager, Page, RandomWalkStore


class GpuRandomWalk:
    def __init__(self, cfg, sm: LocalStateManager, rw: RandomWalkStore,
                 feed: SyntheticFeed, device="cuda:0",
                 posts_per_hop: int = 500, walkers: int = 256, rng=None):
        from ..ops import gpu as gpu_mod

        self.gpu = gpu_mod
        gpu_mod.require_lib()
        self.cfg = cfg
        self.sm = sm
        self.rw = rw
        self.feed = feed
        self.device = torch.device(device)
        self.ppc = posts_per_hop
        self.walkers = walkers
        self.rng = rng or random.Random()
        self.seen = gpu_mod.SeenSet(self.device)
        self.stats = {"pages": 0, "posts": 0, "invalid_400": 0,
                      "walkback_exhausted": 0, "edges": 0}
        self._pin_ring = [None, None]       # pinned host JSONL slots
        self._spill_tickets = [None, None]  # native-sink tickets
        self._inflight_paths = {}           # slot -> channel set
        self._hop_idx = 0
        # cross-hop validation cache keyed by link hash — a vectorized
        # open-addressing table (linear probe, power-of-two size):
        # validation is a pure function of the name, so hits skip
        # validate_names AND the decode. One table probe per hop
        # replaced the earlier two-level sorted design's per-hop
        # argsort+searchsorted (~90 ms/hop at steady state -> ~15 ms).
        self._vc_init(1 << 20)
        import collections
        self.timings = collections.defaultdict(float)  # phase seconds
        self.hop_log = []  # (pages, posts, seconds) per hop

    def seed(self, urls: List[str]):
        import uuid

        for u in urls:
            self.sm.add_discovered_channel(u)
            self.rw.upsert_seed_channel(u)
            self.rw.add_page(Page(
                id=str(uuid.uuid4()), url=u, depth=0,
                sequence_id=str(uuid.uuid4()), status="unfetched",
            ))

    def _cid_of(self, username: str) -> Optional[int]:
        if username.startswith("c") and username[1:].isdigit():
            cid = int(username[1:])
            if cid < self.feed.cfg.universe:
                return cid
        return None

    def _inv_snapshot(self, now, width: int):
        """Sorted S<width> snapshot of the live (non-expired) invalid
        names, rebuilt only when the store's version stamp moves or the
        oldest included entry crosses its 30-day TTL — the per-hop
        sorted()+np.isin rebuild cost ~0.5 s/10 hops once the cache
        grew past ~100k names."""
        ver = getattr(self.rw, "invalid_version", 0)
        cached = getattr(self, "_inv_snap", None)
        if cached is not None:
            cver, expire_at, arr = cached
            if cver == ver and (expire_at is None or now < expire_at):
                return arr
        ttl_cut = now - _dt.timedelta(days=30)
        names, oldest = [], None
        for u, t in self.rw.invalid_channels.items():
            if t > ttl_cut:
                names.append(u.encode())
                if oldest is None or t < oldest:
                    oldest = t
        arr = (np.sort(np.array(names, dtype=f"S{width}")) if names
               else np.zeros(0, dtype=f"S{width}"))
        expire_at = (oldest + _dt.timedelta(days=30)
                     if oldest is not None else None)
        self._inv_snap = (ver, expire_at, arr)
        return arr

    @staticmethod
    def _in_sorted(sorted_arr, vals):
        """Membership of vals in a sorted array via searchsorted."""
        pos = np.searchsorted(sorted_arr, vals)
        pos_c = np.clip(pos, 0, len(sorted_arr) - 1)
        return sorted_arr[pos_c] == vals

    # ---- validation cache: vectorized open-addressing hash table ----
    # Keys are the device fnv1a64 link hashes (int64); values index an
    # append-only admission-bit store (the NAME need not be cached —
    # every query carries its own bytes row, and validation is a pure
    # function of the name, so a hit only answers "admitted?").
    # Inserts only ever add NEW keys (misses), so there are no updates
    # or deletes; the sentinel key is int64.min, which fnv1a64 of a
    # <=32-byte ASCII name never hits.

    _VC_EMPTY = np.int64(-2 ** 63)

    def _vc_init(self, slots: int):
        self._vc_keys = np.full(slots, self._VC_EMPTY, dtype=np.int64)
        self._vc_vals = np.zeros(slots, dtype=np.int64)
        self._vc_mask = slots - 1
        self._vc_n = 0
        self._vc_adm = np.zeros(0, dtype=bool)

    @staticmethod
    def _vc_mix(h):
        """64-bit finalizer so linear probing sees uniform slots even
        for correlated fnv values."""
        h = h.astype(np.uint64)
        h ^= h >> np.uint64(33)
        h *= np.uint64(0xFF51AFD7ED558CCD)
        h ^= h >> np.uint64(33)
        return h

    def _vc_slots_of(self, keys):
        return (self._vc_mix(keys) & np.uint64(self._vc_mask)).astype(
            np.int64)

    def _vc_lookup(self, hashes, out_adm):
        """Fill out_adm for cached keys; returns the missing mask.
        Vectorized linear probe: each round gathers one slot per
        still-active query."""
        n = len(hashes)
        miss = np.ones(n, dtype=bool)
        if self._vc_n == 0 or n == 0:
            return miss
        slot = self._vc_slots_of(hashes)
        active = np.arange(n)
        while len(active):
            cur = self._vc_keys[slot[active]]
            hit = cur == hashes[active]
            empty = cur == self._VC_EMPTY
            if hit.any():
                ai = active[hit]
                vi = self._vc_vals[slot[ai]]
                out_adm[ai] = self._vc_adm[vi]
                miss[ai] = False
            active = active[~(hit | empty)]
            slot[active] = (slot[active] + 1) & self._vc_mask
        return miss

    def _vc_insert(self, new_h, new_a):
        """Insert NEW unique keys (the hop's cache misses). Grows the
        table at 50% load (rebuild is a bulk re-insert, amortized)."""
        k = len(new_h)
        if not k:
            return
        base = len(self._vc_adm)
        self._vc_adm = np.concatenate([self._vc_adm, new_a])
        if (self._vc_n + k) * 2 > self._vc_mask + 1:
            slots = (self._vc_mask + 1) * 2
            while (self._vc_n + k) * 2 > slots:
                slots *= 2
            old_keys = self._vc_keys[self._vc_keys != self._VC_EMPTY]
            old_vals = self._vc_vals[self._vc_keys != self._VC_EMPTY]
            self._vc_keys = np.full(slots, self._VC_EMPTY,
                                    dtype=np.int64)
            self._vc_vals = np.zeros(slots, dtype=np.int64)
            self._vc_mask = slots - 1
            self._vc_place(old_keys, old_vals)
        self._vc_place(new_h, base + np.arange(k, dtype=np.int64))
        self._vc_n += k

    def _vc_place(self, keys, vals):
        """Claim empty slots for unique keys (vectorized linear probe;
        same-slot collisions within a batch are resolved by letting the
        first claimant win each round and advancing the rest)."""
        slot = self._vc_slots_of(keys)
        pending = np.arange(len(keys))
        while len(pending):
            s = slot[pending]
            free = self._vc_keys[s] == self._VC_EMPTY
            pf = pending[free]
            sf = s[free]
            uniq_s, first = np.unique(sf, return_index=True)
            self._vc_keys[uniq_s] = keys[pf[first]]
            self._vc_vals[uniq_s] = vals[pf[first]]
            placed = np.zeros(len(pending), dtype=bool)
            placed[np.flatnonzero(free)[first]] = True
            pending = pending[~placed]
            slot[pending] = (slot[pending] + 1) & self._vc_mask

    def _hop(self, pages: List[Page],
             now: Optional[_dt.datetime] = None) -> int:
        """One batched hop over up to W frontier pages (serial form:
        claim -> GPU stage -> host tail)."""
        live = self._claim_400(pages)
        if not live:
            return 0
        slot = self._ring_slot()
        payload = self._hop_gpu(live, slot, now)
        self._hop_host(live, payload)
        return len(live)

    def _claim_400(self, pages: List[Page]) -> List[Page]:
        """400s: invalid usernames get the replacement machinery
        (host-side; consumes rng, so it stays on the main thread)."""
        live: List[Page] = []
        for p in pages:
            if self._cid_of(p.url) is None:
                self.stats["invalid_400"] += 1
                try:
                    randomwalk.handle_400_replacement(
                        self.sm, self.rw, p, self.cfg, self.rng
                    )
                except Exception:
                    self.stats["walkback_exhausted"] += 1
                self.rw.delete_pages([p.id])
            else:
                live.append(p)
        return live

    def _ring_slot(self) -> int:
        """Claim the next pinned-ring slot and retire its old spill
        ticket. MAIN THREAD ONLY: the native sink's ticket wait also
        evicts surplus fds and must not race a concurrent submit."""
        slot = self._hop_idx % 2
        self._hop_idx += 1
        self.sm.wait_post_write(self._spill_tickets[slot])
        self._spill_tickets[slot] = None
        return slot

    def _hop_gpu(self, live: List[Page], slot: int,
                 now: Optional[_dt.datetime] = None) -> dict:
        """Device stage: feedgen -> parse/encode -> seen claim ->
        device dedup + validation -> host pulls + async JSONL D2H.
        Returns the host-side payload for _hop_host. Safe to run on a
        worker thread (GPU + numpy only; no rng, no store writes — the
        pinned-ring slot is claimed by the caller)."""
        now = now or _dt.datetime.now(_dt.timezone.utc)
        import time as _time
        t0 = _time.perf_counter()
        cids = np.array([self._cid_of(p.url) for p in live],
                        dtype=np.int64)
        batch = self.feed.build_batch_device(
            cids, self.device, posts_per_channel=self.ppc
        )
        torch.cuda.synchronize()
        self.timings["feedgen"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()
        res = self.gpu.parse_encode(batch, now=now)
        self.seen.claim(res)
        torch.cuda.synchronize()
        self.timings["kernels"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()

        # Per-walker link dedup ON DEVICE: an
        # exact byte-lexicographic sort over (walker, name) pairs, keep
        # the first of each run — the host never sees the ~1M raw link
        # slots, only ~150k unique name rows (5 MB instead of 65 MB
        # D2H, no host np.unique), already name-sorted per walker.
        lens_g = res.link_len
        names_g = res.link_name
        w = names_g.shape[2]
        P = self.ppc
        dev = names_g.device
        L = names_g.shape[1]
        cnt_g = res.link_cnt
        mask_g = (torch.arange(L, device=dev)[None, :]
                  < cnt_g[:, None])
        nz = mask_g.nonzero()
        mi_g, si_g = nz[:, 0], nz[:, 1]
        wk_g = torch.div(mi_g, P, rounding_mode="floor")
        # zero-padded rows for ALL pairs, then a byte-LEXICOGRAPHIC sort
        # within each walker (names pack into 4 big-endian int64 keys:
        # lexicographic byte order == big-endian integer order). Exact
        # name dedup AND pre-sorted per-walker output: the host's
        # per-walker sorted() over ~300-name lists disappears, and the
        # cross-hop cache's hash keying no longer decides dedup.
        col = torch.arange(w, device=dev, dtype=lens_g.dtype)
        all_lens = lens_g[mi_g, si_g]
        all_rows = torch.where(col[None, :] < all_lens[:, None],
                               names_g[mi_g, si_g],
                               torch.zeros_like(names_g[mi_g, si_g]))
        be = (256 ** torch.arange(7, -1, -1, device=dev,
                                  dtype=torch.int64))
        keys = (all_rows.view(-1, 4, 8).to(torch.int64) * be).sum(-1)
        perm = torch.argsort(keys[:, 3])
        for kcol in (2, 1, 0):
            perm = perm[torch.argsort(keys[perm, kcol], stable=True)]
        perm = perm[torch.argsort(wk_g[perm], stable=True)]
        ws_g = wk_g[perm]
        ks_g = keys[perm]
        keep = torch.ones(ws_g.numel(), dtype=torch.bool, device=dev)
        if ws_g.numel() > 1:
            keep[1:] = ((ws_g[1:] != ws_g[:-1])
                        | (ks_g[1:] != ks_g[:-1]).any(dim=1))
        sel = perm[keep]
        h_g = res.link_hash[mi_g, si_g]
        hs_sel_g = h_g[sel]
        u_rows_g = all_rows[sel]
        # validate the deduped rows ON DEVICE (oracle-equal torch port
        # of FilterUsername + universe decode); the host miss path just
        # indexes these instead of re-scanning bytes
        ok_g, cid_ok_g, _ = vecvalidate.validate_names_torch(
            u_rows_g, self.feed.cfg.universe)

        # D2H through a reusable pinned ring (2-deep: this hop's disk
        # writes overlap the next hop's kernels via the native sink).
        # Small control tensors pull FIRST (each .cpu() syncs the
        # stream up to its producer); the big JSONL copy is issued
        # LAST, non-blocking, behind a recorded event — the host tail
        # waits on the EVENT, not a device-wide synchronize, so a
        # pipelined run never serializes against the other pool's
        # in-flight kernels.
        line_off = res.line_off.cpu().numpy()
        line_len = res.line_len.cpu().numpy()
        u_w = ws_g[keep].to(torch.int32).cpu().numpy()
        u_h = hs_sel_g.cpu().numpy()
        u_rows = u_rows_g.cpu().numpy()
        ok_all = ok_g.cpu().numpy()
        cid_ok_all = cid_ok_g.cpu().numpy()
        need = int(res.out.numel())
        ring = self._pin_ring
        if ring[slot] is None or ring[slot].numel() < need:
            ring[slot] = torch.empty(need + need // 8,
                                     dtype=torch.uint8,
                                     pin_memory=True)
        out_host = ring[slot][:need]
        out_host.copy_(res.out, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record()
        self.timings["d2h"] += _time.perf_counter() - t0
        return {"slot": slot, "out_host": out_host, "ev": ev,
                "res_out": res.out, "line_off": line_off,
                "line_len": line_len, "u_w": u_w, "u_h": u_h,
                "u_rows": u_rows, "ok_all": ok_all,
                "cid_ok_all": cid_ok_all, "w": w, "now": now}

    def _hop_host(self, live: List[Page], payload: dict) -> None:
        """Host tail: spill, validation cache, admission, walk
        decisions. MAIN THREAD ONLY (rng + store writes)."""
        import time as _time
        t0 = _time.perf_counter()
        slot = payload["slot"]
        out_host = payload["out_host"]
        line_off = payload["line_off"]
        line_len = payload["line_len"]
        u_w = payload["u_w"]
        u_h = payload["u_h"]
        u_rows = payload["u_rows"]
        ok_all = payload["ok_all"]
        cid_ok_all = payload["cid_ok_all"]
        w = payload["w"]
        P = self.ppc
        payload["ev"].synchronize()   # JSONL bytes landed in the ring
        payload["res_out"] = None
        self.timings["d2h"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()

        # spill JSONL per channel (K x P layout) in ONE native-sink call
        K = len(live)
        off2 = line_off.reshape(K, P)
        len2 = line_len.reshape(K, P)
        lo_k = off2[:, 0]
        hi_k = off2[:, -1] + len2[:, -1]
        self.stats["posts"] += int((len2 > 0).sum())
        items = [(p.url, int(lo_k[k]), int(hi_k[k]))
                 for k, p in enumerate(live) if hi_k[k] > lo_k[k]]
        # Random-walk allows REVISITS: if this hop touches a channel the
        # other in-flight batch is still appending to, wait it out first
        # so a channel's posts never interleave mid-line (the CPU path
        # appends visits strictly in order).
        other = 1 - slot
        if self._spill_tickets[other] is not None:
            mine = {u for u, _lo, _hi in items}
            if mine & self._inflight_paths.get(other, set()):
                self.sm.wait_post_write(self._spill_tickets[other])
                self._spill_tickets[other] = None
        self._inflight_paths[slot] = {u for u, _lo, _hi in items}
        self._spill_tickets[slot] = self.sm.store_post_lines_batch(
            items, memoryview(out_host.numpy()), ticket=True
        )
        self.timings["spill-issue"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()

        # Outlink validation (SearchPublicChat equivalent + filter +
        # invalid cache; runner.go:1310-1383) on the device-deduped
        # (walker, name) rows. A cross-hop cache keyed by link hash
        # skips validate+decode for names seen in earlier hops (the
        # seen-set/bloom already key on this same fnv1a64).
        hop_now = _dt.datetime.now(_dt.timezone.utc)
        U = len(u_h)
        un = np.ascontiguousarray(u_rows).view(f"S{w}").ravel()
        res_adm = np.zeros(U, dtype=bool)
        miss = self._vc_lookup(u_h, res_adm)
        self.timings["t-cache"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()
        inv_arr = self._inv_snapshot(hop_now, w)
        if miss.any():
            mh, m_first = np.unique(u_h[miss], return_index=True)
            m_rows = un[miss][m_first]
            ok_v = ok_all[miss][m_first]
            cid_ok_v = cid_ok_all[miss][m_first]
            adm_v = ok_v & cid_ok_v
            adm_i = np.nonzero(adm_v)[0]
            new_admitted = vecvalidate.decode_names(m_rows[adm_i])
            # filter-ok but not a live channel -> invalid-cache insert
            # (runner.go:1310-1383's 400 branch); marked once, cached
            for nm in vecvalidate.decode_names(m_rows[ok_v & ~cid_ok_v]):
                self.rw.mark_invalid_channel(nm)
            mpos = np.searchsorted(mh, u_h[miss])
            res_adm[miss] = adm_v[mpos]
            self._vc_insert(mh, adm_v)
            # discovered/seed admission only needs FIRST-SEEN names:
            # every admitted name from an earlier hop was admitted the
            # hop it was first validated (400-marked names are never
            # statically admissible, so the inv filter is a no-op here
            # beyond correctness hygiene)
            if len(inv_arr):
                keep = ~self._in_sorted(inv_arr, m_rows[adm_i])
                new_admitted = [nm for nm, k in zip(new_admitted, keep)
                                if k]
            self.sm.add_discovered_channels_bulk(new_admitted)
            self.rw.upsert_seed_channels_bulk(new_admitted)
        self.timings["t-validate"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()
        # dynamic invalid-channel cache (mutated by 400 handling):
        # searchsorted membership against the cached sorted snapshot
        not_inv = (~self._in_sorted(inv_arr, un) if len(inv_arr)
                   else np.ones(U, dtype=bool))
        owner_b = np.array([p.url for p in live], dtype=un.dtype)
        final = res_adm & not_inv & (un != owner_b[u_w])
        uw_s = u_w[final]
        names_s = un[final]   # names stay S<w> bytes end-to-end
        bounds = np.searchsorted(uw_s, np.arange(K + 1))
        # per-walker slices arrive NAME-SORTED from the device sort
        # (byte-lexicographic == python str sort for ASCII names)
        self.timings["t-group"] += _time.perf_counter() - t0
        t0 = _time.perf_counter()
        for k, p in enumerate(live):
            try:
                self._walk_tail_rows(
                    p, names_s[bounds[k]:bounds[k + 1]], hop_now)
            except randomwalk.E.WalkbackExhausted:
                self.stats["walkback_exhausted"] += 1
                continue  # page left in buffer
            self.rw.mark_channel_crawled(p.url, 0)
            self.rw.delete_pages([p.id])
            self.stats["pages"] += 1
        self.timings["walk-tail"] += _time.perf_counter() - t0
        self.stats["edges"] = self.rw.edge_count()

    def _walk_tail_rows(self, owner: Page, rows, now) -> Page:
        """randomwalk.walk_tail_fast with IDENTICAL decisions and rng
        consumption, taking the walker's sorted S<w> byte rows instead
        of decoded strings: only the followed name is decoded; skipped
        names land in the O(1) edge block as the bytes slice and
        decode lazily on edge_records materialization. Equivalence is
        pinned by tests/test_gpu_randomwalk.py (CPU unit:
        test_walk_tail_rows_matches_fast)."""
        import uuid as _uuid

        from .state import EdgeRecord

        page = Page(
            id=str(_uuid.uuid4()), parent_id=owner.id,
            depth=owner.depth + 1, status="unfetched",
        )
        src = owner.url
        seq = owner.sequence_id
        walkback = len(rows) == 0
        rnd = None
        if not walkback:
            rnd = self.rng.randint(1, 100)
        if walkback or self.cfg.walkback_rate >= rnd:
            # pick_walkback_channel (randomwalk.py:30-41) with a
            # bytes-membership exclusion set (same rng consumption)
            excl = set(rows.tolist())
            url = None
            for _ in range(randomwalk.MAX_WALKBACK_ATTEMPTS):
                cand = self.sm.get_random_discovered_channel(self.rng)
                if cand is None:
                    raise randomwalk.E.WalkbackExhausted(src)
                if cand == src or cand.encode() in excl:
                    continue
                url = cand
                break
            if url is None:
                raise randomwalk.E.WalkbackExhausted(src)
            page.url = url
            page.sequence_id = str(_uuid.uuid4())  # fresh chain
            skipped = rows
            edge = EdgeRecord(url, now, src, True, False, seq, "")
        else:
            pick = self.rng.randrange(len(rows))
            page.url = rows[pick].decode()
            page.sequence_id = seq
            skipped = (np.delete(rows, pick) if len(rows) > 1
                       else rows[:0])
            edge = EdgeRecord(page.url, now, src, False, False, seq, "")
        self.rw.add_page(page)
        self.rw.save_edge_records_fast([edge])
        self.rw.save_skipped_edges_block(src, seq, skipped, now)
        return page

    def run(self, max_pages: int, now: Optional[_dt.datetime] = None,
            pipelined: bool = False) -> dict:
        """Walk until max_pages pages complete (or the frontier dries
        up). pipelined=True runs two half-size walker pools in a
        software pipeline: pool B's device stage (launched on a worker
        thread — GPU + numpy only) overlaps pool A's host tail, hiding
        the ~65 ms device phase under the ~135 ms host phase. The walk
        REMAINS deterministic for a fixed seed, but consumes the rng in
        a different (interleaved) order than the serial form, so it is
        opt-in; both forms have identical per-page semantics."""
        if pipelined:
            self._run_pipelined(max_pages, now)
        else:
            self._run_serial(max_pages, now)
        for slot in (0, 1):
            self.sm.wait_post_write(self._spill_tickets[slot])
            self._spill_tickets[slot] = None
        self.sm.drain_post_writes()
        self.sm.save_state()
        self.sm.close()
        return dict(self.stats)

    def _run_serial(self, max_pages: int, now) -> None:
        while self.stats["pages"] < max_pages:
            pages = self.rw.get_pages(
                min(self.walkers, max_pages - self.stats["pages"])
            )
            if not pages:
                break
            import time as _time
            t0 = _time.perf_counter()
            p0 = self.stats["posts"]
            ph0 = dict(self.timings)
            n = self._hop(pages, now)
            self.hop_log.append((n, self.stats["posts"] - p0,
                                 round(_time.perf_counter() - t0, 4),
                                 {k: round(v - ph0.get(k, 0.0), 4)
                                  for k, v in self.timings.items()
                                  if v - ph0.get(k, 0.0) > 1e-4}))

    def _run_pipelined(self, max_pages: int, now) -> None:
        import concurrent.futures as cf
        import time as _time

        half = max(1, self.walkers // 2)
        pending = None  # (live_pages, future)
        with cf.ThreadPoolExecutor(max_workers=1) as ex:
            while True:
                t0 = _time.perf_counter()
                p0 = self.stats["posts"]
                inflight = pending[0] if pending else []
                budget = (max_pages - self.stats["pages"]
                          - len(inflight))
                pages: List[Page] = []
                if budget > 0:
                    want = min(half, budget)
                    excl = {p.id for p in inflight}
                    raw = self.rw.get_pages(want + len(excl))
                    pages = [p for p in raw if p.id not in excl][:want]
                live = self._claim_400(pages) if pages else []
                fut = None
                if live:
                    slot = self._ring_slot()
                    fut = ex.submit(self._hop_gpu, live, slot, now)
                processed = pending is not None
                if processed:
                    plive, pfut = pending
                    self._hop_host(plive, pfut.result())
                    self.hop_log.append(
                        (len(plive), self.stats["posts"] - p0,
                         round(_time.perf_counter() - t0, 4), {}))
                pending = (live, fut) if fut is not None else None
                # stop only when nothing was claimed AND nothing was
                # processed this iteration (a processed tail refills
                # the frontier, so re-check before giving up)
                if pending is None and not pages and not processed:
                    break
