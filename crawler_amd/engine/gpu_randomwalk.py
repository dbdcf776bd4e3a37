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
from ..ops.golden import filter_username
from . import randomwalk
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

    def _hop(self, pages: List[Page],
             now: Optional[_dt.datetime] = None) -> int:
        """One batched hop over up to W frontier pages."""
        now = now or _dt.datetime.now(_dt.timezone.utc)
        # 400s: invalid usernames get the replacement machinery
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
        if not live:
            return 0

        cids = np.array([self._cid_of(p.url) for p in live],
                        dtype=np.int64)
        batch = self.feed.build_batch_device(
            cids, self.device, posts_per_channel=self.ppc
        )
        res = self.gpu.parse_encode(batch, now=now)
        self.seen.claim(res)
        torch.cuda.synchronize()

        # spill JSONL per channel (K x P layout)
        out_host = res.out.cpu().numpy()
        line_off = res.line_off.cpu().numpy()
        line_len = res.line_len.cpu().numpy()
        P = self.ppc
        mv = memoryview(out_host)
        for k, p in enumerate(live):
            lo = int(line_off[k * P])
            last = (k + 1) * P - 1
            hi = int(line_off[last] + line_len[last])
            if hi > lo:
                self.sm.store_post_lines(p.url, mv[lo:hi])
            self.stats["posts"] += int((line_len[k * P:(k + 1) * P] > 0).sum())

        # per-walker unique link names (host; the per-hop volume is small)
        names = res.link_name.cpu().numpy().view("S32").reshape(
            res.link_name.shape[0], self.gpu.MAX_LINKS
        )
        lens = res.link_len.cpu().numpy()
        cnts = res.link_cnt.cpu().numpy()
        for k, p in enumerate(live):
            uniq = {}
            for i in range(k * P, (k + 1) * P):
                for s in range(int(cnts[i])):
                    nm = bytes(names[i, s])[: lens[i, s]].decode()
                    uniq.setdefault(nm, True)
            # outlink validation (SearchPublicChat equivalent + filter +
            # invalid cache; runner.go:1310-1383)
            new_channels = {}
            for nm in uniq:
                if nm == p.url:
                    continue
                ok, _ = filter_username(nm)
                if not ok or self.rw.is_invalid_channel(nm):
                    continue
                if self._cid_of(nm) is None:
                    self.rw.mark_invalid_channel(nm)
                    continue
                self.sm.add_discovered_channel(nm)
                self.rw.upsert_seed_channel(nm)
                new_channels[nm] = True
            try:
                randomwalk.walk_tail(p, new_channels, self.sm, self.rw,
                                     self.cfg, self.rng)
            except randomwalk.E.WalkbackExhausted:
                self.stats["walkback_exhausted"] += 1
                continue  # page left in buffer
            self.rw.mark_channel_crawled(p.url, 0)
            self.rw.delete_pages([p.id])
            self.stats["pages"] += 1
        self.stats["edges"] = len(self.rw.edge_records)
        return len(live)

    def run(self, max_pages: int, now: Optional[_dt.datetime] = None) -> dict:
        while self.stats["pages"] < max_pages:
            pages = self.rw.get_pages(
                min(self.walkers, max_pages - self.stats["pages"])
            )
            if not pages:
                break
            self._hop(pages, now)
        self.sm.save_state()
        self.sm.close()
        return dict(self.stats)
