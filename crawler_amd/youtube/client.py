"""Synthetic YouTube Data API client with quota accounting and sampling.

Parity (reference client/youtube_client.go):
- quota economics: search=100 units, videos/channels list=1 unit,
  10k units/day (README "Handling Rate Limits and Errors")
- channel/video caches (youtube_client.go:32-48)
- GetRandomVideos random prefix sampling (:1112-1543): 5-char a-z
  prefixes, quoted "watch?v=" search, validity rule len==11 AND
  starts-with AND id[5]=='-' (:1266-1281), 5 channel-verification
  workers with the >min-videos gate (:1371-1429), 50-ID Videos.List
  batches (:1077-1107), collector merges until the effective limit
- GetSnowballVideos (:1547): seed channels -> uploads -> channel ids
  extracted from descriptions
"""
from __future__ import annotations

import concurrent.futures as cf
import random
import re
from typing import Dict, List, Optional, Set

from .synth import SyntheticYouTubeIndex, YouTubeChannel, YouTubeVideo

QUOTA_SEARCH = 100
QUOTA_LIST = 1
QUOTA_DAILY = 10_000

_CHANNEL_ID_RE = re.compile(r"(UC[0-9A-Za-z_-]{22})")


class QuotaExceeded(Exception):
    pass


class SyntheticYouTubeClient:
    def __init__(self, index: Optional[SyntheticYouTubeIndex] = None,
                 daily_quota: int = QUOTA_DAILY,
                 min_channel_videos: int = 10, rng=None):
        self.index = index or SyntheticYouTubeIndex()
        self.quota_used = 0
        self.daily_quota = daily_quota
        self.min_channel_videos = min_channel_videos
        self.rng = rng or random.Random(0)
        self._channel_cache: Dict[str, YouTubeChannel] = {}
        self._video_cache: Dict[str, YouTubeVideo] = {}
        self.stats = {"searches": 0, "video_batches": 0,
                      "channels_checked": 0, "gate_rejected": 0,
                      "invalid_ids": 0}

    def _spend(self, units: int):
        if self.quota_used + units > self.daily_quota:
            raise QuotaExceeded(
                f"daily quota exceeded: {self.quota_used}+{units}"
            )
        self.quota_used += units

    # ---- raw API surface ----

    def get_channel_info(self, channel_id: str) -> Optional[YouTubeChannel]:
        if channel_id in self._channel_cache:
            return self._channel_cache[channel_id]
        self._spend(QUOTA_LIST)
        n = self.index.channel_index_of(channel_id)
        if n is None:
            return None
        ch = self.index.channel(n)
        self._channel_cache[channel_id] = ch
        return ch

    def list_videos(self, ids: List[str]) -> List[YouTubeVideo]:
        """Videos.List — up to 50 ids per call (one quota unit)."""
        out = []
        for i in range(0, len(ids), 50):
            self._spend(QUOTA_LIST)
            self.stats["video_batches"] += 1
            for vid in ids[i:i + 50]:
                if vid not in self._video_cache:
                    self._video_cache[vid] = self.index.video(vid)
                out.append(self._video_cache[vid])
        return out

    def search(self, prefix: str) -> List[str]:
        self._spend(QUOTA_SEARCH)
        self.stats["searches"] += 1
        return self.index.search_prefix(prefix)

    # ---- random prefix sampling (GetRandomVideos) ----

    @staticmethod
    def generate_random_prefix(rng) -> str:
        """5 random a-z chars (youtube_client.go:886-911)."""
        return "".join(chr(97 + rng.randrange(26)) for _ in range(5))

    @staticmethod
    def is_valid_sample(video_id: str, prefix: str) -> bool:
        """Validity rule (youtube_client.go:1266-1281)."""
        return (len(video_id) == 11 and video_id.startswith(prefix)
                and video_id[5] == "-")

    def get_random_videos(self, limit: int,
                          max_searches: int = 200) -> List[YouTubeVideo]:
        videos: List[YouTubeVideo] = []
        queued_ids: List[str] = []
        seen: Set[str] = set()
        searches = 0
        while len(videos) + len(queued_ids) < limit and \
                searches < max_searches:
            prefix = self.generate_random_prefix(self.rng)
            try:
                ids = self.search(prefix)
            except QuotaExceeded:
                break
            searches += 1
            candidates = []
            for vid in ids:
                if not self.is_valid_sample(vid, prefix):
                    self.stats["invalid_ids"] += 1
                    continue
                if vid in seen:
                    continue
                seen.add(vid)
                candidates.append(vid)
            # 5-worker channel verification pool (youtube_client.go:1371-1429)
            verified = []
            if candidates:
                metas = self.list_videos(candidates)
                with cf.ThreadPoolExecutor(max_workers=5) as ex:
                    infos = list(ex.map(
                        lambda v: self.get_channel_info(v.channel_id), metas
                    ))
                for v, ch in zip(metas, infos):
                    self.stats["channels_checked"] += 1
                    if ch is None or ch.video_count <= self.min_channel_videos:
                        self.stats["gate_rejected"] += 1
                        continue
                    verified.append(v.id)
            queued_ids.extend(verified)
            # 50-ID batch processing gate (ShouldProcessRandomBatch, :1077)
            while len(queued_ids) >= 50 or (
                queued_ids and len(videos) + len(queued_ids) >= limit
            ):
                batch, queued_ids = queued_ids[:50], queued_ids[50:]
                videos.extend(self.list_videos(batch))
                if len(videos) >= limit:
                    return videos[:limit]
        if queued_ids:
            videos.extend(self.list_videos(queued_ids))
        return videos[:limit]

    # ---- snowball (GetSnowballVideos, :1547) ----

    @staticmethod
    def extract_channel_ids_from_text(text: str) -> List[str]:
        """youtube_client.go:1856."""
        seen = []
        for m in _CHANNEL_ID_RE.finditer(text or ""):
            cid = m.group(1)
            if cid not in seen:
                seen.append(cid)
        return seen

    def get_snowball_videos(self, seed_channel_ids: List[str], limit: int,
                            max_depth: int = 2) -> List[YouTubeVideo]:
        videos: List[YouTubeVideo] = []
        frontier = list(seed_channel_ids)
        visited: Set[str] = set()
        depth = 0
        while frontier and len(videos) < limit and depth <= max_depth:
            nxt: List[str] = []
            for cid in frontier:
                if cid in visited:
                    continue
                visited.add(cid)
                ids = self.index.channel_uploads(cid)
                if not ids:
                    continue
                for v in self.list_videos(ids):
                    videos.append(v)
                    for disc in self.extract_channel_ids_from_text(
                        v.description
                    ):
                        if disc not in visited:
                            nxt.append(disc)
                    if len(videos) >= limit:
                        return videos[:limit]
            frontier = nxt
            depth += 1
        return videos[:limit]

    def get_channel_videos(self, channel_id: str,
                           limit: int = 50) -> List[YouTubeVideo]:
        ids = self.index.channel_uploads(channel_id, limit)
        return self.list_videos(ids)


class YouTubeClientPool:
    """Rotating client pool (ytWorker, dapr/standalone.go:1245-1272):
    each client retires after ~50 +/- 10 uses and is replaced by a fresh
    one, spreading quota pressure the way the reference rotates workers."""

    def __init__(self, make_client, retire_at: int = 50,
                 retire_jitter: int = 10, rng=None):
        self.make_client = make_client
        self.retire_at = retire_at
        self.retire_jitter = retire_jitter
        self.rng = rng or random.Random(0)
        self.retired = 0
        self._fresh()

    def _fresh(self):
        self.client = self.make_client()
        self.uses = 0
        self.limit = self.retire_at + self.rng.randint(
            -self.retire_jitter, self.retire_jitter
        )

    def get(self) -> SyntheticYouTubeClient:
        if self.uses >= self.limit:
            self.retired += 1
            self._fresh()
        self.uses += 1
        return self.client
