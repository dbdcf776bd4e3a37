"""Synthetic YouTube index: deterministic channels, videos, comments.

Stand-in for the YouTube Data API v3 (reference client/youtube_client.go);
data shapes mirror model/youtube/types.go (YouTubeChannel / YouTubeVideo).

Video-ID scheme (11 chars, YouTube alphabet):
- a "samplable" subset has ids of the form <5 lowercase chars> + '-' +
  <5 chars>, satisfying the reference's prefix-sampling validity rule
  (len==11 AND starts-with-prefix AND id[5]=='-',
  client/youtube_client.go:1266-1281);
- search(prefix) also returns noise ids that FAIL the rule, exercising
  the filter exactly as live search results do.
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
from typing import Dict, List, Optional

import numpy as np

from ..feed.synth import _splitmix64

_ALPHABET = (
    "abcdefghijklmnopqrstuvwxyz"
    "ABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789-_"
)
UTC = _dt.timezone.utc


@dataclasses.dataclass
class YouTubeChannel:
    """model/youtube/types.go YouTubeChannel."""

    id: str
    title: str
    description: str = ""
    subscriber_count: int = 0
    video_count: int = 0
    view_count: int = 0
    country: str = ""
    published_at: Optional[_dt.datetime] = None
    thumbnails: Dict[str, str] = dataclasses.field(default_factory=dict)


@dataclasses.dataclass
class YouTubeVideo:
    """model/youtube/types.go YouTubeVideo."""

    id: str
    channel_id: str
    title: str = ""
    description: str = ""
    published_at: Optional[_dt.datetime] = None
    view_count: int = 0
    like_count: int = 0
    comment_count: int = 0
    duration: str = "PT5M30S"   # ISO-8601
    language: str = "en"
    thumbnails: Dict[str, str] = dataclasses.field(default_factory=dict)


class SyntheticYouTubeIndex:
    def __init__(self, seed: int = 99, universe_channels: int = 100_000,
                 base_date: int = 1_700_000_000):
        self.seed = seed
        self.universe = universe_channels
        self.base_date = base_date

    def _h(self, *xs) -> int:
        a = np.uint64(self.seed ^ 0xC0FFEE)
        for x in xs:
            if isinstance(x, str):
                for ch in x:
                    a = _splitmix64(a ^ np.uint64(ord(ch)))
            else:
                a = _splitmix64(a ^ np.uint64(int(x) & 0xFFFFFFFFFFFFFFFF))
        return int(a)

    # ---- channels ----

    def channel_id_of(self, n: int) -> str:
        h = self._h("chan", n)
        s = "UC"
        v = h
        for _ in range(22):
            s += _ALPHABET[v % 62]  # UC ids avoid -_ in practice
            v = (v >> 5) | ((v & 31) << 58)
            v = self._h(v) if v < 62 else v
        cid = s[:24]
        self._chan_rev()[cid] = n
        return cid

    def channel_index_of(self, channel_id: str) -> Optional[int]:
        # channel ids are regenerable; maintain a reverse check by probing
        # the deterministic generator (cheap for test-scale universes is
        # wrong at 1e5 — instead embed the index via hash lookup table)
        return self._chan_rev().get(channel_id)

    def _chan_rev(self):
        if not hasattr(self, "_rev"):
            self._rev = {}
        return self._rev

    def channel(self, n: int) -> YouTubeChannel:
        h = self._h("chanmeta", n)
        cid = self.channel_id_of(n)
        self._chan_rev()[cid] = n
        return YouTubeChannel(
            id=cid,
            title=f"Synthetic YT Channel {n}",
            description=f"Channel {n} description — see also UC friends",
            subscriber_count=h % 1_000_000,
            video_count=(h >> 20) % 30,   # some <= 10 -> gate exercised
            view_count=(h >> 25) % 50_000_000,
            country="US" if h % 3 == 0 else "",
            published_at=_dt.datetime.fromtimestamp(
                self.base_date - (h % 100_000_000), UTC
            ),
            thumbnails={"default": f"https://i.ytimg.com/ch/{cid}/default.jpg"},
        )

    # ---- videos ----

    def video_id(self, prefix: str, k: int) -> str:
        """Samplable id: prefix(5 lowercase) + '-' + 5 hash chars."""
        h = self._h("vid", prefix, k)
        tail = "".join(_ALPHABET[(h >> (6 * i)) % 64] for i in range(5))
        return prefix + "-" + tail

    def noise_id(self, salt: int) -> str:
        h = self._h("noise", salt)
        return "".join(_ALPHABET[(h >> (6 * i)) % 64] for i in range(11))[:11]

    def video(self, video_id: str) -> YouTubeVideo:
        h = self._h("vidmeta", video_id)
        n_chan = h % self.universe
        ch = self.channel(n_chan)
        secs = (h >> 8) % 7200
        duration = f"PT{secs // 60}M{secs % 60}S" if secs else "P0D"
        desc = (
            f"Video about topic {h % 1000}. "
            f"More: https://example.com/t{h % 97} "
            f"and channel https://www.youtube.com/channel/"
            f"{self.channel_id_of((h >> 13) % self.universe)}"
        )
        return YouTubeVideo(
            id=video_id,
            channel_id=ch.id,
            title=f"Synthetic video {video_id}",
            description=desc,
            published_at=_dt.datetime.fromtimestamp(
                self.base_date + (h % 10_000_000), UTC
            ),
            view_count=h % 1_000_000,
            like_count=(h >> 12) % 50_000,
            comment_count=(h >> 22) % 5_000,
            duration=duration,
            language="en" if h % 4 else "ru",
            thumbnails={
                "default": f"https://i.ytimg.com/vi/{video_id}/default.jpg",
                "high": f"https://i.ytimg.com/vi/{video_id}/hq.jpg",
            },
        )

    def search_prefix(self, prefix: str, max_results: int = 50
                      ) -> List[str]:
        """Search.List('watch?v=<prefix>'): mixed valid/noise ids
        (client/youtube_client.go:1180-1368 result shape)."""
        h = self._h("search", prefix)
        n_valid = h % 6                  # 0..5 samplable hits
        n_noise = 3 + (h >> 8) % 10
        out = [self.video_id(prefix, k) for k in range(n_valid)]
        out += [self.noise_id(h + s) for s in range(n_noise)]
        return out[:max_results]

    def channel_uploads(self, channel_id: str, limit: int = 50
                        ) -> List[str]:
        n = self._chan_rev().get(channel_id)
        if n is None:
            return []
        ch = self.channel(n)
        ids = []
        for k in range(min(ch.video_count, limit)):
            h = self._h("upload", n, k)
            prefix = "".join(chr(97 + ((h >> (5 * i)) % 26))
                             for i in range(5))
            ids.append(self.video_id(prefix, h % 7))
        return ids

    def comments(self, video_id: str, limit: int) -> List[dict]:
        v = self.video(video_id)
        n = min(v.comment_count, max(0, limit)) if limit >= 0 else min(
            v.comment_count, 100
        )
        out = []
        for k in range(min(n, 100)):
            h = self._h("comment", video_id, k)
            out.append({
                "text": f"comment {k} on {video_id} ({h % 1000})",
                "like_count": h % 500,
                "author": f"user{h % 100000}",
            })
        return out
