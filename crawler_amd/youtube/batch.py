"""Packed YouTube video batch — device record layout for the yt encoder.

SoA mirror of ops/batch.py for the YouTube platform: one row per video,
byte pools for ids/titles/descriptions/channel strings. The GPU emitter
(csrc/yt_encode.hip) consumes this and must produce byte-identical JSONL
to :func:`encode_yt_batch` (which goes through convert_video_to_post).
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
from typing import List, Optional, Tuple

import numpy as np
import torch

from .convert import convert_video_to_post, parse_iso8601_duration
from .synth import SyntheticYouTubeIndex, YouTubeChannel, YouTubeVideo

LANGS = ["en", "ru"]
UTC = _dt.timezone.utc


@dataclasses.dataclass
class YouTubeBatch:
    n: int
    vid_off: torch.Tensor       # int32[N] -> pool (11 bytes each)
    channel_idx: torch.Tensor   # int32[N]
    published: torch.Tensor     # int64[N] unix secs
    views: torch.Tensor         # int64[N]
    likes: torch.Tensor         # int32[N]
    comments: torch.Tensor      # int32[N]
    duration_s: torch.Tensor    # int32[N]; -1 = null (P0D/empty)
    lang: torch.Tensor          # int32[N] -> LANGS
    title_off: torch.Tensor     # int64[N]
    title_len: torch.Tensor     # int32[N]
    desc_off: torch.Tensor      # int64[N]
    desc_len: torch.Tensor      # int32[N]
    pool: torch.Tensor          # uint8
    # channel table
    n_channels: int
    ch_id_off: torch.Tensor     # int32[K] (24 bytes each)
    ch_title_off: torch.Tensor
    ch_title_len: torch.Tensor
    ch_desc_off: torch.Tensor
    ch_desc_len: torch.Tensor
    ch_subs: torch.Tensor       # int64[K]
    ch_videos: torch.Tensor     # int32[K]
    ch_views: torch.Tensor      # int64[K]
    ch_country_off: torch.Tensor
    ch_country_len: torch.Tensor
    ch_published: torch.Tensor  # int64[K]
    crawl_label: str = ""

    def to(self, device) -> "YouTubeBatch":
        kw = {}
        for f in dataclasses.fields(self):
            v = getattr(self, f.name)
            kw[f.name] = v.to(device) if isinstance(v, torch.Tensor) else v
        return YouTubeBatch(**kw)

    @property
    def device(self):
        return self.pool.device


def pack_videos(videos: List[YouTubeVideo],
                channels: List[YouTubeChannel],
                channel_of: List[int],
                crawl_label: str = "") -> YouTubeBatch:
    n = len(videos)
    parts: List[bytes] = []
    off = 0

    def add(b: bytes) -> int:
        nonlocal off
        parts.append(b)
        o = off
        off += len(b)
        return o

    vid_off = np.zeros(n, dtype=np.int32)
    published = np.zeros(n, dtype=np.int64)
    views = np.zeros(n, dtype=np.int64)
    likes = np.zeros(n, dtype=np.int32)
    comments = np.zeros(n, dtype=np.int32)
    duration_s = np.zeros(n, dtype=np.int32)
    lang = np.zeros(n, dtype=np.int32)
    title_off = np.zeros(n, dtype=np.int64)
    title_len = np.zeros(n, dtype=np.int32)
    desc_off = np.zeros(n, dtype=np.int64)
    desc_len = np.zeros(n, dtype=np.int32)
    for i, v in enumerate(videos):
        vb = v.id.encode()
        assert len(vb) == 11
        vid_off[i] = add(vb)
        published[i] = int(v.published_at.timestamp()) if v.published_at else 0
        views[i] = v.view_count
        likes[i] = v.like_count
        comments[i] = v.comment_count
        if not v.duration or v.duration == "P0D":
            duration_s[i] = -1
        else:
            d = parse_iso8601_duration(v.duration)
            duration_s[i] = -1 if d is None else d
        lang[i] = LANGS.index(v.language) if v.language in LANGS else 0
        tb = v.title.encode()
        title_off[i] = add(tb)
        title_len[i] = len(tb)
        db = v.description.encode()
        desc_off[i] = add(db)
        desc_len[i] = len(db)

    k = len(channels)
    ch = {f: np.zeros(k, dtype=np.int64 if f in
                      ("subs", "views", "published") else np.int32)
          for f in ["id_off", "title_off", "title_len", "desc_off",
                    "desc_len", "subs", "videos", "views", "country_off",
                    "country_len", "published"]}
    for c, row in enumerate(channels):
        cb = row.id.encode()
        assert len(cb) == 24
        ch["id_off"][c] = add(cb)
        tb = row.title.encode()
        ch["title_off"][c] = add(tb)
        ch["title_len"][c] = len(tb)
        db = row.description.encode()
        ch["desc_off"][c] = add(db)
        ch["desc_len"][c] = len(db)
        ch["subs"][c] = row.subscriber_count
        ch["videos"][c] = row.video_count
        ch["views"][c] = row.view_count
        nb = row.country.encode()
        ch["country_off"][c] = add(nb)
        ch["country_len"][c] = len(nb)
        ch["published"][c] = (int(row.published_at.timestamp())
                              if row.published_at else 0)

    blob = b"".join(parts) or b"\0"
    t = torch.from_numpy
    return YouTubeBatch(
        n=n,
        vid_off=t(vid_off),
        channel_idx=t(np.array(channel_of, dtype=np.int32)),
        published=t(published), views=t(views), likes=t(likes),
        comments=t(comments), duration_s=t(duration_s), lang=t(lang),
        title_off=t(title_off), title_len=t(title_len),
        desc_off=t(desc_off), desc_len=t(desc_len),
        pool=torch.frombuffer(bytearray(blob), dtype=torch.uint8),
        n_channels=k,
        ch_id_off=t(ch["id_off"].astype(np.int32)),
        ch_title_off=t(ch["title_off"].astype(np.int32)),
        ch_title_len=t(ch["title_len"].astype(np.int32)),
        ch_desc_off=t(ch["desc_off"].astype(np.int32)),
        ch_desc_len=t(ch["desc_len"].astype(np.int32)),
        ch_subs=t(ch["subs"]), ch_videos=t(ch["videos"].astype(np.int32)),
        ch_views=t(ch["views"]),
        ch_country_off=t(ch["country_off"].astype(np.int32)),
        ch_country_len=t(ch["country_len"].astype(np.int32)),
        ch_published=t(ch["published"]),
        crawl_label=crawl_label,
    )


def build_corpus(index: SyntheticYouTubeIndex, n_videos: int,
                 crawl_label: str = "") -> YouTubeBatch:
    """Deterministic corpus: sequential samplable video ids + their
    channels (host-side; setup cost only)."""
    import string

    videos: List[YouTubeVideo] = []
    chan_rows: List[YouTubeChannel] = []
    chan_pos = {}
    channel_of: List[int] = []
    k = 0
    i = 0
    while len(videos) < n_videos:
        # deterministic prefix walk over aaaaa..zzzzz space
        p = ""
        v = i
        for _ in range(5):
            p = string.ascii_lowercase[v % 26] + p
            v //= 26
        vid = index.video_id(p, i % 7)
        video = index.video(vid)
        videos.append(video)
        cidx = chan_pos.get(video.channel_id)
        if cidx is None:
            cidx = len(chan_rows)
            chan_pos[video.channel_id] = cidx
            n = index.channel_index_of(video.channel_id)
            chan_rows.append(index.channel(n))
        channel_of.append(cidx)
        i += 1
        k += 1
    return pack_videos(videos, chan_rows, channel_of, crawl_label)


def unpack_video(b: YouTubeBatch, i: int
                 ) -> Tuple[YouTubeVideo, YouTubeChannel]:
    pool = b.pool.numpy()

    def s(o, ln):
        return bytes(pool[o:o + ln]).decode()

    vid = s(int(b.vid_off[i]), 11)
    c = int(b.channel_idx[i])
    dur = int(b.duration_s[i])
    video = YouTubeVideo(
        id=vid,
        channel_id=s(int(b.ch_id_off[c]), 24),
        title=s(int(b.title_off[i]), int(b.title_len[i])),
        description=s(int(b.desc_off[i]), int(b.desc_len[i])),
        published_at=_dt.datetime.fromtimestamp(int(b.published[i]), UTC),
        view_count=int(b.views[i]),
        like_count=int(b.likes[i]),
        comment_count=int(b.comments[i]),
        duration="P0D" if dur < 0 else f"PT{dur}S",
        language=LANGS[int(b.lang[i])],
        thumbnails={
            "default": f"https://i.ytimg.com/vi/{vid}/default.jpg",
            "high": f"https://i.ytimg.com/vi/{vid}/hq.jpg",
        },
    )
    channel = YouTubeChannel(
        id=video.channel_id,
        title=s(int(b.ch_title_off[c]), int(b.ch_title_len[c])),
        description=s(int(b.ch_desc_off[c]), int(b.ch_desc_len[c])),
        subscriber_count=int(b.ch_subs[c]),
        video_count=int(b.ch_videos[c]),
        view_count=int(b.ch_views[c]),
        country=s(int(b.ch_country_off[c]), int(b.ch_country_len[c])),
        published_at=_dt.datetime.fromtimestamp(int(b.ch_published[c]), UTC),
        thumbnails={"default": (
            f"https://i.ytimg.com/ch/{video.channel_id}/default.jpg")},
    )
    return video, channel


def encode_yt_batch(b: YouTubeBatch,
                    now: Optional[_dt.datetime] = None) -> List[bytes]:
    """CPU oracle: batch -> JSONL lines via convert_video_to_post."""
    now = now or _dt.datetime.now(UTC)
    out = []
    for i in range(b.n):
        video, channel = unpack_video(b, i)
        post = convert_video_to_post(video, channel,
                                     crawl_label=b.crawl_label, now=now)
        out.append(post.to_jsonl().encode())
    return out
