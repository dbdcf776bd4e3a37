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


# ---------- vectorized corpus builder ----------

_ALPH64 = ("abcdefghijklmnopqrstuvwxyz"
           "ABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789-_")
_ALPH64_U8 = np.frombuffer(_ALPH64.encode(), dtype=np.uint8)


def _smx(x):
    from ..feed.synth import _splitmix64

    return _splitmix64(x.astype(np.uint64))


def _chain_scalar(seed: int, s: str) -> np.uint64:
    """Scalar splitmix chain over a constant string (synth._h prefix)."""
    a = np.uint64(seed ^ 0xC0FFEE)
    for ch in s:
        a = _smx(np.asarray(a ^ np.uint64(ord(ch))))
    return np.uint64(a)


def _chain_cols(a0: np.uint64, cols: np.ndarray) -> np.ndarray:
    """Vector splitmix chain: cols is uint64[n, m]; one round per col."""
    a = np.full(cols.shape[0], a0, dtype=np.uint64)
    for j in range(cols.shape[1]):
        a = _smx(a ^ cols[:, j])
    return a


def _cid_bytes_vec(index: SyntheticYouTubeIndex,
                   n_vec: np.ndarray) -> np.ndarray:
    """Vectorized channel_id_of: uint8[n, 24] 'UC' + 22 chars
    (synth.py:82-91 incl. the rotate + rare re-hash step)."""
    a0 = _chain_scalar(index.seed, "chan")
    h = _smx(np.full(len(n_vec), a0, dtype=np.uint64)
             ^ n_vec.astype(np.uint64))
    out = np.empty((len(n_vec), 24), dtype=np.uint8)
    out[:, 0] = ord("U")
    out[:, 1] = ord("C")
    v = h.copy()
    seed_base = np.uint64(index.seed ^ 0xC0FFEE)
    for j in range(22):
        out[:, 2 + j] = _ALPH64_U8[(v % np.uint64(62)).astype(np.int64)]
        v = (v >> np.uint64(5)) | ((v & np.uint64(31)) << np.uint64(58))
        small = v < np.uint64(62)
        if small.any():
            v = np.where(small, _smx(seed_base ^ v), v)
    return out


def build_corpus_fast(index: SyntheticYouTubeIndex, n_videos: int,
                      crawl_label: str = "") -> YouTubeBatch:
    """Vectorized build_corpus: byte-identical encode output (pinned by
    tests/test_youtube.py::test_build_corpus_fast_matches_slow), ~20x
    faster (the python path costs ~140 us/video; 1.2M-video bench
    corpora took minutes of setup)."""
    n = n_videos
    i_arr = np.arange(n, dtype=np.uint64)

    # prefix chars: 5-digit base-26 of i, MSB first (build_corpus walk)
    pchars = np.empty((n, 5), dtype=np.uint64)
    for j in range(5):
        pchars[:, j] = (i_arr // (26 ** (4 - j))) % 26 + ord("a")

    # h1 = _h("vid", prefix, i % 7); tail chars from h1
    a_vid = _chain_scalar(index.seed, "vid")
    h1 = _chain_cols(a_vid, pchars)
    h1 = _smx(h1 ^ (i_arr % np.uint64(7)))
    vid_u8 = np.empty((n, 11), dtype=np.uint8)
    vid_u8[:, :5] = pchars.astype(np.uint8)
    vid_u8[:, 5] = ord("-")
    for j in range(5):
        vid_u8[:, 6 + j] = _ALPH64_U8[
            ((h1 >> np.uint64(6 * j)) % np.uint64(64)).astype(np.int64)]

    # h = _h("vidmeta", video_id)
    a_meta = _chain_scalar(index.seed, "vidmeta")
    h = _chain_cols(a_meta, vid_u8.astype(np.uint64))

    universe = np.uint64(index.universe)
    n_chan = (h % universe).astype(np.int64)
    secs = ((h >> np.uint64(8)) % np.uint64(7200)).astype(np.int32)
    duration_s = np.where(secs == 0, np.int32(-1), secs)
    published = (np.uint64(index.base_date)
                 + h % np.uint64(10_000_000)).astype(np.int64)
    views = (h % np.uint64(1_000_000)).astype(np.int64)
    likes = ((h >> np.uint64(12)) % np.uint64(50_000)).astype(np.int32)
    comments = ((h >> np.uint64(22)) % np.uint64(5_000)).astype(np.int32)
    lang = np.where((h % np.uint64(4)) == 0, np.int32(LANGS.index("ru")),
                    np.int32(LANGS.index("en")))

    # channel table: first-appearance order (build_corpus chan_pos)
    uniq, first_idx, inv = np.unique(n_chan, return_index=True,
                                     return_inverse=True)
    order = np.argsort(first_idx, kind="stable")
    chan_ns = uniq[order]                       # table rows, in order
    rank_of_uniq = np.empty_like(order)
    rank_of_uniq[order] = np.arange(len(order))
    channel_idx = rank_of_uniq[inv].astype(np.int32)

    # strings (the only python-loop part: ~2-3 us/video)
    topic = (h % np.uint64(1000)).astype(np.int64)
    tnum = (h % np.uint64(97)).astype(np.int64)
    link_cid = _cid_bytes_vec(index,
                              ((h >> np.uint64(13)) % universe))
    d1 = b"Video about topic "
    d2 = b". More: https://example.com/t"
    d3 = b" and channel https://www.youtube.com/channel/"
    title_prefix = b"Synthetic video "
    parts: List[bytes] = []
    off = 0
    vid_off = np.zeros(n, dtype=np.int32)
    title_off = np.zeros(n, dtype=np.int64)
    title_len = np.zeros(n, dtype=np.int32)
    desc_off = np.zeros(n, dtype=np.int64)
    desc_len = np.zeros(n, dtype=np.int32)
    vid_bytes = [bytes(vid_u8[i]) for i in range(n)]
    link_bytes = [bytes(link_cid[i]) for i in range(n)]
    for i in range(n):
        vb = vid_bytes[i]
        vid_off[i] = off
        parts.append(vb)
        off += 11
        tb = title_prefix + vb
        title_off[i] = off
        title_len[i] = 27
        parts.append(tb)
        off += 27
        db = b"%s%d%s%d%s%s" % (d1, topic[i], d2, tnum[i], d3,
                                link_bytes[i])
        desc_off[i] = off
        desc_len[i] = len(db)
        parts.append(db)
        off += len(db)

    k = len(chan_ns)
    a_chan = _chain_scalar(index.seed, "chanmeta")
    hc = _smx(np.full(k, a_chan, dtype=np.uint64)
              ^ chan_ns.astype(np.uint64))
    ch_id_u8 = _cid_bytes_vec(index, chan_ns)
    ch = {f: np.zeros(k, dtype=np.int64 if f in
                      ("subs", "views", "published") else np.int32)
          for f in ["id_off", "title_off", "title_len", "desc_off",
                    "desc_len", "subs", "videos", "views", "country_off",
                    "country_len", "published"]}
    ch["subs"] = (hc % np.uint64(1_000_000)).astype(np.int64)
    ch["videos"] = ((hc >> np.uint64(20)) % np.uint64(30)).astype(np.int32)
    ch["views"] = ((hc >> np.uint64(25))
                   % np.uint64(50_000_000)).astype(np.int64)
    ch["published"] = (np.int64(index.base_date)
                       - (hc % np.uint64(100_000_000)).astype(np.int64))
    us = (hc % np.uint64(3)) == 0
    for c in range(k):
        cb = bytes(ch_id_u8[c])
        ch["id_off"][c] = off
        parts.append(cb)
        off += 24
        tb = b"Synthetic YT Channel %d" % chan_ns[c]
        ch["title_off"][c] = off
        ch["title_len"][c] = len(tb)
        parts.append(tb)
        off += len(tb)
        db = ("Channel %d description — see also UC friends"
              % chan_ns[c]).encode()
        ch["desc_off"][c] = off
        ch["desc_len"][c] = len(db)
        parts.append(db)
        off += len(db)
        nb = b"US" if us[c] else b""
        ch["country_off"][c] = off
        ch["country_len"][c] = len(nb)
        parts.append(nb)
        off += len(nb)

    blob = b"".join(parts) or b"\0"
    t = torch.from_numpy
    return YouTubeBatch(
        n=n,
        vid_off=t(vid_off), channel_idx=t(channel_idx),
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


def build_corpus_device(index: SyntheticYouTubeIndex, n_videos: int,
                        device, crawl_label: str = "",
                        i0: int = 0) -> YouTubeBatch:
    """GPU-resident corpus generation (csrc/yt_feedgen.hip) — the
    YouTube half of the device synthetic-API engine. Byte-identical to
    build_corpus_fast (GPU test pins it); generation never leaves HBM."""
    import ctypes

    from ..feed.synth import _splitmix64
    from ..ops import gpu as _g

    lib = _g.require_lib()
    dev = torch.device(device)
    n = n_videos
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    z64 = lambda m, dt_: torch.empty(m, dtype=dt_, device=dev)
    vid_pool = z64(n * 11, torch.uint8)
    published = z64(n, torch.int64)
    views = z64(n, torch.int64)
    likes = z64(n, torch.int32)
    comments = z64(n, torch.int32)
    duration_s = z64(n, torch.int32)
    lang = z64(n, torch.int32)
    n_chan = z64(n, torch.int64)
    topic = z64(n, torch.int32)
    tnum = z64(n, torch.int32)
    desc_len = z64(n, torch.int32)
    cv = ctypes.c_void_p
    rc = lib.crawl_yt_gen_meta(
        index.seed, index.universe, index.base_date, n, i0,
        cv(vid_pool.data_ptr()), cv(published.data_ptr()),
        cv(views.data_ptr()), cv(likes.data_ptr()),
        cv(comments.data_ptr()), cv(duration_s.data_ptr()),
        cv(lang.data_ptr()), cv(n_chan.data_ptr()),
        cv(topic.data_ptr()), cv(tnum.data_ptr()),
        cv(desc_len.data_ptr()), stream)
    if rc != 0:
        raise RuntimeError(f"crawl_yt_gen_meta failed: hip {rc}")

    # layout: [vid n*11 | title n*27 | desc var | channel region]
    vid_off = (torch.arange(n, dtype=torch.int64, device=dev) * 11)
    title_off = n * 11 + torch.arange(n, dtype=torch.int64,
                                      device=dev) * 27
    desc_base = n * 11 + n * 27
    dl64 = desc_len.to(torch.int64)
    desc_off = torch.zeros(n, dtype=torch.int64, device=dev)
    if n > 1:
        torch.cumsum(dl64[:-1], 0, out=desc_off[1:])
    desc_off += desc_base
    desc_total = int(desc_base + dl64.sum().item())

    # channel table in FIRST-APPEARANCE order
    uniq, inverse = torch.unique(n_chan, sorted=True,
                                 return_inverse=True)
    K = uniq.numel()
    first = torch.full((K,), n, dtype=torch.int64, device=dev)
    first.scatter_reduce_(0, inverse,
                          torch.arange(n, dtype=torch.int64, device=dev),
                          reduce="amin")
    order = torch.argsort(first)
    rank = torch.empty_like(order)
    rank[order] = torch.arange(K, dtype=torch.int64, device=dev)
    channel_idx = rank[inverse].to(torch.int32)
    chan_ns = uniq[order]

    # channel string lengths (digits via comparisons; hc on host numpy
    # for the country flag — torch lacks uint64 bit ops)
    digs = torch.ones(K, dtype=torch.int64, device=dev)
    p10 = 10
    for _ in range(18):
        digs += (chan_ns >= p10).to(torch.int64)
        p10 *= 10
    title_len_ch = 21 + digs
    desc_len_ch = 8 + digs + 36
    ns_host = chan_ns.cpu().numpy().astype(np.uint64)
    a = np.uint64(index.seed ^ 0xC0FFEE)
    for chf in "chanmeta":
        a = _splitmix64(np.asarray(a ^ np.uint64(ord(chf))))
    hc = _splitmix64(np.full(K, a, dtype=np.uint64) ^ ns_host)
    country_len_np = np.where(hc % np.uint64(3) == 0, 2, 0)
    country_len = torch.from_numpy(
        country_len_np.astype(np.int64)).to(dev)
    block = 24 + title_len_ch + desc_len_ch + country_len
    ch_base = torch.zeros(K, dtype=torch.int64, device=dev)
    if K > 1:
        torch.cumsum(block[:-1], 0, out=ch_base[1:])
    ch_base += desc_total
    id_off = ch_base
    t_off = id_off + 24
    d_off = t_off + title_len_ch
    c_off = d_off + desc_len_ch
    total = int(desc_total + block.sum().item())

    pool = torch.zeros(total, dtype=torch.uint8, device=dev)
    rc = lib.crawl_yt_gen_fill(
        index.seed, index.universe, index.base_date, n, i0,
        cv(vid_pool.data_ptr()), cv(topic.data_ptr()),
        cv(tnum.data_ptr()), cv(title_off.data_ptr()),
        cv(desc_off.data_ptr()), cv(pool.data_ptr()), stream)
    if rc != 0:
        raise RuntimeError(f"crawl_yt_gen_fill failed: hip {rc}")
    subs = z64(K, torch.int64)
    videos_t = z64(K, torch.int32)
    ch_views = z64(K, torch.int64)
    ch_published = z64(K, torch.int64)
    rc = lib.crawl_yt_gen_channels(
        index.seed, index.universe, index.base_date,
        cv(chan_ns.data_ptr()), K, cv(id_off.data_ptr()),
        cv(t_off.data_ptr()), cv(d_off.data_ptr()),
        cv(c_off.data_ptr()), cv(subs.data_ptr()),
        cv(videos_t.data_ptr()), cv(ch_views.data_ptr()),
        cv(ch_published.data_ptr()), cv(pool.data_ptr()), stream)
    if rc != 0:
        raise RuntimeError(f"crawl_yt_gen_channels failed: hip {rc}")

    return YouTubeBatch(
        n=n, vid_off=vid_off.to(torch.int32), channel_idx=channel_idx,
        published=published, views=views, likes=likes,
        comments=comments, duration_s=duration_s, lang=lang,
        title_off=title_off, title_len=torch.full(
            (n,), 27, dtype=torch.int32, device=dev),
        desc_off=desc_off, desc_len=desc_len, pool=pool,
        n_channels=int(K),
        ch_id_off=id_off.to(torch.int32),
        ch_title_off=t_off.to(torch.int32),
        ch_title_len=title_len_ch.to(torch.int32),
        ch_desc_off=d_off.to(torch.int32),
        ch_desc_len=desc_len_ch.to(torch.int32),
        ch_subs=subs, ch_videos=videos_t, ch_views=ch_views,
        ch_country_off=c_off.to(torch.int32),
        ch_country_len=country_len.to(torch.int32),
        ch_published=ch_published,
        crawl_label=crawl_label,
    )
