"""ctypes driver for the CDNA4 hot-path kernels (ops/csrc/libcrawlhip.so).

The HIP extension is built in-tree by ``__graft_entry__.build()`` (or
``python -m crawler_amd.ops.build``) with hipcc --offload-arch=gfx950.
On a GPU host the extension is REQUIRED: :func:`require_lib` raises rather
than letting callers silently fall back to the (1000x slower) Python path.
"""
from __future__ import annotations

import ctypes
import dataclasses
import datetime as _dt
import os
from typing import List, Optional, Tuple

import torch

from ..models.post import format_go_time
from . import batch as B

_CSRC = os.path.dirname(os.path.abspath(__file__)) + "/csrc"
_LIB_PATH = os.path.join(_CSRC, "libcrawlhip.so")

MAX_LINKS = 8

# Must match crawl::make_view() field order in csrc/parse_encode.hip.
_BATCH_PTR_ORDER = [
    "chat_id", "msg_id", "text_off",
    # meta int32 fields, in BatchView order:
    "m:date", "m:content_type", "m:views", "m:forwards", "m:media_album_id",
    "m:channel_idx", "m:flags", "m:text_len", "m:aux_off", "m:aux_len",
    "m:ent_off", "m:ent_cnt", "m:react_off", "m:react_cnt", "m:com_off",
    "m:com_cnt", "m:poster_off", "m:poster_len",
    "text_pool", "entities", "react_emoji", "react_count",
    "com_text_off", "com_text_len", "com_handle_off", "com_handle_len",
    "com_views", "com_replies", "com_react_off", "com_react_cnt",
    "ch_chat_id", "ch_member", "ch_postcount", "ch_totalviews",
    "ch_user_off", "ch_user_len", "ch_title_off", "ch_title_len",
]

_lib = None


def lib_available() -> bool:
    return os.path.exists(_LIB_PATH)


def load_lib():
    global _lib
    if _lib is not None:
        return _lib
    lib = ctypes.CDLL(_LIB_PATH)
    lib.crawl_batch_ptr_count.restype = ctypes.c_int
    lib.crawl_measure_extract.restype = ctypes.c_int
    lib.crawl_measure_extract.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_int,
        ctypes.c_void_p,
    ]
    lib.crawl_write.restype = ctypes.c_int
    lib.crawl_write.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_write_staged.restype = ctypes.c_int
    lib.crawl_write_staged.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_stage_budget.restype = ctypes.c_int
    _bind_dedup(lib)
    _bind_feedgen(lib)
    _bind_v2(lib)
    _bind_yt(lib)
    _bind_aux(lib)
    _lib = lib
    return lib


def _bind_v2(lib):
    lib.crawl_write_lds.restype = ctypes.c_int
    lib.crawl_write_lds.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_write_scratch.restype = ctypes.c_int
    lib.crawl_write_scratch.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_long,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_compact.restype = ctypes.c_int
    lib.crawl_compact.argtypes = [
        ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
    ]


def _bind_yt(lib):
    for fn in ("crawl_yt_gen_meta", "crawl_yt_gen_fill",
               "crawl_yt_gen_channels"):
        getattr(lib, fn).restype = ctypes.c_int
    lib.crawl_yt_gen_meta.argtypes = (
        [ctypes.c_longlong] * 5 + [ctypes.c_void_p] * 11
        + [ctypes.c_void_p])
    lib.crawl_yt_gen_fill.argtypes = (
        [ctypes.c_longlong] * 5 + [ctypes.c_void_p] * 6
        + [ctypes.c_void_p])
    lib.crawl_yt_gen_channels.argtypes = (
        [ctypes.c_longlong] * 3 + [ctypes.c_void_p, ctypes.c_longlong]
        + [ctypes.c_void_p] * 9 + [ctypes.c_void_p])
    lib.crawl_yt_ptr_count.restype = ctypes.c_int
    lib.crawl_yt_measure.restype = ctypes.c_int
    lib.crawl_yt_measure.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_yt_write.restype = ctypes.c_int
    lib.crawl_yt_write.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]


# Must match crawl::yt_make_view() in csrc/yt_encode.hip.
_YT_PTR_ORDER = [
    "vid_off", "channel_idx", "published", "views", "likes", "comments",
    "duration_s", "lang", "title_off", "title_len", "desc_off", "desc_len",
    "pool", "ch_id_off", "ch_title_off", "ch_title_len", "ch_desc_off",
    "ch_desc_len", "ch_subs", "ch_videos", "ch_views", "ch_country_off",
    "ch_country_len", "ch_published",
]


def yt_parse_encode(batch, now: Optional[_dt.datetime] = None,
                    grid: int = 0) -> Tuple[torch.Tensor, torch.Tensor,
                                            torch.Tensor]:
    """YouTube batch -> (out bytes, line_off, line_len); byte-identical to
    youtube.batch.encode_yt_batch."""
    lib = require_lib()
    dev = batch.device
    assert dev.type == "cuda"
    n = batch.n
    now = now or _dt.datetime.now(_dt.timezone.utc)
    created = format_go_time(now.replace(microsecond=0)).encode()
    capture = format_go_time(now).encode()
    to_dev = lambda b: torch.frombuffer(bytearray(b or b"\0"),
                                        dtype=torch.uint8).to(dev)
    label_t = to_dev(batch.crawl_label.encode())
    from ..youtube.batch import LANGS

    lang_t = to_dev("".join(LANGS).encode())
    created_t, capture_t = to_dev(created), to_dev(capture)
    tensors = [getattr(batch, name) for name in _YT_PTR_ORDER]
    tensors += [label_t, lang_t, created_t, capture_t]
    ptrs = _ptr_array(tensors)
    assert len(tensors) == lib.crawl_yt_ptr_count()
    scalars = (ctypes.c_long * 4)(
        n, len(batch.crawl_label.encode()), len(created), len(capture)
    )
    if grid <= 0:
        grid = min((n + 3) // 4, 8192)
    stream_ptr = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    line_len = torch.zeros(n, dtype=torch.int32, device=dev)
    rc = lib.crawl_yt_measure(ptrs, scalars,
                              ctypes.c_void_p(line_len.data_ptr()),
                              grid, stream_ptr)
    if rc != 0:
        raise RuntimeError(f"crawl_yt_measure failed: hip {rc}")
    line_off = torch.zeros(n, dtype=torch.int64, device=dev)
    torch.cumsum(line_len.to(torch.int64)[:-1], 0, out=line_off[1:])
    total = int(line_off[-1].item() + line_len[-1].item()) if n else 0
    out = torch.empty(total, dtype=torch.uint8, device=dev)
    rc = lib.crawl_yt_write(ptrs, scalars,
                            ctypes.c_void_p(line_off.data_ptr()),
                            ctypes.c_void_p(out.data_ptr()),
                            grid, stream_ptr)
    if rc != 0:
        raise RuntimeError(f"crawl_yt_write failed: hip {rc}")
    return out, line_off, line_len




def _bind_aux(lib):
    lib.crawl_channel_stats.restype = ctypes.c_int
    lib.crawl_channel_stats.argtypes = [ctypes.c_void_p] * 4 + [
        ctypes.c_int, ctypes.c_int] + [ctypes.c_void_p] * 5 + [
        ctypes.c_void_p]
    lib.crawl_reservoir_sample.restype = ctypes.c_int
    lib.crawl_reservoir_sample.argtypes = [
        ctypes.c_long, ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_html_classify.restype = ctypes.c_int
    lib.crawl_html_classify.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
    ]


def channel_stats(batch: B.MessageBatch, line_len=None):
    """Segmented per-channel engagement aggregation (SURVEY §2.6:
    GetTotalChannelViews etc. -> one workgroup per channel segment).
    Returns dict of device tensors: views/forwards/replies int64[K],
    posts int32[K], totals int64[4] (views,forwards,replies,posts)."""
    lib = require_lib()
    dev = batch.device
    K = batch.n_channels
    P = batch.n // K
    assert K * P == batch.n, "batch must be K x P channel-grouped"
    out = {
        "views": torch.zeros(K, dtype=torch.int64, device=dev),
        "forwards": torch.zeros(K, dtype=torch.int64, device=dev),
        "replies": torch.zeros(K, dtype=torch.int64, device=dev),
        "posts": torch.zeros(K, dtype=torch.int32, device=dev),
        "totals": torch.zeros(4, dtype=torch.int64, device=dev),
    }
    m = batch.meta
    rc = lib.crawl_channel_stats(
        ctypes.c_void_p(m["views"].data_ptr()),
        ctypes.c_void_p(m["forwards"].data_ptr()),
        ctypes.c_void_p(m["reply_count"].data_ptr()),
        ctypes.c_void_p(line_len.data_ptr()) if line_len is not None
        else None,
        P, K,
        ctypes.c_void_p(out["views"].data_ptr()),
        ctypes.c_void_p(out["forwards"].data_ptr()),
        ctypes.c_void_p(out["replies"].data_ptr()),
        ctypes.c_void_p(out["posts"].data_ptr()),
        ctypes.c_void_p(out["totals"].data_ptr()),
        ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
    )
    if rc != 0:
        raise RuntimeError(f"crawl_channel_stats failed: hip {rc}")
    return out


def reservoir_sample(batch: B.MessageBatch, k: int, seed: int = 0):
    """Per-channel uniform sample without replacement (SURVEY §2.6
    Fisher-Yates row). Returns int32[K, k] global row indices (device)."""
    lib = require_lib()
    dev = batch.device
    K = batch.n_channels
    P = batch.n // K
    out = torch.zeros((K, k), dtype=torch.int32, device=dev)
    rc = lib.crawl_reservoir_sample(
        seed, P, K, k, ctypes.c_void_p(out.data_ptr()),
        min(max(1, (K + 3) // 4), 8192),
        ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
    )
    if rc != 0:
        raise RuntimeError(f"crawl_reservoir_sample failed: hip {rc}")
    return out


def reservoir_sample_oracle(P: int, K: int, k: int, seed: int = 0):
    """Python replay of the device reservoir (same splitmix stream)."""
    def sm64(x):
        z = (x + 0x9E3779B97F4A7C15) & (2**64 - 1)
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & (2**64 - 1)
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & (2**64 - 1)
        return z ^ (z >> 31)

    out = []
    for c in range(K):
        slot = list(range(c * P, c * P + k))
        for i in range(k, P):
            r = sm64((seed ^ ((c * 0x9E3779B1 + i) & (2**64 - 1)))
                     & (2**64 - 1))
            j = r % (i + 1)
            if j < k:
                slot[j] = c * P + i
        out.append(slot)
    return out


_HTML_STATUS = ["valid", "not_channel", "invalid"]
_HTML_REASON = ["", "not_supergroup", "not_found", "not_found",
                "unrecognized"]


def html_classify(docs, device="cuda:0"):
    """Batch t.me HTML classification on GPU (oracle:
    engine.htmlvalidator.parse_channel_html). docs: list[bytes].
    Returns list of (status, reason) strings."""
    lib = require_lib()
    dev = torch.device(device)
    n = len(docs)
    blob = b"".join(docs) or b"\0"
    pool = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(dev)
    offs, cur = [], 0
    lens = []
    for d in docs:
        offs.append(cur)
        lens.append(len(d))
        cur += len(d)
    doc_off = torch.tensor(offs, dtype=torch.int64, device=dev)
    doc_len = torch.tensor(lens, dtype=torch.int32, device=dev)
    status = torch.zeros(n, dtype=torch.int32, device=dev)
    reason = torch.zeros(n, dtype=torch.int32, device=dev)
    rc = lib.crawl_html_classify(
        ctypes.c_void_p(pool.data_ptr()),
        ctypes.c_void_p(doc_off.data_ptr()),
        ctypes.c_void_p(doc_len.data_ptr()), n,
        ctypes.c_void_p(status.data_ptr()),
        ctypes.c_void_p(reason.data_ptr()),
        min(max(1, (n + 3) // 4), 8192),
        ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
    )
    if rc != 0:
        raise RuntimeError(f"crawl_html_classify failed: hip {rc}")
    st = status.cpu().numpy()
    rs = reason.cpu().numpy()
    return [(_HTML_STATUS[int(st[i])], _HTML_REASON[int(rs[i])])
            for i in range(n)]


def _bind_dedup(lib):
    lib.crawl_claim_links.restype = ctypes.c_int
    lib.crawl_claim_links.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_longlong, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_bloom_update.restype = ctypes.c_int
    lib.crawl_bloom_update.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_longlong, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.crawl_insert_hashes.restype = ctypes.c_int
    lib.crawl_insert_hashes.argtypes = [
        ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p, ctypes.c_longlong,
        ctypes.c_int, ctypes.c_void_p,
    ]


def _bind_feedgen(lib):
    lib.crawl_feed_meta.restype = ctypes.c_int
    lib.crawl_feed_meta.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int,
        ctypes.c_void_p,
    ]
    lib.crawl_feed_fill.restype = ctypes.c_int
    lib.crawl_feed_fill.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int,
        ctypes.c_void_p,
    ]
    lib.crawl_feed_comments.restype = ctypes.c_int
    lib.crawl_feed_comments.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_long),
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p,
        ctypes.c_long, ctypes.c_long, ctypes.POINTER(ctypes.c_void_p),
        ctypes.c_int, ctypes.c_void_p,
    ]


def require_lib():
    """On a GPU host, a missing/failed extension is an ERROR, not a fallback."""
    if not lib_available():
        raise RuntimeError(
            "libcrawlhip.so not built — run __graft_entry__.build() "
            "(hipcc --offload-arch=gfx950) before GPU execution; the Python "
            "golden path is not a substitute on GPU hosts"
        )
    return load_lib()


# Emoji / content-type constant tables shipped to the device once.
def _const_tables(device):
    epool = b"".join(e.encode("utf-8") for e in B.EMOJI_TABLE)
    eoff, cur = [], 0
    elen = []
    for e in B.EMOJI_TABLE:
        n = len(e.encode("utf-8"))
        eoff.append(cur)
        elen.append(n)
        cur += n
    cpool = b"".join(c.encode() for c in B.CONTENT_TYPES)
    coff, ccur = [], 0
    clen = []
    for c in B.CONTENT_TYPES:
        coff.append(ccur)
        clen.append(len(c))
        ccur += len(c)
    t = lambda x, dt_: torch.tensor(x, dtype=dt_, device=device)
    return {
        "emoji_pool": torch.frombuffer(bytearray(epool), dtype=torch.uint8).to(device),
        "emoji_off": t(eoff, torch.int32),
        "emoji_len": t(elen, torch.int32),
        "ctname_pool": torch.frombuffer(bytearray(cpool), dtype=torch.uint8).to(device),
        "ctname_off": t(coff, torch.int32),
        "ctname_len": t(clen, torch.int32),
    }


_tables_cache = {}


def const_tables(device):
    key = str(device)
    if key not in _tables_cache:
        _tables_cache[key] = _const_tables(device)
    return _tables_cache[key]


@dataclasses.dataclass
class EncodeResult:
    """Device-side results of one parse+encode pass."""

    out: torch.Tensor        # uint8[total_bytes] JSONL (concatenated lines)
    line_off: torch.Tensor   # int64[N] start offset of each line
    line_len: torch.Tensor   # int32[N] (0 = filtered by min_post_date)
    link_name: torch.Tensor  # uint8[N, MAX_LINKS, 32]
    link_len: torch.Tensor   # uint8[N, MAX_LINKS]
    link_src: torch.Tensor   # uint8[N, MAX_LINKS]
    link_cnt: torch.Tensor   # int32[N]
    link_hash: torch.Tensor  # int64[N, MAX_LINKS] (fnv1a64 bits)


def _ptr_array(tensors: List[torch.Tensor]):
    arr = (ctypes.c_void_p * len(tensors))()
    for k, t in enumerate(tensors):
        arr[k] = ctypes.c_void_p(t.data_ptr())
    return arr


def _stride_bound(batch: B.MessageBatch) -> int:
    """Sound per-message upper bound on the JSONL line length.

    Worst-case expansion of any escaped byte is 6x; the constant skeleton,
    numeric fields and timestamps fit in the 2200-byte base. Comment text /
    handle / reaction contributions are segment-summed per message."""
    m = batch.meta
    dev = batch.device
    i64 = lambda t: t.to(torch.int64)
    c = m["channel_idx"].to(torch.int64)
    bound = (
        2200
        + 6 * i64(m["text_len"]) + 6 * i64(m["aux_len"])
        + 12 * i64(batch.ch_title_len[c])
        + 36 * i64(batch.ch_user_len[c])
        + 6 * i64(m["poster_len"])
        + 45 * i64(m["react_cnt"])
        + 170 * i64(m["com_cnt"])
        + 35 * MAX_LINKS
    )
    ccnt = i64(m["com_cnt"])
    total_c = int(ccnt.sum().item())
    if total_c:
        cmsg = torch.repeat_interleave(
            torch.arange(batch.n, dtype=torch.int64, device=dev), ccnt
        )
        extra = (6 * (i64(batch.com_text_len) + i64(batch.com_handle_len))
                 + 45 * i64(batch.com_react_cnt))
        bound = bound.index_add(0, cmsg, extra)
    return int(bound.max().item())


# per-line staged extras beyond the text fields: outlink names
# (8 x 32 + 8) + reaction id/count ints (2 x 64 x 4) + alignment slack
_STAGE_FIXED = 812
# staged kernel's block-table caps (parse_encode.hip TG_*_CAP)
_EMOJI_POOL_CAP, _EMOJI_CAP, _TS_CAP = 512, 64, 64
_CTNAME_POOL_CAP, _CTNAME_CAP = 512, 32


def _stage_fits(batch, lib, tables, created, capture) -> bool:
    """True when every line's staged fields (user+title+poster+desc +
    links/reactions) fit the staged writer's per-wave LDS budget AND
    the per-batch tables fit its block caps; cached per batch."""
    if os.environ.get("CRAWL_NO_STAGED") == "1":
        return False
    if (tables["emoji_pool"].numel() > _EMOJI_POOL_CAP
            or tables["emoji_off"].numel() > _EMOJI_CAP
            or tables["ctname_pool"].numel() > _CTNAME_POOL_CAP
            or tables["ctname_off"].numel() > _CTNAME_CAP
            or len(created) > _TS_CAP or len(capture) > _TS_CAP):
        return False
    fit = getattr(batch, "_stage_fit", None)
    if fit is None:
        m = batch.meta
        mx = _STAGE_FIXED
        if batch.n:
            mx += (max(int(m["text_len"].max().item()),
                       int(m["aux_len"].max().item()))
                   + int(m["poster_len"].max().item()))
        if batch.n_channels:
            mx += (int(batch.ch_user_len.max().item())
                   + int(batch.ch_title_len.max().item()))
        fit = mx <= int(lib.crawl_stage_budget())
        try:
            batch._stage_fit = fit
        except AttributeError:
            pass
    return fit


def parse_encode(
    batch: B.MessageBatch,
    now: Optional[_dt.datetime] = None,
    min_post_date: Optional[_dt.datetime] = None,
    grid: int = 0,
    stream: Optional[torch.cuda.Stream] = None,
    single_pass: bool = False,
) -> EncodeResult:
    """Parse + JSONL-encode a batch on the current CUDA device.

    Default path: measure+extract then write at exact offsets, with the
    D2H overlap handled by the caller. The single_pass variant (scratch +
    compaction) measured slower on 1.25M-post batches (strided scratch
    thrashes L2) and is kept for experimentation.
    Returns device tensors; the caller D2H-copies ``out`` for the host
    writer. Byte-for-byte equal to golden_batch.encode_batch(...).
    """
    lib = require_lib()
    dev = batch.device
    assert dev.type == "cuda", "parse_encode requires a CUDA (ROCm) batch"
    n = batch.n
    tables = const_tables(dev)

    now = now or _dt.datetime.now(_dt.timezone.utc)
    created = format_go_time(now.replace(microsecond=0)).encode()
    capture = format_go_time(now).encode()
    created_t = torch.frombuffer(bytearray(created), dtype=torch.uint8).to(dev)
    capture_t = torch.frombuffer(bytearray(capture), dtype=torch.uint8).to(dev)

    tensors = []
    for name in _BATCH_PTR_ORDER:
        if name.startswith("m:"):
            tensors.append(batch.meta[name[2:]])
        else:
            tensors.append(getattr(batch, name))
    tensors += [
        tables["emoji_pool"], tables["emoji_off"], tables["emoji_len"],
        created_t, capture_t,
        tables["ctname_pool"], tables["ctname_off"], tables["ctname_len"],
    ]
    names = _BATCH_PTR_ORDER + [
        "emoji_pool", "emoji_off", "emoji_len", "created", "capture",
        "ctname_pool", "ctname_off", "ctname_len",
    ]
    for nm, t in zip(names, tensors):
        assert t.device == dev, f"{nm} on {t.device}, batch on {dev}"
        assert t.is_contiguous(), f"{nm} not contiguous"
    batch_ptrs = _ptr_array(tensors)
    assert len(tensors) == lib.crawl_batch_ptr_count()

    if min_post_date is not None:
        mpd = int(min_post_date.timestamp())
    else:
        mpd = -(1 << 62)
    scalars = (ctypes.c_long * 5)(n, 1, mpd, len(created), len(capture))

    link_name = torch.empty((n, MAX_LINKS, 32), dtype=torch.uint8, device=dev)
    link_len = torch.zeros((n, MAX_LINKS), dtype=torch.uint8, device=dev)
    link_src = torch.zeros((n, MAX_LINKS), dtype=torch.uint8, device=dev)
    link_cnt = torch.zeros(n, dtype=torch.int32, device=dev)
    link_hash = torch.zeros((n, MAX_LINKS), dtype=torch.int64, device=dev)
    line_len = torch.zeros(n, dtype=torch.int32, device=dev)
    link_ptrs = _ptr_array([link_name, link_len, link_src, link_cnt, link_hash])

    if grid <= 0:
        grid = min((n + 3) // 4, 8192)
    stream_ptr = ctypes.c_void_p(
        torch.cuda.current_stream().cuda_stream if stream is None
        else stream.cuda_stream
    )

    if single_pass:
        stride = (_stride_bound(batch) + 31) & ~15
        scratch = torch.empty(n * stride, dtype=torch.uint8, device=dev)
        overflow = torch.zeros(1, dtype=torch.int32, device=dev)
        rc = lib.crawl_write_scratch(
            batch_ptrs, scalars, link_ptrs,
            ctypes.c_void_p(scratch.data_ptr()), stride,
            ctypes.c_void_p(line_len.data_ptr()),
            ctypes.c_void_p(overflow.data_ptr()), grid, stream_ptr,
        )
        if rc != 0:
            raise RuntimeError(f"crawl_write_scratch failed: hip error {rc}")
        line_off = torch.zeros(n, dtype=torch.int64, device=dev)
        torch.cumsum(line_len.to(torch.int64)[:-1], 0, out=line_off[1:])
        total = int(line_off[-1].item() + line_len[-1].item()) if n else 0
        ovf = int(overflow.item())
        if ovf:
            raise RuntimeError(
                f"line overflowed scratch stride ({ovf} > {stride}); "
                "stride bound is unsound for this batch"
            )
        out = torch.empty(total, dtype=torch.uint8, device=dev)
        rc = lib.crawl_compact(
            ctypes.c_void_p(scratch.data_ptr()), stride,
            ctypes.c_void_p(line_off.data_ptr()),
            ctypes.c_void_p(line_len.data_ptr()),
            ctypes.c_void_p(out.data_ptr()), n, grid, stream_ptr,
        )
        if rc != 0:
            raise RuntimeError(f"crawl_compact failed: hip error {rc}")
    else:
        rc = lib.crawl_measure_extract(
            batch_ptrs, scalars, link_ptrs,
            ctypes.c_void_p(line_len.data_ptr()), grid, stream_ptr,
        )
        if rc != 0:
            raise RuntimeError(
                f"crawl_measure_extract launch failed: hip error {rc}"
            )
        line_off = torch.zeros(n, dtype=torch.int64, device=dev)
        torch.cumsum(line_len.to(torch.int64)[:-1], 0, out=line_off[1:])
        total = int(line_off[-1].item() + line_len[-1].item()) if n else 0
        out = torch.empty(total, dtype=torch.uint8, device=dev)
        if (not os.environ.get("CRAWL_LDS_WRITE")
                and _stage_fits(batch, lib, tables, created, capture)):
            # staged writer: escape-scanned fields + reactions +
            # outlinks + per-batch tables go through LDS so in-line
            # loads never wait the store FIFO (profiles/
            # r02_valu_diet.md); host verified everything fits
            rc = lib.crawl_write_staged(
                batch_ptrs, scalars, link_ptrs,
                ctypes.c_void_p(line_off.data_ptr()),
                ctypes.c_void_p(line_len.data_ptr()),
                ctypes.c_void_p(out.data_ptr()),
                tables["emoji_off"].numel(),
                tables["emoji_pool"].numel(),
                tables["ctname_off"].numel(),
                tables["ctname_pool"].numel(), grid, stream_ptr,
            )
        else:
            writer = (lib.crawl_write_lds
                      if os.environ.get("CRAWL_LDS_WRITE")
                      else lib.crawl_write)
            rc = writer(
                batch_ptrs, scalars, link_ptrs,
                ctypes.c_void_p(line_off.data_ptr()),
                ctypes.c_void_p(line_len.data_ptr()),
                ctypes.c_void_p(out.data_ptr()), grid, stream_ptr,
            )
        if rc != 0:
            raise RuntimeError(f"crawl_write launch failed: hip error {rc}")

    return EncodeResult(
        out=out, line_off=line_off, line_len=line_len,
        link_name=link_name, link_len=link_len, link_src=link_src,
        link_cnt=link_cnt, link_hash=link_hash,
    )


class SeenSet:
    """Device-resident seen-channel set (hash table + bloom).

    Replaces the reference's DiscoveredChannels map / URL dedup cache
    (state/datamodels.go:118-162, daprstate.go:550-657) with an HBM
    open-addressing table claimed via atomicCAS (exactly-once discovery)
    plus a bitwise-OR-mergeable bloom for cross-rank union over RCCL.
    """

    def __init__(self, device, slots_log2: int = 22, bloom_bits_log2: int = 26):
        self.device = device
        self.slots = 1 << slots_log2
        self.bloom_bits = 1 << bloom_bits_log2
        self.table = torch.zeros(self.slots, dtype=torch.int64, device=device)
        self.bloom = torch.zeros(
            self.bloom_bits // 32, dtype=torch.int32, device=device
        )
        self._n_new = torch.zeros(1, dtype=torch.int32, device=device)
        self.lib = require_lib()

    def _stream(self):
        return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)

    def claim(self, res: EncodeResult) -> torch.Tensor:
        """Claim every extracted link; returns new_mask uint8[N, MAX_LINKS].

        Also updates the bloom. Reads self.new_count() for the number of
        first-time discoveries in this batch.
        """
        n = res.link_cnt.shape[0]
        new_mask = torch.zeros(
            (n, MAX_LINKS), dtype=torch.uint8, device=self.device
        )
        self._n_new.zero_()
        grid = min((n * MAX_LINKS + 255) // 256, 4096)
        rc = self.lib.crawl_claim_links(
            ctypes.c_void_p(res.link_hash.data_ptr()),
            ctypes.c_void_p(res.link_cnt.data_ptr()),
            n, MAX_LINKS,
            ctypes.c_void_p(self.table.data_ptr()),
            ctypes.c_longlong(self.slots),
            ctypes.c_void_p(new_mask.data_ptr()),
            ctypes.c_void_p(self._n_new.data_ptr()),
            grid, self._stream(),
        )
        if rc != 0:
            raise RuntimeError(f"crawl_claim_links failed: hip error {rc}")
        rc = self.lib.crawl_bloom_update(
            ctypes.c_void_p(res.link_hash.data_ptr()),
            ctypes.c_void_p(res.link_cnt.data_ptr()),
            n, MAX_LINKS,
            ctypes.c_void_p(self.bloom.data_ptr()),
            ctypes.c_longlong(self.bloom_bits),
            grid, self._stream(),
        )
        if rc != 0:
            raise RuntimeError(f"crawl_bloom_update failed: hip error {rc}")
        return new_mask

    def new_count(self) -> int:
        return int(self._n_new.item())

    def compact_claimed(self, res: EncodeResult, new_mask: torch.Tensor):
        """Densify the claim winners ON DEVICE: returns
        (names uint8[M, 32] zero-padded, hashes int64[M]) — row order is
        the atomic claim order (consumers sort host-side). Replaces a
        host nonzero+gather+pad over N*MAX_LINKS candidate slots."""
        n = res.link_cnt.shape[0]
        cap = n * MAX_LINKS
        out_names = torch.empty((cap, 32), dtype=torch.uint8,
                                device=self.device)
        out_hashes = torch.empty(cap, dtype=torch.int64,
                                 device=self.device)
        cursor = torch.zeros(1, dtype=torch.int32, device=self.device)
        grid = min((cap + 255) // 256, 4096)
        rc = self.lib.crawl_claim_compact(
            ctypes.c_void_p(new_mask.data_ptr()),
            ctypes.c_void_p(res.link_name.data_ptr()),
            ctypes.c_void_p(res.link_len.data_ptr()),
            ctypes.c_void_p(res.link_hash.data_ptr()),
            n, MAX_LINKS,
            ctypes.c_void_p(out_names.data_ptr()),
            ctypes.c_void_p(out_hashes.data_ptr()),
            ctypes.c_void_p(cursor.data_ptr()),
            ctypes.c_uint(cap), grid, self._stream(),
        )
        if rc != 0:
            raise RuntimeError(
                f"crawl_claim_compact failed: hip error {rc}")
        m = int(cursor.item())  # syncs the stream
        return out_names[:m], out_hashes[:m]

    def merge_remote(self, res_or_hashes, dist, world: int,
                     group=None) -> int:
        """Cross-rank seen-set union: exchange this
        rank's newly-claimed hashes with every peer (count-sized
        all-gather over RCCL — NO cap, round 1's silent 64k truncation
        is gone) and insert the remote ones into the local table. Also
        OR-unions the bloom pre-filter. Returns the number of remote
        hashes inserted."""
        from ..parallel import collectives as C

        local = (res_or_hashes if isinstance(res_or_hashes, torch.Tensor)
                 else res_or_hashes.link_hash.flatten())
        per_rank = C.allgather_hashes(local, dist, world,
                                      device=self.device, group=group)
        rank = C._safe_rank(dist, group)
        inserted = 0
        for r, h in enumerate(per_rank):
            if r == rank:
                continue
            h = h[h != 0].to(self.device)
            self.insert_hashes(h)
            inserted += int(h.numel())
        C.bloom_union(self.bloom, dist, world, group=group)
        return inserted

    def insert_hashes(self, hashes: torch.Tensor) -> None:
        """Bulk-insert merged remote hashes (post all-gather)."""
        n = hashes.numel()
        if n == 0:
            return
        grid = min((n + 255) // 256, 4096)
        rc = self.lib.crawl_insert_hashes(
            ctypes.c_void_p(hashes.data_ptr()), ctypes.c_long(n),
            ctypes.c_void_p(self.table.data_ptr()),
            ctypes.c_longlong(self.slots), grid, self._stream(),
        )
        if rc != 0:
            raise RuntimeError(f"crawl_insert_hashes failed: hip error {rc}")


def fnv1a64(data: bytes) -> int:
    """Python oracle for the device hash (csrc/common.h fnv1a64)."""
    h = 1469598103934665603
    for b in data:
        h ^= b
        h = (h * 1099511628211) & 0xFFFFFFFFFFFFFFFF
    return h


def links_to_python(res: EncodeResult) -> List[List[Tuple[str, str]]]:
    """Decode the device link output to [(name, source_type), ...] per msg."""
    srcs = ["mention", "text_url", "url", "plaintext"]
    name = res.link_name.cpu().numpy()
    ln = res.link_len.cpu().numpy()
    src = res.link_src.cpu().numpy()
    cnt = res.link_cnt.cpu().numpy()
    out = []
    for i in range(len(cnt)):
        row = []
        for k in range(int(cnt[i])):
            row.append(
                (bytes(name[i, k, : ln[i, k]]).decode(), srcs[int(src[i, k])])
            )
        out.append(row)
    return out
