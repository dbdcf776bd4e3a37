"""Synthetic Telegram feed engine — the TDLib stand-in.

Generates deterministic, TDLib-shaped message batches directly in the packed
SoA format (ops/batch.py) that the HIP parse/encode kernels consume. The
design goal is vectorized generation (numpy, no per-message Python) so a
10M-post corpus materializes in seconds at bench setup.

Mechanics:
- channel usernames are FIXED WIDTH: "c" + 10 digits (11 bytes), so link /
  mention slots inside text templates sit at constant byte+UTF-16 offsets;
- each message picks one of ~16 byte templates (weighted); a template
  carries: pool bytes (text + optional url tail for text_url entities),
  entity rows at constant offsets, digit slots to patch with the target
  channel number, its content type, and aux strings;
- all scalar fields derive from splitmix64 hashes of (seed, msg index), so
  any message is reproducible in isolation (GPU and CPU agree);
- templates include non-ASCII (Cyrillic, emoji incl. surrogate pairs),
  JSON-escaping hazards (quotes, backslash, <, >, &, newline, U+2028) and
  reserved t.me paths, to exercise the UTF-16 entity math and the Go-style
  escaper end to end.

FLOOD_WAIT / TDLib-400 injection and cache-vs-server latency classes are
implemented at the client facade level (feed/client.py), not in the batch
data itself (reference: telegramhelper/rate_limiter.go, crawl/runner.go:55-104).
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional, Tuple

import numpy as np
import torch

from ..ops import batch as B

USERNAME_WIDTH = 11  # "c" + 10 digits
HANDLE_WIDTH = 9     # "u" + 8 digits


def _utf16_len(s: str) -> int:
    return sum(2 if ord(c) >= 0x10000 else 1 for c in s)


@dataclasses.dataclass
class Template:
    name: str
    content_type: str
    text: str                      # with one "{U}" per link slot (11-wide when filled)
    entity_specs: List[Tuple[str, int]] = dataclasses.field(default_factory=list)
    # entity_specs: (etype, slot_index) — offsets computed after slot fill
    aux: str = ""                  # emoji / poll question / prize / doc name
    has_thumb: bool = False
    has_video: bool = False
    weight: float = 1.0

    # computed by _compile:
    pool: bytes = b""
    text_len: int = 0
    slot_positions: List[int] = dataclasses.field(default_factory=list)
    entities: List[Tuple[int, int, int, int, int]] = dataclasses.field(
        default_factory=list
    )


# Placeholder username used during compilation; every real username has the
# same width so offsets survive substitution.
_PH = "c" + "0" * 10


def _compile(t: Template) -> Template:
    """Fill slots with the placeholder, compute byte/UTF-16 entity offsets."""
    parts = t.text.split("{U}")
    text = _PH.join(parts)
    tb = text.encode("utf-8")

    # byte position of each slot
    slot_pos: List[int] = []
    bpos = 0
    for k, p in enumerate(parts[:-1]):
        bpos += len(p.encode("utf-8"))
        slot_pos.append(bpos + 1)  # +1 skips the 'c' prefix; digits only
        bpos += len(_PH.encode("utf-8"))

    # utf16 position of each slot start (including the char before, e.g. '@')
    # entity offsets are computed from entity_specs: each spec names a slot
    # and a kind; kinds determine how the entity covers the slot.
    def u16_at(byte_off: int) -> int:
        return _utf16_len(tb[:byte_off].decode("utf-8"))

    entities = []
    url_tail = b""
    url_off_base = len(tb)
    for (etype, slot) in t.entity_specs:
        # text_url URLs live in the tail, not at a text slot
        spos = (slot_pos[slot] - 1) if etype != "text_url" else -1
        if etype == "mention":
            # template text must have '@' right before the slot
            assert tb[spos - 1:spos] == b"@"
            off16 = u16_at(spos - 1)
            entities.append((B.ENTITY_TYPE_IDX["mention"], off16,
                             1 + USERNAME_WIDTH, 0, 0))
        elif etype == "url":
            # entity covers "https://t.me/<user>" present in the text
            prefix = b"https://t.me/"
            assert tb[spos - len(prefix):spos] == prefix
            off16 = u16_at(spos - len(prefix))
            entities.append((B.ENTITY_TYPE_IDX["url"], off16,
                             len(prefix) + USERNAME_WIDTH, 0, 0))
        elif etype == "text_url":
            # URL lives in the tail, not the text; entity covers the anchor
            # text between «» markers placed in the template around a word.
            anchor = "click here"
            a = text.index(anchor)
            off16 = _utf16_len(text[:a])
            url = "https://t.me/" + _PH
            uoff = url_off_base + len(url_tail) + len(b"https://t.me/")
            slot_pos.append(uoff + 1)  # url slot also patched with digits
            entities.append((B.ENTITY_TYPE_IDX["text_url"], off16,
                             _utf16_len(anchor),
                             url_off_base + len(url_tail), len(url)))
            url_tail += url.encode("utf-8")
        else:
            raise ValueError(etype)

    out = dataclasses.replace(t)
    out.pool = tb + url_tail
    out.text_len = len(tb)
    out.slot_positions = slot_pos
    out.entities = entities
    return out


def default_templates() -> List[Template]:
    """The standard synthetic corpus template set."""
    ts = [
        Template(
            name="plain_short", content_type="messageText", weight=3.0,
            text="Сегодня отличные новости по рынку. Подробности позже.",
        ),
        Template(
            name="plain_long", content_type="messageText", weight=2.0,
            text=(
                "Аналитика за неделю: рост продолжился, несмотря на "
                "волатильность. Ключевые уровни удержались, объём вырос на "
                "12%. Detailed breakdown & charts attached — see the thread "
                "below for \"context\" and <notes>.\nStay tuned! 🚀🚀"
            ),
        ),
        Template(
            name="one_link", content_type="messageText", weight=2.5,
            text="Подписывайтесь на наш резервный канал: https://t.me/{U} 🔥",
        ),
        Template(
            name="bare_link", content_type="messageText", weight=1.5,
            text="репост от t.me/{U} — читайте первыми",
        ),
        Template(
            name="mention", content_type="messageText", weight=2.0,
            text="спасибо @{U} за наводку 🙏 детали в закрепе",
            entity_specs=[("mention", 0)],
        ),
        Template(
            name="mention_emoji_prefix", content_type="messageText", weight=1.0,
            text="🤯🤯 срочно! @{U} опубликовал данные",
            entity_specs=[("mention", 0)],
        ),
        Template(
            name="url_entity", content_type="messageText", weight=1.0,
            text="источник: https://t.me/{U} (проверено)",
            entity_specs=[("url", 0)],
        ),
        Template(
            name="text_url", content_type="messageText", weight=1.0,
            text="Новый разбор — click here чтобы открыть",
            entity_specs=[("text_url", 0)],
        ),
        Template(
            name="two_links", content_type="messageText", weight=1.0,
            text="зеркала: t.me/{U} и https://t.me/{U}, сохраняйте",
        ),
        Template(
            name="dup_link", content_type="messageText", weight=0.5,
            text="канал https://t.me/{U} — да, именно https://t.me/{U}",
        ),
        Template(
            name="reserved_path", content_type="messageText", weight=0.5,
            text="вход: https://t.me/joinchat/AbCdEf123 или t.me/share/url",
        ),
        Template(
            name="escapes", content_type="messageText", weight=0.5,
            text="a<b & b>c \\ \"quoted\"\nline2\ttab after-ls",
        ),
        Template(
            name="photo_caption", content_type="messagePhoto", weight=1.5,
            text="фото дня 📸 подпись с ссылкой t.me/{U}",
            has_thumb=True,
        ),
        Template(
            name="video_caption", content_type="messageVideo", weight=1.0,
            text="видео: главное за день",
            has_thumb=True,
        ),
        Template(
            name="document", content_type="messageDocument", weight=0.5,
            text="отчёт во вложении", aux="report_2024_final.pdf",
            has_thumb=True, has_video=True,
        ),
        Template(
            name="sticker", content_type="messageSticker", weight=0.5,
            text="", has_thumb=True,
        ),
        Template(
            name="animated_emoji", content_type="messageAnimatedEmoji",
            weight=0.3, text="", aux="🎉",
        ),
        Template(
            name="poll", content_type="messagePoll", weight=0.3,
            text="", aux="Как вам обновление?",
        ),
        Template(
            name="giveaway", content_type="messageGiveaway", weight=0.2,
            text="", aux="premium",
        ),
    ]
    return [_compile(t) for t in ts]


COMMENT_TEXTS = [
    "согласен полностью", "интересно 🤔", "first!", "спасибо за инфу",
    "это уже было вчера", "не согласен, но ок", "топ контент 🔥🔥",
    "а пруфы будут?",
]


def _splitmix64(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 over uint64 arrays (wrapping is intentional)."""
    with np.errstate(over="ignore"):
        z = (x + np.uint64(0x9E3779B97F4A7C15)).astype(np.uint64)
        z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        return z ^ (z >> np.uint64(31))


@dataclasses.dataclass
class FeedConfig:
    seed: int = 1234
    universe: int = 1_000_000       # channel-id namespace for outlink targets
    posts_per_channel: int = 10_000
    base_date: int = 1_700_000_000  # unix seconds of post 0
    date_step: int = 60             # seconds between posts in a channel
    comment_rate: float = 0.02      # fraction of posts with comments
    max_comments_per_post: int = 3


class SyntheticFeed:
    """Deterministic channel/history generator in packed-batch form."""

    def __init__(self, cfg: Optional[FeedConfig] = None):
        self.cfg = cfg or FeedConfig()
        self.templates = default_templates()
        w = np.array([t.weight for t in self.templates])
        self._cdf = np.cumsum(w) / w.sum()
        # Precompute template arrays
        self._pool_bytes = [np.frombuffer(t.pool, dtype=np.uint8)
                            for t in self.templates]
        self._pool_len = np.array([len(t.pool) for t in self.templates])
        self._text_len = np.array([t.text_len for t in self.templates])
        self._ctype = np.array(
            [B.CONTENT_TYPE_IDX[t.content_type] for t in self.templates]
        )
        self._flags = np.array(
            [(B.FLAG_HAS_THUMB if t.has_thumb else 0)
             | (B.FLAG_HAS_VIDEO if t.has_video else 0)
             for t in self.templates]
        )
        self._aux = [t.aux.encode("utf-8") for t in self.templates]
        self._ents = [np.array(t.entities, dtype=np.int32).reshape(-1, 5)
                      for t in self.templates]
        self._n_slots = np.array([len(t.slot_positions)
                                  for t in self.templates])
        self._comment_bytes = [c.encode("utf-8") for c in COMMENT_TEXTS]

    # ---- device-side generation (csrc/feedgen.hip) ----

    def _device_tables(self, device):
        """Upload the compiled template tables once per device."""
        if not hasattr(self, "_dev_tables"):
            self._dev_tables = {}
        key = str(device)
        if key in self._dev_tables:
            return self._dev_tables[key]
        import torch as T

        def cat_u8(parts):
            blob = b"".join(parts)
            return T.frombuffer(bytearray(blob or b"\0"), dtype=T.uint8).to(device)

        def offs(parts):
            off, cur = [], 0
            for p in parts:
                off.append(cur)
                cur += len(p)
            return off

        t32 = lambda x: T.tensor(x, dtype=T.int32, device=device)
        tmpl_pool = [t.pool for t in self.templates]
        aux_pool = [a for a in self._aux]
        ent_rows = []
        ent_off, ent_cnt = [], []
        nrows = 0
        for te in self._ents:
            ent_off.append(nrows)  # ROW offset (kernel indexes rows * 5)
            ent_cnt.append(len(te))
            nrows += len(te)
            ent_rows.extend(int(v) for row in te for v in row)
        slot_flat, slot_off, slot_cnt = [], [], []
        for t in self.templates:
            slot_off.append(len(slot_flat))
            slot_cnt.append(len(t.slot_positions))
            slot_flat.extend(t.slot_positions)
        tables = {
            "pool": cat_u8(tmpl_pool),
            "pool_off": t32(offs(tmpl_pool)),
            "pool_len": t32([len(p) for p in tmpl_pool]),
            "text_len": t32([t.text_len for t in self.templates]),
            "ctype": t32([int(v) for v in self._ctype]),
            "tflags": t32([int(v) for v in self._flags]),
            "aux": cat_u8(aux_pool),
            "aux_off": t32(offs(aux_pool)),
            "aux_len": t32([len(a) for a in aux_pool]),
            "ents": t32(ent_rows or [0]),
            "ent_off": t32(ent_off),
            "ent_cnt": t32(ent_cnt),
            "slots": t32(slot_flat or [0]),
            "slot_off": t32(slot_off),
            "slot_cnt": t32(slot_cnt),
            "cdf": T.tensor(self._cdf, dtype=T.float64, device=device),
            "ctext": cat_u8(self._comment_bytes),
            "ctext_off": t32(offs(self._comment_bytes)),
            "ctext_len": t32([len(b) for b in self._comment_bytes]),
        }
        self._dev_tables[key] = tables
        return tables

    _TBL_ORDER = [
        "pool", "pool_off", "pool_len", "text_len", "ctype", "tflags",
        "aux", "aux_off", "aux_len", "ents", "ent_off", "ent_cnt",
        "slots", "slot_off", "slot_cnt", "cdf", "ctext", "ctext_off",
        "ctext_len",
    ]

    def build_batch_device(self, channel_ids: np.ndarray, device,
                           posts_per_channel: Optional[int] = None
                           ) -> B.MessageBatch:
        """GPU-side batch generation, bit-identical to build_batch()."""
        import ctypes
        import torch as T

        from ..ops import gpu as gpu_mod

        lib = gpu_mod.require_lib()
        cfg = self.cfg
        P = posts_per_channel or cfg.posts_per_channel
        K = len(channel_ids)
        N = K * P
        permille = cfg.comment_rate * 1000
        assert permille == int(permille), (
            "comment_rate must be an integer number of permille for the "
            "device generator"
        )
        tables = self._device_tables(device)
        tbl_ptrs = (ctypes.c_void_p * len(self._TBL_ORDER))()
        for k, name in enumerate(self._TBL_ORDER):
            tbl_ptrs[k] = ctypes.c_void_p(tables[name].data_ptr())
        scalars = (ctypes.c_long * 10)(
            len(self.templates), len(self._comment_bytes), cfg.seed,
            cfg.universe, cfg.base_date, cfg.date_step, int(permille),
            cfg.max_comments_per_post, K, P,
        )
        cids_t = T.tensor(np.asarray(channel_ids, dtype=np.int64),
                          device=device)
        stream = ctypes.c_void_p(T.cuda.current_stream().cuda_stream)

        z64 = lambda n: T.zeros(n, dtype=T.int64, device=device)
        z32 = lambda n: T.zeros(n, dtype=T.int32, device=device)
        chat_id, msg_id, block_len = z64(N), z64(N), z64(N)
        m = {f: z32(N) for f in B._META_FIELDS_I32}
        tidx = z32(N)
        out_ptrs = (ctypes.c_void_p * 18)(
            *[ctypes.c_void_p(x.data_ptr()) for x in [
                chat_id, msg_id, m["date"], m["content_type"], m["views"],
                m["forwards"], m["media_album_id"], m["channel_idx"],
                m["flags"], m["text_len"], m["aux_len"], m["ent_cnt"],
                m["react_cnt"], m["com_cnt"], m["poster_len"],
                m["reply_count"], block_len, tidx,
            ]]
        )
        grid = min((N + 255) // 256, 8192)
        rc = lib.crawl_feed_meta(tbl_ptrs, scalars,
                                 ctypes.c_void_p(cids_t.data_ptr()),
                                 out_ptrs, grid, stream)
        if rc != 0:
            raise RuntimeError(f"crawl_feed_meta failed: hip {rc}")

        def excl_cumsum(x64):
            out = T.zeros(x64.numel() + 1, dtype=T.int64, device=device)
            T.cumsum(x64, 0, out=out[1:])
            return out

        block_off = excl_cumsum(block_len)
        ent_off_l = excl_cumsum(m["ent_cnt"].to(T.int64))
        react_off_l = excl_cumsum(m["react_cnt"].to(T.int64))
        com_off_l = excl_cumsum(m["com_cnt"].to(T.int64))
        msg_total = int(block_off[-1].item())
        E = int(ent_off_l[-1].item())
        R = int(react_off_l[-1].item())
        C = int(com_off_l[-1].item())
        cblock = max(len(b) for b in self._comment_bytes) + HANDLE_WIDTH

        # channel table bytes (host-small)
        rows = self.channel_rows(np.asarray(channel_ids), P)
        ch_parts = []
        ch_user_off = np.zeros(K, dtype=np.int32)
        ch_user_len = np.zeros(K, dtype=np.int32)
        ch_title_off = np.zeros(K, dtype=np.int32)
        ch_title_len = np.zeros(K, dtype=np.int32)
        koff = msg_total + C * cblock
        for c, row in enumerate(rows):
            ub, tb2 = row.username.encode(), row.title.encode()
            ch_user_off[c] = koff; ch_user_len[c] = len(ub)
            ch_parts.append(ub); koff += len(ub)
            ch_title_off[c] = koff; ch_title_len[c] = len(tb2)
            ch_parts.append(tb2); koff += len(tb2)
        ch_blob = b"".join(ch_parts)
        total_pool = msg_total + C * cblock + len(ch_blob)
        if total_pool >= (1 << 31):
            raise ValueError("pool exceeds int32 offsets; shrink the batch")
        pool = T.zeros(total_pool, dtype=T.uint8, device=device)
        pool[msg_total + C * cblock:] = T.frombuffer(
            bytearray(ch_blob), dtype=T.uint8
        ).to(device)

        text_off = z64(N)
        entities = z32(max(E, 1) * 5).view(-1, 5)[:E] if E else T.zeros(
            (0, 5), dtype=T.int32, device=device)
        ent_store = entities if E else z32(5).view(1, 5)
        react_emoji, react_count = z32(max(R, 1)), z32(max(R, 1))
        fill_ptrs = (ctypes.c_void_p * 9)(
            *[ctypes.c_void_p(x.data_ptr()) for x in [
                pool, text_off, m["aux_off"], m["poster_off"], m["ent_off"],
                m["react_off"], ent_store, react_emoji, react_count,
            ]]
        )
        rc = lib.crawl_feed_fill(
            tbl_ptrs, scalars, ctypes.c_void_p(cids_t.data_ptr()),
            ctypes.c_void_p(tidx.data_ptr()),
            ctypes.c_void_p(block_off.data_ptr()),
            ctypes.c_void_p(ent_off_l.data_ptr()),
            ctypes.c_void_p(react_off_l.data_ptr()),
            fill_ptrs, min((N + 3) // 4, 8192), stream,
        )
        if rc != 0:
            raise RuntimeError(f"crawl_feed_fill failed: hip {rc}")

        com = {f: z32(max(C, 1)) for f in [
            "text_off", "text_len", "handle_off", "handle_len", "views",
            "replies", "react_off", "react_cnt"]}
        if C:
            cmsg_of = T.repeat_interleave(
                T.arange(N, dtype=T.int64, device=device),
                m["com_cnt"].to(T.int64))
            com_ptrs = (ctypes.c_void_p * 9)(
                *[ctypes.c_void_p(x.data_ptr()) for x in [
                    pool, com["text_off"], com["text_len"],
                    com["handle_off"], com["handle_len"], com["views"],
                    com["replies"], com["react_off"], com["react_cnt"],
                ]]
            )
            rc = lib.crawl_feed_comments(
                tbl_ptrs, scalars, ctypes.c_void_p(cids_t.data_ptr()),
                ctypes.c_void_p(com_off_l.data_ptr()), cblock,
                ctypes.c_void_p(cmsg_of.data_ptr()), msg_total, C,
                com_ptrs, min((C + 255) // 256, 8192), stream,
            )
            if rc != 0:
                raise RuntimeError(f"crawl_feed_comments failed: hip {rc}")

        m["com_off"] = com_off_l[:-1].to(T.int32)
        t = lambda arr, dt_: T.tensor(arr, dtype=dt_, device=device)
        return B.MessageBatch(
            n=N, chat_id=chat_id, msg_id=msg_id, text_off=text_off,
            meta=m, text_pool=pool,
            entities=entities if E else T.zeros((0, 5), dtype=T.int32,
                                                device=device),
            react_emoji=react_emoji[:R], react_count=react_count[:R],
            com_text_off=com["text_off"][:C], com_text_len=com["text_len"][:C],
            com_handle_off=com["handle_off"][:C],
            com_handle_len=com["handle_len"][:C],
            com_views=com["views"][:C], com_replies=com["replies"][:C],
            com_react_off=com["react_off"][:C],
            com_react_cnt=com["react_cnt"][:C],
            n_channels=K,
            ch_chat_id=t(np.array([r.chat_id for r in rows]), T.int64),
            ch_member=t(np.array([r.member_count for r in rows]), T.int32),
            ch_postcount=t(np.array([r.post_count for r in rows]), T.int32),
            ch_totalviews=t(np.array([r.total_views for r in rows]), T.int32),
            ch_user_off=t(ch_user_off, T.int32),
            ch_user_len=t(ch_user_len, T.int32),
            ch_title_off=t(ch_title_off, T.int32),
            ch_title_len=t(ch_title_len, T.int32),
        )

    # ---- channel metadata ----

    @staticmethod
    def username_of(cid: int) -> str:
        return "c%010d" % cid

    def chat_id_of(self, cid: int) -> int:
        return -1001000000000 - cid

    def channel_rows(self, cids: np.ndarray,
                     posts_per_channel: Optional[int] = None) -> List[B.ChannelRow]:
        h = _splitmix64(np.uint64(self.cfg.seed) ^ (cids.astype(np.uint64)
                                                    * np.uint64(0x51ED)))
        rows = []
        for k, cid in enumerate(cids):
            rows.append(B.ChannelRow(
                chat_id=self.chat_id_of(int(cid)),
                username=self.username_of(int(cid)),
                title="Synthetic Channel %d" % int(cid),
                member_count=int(h[k] % np.uint64(500_000)) + 100,
                post_count=posts_per_channel or self.cfg.posts_per_channel,
                total_views=int(h[k] % np.uint64(10_000_000)),
            ))
        return rows

    # ---- batched history generation (the hot input path) ----

    def build_batch(
        self, channel_ids: np.ndarray, posts_per_channel: Optional[int] = None
    ) -> B.MessageBatch:
        """Generate all messages for `channel_ids` as one packed batch.

        Layout: messages grouped by channel, newest-last; channel_idx points
        into the batch channel table (ordered as channel_ids).
        """
        cfg = self.cfg
        P = posts_per_channel or cfg.posts_per_channel
        K = len(channel_ids)
        N = K * P
        seed = np.uint64(cfg.seed)

        cid = np.repeat(channel_ids.astype(np.int64), P)
        pidx = np.tile(np.arange(P, dtype=np.int64), K)
        gidx = (cid.astype(np.uint64) * np.uint64(1_000_003)
                + pidx.astype(np.uint64))
        h0 = _splitmix64(seed ^ gidx)
        h1 = _splitmix64(h0)
        h2 = _splitmix64(h1)
        h3 = _splitmix64(h2)

        # template choice
        u = (h0 >> np.uint64(11)).astype(np.float64) / float(1 << 53)
        tidx = np.searchsorted(self._cdf, u, side="right").astype(np.int32)
        tidx = np.minimum(tidx, len(self.templates) - 1)

        # fixed-size meta
        meta = {f: np.zeros(N, dtype=np.int32) for f in B._META_FIELDS_I32}
        meta["content_type"] = self._ctype[tidx].astype(np.int32)
        meta["flags"] = self._flags[tidx].astype(np.int32)
        meta["views"] = (h1 % np.uint64(100_000)).astype(np.int32)
        meta["forwards"] = ((h1 >> np.uint64(17)) % np.uint64(1000)).astype(np.int32)
        meta["media_album_id"] = np.where(
            (h2 % np.uint64(16)) == 0, (h2 % np.uint64(1 << 31)).astype(np.int64), 0
        ).astype(np.int32)
        meta["channel_idx"] = np.repeat(
            np.arange(K, dtype=np.int32), P
        )
        meta["date"] = (
            cfg.base_date + pidx * cfg.date_step + (h2 % np.uint64(30)).astype(np.int64)
        ).astype(np.int32)

        msg_id = ((pidx + 1) << 20).astype(np.int64)
        chat_id = -1001000000000 - cid

        # ---- pool assembly ----
        # per-message pool block = template pool + aux + handle(9B)
        aux_len = np.array([len(a) for a in self._aux])[tidx]
        block_len = self._pool_len[tidx] + aux_len + HANDLE_WIDTH
        block_off = np.zeros(N + 1, dtype=np.int64)
        np.cumsum(block_len, out=block_off[1:])
        total = int(block_off[-1])
        pool = np.zeros(total, dtype=np.uint8)

        if total >= (1 << 31):
            raise ValueError("pool exceeds int32 offsets; shrink the batch")
        text_off = block_off[:-1].copy()
        meta["text_len"] = self._text_len[tidx].astype(np.int32)
        meta["aux_off"] = (block_off[:-1] + self._pool_len[tidx]).astype(np.int32)
        meta["aux_len"] = aux_len.astype(np.int32)
        meta["poster_off"] = (block_off[:-1] + self._pool_len[tidx] + aux_len).astype(np.int32)
        meta["poster_len"] = np.full(N, HANDLE_WIDTH, dtype=np.int32)

        # write template+aux bytes per template id (vectorized per template)
        for ti, tmpl in enumerate(self.templates):
            rows = np.nonzero(tidx == ti)[0]
            if len(rows) == 0:
                continue
            tb = self._pool_bytes[ti]
            if len(tb):
                idx = block_off[rows][:, None] + np.arange(len(tb))[None, :]
                pool[idx] = tb[None, :]
            ab = np.frombuffer(self._aux[ti], dtype=np.uint8)
            if len(ab):
                idx = (block_off[rows] + len(tb))[:, None] + np.arange(len(ab))[None, :]
                pool[idx] = ab[None, :]
            # handle: "u" + 8 digits of (h3 % 1e8)
            hoff = block_off[rows] + len(tb) + len(ab)
            pool[hoff] = ord("u")
            hv = (h3[rows] % np.uint64(100_000_000)).astype(np.int64)
            for d in range(8):
                digit = (hv // (10 ** (7 - d))) % 10
                pool[hoff + 1 + d] = (48 + digit).astype(np.uint8)
            # patch link slots with target digits
            for s, spos in enumerate(tmpl.slot_positions):
                tgt = (_splitmix64(h0[rows] + np.uint64(7919 * (s + 1)))
                       % np.uint64(self.cfg.universe)).astype(np.int64)
                for d in range(10):
                    digit = (tgt // (10 ** (9 - d))) % 10
                    pool[block_off[rows] + spos + d] = (48 + digit).astype(np.uint8)

        # ---- entities ----
        ent_cnt = np.array([len(e) for e in self._ents])[tidx]
        ent_off = np.zeros(N + 1, dtype=np.int64)
        np.cumsum(ent_cnt, out=ent_off[1:])
        E = int(ent_off[-1])
        entities = np.zeros((E, 5), dtype=np.int32)
        for ti, tmpl in enumerate(self.templates):
            te = self._ents[ti]
            if len(te) == 0:
                continue
            rows = np.nonzero(tidx == ti)[0]
            if len(rows) == 0:
                continue
            dst = (ent_off[rows][:, None] + np.arange(len(te))[None, :]).ravel()
            tiled = np.tile(te, (len(rows), 1))
            # url offsets are relative to the message block start
            tiled_abs = tiled.copy()
            has_url = tiled[:, 4] > 0
            tiled_abs[:, 3] = tiled[:, 3] + np.where(
                has_url, np.repeat(block_off[rows], len(te)), 0
            ).astype(np.int64)
            entities[dst] = tiled_abs
        meta["ent_off"] = ent_off[:-1].astype(np.int32)
        meta["ent_cnt"] = ent_cnt.astype(np.int32)

        # ---- reactions ----
        nem = len(B.EMOJI_TABLE)
        rmask = (h2 & h2 >> np.uint64(13) & np.uint64((1 << nem) - 1)).astype(
            np.int64
        )
        bits = ((rmask[:, None] >> np.arange(nem)[None, :]) & 1).astype(bool)
        react_cnt = bits.sum(axis=1).astype(np.int32)
        react_off = np.zeros(N + 1, dtype=np.int64)
        np.cumsum(react_cnt, out=react_off[1:])
        rmsg, remoji = np.nonzero(bits)  # ordered by (msg, emoji idx asc)
        react_emoji = remoji.astype(np.int32)
        rh = _splitmix64(h2[rmsg] + remoji.astype(np.uint64))
        react_count = ((rh % np.uint64(500)) + np.uint64(1)).astype(np.int32)
        meta["react_off"] = react_off[:-1].astype(np.int32)
        meta["react_cnt"] = react_cnt

        # ---- comments (packed in a secondary pool appended to pool) ----
        has_c = (h3 % np.uint64(1000)).astype(np.float64) < (
            self.cfg.comment_rate * 1000
        )
        ccnt = np.where(
            has_c, (h3 >> np.uint64(32)) % np.uint64(
                self.cfg.max_comments_per_post
            ) + np.uint64(1), 0
        ).astype(np.int32)
        meta["reply_count"] = ccnt
        com_off = np.zeros(N + 1, dtype=np.int64)
        np.cumsum(ccnt, out=com_off[1:])
        C = int(com_off[-1])
        meta["com_off"] = com_off[:-1].astype(np.int32)
        meta["com_cnt"] = ccnt

        cmsg = np.repeat(np.arange(N), ccnt)
        cslot = (np.arange(C) - com_off[cmsg]).astype(np.int64)
        ch = _splitmix64(h3[cmsg] + cslot.astype(np.uint64) * np.uint64(104729))
        ctext_idx = (ch % np.uint64(len(COMMENT_TEXTS))).astype(np.int64)
        clens = np.array([len(b) for b in self._comment_bytes])
        ctext_len = clens[ctext_idx]
        # comment pool: FIXED-WIDTH blocks (max text len + handle) appended
        # after the message pool — fixed stride keeps the device generator
        # (csrc/feedgen.hip) cumsum-free for comments; unused tail bytes of a
        # block are zero and never referenced.
        cblock = int(max(len(b) for b in self._comment_bytes)) + HANDLE_WIDTH
        cblock_off = np.arange(C + 1, dtype=np.int64) * cblock
        cpool = np.zeros(int(cblock_off[-1]), dtype=np.uint8)
        for k, cb in enumerate(self._comment_bytes):
            rows = np.nonzero(ctext_idx == k)[0]
            if len(rows) == 0:
                continue
            arr = np.frombuffer(cb, dtype=np.uint8)
            idx = cblock_off[rows][:, None] + np.arange(len(arr))[None, :]
            cpool[idx] = arr[None, :]
        hoff = cblock_off[:-1] + ctext_len
        if C:
            cpool[hoff] = ord("u")
            hv = (ch >> np.uint64(8)) % np.uint64(100_000_000)
            hv = hv.astype(np.int64)
            for d in range(8):
                digit = (hv // (10 ** (7 - d))) % 10
                cpool[hoff + 1 + d] = (48 + digit).astype(np.uint8)
        com_text_off = (cblock_off[:-1] + total).astype(np.int32)
        com_handle_off = (hoff + total).astype(np.int32)
        if total + int(cblock_off[-1]) >= (1 << 31):
            raise ValueError("pool exceeds int32 offsets; shrink the batch")

        # comment reactions: none (empty maps) — matches typical thread data
        com_react_off = np.zeros(C, dtype=np.int32)
        com_react_cnt = np.zeros(C, dtype=np.int32)
        com_views = ((ch >> np.uint64(16)) % np.uint64(10_000)).astype(np.int32)
        com_replies = ((ch >> np.uint64(24)) % np.uint64(50)).astype(np.int32)

        full_pool = np.concatenate([pool, cpool]) if C else pool

        # ---- channel table ----
        rows = self.channel_rows(channel_ids, P)
        kpool_parts = []
        ch_user_off = np.zeros(K, dtype=np.int32)
        ch_user_len = np.zeros(K, dtype=np.int32)
        ch_title_off = np.zeros(K, dtype=np.int32)
        ch_title_len = np.zeros(K, dtype=np.int32)
        koff = len(full_pool)
        for c, row in enumerate(rows):
            ub = row.username.encode()
            tb2 = row.title.encode()
            ch_user_off[c] = koff
            ch_user_len[c] = len(ub)
            kpool_parts.append(ub)
            koff += len(ub)
            ch_title_off[c] = koff
            ch_title_len[c] = len(tb2)
            kpool_parts.append(tb2)
            koff += len(tb2)
        full_pool = np.concatenate(
            [full_pool, np.frombuffer(b"".join(kpool_parts), dtype=np.uint8)]
        )

        t = torch.from_numpy
        return B.MessageBatch(
            n=N,
            chat_id=t(chat_id),
            msg_id=t(msg_id),
            text_off=t(text_off),
            meta={f: t(np.ascontiguousarray(v)) for f, v in meta.items()},
            text_pool=t(full_pool),
            entities=t(entities),
            react_emoji=t(react_emoji),
            react_count=t(react_count),
            com_text_off=t(com_text_off),
            com_text_len=t(ctext_len.astype(np.int32)),
            com_handle_off=t(com_handle_off),
            com_handle_len=t(np.full(C, HANDLE_WIDTH, dtype=np.int32)),
            com_views=t(com_views),
            com_replies=t(com_replies),
            com_react_off=t(com_react_off),
            com_react_cnt=t(com_react_cnt),
            n_channels=K,
            ch_chat_id=t(np.array([r.chat_id for r in rows], dtype=np.int64)),
            ch_member=t(np.array([r.member_count for r in rows], dtype=np.int32)),
            ch_postcount=t(np.array([r.post_count for r in rows], dtype=np.int32)),
            ch_totalviews=t(np.array([r.total_views for r in rows], dtype=np.int32)),
            ch_user_off=t(ch_user_off),
            ch_user_len=t(ch_user_len),
            ch_title_off=t(ch_title_off),
            ch_title_len=t(ch_title_len),
        )
