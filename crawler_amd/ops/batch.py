"""Packed message-batch format — the device-side record layout.

The GPU hot path consumes message records in struct-of-arrays form: fixed
per-message metadata in int tensors plus byte pools for variable-length
text. This is the MI355X-native replacement for TDLib's per-message object
graph (reference telegramhelper/tdutils.go ParseMessage input): one batch
holds up to ~1M messages and lives in HBM.

All tensors are torch tensors (CPU or CUDA). Field map (per message i):

  chat_id        int64  TDLib chat id
  msg_id         int64  internal id (public id << 20, tdutils.go:1016)
  date           int32  unix seconds
  content_type   int32  enum CONTENT_TYPES index
  views          int32  InteractionInfo.ViewCount
  forwards       int32  InteractionInfo.ForwardCount
  reply_count    int32  ReplyInfo.ReplyCount
  media_album_id int32  0 = none ("?single" suffix rule tdutils.go:1024)
  channel_idx    int32  row into the batch's channel table
  flags          int32  bit0: has thumb remote id; bit1: has video remote id
  text_off/len   int64/int32  FormattedText bytes in text_pool (body/caption)
  aux_off/len    int32  type-specific aux string (emoji/poll q/prize/docname)
  ent_off/cnt    int32  rows into entities
  react_off/cnt  int32  rows into react_emoji/react_count (sorted by emoji!)
  com_off/cnt    int32  rows into the comment sub-batch
  poster_off/len int32  poster handle bytes in text_pool

entities: int32[E, 5] = (etype, utf16_off, utf16_len, url_off, url_len)
  etype: 0=mention 1=text_url 2=url; url_* index text_pool (text_url only)
react_emoji int32[R] -> EMOJI_TABLE index; react_count int32[R]
comments: com_text_off/len, com_handle_off/len, com_view/com_reply int32[C],
  com_react_off/cnt -> same react arrays

channel table (per channel row c):
  ch_chat_id int64; ch_member/ch_postcount/ch_totalviews int32;
  ch_user_off/len, ch_title_off/len -> text_pool
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np
import torch

from . import golden as G

CONTENT_TYPES = [
    "messageText", "messagePhoto", "messageVideo", "messageDocument",
    "messageAnimation", "messageAudio", "messageVoiceNote", "messageSticker",
    "messageVideoNote", "messageAnimatedEmoji", "messagePoll",
    "messageGiveaway", "messageGiveawayWinners", "messageGiveawayCompleted",
    "messagePaidMedia",
]
CONTENT_TYPE_IDX = {name: i for i, name in enumerate(CONTENT_TYPES)}

ENTITY_TYPES = ["mention", "text_url", "url"]
ENTITY_TYPE_IDX = {name: i for i, name in enumerate(ENTITY_TYPES)}

# Reaction emoji vocabulary, sorted by UTF-8 byte order so per-message
# reaction lists generated in table order are already in Go map-key order
# (post.py _enc_int_map sorts keys; sorted() on str == UTF-8 byte order
# for these single-codepoint emoji).
EMOJI_TABLE = sorted(["❤", "⚡", "👍", "👎", "🔥", "🤔", "😁", "😢", "🤯", "🙏"])

FLAG_HAS_THUMB = 1
FLAG_HAS_VIDEO = 2

_META_FIELDS_I32 = [
    "date", "content_type", "views", "forwards", "reply_count",
    "media_album_id", "channel_idx", "flags", "text_len", "aux_off",
    "aux_len", "ent_off", "ent_cnt", "react_off", "react_cnt", "com_off",
    "com_cnt", "poster_off", "poster_len",
]


@dataclasses.dataclass
class ChannelRow:
    chat_id: int
    username: str
    title: str
    member_count: int = 0
    post_count: int = 0
    total_views: int = 0


@dataclasses.dataclass
class MessageBatch:
    """Struct-of-arrays message batch. Construct via :func:`pack`."""

    n: int
    chat_id: torch.Tensor       # int64[N]
    msg_id: torch.Tensor        # int64[N]
    text_off: torch.Tensor      # int64[N]
    meta: dict                  # name -> int32[N] (see _META_FIELDS_I32)
    text_pool: torch.Tensor     # uint8[T]
    entities: torch.Tensor      # int32[E,5]
    react_emoji: torch.Tensor   # int32[R]
    react_count: torch.Tensor   # int32[R]
    com_text_off: torch.Tensor  # int32[C]
    com_text_len: torch.Tensor
    com_handle_off: torch.Tensor
    com_handle_len: torch.Tensor
    com_views: torch.Tensor
    com_replies: torch.Tensor
    com_react_off: torch.Tensor
    com_react_cnt: torch.Tensor
    # channel table
    n_channels: int
    ch_chat_id: torch.Tensor    # int64[K]
    ch_member: torch.Tensor     # int32[K]
    ch_postcount: torch.Tensor
    ch_totalviews: torch.Tensor
    ch_user_off: torch.Tensor
    ch_user_len: torch.Tensor
    ch_title_off: torch.Tensor
    ch_title_len: torch.Tensor

    def to(self, device) -> "MessageBatch":
        kw = {}
        for f in dataclasses.fields(self):
            v = getattr(self, f.name)
            if isinstance(v, torch.Tensor):
                kw[f.name] = v.to(device)
            elif isinstance(v, dict):
                kw[f.name] = {k: t.to(device) for k, t in v.items()}
            else:
                kw[f.name] = v
        return MessageBatch(**kw)

    @property
    def device(self):
        return self.text_pool.device


class _PoolBuilder:
    def __init__(self):
        self.parts: List[bytes] = []
        self.off = 0

    def add(self, data: bytes) -> int:
        off = self.off
        self.parts.append(data)
        self.off += len(data)
        return off

    def tensor(self) -> torch.Tensor:
        blob = b"".join(self.parts)
        return torch.frombuffer(bytearray(blob), dtype=torch.uint8)


def pack(
    messages: List[G.SynthMessage],
    channels: List[ChannelRow],
    channel_of_msg: List[int],
    comments_of_msg: Optional[List[List]] = None,
) -> MessageBatch:
    """Pack python-object messages into the SoA batch (test/CPU-scale path).

    The synthetic feed builds big batches vectorized (feed/synth.py); this
    packer is the readable reference used by tests and the fixtures path.
    ``comments_of_msg[i]`` is a list of (text, handle, views, replies,
    reactions_dict) tuples.
    """
    n = len(messages)
    pool = _PoolBuilder()
    meta = {f: np.zeros(n, dtype=np.int32) for f in _META_FIELDS_I32}
    chat_id = np.zeros(n, dtype=np.int64)
    msg_id = np.zeros(n, dtype=np.int64)
    text_off = np.zeros(n, dtype=np.int64)
    ents: List[List[int]] = []
    remoji: List[int] = []
    rcount: List[int] = []
    com_rows: List[List[int]] = []

    def add_reactions(d: dict) -> (int, int):
        off = len(remoji)
        for emoji in sorted(d):
            remoji.append(EMOJI_TABLE.index(emoji))
            rcount.append(d[emoji])
        return off, len(d)

    for i, m in enumerate(messages):
        chat_id[i] = m.chat_id
        msg_id[i] = m.msg_id
        meta["date"][i] = m.date
        meta["content_type"][i] = CONTENT_TYPE_IDX[m.content_type]
        meta["views"][i] = m.views
        meta["forwards"][i] = m.forwards
        meta["reply_count"][i] = m.reply_count
        meta["media_album_id"][i] = m.media_album_id
        meta["channel_idx"][i] = channel_of_msg[i]
        flags = 0
        if m.thumb_remote_id:
            flags |= FLAG_HAS_THUMB
        if m.video_remote_id:
            flags |= FLAG_HAS_VIDEO
        meta["flags"][i] = flags

        ft = m.text if m.content_type == "messageText" else m.caption
        text = ft.text if ft else ""
        tb = text.encode("utf-8")
        text_off[i] = pool.add(tb)
        meta["text_len"][i] = len(tb)

        aux = ""
        if m.content_type == "messageAnimatedEmoji":
            aux = m.emoji
        elif m.content_type == "messagePoll":
            aux = m.poll_question
        elif m.content_type == "messageGiveaway":
            aux = m.giveaway_prize
        elif m.content_type == "messageDocument":
            aux = m.document_name
        ab = aux.encode("utf-8")
        meta["aux_off"][i] = pool.add(ab)
        meta["aux_len"][i] = len(ab)

        meta["ent_off"][i] = len(ents)
        if ft:
            for e in ft.entities:
                ub = e.url.encode("utf-8")
                uoff = pool.add(ub) if ub else 0
                ents.append(
                    [ENTITY_TYPE_IDX[e.type], e.offset, e.length, uoff, len(ub)]
                )
        meta["ent_cnt"][i] = len(ents) - meta["ent_off"][i]

        off, cnt = add_reactions(m.reactions)
        meta["react_off"][i] = off
        meta["react_cnt"][i] = cnt

        meta["com_off"][i] = len(com_rows)
        for (ctext, chandle, cviews, creplies, creacts) in (
            comments_of_msg[i] if comments_of_msg else []
        ):
            cb = ctext.encode("utf-8")
            hb = chandle.encode("utf-8")
            roff, rcnt = add_reactions(creacts)
            com_rows.append(
                [pool.add(cb), len(cb), pool.add(hb), len(hb), cviews,
                 creplies, roff, rcnt]
            )
        meta["com_cnt"][i] = len(com_rows) - meta["com_off"][i]

        hb = m.poster_handle.encode("utf-8")
        meta["poster_off"][i] = pool.add(hb)
        meta["poster_len"][i] = len(hb)

    k = len(channels)
    ch_chat_id = np.zeros(k, dtype=np.int64)
    ch_i32 = {f: np.zeros(k, dtype=np.int32) for f in
              ["member", "postcount", "totalviews", "user_off", "user_len",
               "title_off", "title_len"]}
    for c, ch in enumerate(channels):
        ch_chat_id[c] = ch.chat_id
        ub = ch.username.encode("utf-8")
        tb = ch.title.encode("utf-8")
        ch_i32["user_off"][c] = pool.add(ub)
        ch_i32["user_len"][c] = len(ub)
        ch_i32["title_off"][c] = pool.add(tb)
        ch_i32["title_len"][c] = len(tb)
        ch_i32["member"][c] = ch.member_count
        ch_i32["postcount"][c] = ch.post_count
        ch_i32["totalviews"][c] = ch.total_views

    ent_arr = np.array(ents, dtype=np.int32).reshape(-1, 5)
    com_arr = np.array(com_rows, dtype=np.int32).reshape(-1, 8)
    t = torch.from_numpy
    return MessageBatch(
        n=n,
        chat_id=t(chat_id),
        msg_id=t(msg_id),
        text_off=t(text_off),
        meta={f: t(v) for f, v in meta.items()},
        text_pool=pool.tensor() if pool.off else torch.zeros(1, dtype=torch.uint8),
        entities=t(ent_arr),
        react_emoji=t(np.array(remoji, dtype=np.int32)),
        react_count=t(np.array(rcount, dtype=np.int32)),
        com_text_off=t(com_arr[:, 0].copy()),
        com_text_len=t(com_arr[:, 1].copy()),
        com_handle_off=t(com_arr[:, 2].copy()),
        com_handle_len=t(com_arr[:, 3].copy()),
        com_views=t(com_arr[:, 4].copy()),
        com_replies=t(com_arr[:, 5].copy()),
        com_react_off=t(com_arr[:, 6].copy()),
        com_react_cnt=t(com_arr[:, 7].copy()),
        n_channels=k,
        ch_chat_id=t(ch_chat_id),
        ch_member=t(ch_i32["member"]),
        ch_postcount=t(ch_i32["postcount"]),
        ch_totalviews=t(ch_i32["totalviews"]),
        ch_user_off=t(ch_i32["user_off"]),
        ch_user_len=t(ch_i32["user_len"]),
        ch_title_off=t(ch_i32["title_off"]),
        ch_title_len=t(ch_i32["title_len"]),
    )


def _pool_str(batch: MessageBatch, off: int, length: int) -> str:
    return bytes(batch.text_pool[off:off + length].numpy()).decode("utf-8")


def unpack_message(batch: MessageBatch, i: int) -> G.SynthMessage:
    """Decode message i back into a SynthMessage (golden-validation path)."""
    m = batch.meta
    ct = CONTENT_TYPES[int(m["content_type"][i])]
    text = _pool_str(batch, int(batch.text_off[i]), int(m["text_len"][i]))
    ents = []
    for e in range(int(m["ent_off"][i]), int(m["ent_off"][i]) + int(m["ent_cnt"][i])):
        etype, off16, len16, uoff, ulen = (int(x) for x in batch.entities[e])
        ents.append(G.Entity(
            type=ENTITY_TYPES[etype], offset=off16, length=len16,
            url=_pool_str(batch, uoff, ulen) if ulen else "",
        ))
    ft = G.FormattedText(text=text, entities=ents)
    reactions = {}
    for r in range(int(m["react_off"][i]),
                   int(m["react_off"][i]) + int(m["react_cnt"][i])):
        reactions[EMOJI_TABLE[int(batch.react_emoji[r])]] = int(batch.react_count[r])
    aux = _pool_str(batch, int(m["aux_off"][i]), int(m["aux_len"][i]))
    flags = int(m["flags"][i])
    return G.SynthMessage(
        chat_id=int(batch.chat_id[i]),
        msg_id=int(batch.msg_id[i]),
        date=int(m["date"][i]),
        content_type=ct,
        text=ft if ct == "messageText" else None,
        caption=ft if ct != "messageText" else None,
        views=int(m["views"][i]),
        forwards=int(m["forwards"][i]),
        reply_count=int(m["reply_count"][i]),
        reactions=reactions,
        media_album_id=int(m["media_album_id"][i]),
        # Canonical synthetic remote-id scheme (shared with the HIP encoder):
        # derived from the public message id so kernels can regenerate them
        # without extra pool strings.
        thumb_remote_id=(
            f"AgAD{int(batch.msg_id[i]) >> 20}t" if flags & FLAG_HAS_THUMB else ""
        ),
        video_remote_id=(
            f"AgAD{int(batch.msg_id[i]) >> 20}v" if flags & FLAG_HAS_VIDEO else ""
        ),
        document_name=aux if ct == "messageDocument" else "",
        emoji=aux if ct == "messageAnimatedEmoji" else "",
        poll_question=aux if ct == "messagePoll" else "",
        giveaway_prize=aux if ct == "messageGiveaway" else "",
        poster_handle=_pool_str(batch, int(m["poster_off"][i]),
                                int(m["poster_len"][i])),
    )


def unpack_comments(batch: MessageBatch, i: int):
    """Decode message i's comments into model Comment objects."""
    from ..models.post import Comment

    m = batch.meta
    out = []
    for c in range(int(m["com_off"][i]), int(m["com_off"][i]) + int(m["com_cnt"][i])):
        reactions = {}
        for r in range(int(batch.com_react_off[c]),
                       int(batch.com_react_off[c]) + int(batch.com_react_cnt[c])):
            reactions[EMOJI_TABLE[int(batch.react_emoji[r])]] = int(
                batch.react_count[r]
            )
        out.append(Comment(
            text=_pool_str(batch, int(batch.com_text_off[c]),
                           int(batch.com_text_len[c])),
            reactions=reactions,
            view_count=int(batch.com_views[c]),
            reply_count=int(batch.com_replies[c]),
            handle=_pool_str(batch, int(batch.com_handle_off[c]),
                             int(batch.com_handle_len[c])),
        ))
    return out


def channel_row(batch: MessageBatch, c: int) -> ChannelRow:
    return ChannelRow(
        chat_id=int(batch.ch_chat_id[c]),
        username=_pool_str(batch, int(batch.ch_user_off[c]),
                           int(batch.ch_user_len[c])),
        title=_pool_str(batch, int(batch.ch_title_off[c]),
                        int(batch.ch_title_len[c])),
        member_count=int(batch.ch_member[c]),
        post_count=int(batch.ch_postcount[c]),
        total_views=int(batch.ch_totalviews[c]),
    )

def select_rows(batch: MessageBatch, idx: torch.Tensor) -> MessageBatch:
    """Row-subset view of a batch (pools shared; per-message tensors
    gathered). Offsets reference the shared pool, so any row order is
    valid — used by the sampling path (date-between + sample-size)."""
    idx = idx.to(batch.chat_id.device, torch.long)
    return dataclasses.replace(
        batch,
        n=int(idx.numel()),
        chat_id=batch.chat_id[idx].contiguous(),
        msg_id=batch.msg_id[idx].contiguous(),
        text_off=batch.text_off[idx].contiguous(),
        meta={k: v[idx].contiguous() for k, v in batch.meta.items()},
    )
