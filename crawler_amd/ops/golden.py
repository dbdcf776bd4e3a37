"""Golden (reference-semantics) implementations of the per-post hot path.

These are the oracle the HIP kernels are validated against, and the CPU
execution path for the no-GPU configuration (BASELINE config #1).

Parity targets (reference file:line):
- channel link regex + reserved paths: telegramhelper/tdutils.go:23-45
- utf16OffsetToBytes: telegramhelper/tdutils.go:55-78
- entity walk (entity order, then plaintext scan; first-wins):
  telegramhelper/tdutils.go:897-949
- username regex: telegramhelper/tdutils.go:82
- FilterUsername rules: telegramhelper/username_filter.go:26-81
- BuildTelegramLinkAndMessageID (public id = internal >> 20):
  telegramhelper/tdutils.go:1005-1031
- ParseMessage content-type switch + Post assembly:
  telegramhelper/tdutils.go:380-732
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import re
from typing import Dict, List, Optional, Tuple

from ..models.post import ChannelData, Comment, EngagementData, Post

# --- link / username regexes (tdutils.go:23, :82) ---

CHANNEL_LINK_RE = re.compile(r"(https?://)?t\.me/([a-zA-Z][a-zA-Z0-9_]{4,31})")
USERNAME_RE = re.compile(r"(?:@)?([a-zA-Z][a-zA-Z0-9_]{4,31})")

RESERVED_PATHS = {
    "joinchat", "addlist", "addstickers", "addtheme", "setlanguage",
    "share", "c", "s", "iv", "proxy", "socks", "login", "confirm", "bg",
}

SOURCE_MENTION = "mention"
SOURCE_TEXT_URL = "text_url"
SOURCE_URL = "url"
SOURCE_PLAINTEXT = "plaintext"


@dataclasses.dataclass
class Entity:
    """A TDLib text entity: offsets/lengths are UTF-16 code units."""

    type: str  # "mention" | "text_url" | "url"
    offset: int
    length: int
    url: str = ""  # only for text_url


@dataclasses.dataclass
class FormattedText:
    text: str = ""
    entities: List[Entity] = dataclasses.field(default_factory=list)


@dataclasses.dataclass
class DiscoveredLink:
    name: str
    source_type: str


# Content types whose FormattedText feeds link extraction
# (extractFormattedTextFromMessage, tdutils.go:953-975).
_LINKABLE_TYPES = {
    "messageText", "messagePhoto", "messageVideo", "messageDocument",
    "messageAnimation", "messageAudio", "messageVoiceNote",
}


@dataclasses.dataclass
class SynthMessage:
    """A TDLib-shaped message record as served by the synthetic feed."""

    chat_id: int = 0
    msg_id: int = 0          # internal TDLib id (public id << 20)
    date: int = 0            # unix seconds
    content_type: str = "messageText"
    text: Optional[FormattedText] = None      # body (messageText)
    caption: Optional[FormattedText] = None   # media caption
    views: int = 0
    forwards: int = 0
    reply_count: int = 0
    reactions: Dict[str, int] = dataclasses.field(default_factory=dict)
    media_album_id: int = 0
    thumb_remote_id: str = ""
    video_remote_id: str = ""
    document_name: str = ""
    emoji: str = ""
    poll_question: str = ""
    giveaway_prize: str = ""
    poster_handle: str = ""


def utf16_offset_to_bytes(s_bytes: bytes, off16: int, len16: int) -> Tuple[int, int]:
    """UTF-16 code-unit (offset, length) -> byte (start, end) in UTF-8 text.

    Exact mirror of tdutils.go:55-78 including its edge cases:
    - returns (0, 0) if the offset was never reached;
    - returns (start, len) if the end lands past the end of text.
    """
    i = 0
    u16pos = 0
    rune_start = -1
    n = len(s_bytes)
    while i < n:
        if u16pos == off16:
            rune_start = i
        if u16pos == off16 + len16:
            return rune_start, i
        b = s_bytes[i]
        if b < 0x80:
            size, units = 1, 1
        elif b < 0xE0:
            size, units = 2, 1
        elif b < 0xF0:
            size, units = 3, 1
        else:
            size, units = 4, 2  # astral plane -> surrogate pair
        u16pos += units
        i += size
    if rune_start == -1:
        return 0, 0
    return rune_start, n


def channel_name_from_match(m: Optional[re.Match]) -> Optional[str]:
    """channelNameFromMatch (tdutils.go:37-46): reserved-path filter + lower."""
    if m is None:
        return None
    name = m.group(2)
    if name.lower() in RESERVED_PATHS:
        return None
    return name.lower()


def extract_links_from_formatted_text(ft: Optional[FormattedText], source_map):
    """Entity walk then plaintext scan, first-wins (tdutils.go:897-949)."""
    if ft is None:
        return
    text_bytes = ft.text.encode("utf-8")

    def add_if_new(name: str, src: str):
        if name not in source_map:
            source_map[name] = src

    for ent in ft.entities:
        if ent.type == "text_url":
            name = channel_name_from_match(CHANNEL_LINK_RE.search(ent.url))
            if name:
                add_if_new(name, SOURCE_TEXT_URL)
        elif ent.type == "mention":
            start, end = utf16_offset_to_bytes(text_bytes, ent.offset, ent.length)
            if start < end and end <= len(text_bytes):
                mention = text_bytes[start:end].decode("utf-8", "replace")
                m = USERNAME_RE.search(mention)
                if m:
                    add_if_new(m.group(1).lower(), SOURCE_MENTION)
        elif ent.type == "url":
            start, end = utf16_offset_to_bytes(text_bytes, ent.offset, ent.length)
            if start < end and end <= len(text_bytes):
                url = text_bytes[start:end].decode("utf-8", "replace")
                name = channel_name_from_match(CHANNEL_LINK_RE.search(url))
                if name:
                    add_if_new(name, SOURCE_URL)

    for m in CHANNEL_LINK_RE.finditer(ft.text):
        name = channel_name_from_match(m)
        if name:
            add_if_new(name, SOURCE_PLAINTEXT)


def _formatted_text_of(msg: SynthMessage) -> Optional[FormattedText]:
    if msg.content_type == "messageText":
        return msg.text
    if msg.content_type in _LINKABLE_TYPES:
        return msg.caption
    return None


def extract_links_with_source(msg: SynthMessage) -> List[DiscoveredLink]:
    """ExtractChannelLinksWithSource (tdutils.go:978-988).

    Returns links in deterministic insertion order (the Go original iterates
    a map, so ITS order is random — order is not part of the contract; the
    set of (name, source) pairs is).
    """
    source_map: Dict[str, str] = {}
    extract_links_from_formatted_text(_formatted_text_of(msg), source_map)
    return [DiscoveredLink(n, s) for n, s in source_map.items()]


def extract_channel_links(msg: SynthMessage) -> List[str]:
    """extractChannelLinksFromMessage (tdutils.go:989-1002)."""
    ft = _formatted_text_of(msg)
    if ft is None:
        return []
    source_map: Dict[str, str] = {}
    extract_links_from_formatted_text(ft, source_map)
    return list(source_map.keys())


def filter_username(username: str) -> Tuple[bool, str]:
    """FilterUsername (username_filter.go:26-81). Returns (valid, reason)."""
    b = username.encode("utf-8", "surrogatepass")
    if len(b) < 5:
        return False, "too_short"
    if len(b) > 32:
        return False, "too_long"
    first = b[0]
    if not (65 <= first <= 90 or 97 <= first <= 122):
        return False, "invalid_start_char"
    if b[-1] == 0x5F:  # '_'
        return False, "ends_with_underscore"
    for ch in username:
        if not (ch.isascii() and (ch.isalnum() or ch == "_")):
            return False, "invalid_char"
    if any(c in username for c in "/\\~."):
        return False, "looks_like_path"
    lower = username.lower()
    if lower.endswith("bot"):
        # "_bot" and bare "bot" suffixes both reject (username_filter.go:60-67)
        return False, "bot_suffix"
    return True, ""


def build_telegram_link_and_message_id(
    active_username: str, msg: SynthMessage
) -> Tuple[str, int]:
    """BuildTelegramLinkAndMessageID (tdutils.go:1005-1031).

    public id = internal id >> 20 (TDLib packs the server message id in the
    upper bits); private channels (no username) yield an empty link.
    """
    public_id = msg.msg_id >> 20
    if active_username:
        link = f"https://t.me/{active_username}/{public_id}"
        if msg.media_album_id != 0:
            link += "?single"
        return link, public_id
    return "", public_id


def _description_of(msg: SynthMessage) -> str:
    """Content-type switch (tdutils.go:443-587)."""
    ct = msg.content_type
    if ct == "messageText":
        return msg.text.text if msg.text else ""
    if ct in ("messageVideo", "messagePhoto", "messageAnimation",
              "messagePaidMedia"):
        return msg.caption.text if msg.caption else ""
    if ct == "messageAnimatedEmoji":
        return msg.emoji
    if ct == "messagePoll":
        return msg.poll_question
    if ct == "messageGiveaway":
        return msg.giveaway_prize
    if ct == "messageDocument":
        return msg.document_name
    # messageSticker / messageVideoNote / giveaway winners+completed / unknown
    return ""


# Content types that carry a thumbnail remote id (tdutils.go:443-587).
_THUMB_TYPES = {
    "messageVideo", "messagePhoto", "messageAnimation", "messageSticker",
    "messageVideoNote", "messageDocument",
}
# Content types that set MediaURL (video path) WITHOUT downloading
# (tdutils.go:545-587 — VideoNote / Document set videoPath from remote id).
_VIDEO_PATH_TYPES = {"messageVideoNote", "messageDocument"}


def parse_message(
    msg: SynthMessage,
    *,
    crawl_id: str = "",
    channel_username: str = "",
    chat_title: str = "",
    member_count: int = 0,
    post_count: int = 0,
    total_views: int = 0,
    comments: Optional[List[Comment]] = None,
    min_post_date: Optional[_dt.datetime] = None,
    skip_media: bool = True,
    media_path_resolver=None,
    now: Optional[_dt.datetime] = None,
) -> Optional[Post]:
    """Assemble a model.Post from a TDLib-shaped message.

    Mirror of ParseMessage (tdutils.go:380-732) with the synthetic feed in
    place of TDLib RPCs:
    - returns None when published_at < min_post_date (tdutils.go:419-421);
    - description / thumb / media per the content-type switch;
    - media paths: "" when skip_media (fetchAndUploadMedia, tdutils.go:233-239),
      otherwise resolved by media_path_resolver(remote_id) -> storage path;
    - views/forwards from interaction info; comment count = len(comments).
    """
    published_at = _dt.datetime.fromtimestamp(msg.date, _dt.timezone.utc)
    if min_post_date is not None and published_at < min_post_date:
        return None

    generated_link, public_id = build_telegram_link_and_message_id(
        channel_username, msg
    )
    comments = comments if comments is not None else []

    thumb_path = ""
    video_path = ""
    if msg.content_type in _THUMB_TYPES and msg.thumb_remote_id:
        if not skip_media and media_path_resolver is not None:
            thumb_path = media_path_resolver(msg.thumb_remote_id)
    if msg.content_type in _VIDEO_PATH_TYPES and msg.video_remote_id:
        video_path = msg.video_remote_id

    now = now or _dt.datetime.now(_dt.timezone.utc)
    created_at = now.replace(microsecond=0)

    vc = msg.views
    share_count = msg.forwards

    post = Post(
        post_link=generated_link,
        channel_id=str(msg.chat_id),
        post_uid=f"{public_id}-{channel_username or chat_title}",
        url=generated_link,
        published_at=published_at,
        created_at=created_at,
        language_code="",
        engagement=vc,
        view_count=vc,
        like_count=0,
        share_count=share_count,
        comment_count=len(comments),
        channel_name=chat_title,
        is_ad=False,
        transcript_text="",
        image_text="",
        channel_data=ChannelData(
            channel_id=str(msg.chat_id),
            channel_name=chat_title,
            channel_description="",
            channel_profile_image="",
            country_code="",
            channel_engagement_data=EngagementData(
                follower_count=member_count,
                following_count=0,
                like_count=0,
                post_count=post_count,
                views_count=total_views,
                comment_count=0,
                share_count=0,
            ),
            channel_url_external=f"https://t.me/c/{channel_username or chat_title}",
            channel_url=f"https://t.me/c/{channel_username or chat_title}",
        ),
        platform_name="Telegram",
        description=_description_of(msg),
        post_type=[msg.content_type],
        likes_count=0,
        shares_count=share_count,
        comments_count=len(comments),
        views_count=vc,
        searchable_text="",
        all_text="",
        thumb_url=thumb_path,
        media_url=video_path,
        comments=comments,
        reactions=dict(msg.reactions),
        outlinks=extract_channel_links(msg),
        capture_time=now,
        handle=msg.poster_handle,
    )
    return post
