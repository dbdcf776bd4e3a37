r"""Unified Post / ChannelData JSONL schema with Go-compatible encoding.

Schema parity: reference model/data.go:9-149 — field names, field ORDER and
JSON encoding must match what Go's ``encoding/json`` produces, because the
JSONL output files are the reference's primary data contract and the GPU
JSONL-encode kernel is validated byte-for-byte against this module.

Go encoding rules implemented here (and in ops/csrc/jsonl kernels):
- struct fields emitted in declaration order;
- strings escaped with ``\"``, ``\\``, ``\n``, ``\r``, ``\t``; other control
  chars as ``\u00xx``; HTML-unsafe ``<``, ``>``, ``&`` as ``<`` etc.;
  U+2028/U+2029 escaped (Go escapes them inside JS-unsafe strings);
- ``time.Time`` as RFC3339Nano, trailing-zero-trimmed, ``Z`` for UTC; the
  zero time renders "0001-01-01T00:00:00Z";
- nil slices -> ``null``, empty slices -> ``[]``; nil pointers -> ``null``;
- maps with keys sorted lexicographically (Go sorts map keys).
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
from typing import Dict, List, Optional

ZERO_TIME = _dt.datetime(1, 1, 1, tzinfo=_dt.timezone.utc)

_ESCAPES = {
    '"': '\\"',
    "\\": "\\\\",
    "\n": "\\n",
    "\r": "\\r",
    "\t": "\\t",
    "<": "\\u003c",
    ">": "\\u003e",
    "&": "\\u0026",
    "\u2028": "\\u2028",
    "\u2029": "\\u2029",
}


def go_json_escape(s: str) -> str:
    """Escape a string exactly as Go encoding/json does (HTML escaping on)."""
    out = []
    for ch in s:
        esc = _ESCAPES.get(ch)
        if esc is not None:
            out.append(esc)
        elif ch < " ":
            out.append("\\u%04x" % ord(ch))
        else:
            out.append(ch)
    return "".join(out)


def format_go_time(t: Optional[_dt.datetime]) -> str:
    """Format a datetime the way Go marshals time.Time (RFC3339Nano).

    Nanosecond precision with trailing zeros trimmed; "Z" when the UTC offset
    is zero. Naive datetimes are treated as UTC (the synthetic feed produces
    UTC timestamps; the reference container runs with TZ=UTC).
    """
    if t is None:
        t = ZERO_TIME
    if t.tzinfo is None:
        t = t.replace(tzinfo=_dt.timezone.utc)
    base = "%04d-%02d-%02dT%02d:%02d:%02d" % (
        t.year, t.month, t.day, t.hour, t.minute, t.second
    )
    frac = ""
    if t.microsecond:
        frac = ("%.9f" % (t.microsecond / 1e6))[1:].rstrip("0")
        if frac == ".":
            frac = ""
    off = t.utcoffset()
    if off is None or off == _dt.timedelta(0):
        zone = "Z"
    else:
        total = int(off.total_seconds())
        sign = "+" if total >= 0 else "-"
        total = abs(total)
        zone = "%s%02d:%02d" % (sign, total // 3600, (total % 3600) // 60)
    return base + frac + zone


def _enc_str(s: str) -> str:
    return '"' + go_json_escape(s) + '"'


def _enc_time(t: Optional[_dt.datetime]) -> str:
    return '"' + format_go_time(t) + '"'


def _enc_float(v: float) -> str:
    # Go encoding/json floatEncoder (encode.go): shortest round-trip
    # digits; 'f' format for 1e-6 <= |v| < 1e21, 'e' otherwise with the
    # e-0X -> e-X exponent cleanup. Integral floats under 1e21 print
    # bare. Python's repr produces the same shortest digit string but
    # its own format choice (switches to exponent at |v| < 1e-4), so
    # re-format per Go's rules.
    if v != v or v in (float("inf"), float("-inf")):
        raise ValueError("json: unsupported value: " + repr(v))
    v = float(v)
    if v.is_integer() and abs(v) < 1e21:
        return str(int(v))
    s = repr(v)
    if "e" not in s:
        return s  # python chose 'f' here; Go agrees in this range
    mant, _, es = s.partition("e")
    exp = int(es)
    if 1e-6 <= abs(v) < 1e21:
        # python chose 'e' (only happens for |v| < 1e-4 among
        # non-integral floats) but Go uses 'f': expand manually.
        sign = "-" if mant.startswith("-") else ""
        core = mant.lstrip("-")
        digits = core.replace(".", "")
        point = len(core.partition(".")[0]) + exp  # always <= 0 here
        return sign + "0." + "0" * (-point) + digits
    if -9 <= exp < 0:
        # Go's exponent cleanup: e-07 -> e-7 (2-digit negatives only)
        return f"{mant}e-{-exp}"
    return s


def _enc_opt_int(v: Optional[int]) -> str:
    return "null" if v is None else str(int(v))


def _enc_opt_str(v: Optional[str]) -> str:
    return "null" if v is None else _enc_str(v)


def _enc_opt_bool(v: Optional[bool]) -> str:
    if v is None:
        return "null"
    return "true" if v else "false"


def _enc_iface_list(v: Optional[List]) -> str:
    # []interface{} — nil slices marshal to null; we only ever emit
    # primitives (str/int/float) here.
    if v is None:
        return "null"
    parts = []
    for item in v:
        if item is None:
            parts.append("null")
        elif isinstance(item, bool):
            parts.append("true" if item else "false")
        elif isinstance(item, str):
            parts.append(_enc_str(item))
        elif isinstance(item, int):
            parts.append(str(item))
        elif isinstance(item, float):
            parts.append(_enc_float(item))
        else:
            raise TypeError(f"unsupported interface{{}} element: {type(item)}")
    return "[" + ",".join(parts) + "]"


def _enc_str_list(v: Optional[List[str]]) -> str:
    if v is None:
        return "null"
    return "[" + ",".join(_enc_str(s) for s in v) + "]"


def _enc_int_map(v: Optional[Dict[str, int]]) -> str:
    if v is None:
        return "null"
    return "{" + ",".join(
        f"{_enc_str(k)}:{int(v[k])}" for k in sorted(v)
    ) + "}"


@dataclasses.dataclass
class EngagementData:
    """model/data.go:107-115."""

    follower_count: int = 0
    following_count: int = 0
    like_count: int = 0
    post_count: int = 0
    views_count: int = 0
    comment_count: int = 0
    share_count: int = 0

    def to_json(self) -> str:
        return (
            '{"follower_count":%d,"following_count":%d,"like_count":%d,'
            '"post_count":%d,"views_count":%d,"comment_count":%d,'
            '"share_count":%d}'
            % (
                self.follower_count,
                self.following_count,
                self.like_count,
                self.post_count,
                self.views_count,
                self.comment_count,
                self.share_count,
            )
        )


@dataclasses.dataclass
class ChannelData:
    """model/data.go:93-104."""

    channel_id: str = ""
    channel_name: str = ""
    channel_description: str = ""
    channel_profile_image: str = ""
    channel_engagement_data: EngagementData = dataclasses.field(
        default_factory=EngagementData
    )
    channel_url_external: str = ""
    channel_url: str = ""
    country_code: str = ""
    published_at: Optional[_dt.datetime] = None  # zero time for Telegram

    def to_json(self) -> str:
        return (
            '{"channel_id":%s,"channel_name":%s,"channel_description":%s,'
            '"channel_profile_image":%s,"channel_engagement_data":%s,'
            '"channel_url_external":%s,"channel_url":%s,"country_code":%s,'
            '"published_at":%s}'
            % (
                _enc_str(self.channel_id),
                _enc_str(self.channel_name),
                _enc_str(self.channel_description),
                _enc_str(self.channel_profile_image),
                self.channel_engagement_data.to_json(),
                _enc_str(self.channel_url_external),
                _enc_str(self.channel_url),
                _enc_str(self.country_code),
                _enc_time(self.published_at),
            )
        )


@dataclasses.dataclass
class Comment:
    """model/data.go:79-87."""

    text: str = ""
    reactions: Optional[Dict[str, int]] = None
    view_count: int = 0
    reply_count: int = 0
    handle: str = ""

    def to_json(self) -> str:
        return (
            '{"text":%s,"reactions":%s,"view_count":%d,"reply_count":%d,'
            '"handle":%s}'
            % (
                _enc_str(self.text),
                _enc_int_map(self.reactions),
                self.view_count,
                self.reply_count,
                _enc_str(self.handle),
            )
        )


@dataclasses.dataclass
class OCRData:
    """model/data.go:118-121."""

    ocr_text: str = ""
    thumb_url: str = ""

    def to_json(self) -> str:
        return '{"ocr_text":%s,"thumb_url":%s}' % (
            _enc_str(self.ocr_text),
            _enc_str(self.thumb_url),
        )


@dataclasses.dataclass
class PerformanceScores:
    """model/data.go:125-130."""

    likes: Optional[int] = None
    shares: Optional[int] = None
    comments: Optional[int] = None
    views: float = 0.0

    def to_json(self) -> str:
        return '{"likes":%s,"shares":%s,"comments":%s,"views":%s}' % (
            _enc_opt_int(self.likes),
            _enc_opt_int(self.shares),
            _enc_opt_int(self.comments),
            _enc_float(self.views),
        )


@dataclasses.dataclass
class InnerLink:
    """model/data.go:134 — empty placeholder struct."""

    def to_json(self) -> str:
        return "{}"


@dataclasses.dataclass
class MediaData:
    """model/data.go:139-142."""

    document_name: str = ""

    def to_json(self) -> str:
        return '{"document_name":%s}' % _enc_str(self.document_name)


@dataclasses.dataclass
class Post:
    """The unified ~70-field post schema (model/data.go:9-75).

    Field order below IS the JSON emission order — do not reorder.
    """

    post_link: str = ""
    channel_id: str = ""
    post_uid: str = ""
    url: str = ""
    published_at: Optional[_dt.datetime] = None
    created_at: Optional[_dt.datetime] = None
    language_code: str = ""
    engagement: int = 0
    view_count: int = 0
    like_count: int = 0
    share_count: int = 0
    comment_count: int = 0
    crawl_label: str = ""
    list_ids: Optional[List] = None
    channel_name: str = ""
    search_terms: Optional[List] = None
    search_term_ids: Optional[List] = None
    project_ids: Optional[List] = None
    exercise_ids: Optional[List] = None
    label_data: Optional[List] = None
    labels_metadata: Optional[List] = None
    project_labeled_post_ids: Optional[List] = None
    labeler_ids: Optional[List] = None
    all_labels: Optional[List] = None
    label_ids: Optional[List] = None
    is_ad: bool = False
    transcript_text: str = ""
    image_text: str = ""
    video_length: Optional[int] = None
    is_verified: Optional[bool] = None
    channel_data: ChannelData = dataclasses.field(default_factory=ChannelData)
    platform_name: str = ""
    shared_id: Optional[str] = None
    quoted_id: Optional[str] = None
    replied_id: Optional[str] = None
    ai_label: Optional[str] = None
    root_post_id: Optional[str] = None
    engagement_steps_count: int = 0
    ocr_data: Optional[List[OCRData]] = None
    performance_scores: PerformanceScores = dataclasses.field(
        default_factory=PerformanceScores
    )
    has_embed_media: Optional[bool] = None
    description: str = ""
    repost_channel_data: Optional[str] = None
    post_type: Optional[List[str]] = None
    inner_link: InnerLink = dataclasses.field(default_factory=InnerLink)
    post_title: Optional[str] = None
    media_data: MediaData = dataclasses.field(default_factory=MediaData)
    is_reply: Optional[bool] = None
    ad_fields: Optional[str] = None
    likes_count: int = 0
    shares_count: int = 0
    comments_count: int = 0
    views_count: int = 0
    searchable_text: str = ""
    all_text: str = ""
    contrast_agent_project_ids: Optional[List] = None
    agent_ids: Optional[List] = None
    segment_ids: Optional[List] = None
    thumb_url: str = ""
    media_url: str = ""
    comments: Optional[List[Comment]] = None
    reactions: Optional[Dict[str, int]] = None
    outlinks: Optional[List[str]] = None
    capture_time: Optional[_dt.datetime] = None
    handle: str = ""

    def to_json(self) -> str:
        ocr = (
            "null"
            if self.ocr_data is None
            else "[" + ",".join(o.to_json() for o in self.ocr_data) + "]"
        )
        comments = (
            "null"
            if self.comments is None
            else "[" + ",".join(c.to_json() for c in self.comments) + "]"
        )
        parts = [
            '"post_link":' + _enc_str(self.post_link),
            '"channel_id":' + _enc_str(self.channel_id),
            '"post_uid":' + _enc_str(self.post_uid),
            '"url":' + _enc_str(self.url),
            '"published_at":' + _enc_time(self.published_at),
            '"created_at":' + _enc_time(self.created_at),
            '"language_code":' + _enc_str(self.language_code),
            '"engagement":' + str(self.engagement),
            '"view_count":' + str(self.view_count),
            '"like_count":' + str(self.like_count),
            '"share_count":' + str(self.share_count),
            '"comment_count":' + str(self.comment_count),
            '"crawl_label":' + _enc_str(self.crawl_label),
            '"list_ids":' + _enc_iface_list(self.list_ids),
            '"channel_name":' + _enc_str(self.channel_name),
            '"search_terms":' + _enc_iface_list(self.search_terms),
            '"search_term_ids":' + _enc_iface_list(self.search_term_ids),
            '"project_ids":' + _enc_iface_list(self.project_ids),
            '"exercise_ids":' + _enc_iface_list(self.exercise_ids),
            '"label_data":' + _enc_iface_list(self.label_data),
            '"labels_metadata":' + _enc_iface_list(self.labels_metadata),
            '"project_labeled_post_ids":'
            + _enc_iface_list(self.project_labeled_post_ids),
            '"labeler_ids":' + _enc_iface_list(self.labeler_ids),
            '"all_labels":' + _enc_iface_list(self.all_labels),
            '"label_ids":' + _enc_iface_list(self.label_ids),
            '"is_ad":' + ("true" if self.is_ad else "false"),
            '"transcript_text":' + _enc_str(self.transcript_text),
            '"image_text":' + _enc_str(self.image_text),
            '"video_length":' + _enc_opt_int(self.video_length),
            '"is_verified":' + _enc_opt_bool(self.is_verified),
            '"channel_data":' + self.channel_data.to_json(),
            '"platform_name":' + _enc_str(self.platform_name),
            '"shared_id":' + _enc_opt_str(self.shared_id),
            '"quoted_id":' + _enc_opt_str(self.quoted_id),
            '"replied_id":' + _enc_opt_str(self.replied_id),
            '"ai_label":' + _enc_opt_str(self.ai_label),
            '"root_post_id":' + _enc_opt_str(self.root_post_id),
            '"engagement_steps_count":' + str(self.engagement_steps_count),
            '"ocr_data":' + ocr,
            '"performance_scores":' + self.performance_scores.to_json(),
            '"has_embed_media":' + _enc_opt_bool(self.has_embed_media),
            '"description":' + _enc_str(self.description),
            '"repost_channel_data":' + _enc_opt_str(self.repost_channel_data),
            '"post_type":' + _enc_str_list(self.post_type),
            '"inner_link":' + self.inner_link.to_json(),
            '"post_title":' + _enc_opt_str(self.post_title),
            '"media_data":' + self.media_data.to_json(),
            '"is_reply":' + _enc_opt_bool(self.is_reply),
            '"ad_fields":' + _enc_opt_str(self.ad_fields),
            '"likes_count":' + str(self.likes_count),
            '"shares_count":' + str(self.shares_count),
            '"comments_count":' + str(self.comments_count),
            '"views_count":' + str(self.views_count),
            '"searchable_text":' + _enc_str(self.searchable_text),
            '"all_text":' + _enc_str(self.all_text),
            '"contrast_agent_project_ids":'
            + _enc_iface_list(self.contrast_agent_project_ids),
            '"agent_ids":' + _enc_iface_list(self.agent_ids),
            '"segment_ids":' + _enc_iface_list(self.segment_ids),
            '"thumb_url":' + _enc_str(self.thumb_url),
            '"media_url":' + _enc_str(self.media_url),
            '"comments":' + comments,
            '"reactions":' + _enc_int_map(self.reactions),
            '"outlinks":' + _enc_str_list(self.outlinks),
            '"capture_time":' + _enc_time(self.capture_time),
            '"handle":' + _enc_str(self.handle),
        ]
        return "{" + ",".join(parts) + "}"

    def to_jsonl(self) -> str:
        return self.to_json() + "\n"
