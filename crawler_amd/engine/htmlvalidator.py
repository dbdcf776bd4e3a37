"""HTTP channel validator: t.me HTML classification + rate limiting.

Parity (reference telegramhelper/channelvalidator.go, validator_rate_limiter.go):
- ParseChannelHTML title rules (channelvalidator.go:132-153):
    "View @..."                         -> valid
    "Contact @..." + meta robots noindex -> invalid / not_found
    "Contact @..."                      -> not_channel / not_supergroup
    "Telegram Messenger"                -> invalid / not_found
- 64 KB body cap (channelvalidator.go:103)
- error taxonomy ErrTransient / ErrBlocked (channelvalidator.go:27-47)
- token bucket + jitter (validator_rate_limiter.go:23-55)

The live t.me + uTLS transport is replaced by an injectable fetcher
(feed/tme.py provides the mock target), keeping the classification state
machine intact.
"""
from __future__ import annotations

import dataclasses
import re
from typing import Callable, Tuple

BODY_CAP = 64 * 1024  # channelvalidator.go:103

ERR_TRANSIENT = "transient"
ERR_BLOCKED = "blocked"


class ValidationHTTPError(Exception):
    def __init__(self, kind: str, msg: str = ""):
        super().__init__(msg or kind)
        self.kind = kind


@dataclasses.dataclass
class ChannelValidationResult:
    status: str  # "valid" | "not_channel" | "invalid"
    reason: str = ""  # "" | "not_supergroup" | "not_found" | ...


_TITLE_RE = re.compile(rb"<title[^>]*>(.*?)</title>", re.S | re.I)
_NOINDEX_RE = re.compile(
    rb"<meta[^>]+name=[\"']robots[\"'][^>]+noindex", re.I
)


def parse_channel_html(body: bytes) -> ChannelValidationResult:
    """Title-rule classification (channelvalidator.go:132-153)."""
    body = body[:BODY_CAP]
    m = _TITLE_RE.search(body)
    title = m.group(1).strip() if m else b""
    if title.startswith(b"View @") or b"Telegram: View @" in title:
        return ChannelValidationResult("valid")
    if title.startswith(b"Contact @") or b"Telegram: Contact @" in title:
        if _NOINDEX_RE.search(body):
            return ChannelValidationResult("invalid", "not_found")
        return ChannelValidationResult("not_channel", "not_supergroup")
    if b"Telegram Messenger" in title:
        return ChannelValidationResult("invalid", "not_found")
    return ChannelValidationResult("invalid", "unrecognized")


def validate_channel_http(
    username: str,
    fetcher: Callable[[str], Tuple[int, bytes]],
) -> ChannelValidationResult:
    """Fetch https://t.me/<username> via `fetcher(username) -> (status,
    body)` and classify. Raises ValidationHTTPError on access problems
    (429/403 -> blocked; 5xx/timeouts -> transient)."""
    try:
        status, body = fetcher(username)
    except TimeoutError as e:
        raise ValidationHTTPError(ERR_TRANSIENT, str(e))
    if status in (403, 429):
        raise ValidationHTTPError(ERR_BLOCKED, f"HTTP {status}")
    if status >= 500:
        raise ValidationHTTPError(ERR_TRANSIENT, f"HTTP {status}")
    if status == 404:
        return ChannelValidationResult("invalid", "not_found")
    return parse_channel_html(body)
