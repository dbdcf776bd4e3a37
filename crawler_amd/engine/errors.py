"""Crawl error taxonomy: FLOOD_WAIT parsing, TDLib-400 detection.

Parity: crawl/runner.go:35-110.
"""
from __future__ import annotations

import re
from typing import Tuple

FLOOD_WAIT_RETIRE_THRESHOLD_SECS = 300  # crawl/runner.go:49


class FloodWaitRetire(Exception):
    """FLOOD_WAIT >= threshold: client retired (crawl/runner.go:35-38)."""


class TDLib400(Exception):
    """Permanently-invalid channel (crawl/runner.go:40-44)."""


class WalkbackExhausted(Exception):
    """No walkback candidate available (layerless loop leaves the page in
    the buffer; dapr/standalone.go:902)."""


class ConnectionDropped(Exception):
    """Transport-level failure: the pooled session must be destroyed and
    recreated (HandleConnectionError, connection_pool.go:346-413)."""


class PoolExhausted(Exception):
    """Connection pool has no available connections."""


_FLOOD = re.compile(r"FLOOD_WAIT_(\d*)")
_RETRY = re.compile(r"retry after (\d*)")


def parse_flood_wait_secs(err_msg: str) -> Tuple[int, bool]:
    """(seconds, is_flood_wait) — crawl/runner.go:55-97.

    Unparseable seconds => (0, True): short transient ban.
    """
    if not err_msg:
        return 0, False
    m = _FLOOD.search(err_msg)
    if m:
        return (int(m.group(1)), True) if m.group(1) else (0, True)
    m = _RETRY.search(err_msg)
    if m:
        return (int(m.group(1)), True) if m.group(1) else (0, True)
    return 0, False


def is_tdlib_400(err_msg: str) -> bool:
    """crawl/runner.go:104-110."""
    if not err_msg:
        return False
    return (
        "[400]" in err_msg
        or "400 USERNAME_NOT_OCCUPIED" in err_msg
        or "400 USERNAME_INVALID" in err_msg
        or "no messages found in the chat" in err_msg
    )


def is_connection_error(err_msg: str) -> bool:
    """Transport-failure classification (the substrings the reference's
    HandleConnectionError call sites key on)."""
    m = err_msg.lower()
    return ("connection" in m or "timeout" in m or "socket" in m
            or "broken pipe" in m)
