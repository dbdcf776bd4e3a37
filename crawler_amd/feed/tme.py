"""Mock t.me HTTP target for the tandem validator.

Serves fixture-shaped HTML per username, derived deterministically from the
synthetic feed universe (replaces the live https://t.me/<u> fetch of
telegramhelper/channelvalidator.go:64-103). Supports blocked-mode injection
to exercise the validator's IP-block state machine
(crawl/validator.go:34-38, 112-169).
"""
from __future__ import annotations

import os
from typing import Tuple

_FIXTURES = os.path.join(
    os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
    "fixtures", "telegram-html",
)


def _load(name: str) -> bytes:
    with open(os.path.join(_FIXTURES, name), "rb") as f:
        return f.read()


class MockTMe:
    """fetcher(username) -> (status, body) with deterministic outcomes.

    Usernames of the synthetic universe ("c" + 10 digits < universe) are
    valid supergroups; a hash slice of them classify as user accounts /
    unoccupied; everything else is not_found. `blocked` simulates an IP
    block (429 until cleared)."""

    def __init__(self, universe: int, user_permille: int = 50,
                 unoccupied_permille: int = 50):
        self.universe = universe
        self.user_permille = user_permille
        self.unoccupied_permille = unoccupied_permille
        self.blocked = False
        self.requests = 0
        self._valid = _load("valid-channel.html")
        self._user = _load("not-a-supergroup.html")
        self._unocc = _load("username-not-occupied.html")
        self._invalid = _load("invalid-channel.html")

    def __call__(self, username: str) -> Tuple[int, bytes]:
        self.requests += 1
        if self.blocked:
            # canary endpoint t.me/telegram stays blocked too until cleared
            return 429, b""
        if username == "telegram":
            return 200, self._valid
        if username.startswith("c") and username[1:].isdigit():
            cid = int(username[1:])
            if cid < self.universe:
                h = (cid * 2654435761) % 1000
                if h < self.user_permille:
                    return 200, self._user
                if h < self.user_permille + self.unoccupied_permille:
                    return 200, self._unocc
                return 200, self._valid
        return 200, self._invalid
