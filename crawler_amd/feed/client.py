"""Synthetic Telegram client facade: the TDLib-shaped API over SyntheticFeed.

Replaces the reference's cgo TDLib stack (crawler/crawler.go:109-126 facade;
telegramhelper/client.go init; connection_pool.go; rate_limiter.go) with a
deterministic in-process engine:

- `SyntheticTelegramClient` implements the TDLib facade surface the crawl
  engine uses (search_public_chat / get_chat_history / get_supergroup_info /
  get_message_comments / get_message) against SyntheticFeed data;
- failure injection with TDLib-shaped error strings: "[429] FLOOD_WAIT_N"
  and "[400] USERNAME_NOT_OCCUPIED" (rates configurable, hash-derived so
  runs are reproducible);
- cache-vs-server latency classes (telegramutils.go:855-879: <5ms = local
  cache) surfaced via `last_call_latency_class` without real sleeps;
- `RateLimitedClient` wraps any client in per-method token buckets with
  jitter (common/utils.go:19-46 defaults; reactive GetMessage throttling
  per rate_limiter.go:145-169);
- `ConnectionPool` reproduces connection_pool.go:21-240,421-439 semantics
  (checkout/checkin, retire on long FLOOD_WAIT, recreate on error,
  PoolExhausted when empty).
"""
from __future__ import annotations

import dataclasses
import random
import threading
import time
from typing import Dict, List, Optional

import numpy as np

from ..engine.errors import PoolExhausted
from ..ops import golden as G
from ..ops import batch as B
from .synth import SyntheticFeed, _splitmix64


class TelegramAPIError(Exception):
    """Carries a TDLib-shaped error string (matched by engine/errors.py)."""


@dataclasses.dataclass
class ChatInfo:
    chat_id: int
    title: str
    username: str
    member_count: int
    message_count: int
    total_views: int


@dataclasses.dataclass
class FaultConfig:
    """Deterministic failure-injection rates (per mille, hash-derived)."""

    flood_wait_permille: int = 0          # odds a search triggers FLOOD_WAIT
    flood_wait_secs: int = 30             # injected retry-after
    long_flood_permille: int = 0          # odds of a >=300s FLOOD_WAIT
    invalid_channel_permille: int = 0     # odds a username 400s
    conn_reset_permille: int = 0          # odds of a transport failure


class SyntheticTelegramClient:
    """One pooled "connection" serving deterministic synthetic data."""

    def __init__(self, feed: SyntheticFeed, conn_id: str = "conn_0",
                 faults: Optional[FaultConfig] = None,
                 posts_per_channel: Optional[int] = None):
        self.feed = feed
        self.conn_id = conn_id
        self.faults = faults or FaultConfig()
        self.posts_per_channel = (
            posts_per_channel or feed.cfg.posts_per_channel
        )
        self.closed = False
        # local "TDLib caches" for cache-vs-server classification
        self._chat_cache: Dict[int, ChatInfo] = {}
        self._history_cache: Dict[int, list] = {}
        self.last_call_latency_class = "server"
        self._call_count = 0

    # -- internals --

    def _cid_of_username(self, username: str) -> int:
        if not username.startswith("c") or not username[1:].isdigit():
            raise TelegramAPIError(
                f"[400] USERNAME_INVALID: {username}"
            )
        cid = int(username[1:])
        if cid >= self.feed.cfg.universe:
            raise TelegramAPIError(
                f"400 USERNAME_NOT_OCCUPIED: {username}"
            )
        return cid

    def _h(self, *xs: int) -> int:
        a = np.uint64(self.feed.cfg.seed ^ 0xFEED)
        for x in xs:
            a = _splitmix64(a ^ np.uint64(x & 0xFFFFFFFFFFFFFFFF))
        return int(a)

    def _maybe_fault(self, cid: int, op: int):
        f = self.faults
        h = self._h(cid, op, self._call_count)
        if f.invalid_channel_permille and op == 1:
            if self._h(cid, 0xBAD) % 1000 < f.invalid_channel_permille:
                raise TelegramAPIError(
                    f"400 USERNAME_NOT_OCCUPIED: c{cid:010d}"
                )
        if f.long_flood_permille and h % 1000 < f.long_flood_permille:
            raise TelegramAPIError("[429] FLOOD_WAIT_3600")
        if f.flood_wait_permille and (h >> 10) % 1000 < f.flood_wait_permille:
            raise TelegramAPIError(
                f"[429] FLOOD_WAIT_{f.flood_wait_secs}"
            )
        if f.conn_reset_permille and (h >> 20) % 1000 < f.conn_reset_permille:
            raise TelegramAPIError("connection reset by peer")

    # -- TDLib facade (crawler/crawler.go:109-126 subset the engine uses) --

    def search_public_chat(self, username: str) -> ChatInfo:
        self._call_count += 1
        cid = self._cid_of_username(username)
        self._maybe_fault(cid, 1)
        if cid in self._chat_cache:
            self.last_call_latency_class = "cache"
            return self._chat_cache[cid]
        self.last_call_latency_class = "server"
        row = self.feed.channel_rows(
            np.array([cid]), self.posts_per_channel
        )[0]
        info = ChatInfo(
            chat_id=row.chat_id, title=row.title, username=row.username,
            member_count=row.member_count,
            message_count=row.post_count, total_views=row.total_views,
        )
        self._chat_cache[cid] = info
        return info

    def get_chat(self, chat_id: int) -> ChatInfo:
        self._call_count += 1
        cid = int(-chat_id - 1001000000000)
        if cid < 0 or cid >= self.feed.cfg.universe:
            raise TelegramAPIError(f"[400] CHANNEL_INVALID: {chat_id}")
        return self.search_public_chat("c%010d" % cid)

    def get_supergroup_info(self, chat_id: int) -> dict:
        self._call_count += 1
        info = self.get_chat(chat_id)
        self._maybe_fault(-chat_id, 2)
        return {
            "member_count": info.member_count,
            "active_usernames": [info.username],
        }

    def _channel_messages(self, cid: int) -> List[G.SynthMessage]:
        if cid in self._history_cache:
            self.last_call_latency_class = "cache"
            return self._history_cache[cid]
        self.last_call_latency_class = "server"
        batch = self.feed.build_batch(
            np.array([cid]), posts_per_channel=self.posts_per_channel
        )
        msgs = [B.unpack_message(batch, i) for i in range(batch.n)]
        coms = [B.unpack_comments(batch, i) for i in range(batch.n)]
        for m, c in zip(msgs, coms):
            m._comments = c  # type: ignore[attr-defined]
        self._history_cache[cid] = msgs
        return msgs

    def get_chat_history(self, chat_id: int, from_message_id: int = 0,
                         limit: int = 100) -> List[G.SynthMessage]:
        """Newest-first pagination, 100/page, from_message_id=0 = latest
        (telegramutils.go:25-157 usage pattern)."""
        self._call_count += 1
        cid = int(-chat_id - 1001000000000)
        self._maybe_fault(cid, 3)
        msgs = self._channel_messages(cid)
        ordered = sorted(msgs, key=lambda m: -m.msg_id)
        if from_message_id:
            ordered = [m for m in ordered if m.msg_id < from_message_id]
        return ordered[: min(limit, 100)]

    def get_message(self, chat_id: int, message_id: int) -> G.SynthMessage:
        self._call_count += 1
        cid = int(-chat_id - 1001000000000)
        for m in self._channel_messages(cid):
            if m.msg_id == message_id:
                return m
        raise TelegramAPIError(f"[404] message not found: {message_id}")

    def get_message_thread_history(self, chat_id: int, message_id: int,
                                   from_message_id: int = 0,
                                   limit: int = 100) -> list:
        """One page of a comment thread, newest-first
        (TDLib GetMessageThreadHistory as used by
        telegramutils.go:594-600): returns [(thread_msg_id, Comment)].
        Thread ids descend from len(comments) to 1; from_message_id=0
        starts at the newest."""
        self._call_count += 1
        m = self.get_message(chat_id, message_id)
        coms = getattr(m, "_comments", [])
        n = len(coms)
        # slot s holds thread id n-s (slot 0 = newest)
        start = 0 if not from_message_id else n - from_message_id + 1
        limit = max(0, min(limit, 100))
        page = []
        for s in range(start, min(n, start + limit)):
            page.append((n - s, coms[s]))
        return page

    def get_message_comments(self, chat_id: int, message_id: int,
                             max_comments: int = -1,
                             comment_count: int = 0) -> list:
        """Paginated thread walk (GetMessageComments,
        telegramutils.go:311-747): 100/batch from the newest, batch
        size shrunk to the remaining need, stop on empty page / no
        progress / enough comments. maxcomments==0 means none;
        comment_count (when known, < maxcomments) caps the walk like
        the reference's commentcount argument."""
        if max_comments == 0:
            return []
        comments: list = []
        from_message_id = 0
        while True:
            batch = 100
            if max_comments and max_comments > 0:
                remaining = max_comments - len(comments)
                if remaining <= 0:
                    break
                if 0 < comment_count < max_comments:
                    remaining = comment_count - len(comments)
                    if remaining <= 0:
                        break
                if remaining < batch:
                    batch = remaining
            page = self.get_message_thread_history(
                chat_id, message_id, from_message_id, batch)
            if not page:
                break
            comments.extend(c for _tid, c in page)
            prev = from_message_id
            from_message_id = page[-1][0]
            if from_message_id == 0 or from_message_id == prev:
                break
        return comments

    def close(self):
        self.closed = True


class TokenBucket:
    """rate/min token bucket with jitter (telegramhelper/rate_limiter.go)."""

    def __init__(self, per_minute: float, jitter_ms: int, rng=None,
                 clock=time.monotonic, sleeper=time.sleep):
        self.rate = per_minute / 60.0
        self.capacity = max(1.0, per_minute / 60.0)
        self.tokens = self.capacity
        self.jitter_ms = jitter_ms
        self.last = clock()
        self.clock = clock
        self.sleep = sleeper
        self.rng = rng or random.Random()
        self.lock = threading.Lock()
        self.waits = 0

    def acquire(self):
        with self.lock:
            now = self.clock()
            self.tokens = min(
                self.capacity, self.tokens + (now - self.last) * self.rate
            )
            self.last = now
            if self.tokens >= 1.0:
                self.tokens -= 1.0
                return
            need = (1.0 - self.tokens) / self.rate
            self.tokens = 0.0
            self.waits += 1
        self.sleep(need + self.rng.random() * self.jitter_ms / 1000.0)


class RateLimitedClient:
    """Per-connection method throttles (rate_limiter.go:76-169).

    GetMessage is REACTIVE: a token is consumed only when the call missed
    TDLib's local cache (latency class 'server')."""

    def __init__(self, client: SyntheticTelegramClient, rl_config,
                 rng=None, sleeper=time.sleep, disabled: bool = False):
        self.c = client
        self.disabled = disabled
        r = rl_config
        mk = lambda rate, jit: TokenBucket(rate, jit, rng=rng, sleeper=sleeper)
        self.history_bucket = mk(r.get_chat_history_rate,
                                 r.get_chat_history_jitter_ms)
        self.search_bucket = mk(r.search_public_chat_rate,
                                r.search_public_chat_jitter_ms)
        self.supergroup_bucket = mk(r.get_supergroup_info_rate,
                                    r.get_supergroup_info_jitter_ms)
        self.message_bucket = mk(r.get_message_server_hit_rate,
                                 r.get_message_server_hit_jitter_ms)

    @property
    def conn_id(self):
        return self.c.conn_id

    def search_public_chat(self, username):
        if not self.disabled:
            self.search_bucket.acquire()
        return self.c.search_public_chat(username)

    def get_chat(self, chat_id):
        return self.c.get_chat(chat_id)

    def get_supergroup_info(self, chat_id):
        if not self.disabled:
            self.supergroup_bucket.acquire()
        return self.c.get_supergroup_info(chat_id)

    def get_chat_history(self, chat_id, from_message_id=0, limit=100):
        if not self.disabled:
            self.history_bucket.acquire()
        return self.c.get_chat_history(chat_id, from_message_id, limit)

    def get_message(self, chat_id, message_id):
        out = self.c.get_message(chat_id, message_id)
        # reactive: consume only on a server hit (rate_limiter.go:145-169)
        if not self.disabled and self.c.last_call_latency_class == "server":
            self.message_bucket.acquire()
        return out

    def get_message_thread_history(self, chat_id, message_id,
                                   from_message_id=0, limit=100):
        return self.c.get_message_thread_history(
            chat_id, message_id, from_message_id, limit)

    def get_message_comments(self, chat_id, message_id, max_comments=-1,
                             comment_count=0):
        return self.c.get_message_comments(chat_id, message_id,
                                           max_comments, comment_count)

    def close(self):
        self.c.close()


class ConnectionPool:
    """connection_pool.go:21-240,421-439 semantics over synthetic clients."""

    def __init__(self, feed: SyntheticFeed, size: int, rl_config,
                 faults: Optional[FaultConfig] = None,
                 posts_per_channel: Optional[int] = None,
                 disable_rate_limits: bool = False, rng=None,
                 sleeper=time.sleep):
        self._feed = feed
        self._rl = rl_config
        self._faults = faults
        self._ppc = posts_per_channel
        self._disable = disable_rate_limits
        self._rng = rng
        self._sleeper = sleeper
        self._lock = threading.Lock()
        self._next = 0
        self.available: Dict[str, RateLimitedClient] = {}
        self.in_use: Dict[str, RateLimitedClient] = {}
        self.retired: List[str] = []
        for _ in range(size):
            c = self._new_client()
            self.available[c.conn_id] = c

    def _new_client(self) -> RateLimitedClient:
        with_id = f"conn_{self._next}"
        self._next += 1
        base = SyntheticTelegramClient(
            self._feed, conn_id=with_id, faults=self._faults,
            posts_per_channel=self._ppc,
        )
        return RateLimitedClient(base, self._rl, rng=self._rng,
                                 sleeper=self._sleeper,
                                 disabled=self._disable)

    def get_connection(self) -> RateLimitedClient:
        with self._lock:
            if not self.available:
                raise PoolExhausted(
                    "connection pool exhausted "
                    f"({len(self.in_use)} in use, "
                    f"{len(self.retired)} retired)"
                )
            cid, client = self.available.popitem()
            self.in_use[cid] = client
            return client

    def release_connection(self, client) -> None:
        with self._lock:
            if client.conn_id in self.in_use:
                del self.in_use[client.conn_id]
                self.available[client.conn_id] = client

    def retire_connection(self, client) -> None:
        """Permanent removal on long FLOOD_WAIT
        (connection_pool.go:421-439)."""
        with self._lock:
            self.in_use.pop(client.conn_id, None)
            self.available.pop(client.conn_id, None)
            self.retired.append(client.conn_id)
            client.close()

    def handle_connection_error(self, client) -> None:
        """Destroy and recreate (connection_pool.go:346-413)."""
        with self._lock:
            self.in_use.pop(client.conn_id, None)
            self.available.pop(client.conn_id, None)
            client.close()
            c = self._new_client()
            self.available[c.conn_id] = c

    def stats(self) -> dict:
        with self._lock:
            return {
                "available": len(self.available),
                "in_use": len(self.in_use),
                "retired": len(self.retired),
            }

    def empty(self) -> bool:
        with self._lock:
            return not self.available and not self.in_use
