"""Work queues + heartbeats over torch.distributed.TCPStore.

MI355X-native replacement for Dapr pub/sub topics (distributed/pubsub.go):
a TCPStore (the same rendezvous store family torch.distributed uses) hosts
append-only topics with atomic head/tail counters:

- publish: tail = add(topic/tail, 1); set(topic/<tail-1>, payload)
- claim  : head = add(topic/claim, 1) - 1; if head < tail: get(topic/<head>)

`add` is atomic on the store server, so concurrent claimers never receive
the same element — the same exactly-once delivery the reference gets from
Redis streams, without a sidecar. Heartbeats are per-worker keys with
timestamps (worker/worker.go:234-252; orchestrator reads them for the
5-min offline rule, orchestrator.go:493-517).
"""
from __future__ import annotations

import datetime as _dt
import time
from typing import List, Optional


class StoreQueue:
    """One named topic over a TCPStore-like object (set/get/add/check).

    Publish order: a slot number is drawn from `pub`, the payload is
    written, THEN `tail` is bumped — so `tail` only ever covers readable
    slots (safe with several concurrent publishers). A claimer that races
    another claimer past `tail` STASHES its ticket instead of blocking:
    the ticket is honored on a later claim() once more elements are
    published, so no element is lost and no claimer ever blocks on a slot
    that may never fill (which would deadlock depth-barriered layers)."""

    def __init__(self, store, topic: str):
        self.store = store
        self.topic = topic
        self._pub_key = f"{topic}/pub"
        self._tail_key = f"{topic}/tail"
        self._claim_key = f"{topic}/claim"
        self._stash: Optional[int] = None

    def publish(self, payload: str) -> int:
        idx = self.store.add(self._pub_key, 1) - 1
        self.store.set(f"{self.topic}/{idx}", payload)
        self.store.add(self._tail_key, 1)
        return idx

    def _counters(self):
        tail = self.store.add(self._tail_key, 0)
        claim = self.store.add(self._claim_key, 0)
        return claim, tail

    def size(self) -> int:
        claim, tail = self._counters()
        return max(0, tail - claim)

    def claim(self, timeout_s: float = 0.0,
              poll_s: float = 0.02) -> Optional[str]:
        """Claim the next element; None if empty past the timeout."""
        deadline = time.monotonic() + timeout_s
        while True:
            _, tail = self._counters()
            if self._stash is not None:
                if self._stash < tail:
                    idx = self._stash
                    self._stash = None
                    return self._get_blocking(idx)
            else:
                claim, tail = self._counters()
                if claim < tail:
                    idx = self.store.add(self._claim_key, 1) - 1
                    if idx < tail:
                        return self._get_blocking(idx)
                    # raced past tail: hold the ticket for the future
                    self._stash = idx
            if time.monotonic() >= deadline:
                return None
            time.sleep(poll_s)

    def _get_blocking(self, idx: int, timeout_s: float = 30.0) -> str:
        key = f"{self.topic}/{idx}"
        deadline = time.monotonic() + timeout_s
        while True:
            try:
                v = self.store.get(key)
                return v.decode() if isinstance(v, (bytes, bytearray)) else v
            except Exception:
                if time.monotonic() >= deadline:
                    raise
                time.sleep(0.02)

    def drain(self, max_items: int = 1_000_000) -> List[str]:
        out = []
        for _ in range(max_items):
            v = self.claim(timeout_s=0.0)
            if v is None:
                break
            out.append(v)
        return out


def _try_get(store, key: str):
    """Non-blocking store read: TCPStore.get WAITS for missing keys (up to
    its timeout), so probe with check() first where supported."""
    try:
        if hasattr(store, "check") and not store.check([key]):
            return None
        v = store.get(key)
        return v.decode() if isinstance(v, (bytes, bytearray)) else v
    except Exception:
        return None


class Heartbeats:
    """Per-worker heartbeat keys (worker/worker.go:234-252)."""

    def __init__(self, store, prefix: str = "hb"):
        self.store = store
        self.prefix = prefix

    def beat(self, worker_id: str, status: str = "active") -> None:
        now = _dt.datetime.now(_dt.timezone.utc).timestamp()
        self.store.set(f"{self.prefix}/{worker_id}", f"{now}|{status}")

    def register(self, worker_id: str) -> None:
        self.store.add(f"{self.prefix}/registry/{worker_id}", 1)
        members = self._members()
        if worker_id not in members:
            members.append(worker_id)
            self.store.set(f"{self.prefix}/members", ",".join(members))

    def _members(self) -> List[str]:
        s = _try_get(self.store, f"{self.prefix}/members")
        return [m for m in s.split(",") if m] if s else []

    def workers(self) -> List[str]:
        return self._members()

    def last_seen(self, worker_id: str):
        s = _try_get(self.store, f"{self.prefix}/{worker_id}")
        if not s:
            return None, "unknown"
        ts, status = s.split("|", 1)
        return float(ts), status

    def offline_workers(self, timeout_s: float) -> List[str]:
        """Workers whose last beat is older than timeout_s
        (orchestrator.go:493-517, default 5 min)."""
        now = _dt.datetime.now(_dt.timezone.utc).timestamp()
        out = []
        for w in self.workers():
            ts, _status = self.last_seen(w)
            if ts is None or now - ts > timeout_s:
                out.append(w)
        return out


class InMemoryStore:
    """TCPStore-compatible shim for unit tests (set/get/add)."""

    def __init__(self):
        import threading

        self._d = {}
        self._lock = threading.Lock()

    def set(self, k: str, v):
        with self._lock:
            self._d[k] = v if isinstance(v, bytes) else str(v).encode()

    def get(self, k: str) -> bytes:
        with self._lock:
            if k not in self._d:
                raise KeyError(k)
            return self._d[k]

    def add(self, k: str, n: int) -> int:
        with self._lock:
            cur = int(self._d.get(k, b"0"))
            cur += n
            self._d[k] = str(cur).encode()
            return cur

    def check(self, keys) -> bool:
        with self._lock:
            return all(k in self._d for k in keys)
