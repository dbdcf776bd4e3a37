"""Distributed wire contracts (reference distributed/messages.go:11-333).

JSON-serializable dataclasses for the work queue, results, status and
control topics, with validation and trace-ID generation. Topic names match
the reference (messages.go:53-58) so operators see familiar terms.
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import json
import uuid
from typing import Dict, List, Optional

# Message types (messages.go:11-29)
MSG_WORK_ITEM = "work_item"
MSG_POISON_PILL = "poison_pill"
MSG_WORK_RESULT = "work_result"
MSG_DISCOVERED_PAGES = "discovered_pages"
MSG_HEARTBEAT = "heartbeat"
MSG_WORKER_STARTED = "worker_started"
MSG_WORKER_STOPPING = "worker_stopping"
MSG_PAUSE = "pause"
MSG_RESUME = "resume"
MSG_STOP = "stop"

STATUS_SUCCESS = "success"
STATUS_ERROR = "error"
STATUS_PARTIAL = "partial"
STATUS_RETRY = "retry"

WORKER_ACTIVE = "active"
WORKER_IDLE = "idle"
WORKER_BUSY = "busy"
WORKER_ERROR = "error"
WORKER_OFFLINE = "offline"

PRIORITY_HIGH = 1
PRIORITY_MEDIUM = 3
PRIORITY_LOW = 5

TOPIC_WORK_QUEUE = "crawl-work-queue"
TOPIC_RESULTS = "crawl-results"
TOPIC_WORKER_STATUS = "worker-status"
TOPIC_ORCHESTRATOR = "orchestrator-commands"


def new_trace_id() -> str:
    """trace_YYYYMMDDHHMMSS_XXXXXXXX (messages.go:239-241)."""
    ts = _dt.datetime.now().strftime("%Y%m%d%H%M%S")
    return f"trace_{ts}_{uuid.uuid4().hex[:8]}"


def _iso(t: Optional[_dt.datetime]) -> Optional[str]:
    return t.isoformat() if t else None


def _from_iso(s: Optional[str]) -> Optional[_dt.datetime]:
    return _dt.datetime.fromisoformat(s) if s else None


@dataclasses.dataclass
class WorkItem:
    """messages.go:72-88."""

    id: str = ""
    url: str = ""
    depth: int = 0
    crawl_id: str = ""
    platform: str = "telegram"
    parent_id: str = ""
    retry_count: int = 0
    assigned_to: str = ""
    priority: int = PRIORITY_MEDIUM
    sequence_id: str = ""
    created_at: Optional[_dt.datetime] = None
    assigned_at: Optional[_dt.datetime] = None
    trace_id: str = ""
    config: Dict = dataclasses.field(default_factory=dict)

    def validate(self) -> None:
        """messages.go validation: id, url, crawl_id and a SUPPORTED
        platform are required (messages_test.go:103-170 matrix)."""
        if not self.id:
            raise ValueError("work item missing id")
        if not self.url:
            raise ValueError("work item missing url")
        if not self.crawl_id:
            raise ValueError("work item missing crawl_id")
        if not self.platform:
            raise ValueError("work item missing platform")
        if self.platform not in ("telegram", "youtube"):
            raise ValueError(
                f"work item has unsupported platform: {self.platform}")
        if self.depth < 0:
            raise ValueError("work item depth must be >= 0")

    def to_json(self) -> str:
        d = dataclasses.asdict(self)
        d["created_at"] = _iso(self.created_at)
        d["assigned_at"] = _iso(self.assigned_at)
        return json.dumps(d)

    @classmethod
    def from_json(cls, s: str) -> "WorkItem":
        d = json.loads(s)
        d["created_at"] = _from_iso(d.get("created_at"))
        d["assigned_at"] = _from_iso(d.get("assigned_at"))
        return cls(**d)


@dataclasses.dataclass
class WorkResult:
    """messages.go:120-140 equivalent."""

    work_item_id: str = ""
    worker_id: str = ""
    status: str = STATUS_SUCCESS   # success|error|partial|retry
    error: str = ""
    page_status: str = "fetched"   # fetched|error|deadend
    posts_stored: int = 0
    discovered: List[str] = dataclasses.field(default_factory=list)
    # count-only form for paths whose next layer travels out-of-band
    # (OrchestratedCrawl all-gathers names; ~1M names as JSON is waste)
    discovered_count: int = 0
    duration_ms: float = 0.0
    trace_id: str = ""

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self))

    @classmethod
    def from_json(cls, s: str) -> "WorkResult":
        return cls(**json.loads(s))


@dataclasses.dataclass
class StatusMessage:
    """Heartbeat / lifecycle (messages.go heartbeat section)."""

    message_type: str = MSG_HEARTBEAT
    worker_id: str = ""
    status: str = WORKER_IDLE
    active_item: str = ""
    processed: int = 0
    timestamp: Optional[_dt.datetime] = None

    def to_json(self) -> str:
        d = dataclasses.asdict(self)
        d["timestamp"] = _iso(self.timestamp)
        return json.dumps(d)

    @classmethod
    def from_json(cls, s: str) -> "StatusMessage":
        d = json.loads(s)
        d["timestamp"] = _from_iso(d.get("timestamp"))
        return cls(**d)


@dataclasses.dataclass
class ControlMessage:
    """pause/resume/stop (messages.go control section)."""

    message_type: str = MSG_STOP
    reason: str = ""
    trace_id: str = ""

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self))

    @classmethod
    def from_json(cls, s: str) -> "ControlMessage":
        return cls(**json.loads(s))
