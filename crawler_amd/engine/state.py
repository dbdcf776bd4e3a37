"""State management: layer map, page/message status, checkpoint/resume,
media cache, random-walk store.

Parity targets (reference file:line):
- Page / Message / EdgeRecord / PendingEdge* / CrawlMetadata / State models:
  state/datamodels.go:41-214
- BaseStateManager in-mem maps, AddLayer URL dedup + MaxPages deadend
  replacement budget: state/base.go:219-322
- DiscoveredChannels O(1) random pick: state/datamodels.go:118-162
- LocalStateManager file layout (state.json / metadata.json /
  media-cache.json under basePath/crawlID; posts ->
  crawlID/channel/posts/posts.jsonl; media -> crawlID/media/channel/):
  state/storageproviders.go:74-647
- FindIncompleteCrawl resume rules (never resume completed; requires a
  non-empty layer map): state/base.go:466-516, daprstate.go:1703-2199
- RandomWalkStore: the Postgres tables (page_buffer, edge_records,
  seed_channels, invalid_channels, pending_edge_batches, pending_edges)
  re-hosted as in-process structures with the same claim semantics
  (FOR UPDATE SKIP LOCKED -> lock + claimed-set; attempt_count poison
  detection): sql/random-walk-schema.sql, sql/validator-schema.sql,
  state/daprstate.go:3150-4391. The per-name exactly-once discovery set
  itself lives on the GPU (ops/gpu.SeenSet) in GPU execution.
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import json
import os
import random
import threading
import uuid
from typing import Dict, List, Optional, Tuple

from ..models.post import Post

UTC = _dt.timezone.utc


def _now() -> _dt.datetime:
    return _dt.datetime.now(UTC)


@dataclasses.dataclass
class PageMessage:
    """state/datamodels.go:67-73."""

    chat_id: int = 0
    message_id: int = 0
    status: str = "unfetched"
    page_id: str = ""
    platform: str = ""

    def to_dict(self):
        return {
            "chatId": self.chat_id, "messageId": self.message_id,
            "status": self.status, "pageId": self.page_id,
            "platform": self.platform,
        }

    @classmethod
    def from_dict(cls, d):
        return cls(
            chat_id=d.get("chatId", 0), message_id=d.get("messageId", 0),
            status=d.get("status", "unfetched"),
            page_id=d.get("pageId", ""), platform=d.get("platform", ""),
        )


@dataclasses.dataclass
class Page:
    """state/datamodels.go:41-64. Status machine: unfetched -> processing ->
    fetched | error | deadend (crawl/runner.go, dapr/standalone.go)."""

    id: str = ""
    url: str = ""
    depth: int = 0
    status: str = "unfetched"
    error: str = ""
    timestamp: Optional[_dt.datetime] = None
    platform: str = ""
    parent_id: str = ""
    messages: List[PageMessage] = dataclasses.field(default_factory=list)
    connection_id: str = ""
    sequence_id: str = ""
    crawl_id: str = ""

    def to_dict(self):
        return {
            "id": self.id, "url": self.url, "depth": self.depth,
            "status": self.status, "error": self.error,
            "timestamp": self.timestamp.isoformat() if self.timestamp else None,
            "platform": self.platform, "parentId": self.parent_id,
            "messages": [m.to_dict() for m in self.messages],
            "LastConnectionID": self.connection_id,
            "sequenceId": self.sequence_id, "crawlId": self.crawl_id,
        }

    @classmethod
    def from_dict(cls, d):
        ts = d.get("timestamp")
        return cls(
            id=d.get("id", ""), url=d.get("url", ""),
            depth=d.get("depth", 0), status=d.get("status", "unfetched"),
            error=d.get("error", ""),
            timestamp=_dt.datetime.fromisoformat(ts) if ts else None,
            platform=d.get("platform", ""), parent_id=d.get("parentId", ""),
            messages=[PageMessage.from_dict(m)
                      for m in d.get("messages", [])],
            connection_id=d.get("LastConnectionID", ""),
            sequence_id=d.get("sequenceId", ""),
            crawl_id=d.get("crawlId", ""),
        )


@dataclasses.dataclass
class EdgeRecord:
    """state/datamodels.go:76-84."""

    destination_channel: str = ""
    discovery_time: Optional[_dt.datetime] = None
    source_channel: str = ""
    walkback: bool = False
    skipped: bool = False
    sequence_id: str = ""
    crawl_id: str = ""


@dataclasses.dataclass
class PendingEdgeBatch:
    """state/datamodels.go:89-98. open -> closed -> processing -> completed."""

    batch_id: str = ""
    crawl_id: str = ""
    source_channel: str = ""
    source_page_id: str = ""
    source_depth: int = 0
    sequence_id: str = ""
    status: str = "open"
    attempt_count: int = 0


@dataclasses.dataclass
class PendingEdge:
    """state/datamodels.go:101-113."""

    pending_id: int = 0
    batch_id: str = ""
    crawl_id: str = ""
    destination_channel: str = ""
    source_channel: str = ""
    sequence_id: str = ""
    discovery_time: Optional[_dt.datetime] = None
    source_type: str = ""
    validation_status: str = "pending"
    validation_reason: str = ""


@dataclasses.dataclass
class CrawlMetadata:
    """state/datamodels.go:172-183."""

    crawl_id: str = ""
    execution_id: str = ""
    start_time: Optional[_dt.datetime] = None
    end_time: Optional[_dt.datetime] = None
    status: str = "running"
    previous_crawl_id: List[str] = dataclasses.field(default_factory=list)
    platform: str = ""
    target_channels: List[str] = dataclasses.field(default_factory=list)
    messages_count: int = 0
    errors_count: int = 0

    def to_dict(self):
        return {
            "crawlId": self.crawl_id, "executionId": self.execution_id,
            "startTime": self.start_time.isoformat() if self.start_time else None,
            "endTime": self.end_time.isoformat() if self.end_time else None,
            "status": self.status, "previousCrawlId": self.previous_crawl_id,
            "platform": self.platform,
            "targetChannels": self.target_channels,
            "messagesCount": self.messages_count,
            "errorsCount": self.errors_count,
        }

    @classmethod
    def from_dict(cls, d):
        def ts(k):
            v = d.get(k)
            return _dt.datetime.fromisoformat(v) if v else None
        return cls(
            crawl_id=d.get("crawlId", ""),
            execution_id=d.get("executionId", ""),
            start_time=ts("startTime"), end_time=ts("endTime"),
            status=d.get("status", "running"),
            previous_crawl_id=d.get("previousCrawlId") or [],
            platform=d.get("platform", ""),
            target_channels=d.get("targetChannels") or [],
            messages_count=d.get("messagesCount", 0),
            errors_count=d.get("errorsCount", 0),
        )


class DiscoveredChannels:
    """Map + key slice for O(1) random pick (state/datamodels.go:118-162)."""

    def __init__(self):
        self._items: Dict[str, bool] = {}
        self._keys: List[str] = []
        self._lock = threading.RLock()

    def add(self, name: str) -> bool:
        with self._lock:
            if name in self._items:
                return False
            self._items[name] = True
            self._keys.append(name)
            return True

    def add_bulk(self, names) -> int:
        """Bulk insert (the GPU walk path adds ~160k names per hop —
        one lock + set ops instead of 160k add() calls)."""
        with self._lock:
            missing = [n for n in names if n not in self._items]
            for n in missing:
                self._items[n] = True
            self._keys.extend(missing)
            return len(missing)

    def contains(self, name: str) -> bool:
        with self._lock:
            return name in self._items

    def random(self, rng: Optional[random.Random] = None) -> Optional[str]:
        with self._lock:
            if not self._keys:
                return None
            r = rng or random
            return self._keys[r.randrange(len(self._keys))]

    def __len__(self):
        with self._lock:
            return len(self._keys)

    def keys(self) -> List[str]:
        with self._lock:
            return list(self._keys)


class BaseStateManager:
    """In-memory layer map / page map core (state/base.go:15-552)."""

    def __init__(self, config):
        self.config = config
        self.pages: Dict[str, Page] = {}
        self.layer_map: Dict[int, List[str]] = {}
        self.metadata = CrawlMetadata(
            crawl_id=getattr(config, "crawl_id", ""),
            execution_id=getattr(config, "crawl_id", ""),
            start_time=_now(),
            platform=getattr(config, "platform", "telegram"),
        )
        self.discovered = DiscoveredChannels()
        self._lock = threading.RLock()
        self._chat_id_cache: Dict[str, int] = {}
        self._seed_channels: Dict[str, bool] = {}

    # ---- layer management ----

    def initialize(self, seed_urls: List[str]) -> None:
        pages = [
            Page(id=str(uuid.uuid4()), url=u, depth=0, status="unfetched",
                 timestamp=_now())
            for u in seed_urls
        ]
        self.add_layer(pages)
        self.metadata.target_channels = list(seed_urls)

    def add_layer(self, pages: List[Page]) -> List[str]:
        """URL-dedup + MaxPages deadend-replacement budget
        (state/base.go:219-322). Returns added page IDs."""
        if not pages:
            return []
        with self._lock:
            total = len(self.pages)
            deadends = sum(1 for p in self.pages.values()
                           if p.status == "deadend")
            max_pages = getattr(self.config, "max_pages", 0) or 0
            replacements = deadends
            existing_urls = {p.url: pid for pid, p in self.pages.items()}
            url_dedup = getattr(self, "url_dedup", {})
            depth = pages[0].depth
            self.layer_map.setdefault(depth, [])
            added = []
            for p in pages:
                if p.url in existing_urls:
                    continue
                if p.url in url_dedup and depth > 0:
                    # crawled by a previous crawl (cross-crawl URL cache);
                    # seeds (depth 0) are always admitted
                    continue
                # MaxPages budget. The reference checks the cap only at
                # layer START (base.go:240-244), so one layer can overshoot
                # it arbitrarily; we enforce it continuously — the stronger
                # guarantee, same replacement semantics at the cap.
                if max_pages > 0 and total + len(added) >= max_pages:
                    if replacements <= 0:
                        continue
                    replacements -= 1
                if not p.id:
                    p.id = str(uuid.uuid4())
                if p.timestamp is None:
                    p.timestamp = _now()
                self.pages[p.id] = p
                existing_urls[p.url] = p.id
                self.layer_map[depth].append(p.id)
                added.append(p.id)
            return added

    def get_page(self, page_id: str) -> Page:
        with self._lock:
            if page_id not in self.pages:
                raise KeyError(f"page not found: {page_id}")
            return self.pages[page_id]

    def update_page(self, page: Page) -> None:
        with self._lock:
            self.pages[page.id] = page

    def update_message(self, page_id: str, chat_id: int, message_id: int,
                       status: str) -> None:
        with self._lock:
            page = self.pages.get(page_id)
            if page is None:
                raise KeyError(f"page not found: {page_id}")
            for m in page.messages:
                if m.chat_id == chat_id and m.message_id == message_id:
                    m.status = status
                    return
            page.messages.append(PageMessage(
                chat_id=chat_id, message_id=message_id, status=status,
                page_id=page_id,
            ))

    def get_layer_by_depth(self, depth: int) -> List[Page]:
        with self._lock:
            return [self.pages[i] for i in self.layer_map.get(depth, [])]

    def get_max_depth(self) -> int:
        with self._lock:
            return max(self.layer_map.keys(), default=-1)

    # ---- discovered channels / seed cache ----

    def is_discovered_channel(self, name: str) -> bool:
        return self.discovered.contains(name)

    def add_discovered_channel(self, name: str) -> None:
        self.discovered.add(name)

    def add_discovered_channels_bulk(self, names) -> int:
        return self.discovered.add_bulk(names)

    def get_random_discovered_channel(self, rng=None) -> Optional[str]:
        return self.discovered.random(rng)

    def upsert_seed_channel_chat_id(self, username: str, chat_id: int):
        with self._lock:
            self._chat_id_cache[username] = chat_id
            self._seed_channels[username] = True

    def get_cached_chat_id(self, username: str) -> Tuple[int, bool]:
        with self._lock:
            v = self._chat_id_cache.get(username)
            return (v, True) if v is not None else (0, False)

    def is_seed_channel(self, username: str) -> bool:
        with self._lock:
            return username in self._seed_channels


class LocalStateManager(BaseStateManager):
    """File-backed state manager (state/storageproviders.go:74-647).

    Layout under base_path/crawl_id/:
      state.json, metadata.json, media-cache.json, progress.json (legacy
      format documented in README "Directory Structure"),
      <channel>/posts/posts.jsonl, media/<channel>/<file>
    """

    def __init__(self, config, base_path: Optional[str] = None):
        super().__init__(config)
        self.base_path = base_path or getattr(
            config, "storage_root", "/tmp/crawl"
        )
        self.media_cache: Dict[str, dict] = {}
        # LRU-capped append-handle cache: crawls touch thousands of
        # channels; keeping every posts.jsonl open exhausts the fd limit
        self._post_files: "collections.OrderedDict" = (
            __import__("collections").OrderedDict()
        )
        self.max_open_post_files = 256
        self.url_dedup: Dict[str, str] = {}

    def load_url_dedup_cache(self) -> int:
        """Cross-crawl URL dedup: url -> "crawlID:pageID" loaded from all
        previous crawls' state files (daprstate.go:550-623). Skipped
        entirely for random-walk (duplicate visits are allowed there,
        daprstate.go:642-657 — callers must not invoke this in that mode).
        """
        cache: Dict[str, str] = {}
        for prev in self.get_previous_crawls():
            path = os.path.join(self.base_path, prev, "state.json")
            if not os.path.exists(path):
                continue
            try:
                with open(path) as f:
                    state = json.load(f)
            except (OSError, json.JSONDecodeError):
                continue
            for layer in state.get("layers", []):
                for pd in layer.get("pages", []):
                    url = pd.get("url")
                    if url and url not in cache:
                        cache[url] = f"{prev}:{pd.get('id', '')}"
        self.url_dedup = cache
        return len(cache)

    # paths (storageproviders.go:634-647)
    def _crawl_dir(self):
        return os.path.join(self.base_path, self.metadata.crawl_id)

    def _state_path(self):
        return os.path.join(self._crawl_dir(), "state.json")

    def _metadata_path(self):
        return os.path.join(self._crawl_dir(), "metadata.json")

    def _media_cache_path(self):
        return os.path.join(self._crawl_dir(), "media-cache.json")

    def _progress_path(self):
        return os.path.join(self._crawl_dir(), "progress.json")

    def _posts_path(self, channel: str):
        return os.path.join(self._crawl_dir(), channel, "posts",
                            "posts.jsonl")

    # ---- persistence ----

    def save_state(self) -> None:
        os.makedirs(self._crawl_dir(), exist_ok=True)
        # Durability contract (reference writes each post synchronously
        # BEFORE status updates, daprstate.go:1106-1147): a checkpoint
        # must not mark pages 'fetched' while their JSONL posts are still
        # sitting in buffered handles or in the native sink's queue — a
        # crash after the checkpoint would lose posts resume never
        # re-emits. Flush everything first.
        for f in self._post_files.values():
            f.flush()
        self.drain_post_writes()
        with self._lock:
            state = {
                "layers": [
                    {
                        "depth": depth,
                        "pages": [self.pages[i].to_dict()
                                  for i in ids],
                    }
                    for depth, ids in sorted(self.layer_map.items())
                ],
                "metadata": self.metadata.to_dict(),
                "lastUpdated": _now().isoformat(),
            }
        tmp = self._state_path() + ".tmp"
        with open(tmp, "w") as f:
            json.dump(state, f)
        os.replace(tmp, self._state_path())
        with open(self._metadata_path() + ".tmp", "w") as f:
            json.dump(self.metadata.to_dict(), f)
        os.replace(self._metadata_path() + ".tmp", self._metadata_path())
        self._save_progress()

    def _save_progress(self) -> None:
        """Legacy progress.json checkpoint (README "Directory Structure",
        docs/architecture.md:166-177): per-depth completion cursor."""
        with self._lock:
            layers = []
            for depth, ids in sorted(self.layer_map.items()):
                done = sum(
                    1 for i in ids
                    if self.pages[i].status in
                    ("fetched", "error", "deadend")
                )
                layers.append({
                    "depth": depth, "total": len(ids), "completed": done,
                })
            progress = {
                "crawlId": self.metadata.crawl_id,
                "executionId": self.metadata.execution_id,
                "status": self.metadata.status,
                "layers": layers,
                "updatedAt": _now().isoformat(),
            }
        with open(self._progress_path() + ".tmp", "w") as f:
            json.dump(progress, f)
        os.replace(self._progress_path() + ".tmp", self._progress_path())

    def load_state(self) -> bool:
        if not os.path.exists(self._state_path()):
            return False
        with open(self._state_path()) as f:
            state = json.load(f)
        with self._lock:
            self.pages.clear()
            self.layer_map.clear()
            for layer in state.get("layers", []):
                ids = []
                for pd in layer.get("pages", []):
                    page = Page.from_dict(pd)
                    self.pages[page.id] = page
                    ids.append(page.id)
                self.layer_map[layer.get("depth", 0)] = ids
            self.metadata = CrawlMetadata.from_dict(
                state.get("metadata", {})
            )
        if os.path.exists(self._media_cache_path()):
            try:
                with open(self._media_cache_path()) as f:
                    self.media_cache = json.load(f)
            except (OSError, json.JSONDecodeError):
                self.media_cache = {}
        return True

    def find_incomplete_crawl(self, crawl_id: str) -> Tuple[str, bool]:
        """Resume rules (state/base.go:466-516): a crawl is resumable iff
        its state exists, has a non-empty layer map, and is not completed.
        Returns (execution_id, exists)."""
        path = os.path.join(self.base_path, crawl_id, "state.json")
        if not os.path.exists(path):
            return "", False
        try:
            with open(path) as f:
                state = json.load(f)
        except (OSError, json.JSONDecodeError):
            return "", False
        meta = state.get("metadata", {})
        if meta.get("status") == "completed":
            return "", False
        if not state.get("layers"):
            return "", False
        return meta.get("executionId") or crawl_id, True

    def get_previous_crawls(self) -> List[str]:
        if not os.path.isdir(self.base_path):
            return []
        return sorted(
            d for d in os.listdir(self.base_path)
            if os.path.isdir(os.path.join(self.base_path, d))
            and d != self.metadata.crawl_id
        )

    def update_crawl_metadata(self, crawl_id: str, md: dict) -> None:
        with self._lock:
            if "status" in md:
                self.metadata.status = md["status"]
            if "endTime" in md:
                v = md["endTime"]
                self.metadata.end_time = (
                    v if isinstance(v, _dt.datetime)
                    else _dt.datetime.fromisoformat(v)
                )
            if "messagesCount" in md:
                self.metadata.messages_count = md["messagesCount"]
            if "errorsCount" in md:
                self.metadata.errors_count = md["errorsCount"]

    # ---- posts / files ----

    # CombineFiles mode: when a chunker sink is attached, post JSONL goes
    # through the temp->watch protocol instead of direct per-channel
    # appends (reference daprstate.go:1106-1248: CombineFiles writes temp
    # files for the chunker rather than the storage binding).
    post_sink = None  # Chunker or None
    _sink_seq = 0

    def attach_chunker(self, chunker) -> None:
        self.post_sink = chunker

    def _sink_write(self, channel: str, data) -> None:
        self._sink_seq += 1
        name = (f"{self.metadata.crawl_id}-{channel}-"
                f"{self._sink_seq:08d}.jsonl")
        self.post_sink.write_temp_then_watch(name, bytes(data))

    def _post_file(self, channel: str):
        path = self._posts_path(channel)
        f = self._post_files.get(path)
        if f is not None:
            self._post_files.move_to_end(path)
            return f
        os.makedirs(os.path.dirname(path), exist_ok=True)
        f = open(path, "ab")
        self._post_files[path] = f
        while len(self._post_files) > self.max_open_post_files:
            _old_path, old_f = self._post_files.popitem(last=False)
            old_f.close()
        return f

    def store_post(self, channel: str, post: Post) -> None:
        data = post.to_jsonl().encode("utf-8")
        if self.post_sink is not None:
            self._sink_write(channel, data)
            return
        self._post_file(channel).write(data)

    def store_post_lines(self, channel: str, data) -> None:
        """Bulk JSONL append (the GPU path hands whole encoded blocks)."""
        if self.post_sink is not None:
            self._sink_write(channel, data)
            return
        self._post_file(channel).write(data)

    _native_sink = None
    _native_tried = False

    def _get_native_sink(self):
        """crawler_amd.native.fanout_native thread-pool writer (the
        runtime-native spill path); None when the extension is absent or
        disabled (CRAWL_NO_NATIVE_SINK=1)."""
        if self._native_tried:
            return self._native_sink
        self._native_tried = True
        if os.environ.get("CRAWL_NO_NATIVE_SINK", "") == "1":
            return None
        try:
            from ..native import load

            self._native_sink = load().FanoutSink(
                8, self.max_open_post_files
            )
        except ImportError:
            self._native_sink = None
        return self._native_sink

    def store_post_lines_batch(self, items, buffer, nowait: bool = False,
                               ticket: bool = False):
        """Fan out one encoded host buffer to many channels' JSONL files:
        items = [(channel, lo, hi)]. Uses the native thread-pool sink
        when built (parallel write(2), GIL released); falls back to the
        per-channel Python path otherwise. Combine-files mode keeps the
        chunker protocol. nowait=True (native only) returns before the
        writes land — hold `buffer` alive until drain_post_writes()."""
        if self.post_sink is not None:
            mv = memoryview(buffer)
            for (channel, lo, hi) in items:
                if hi > lo:
                    self._sink_write(channel, mv[lo:hi])
            return
        sink = self._get_native_sink()
        if sink is None:
            mv = memoryview(buffer)
            for (channel, lo, hi) in items:
                if hi > lo:
                    self.store_post_lines(channel, mv[lo:hi])
            return
        paths, los, his = [], [], []
        for (channel, lo, hi) in items:
            if hi <= lo:
                continue
            path = self._posts_path(channel)
            # never interleave a buffered Python handle with the O_APPEND
            # fd: flush+drop the Python handle first
            f = self._post_files.pop(path, None)
            if f is not None:
                f.close()
            paths.append(path)
            los.append(lo)
            his.append(hi)
        if not paths:
            return None
        if ticket:
            # caller MUST keep `buffer` alive until wait_post_write()
            return sink.write_batch_ticket(paths, buffer, los, his)
        if nowait:
            # caller MUST keep `buffer` alive until drain_post_writes()
            sink.write_batch_nowait(paths, buffer, los, his)
        else:
            sink.write_batch(paths, buffer, los, his)
        return None

    def wait_post_write(self, ticket) -> None:
        """Barrier for ONE store_post_lines_batch(..., ticket=True) batch
        (None tickets — sync fallback paths — are a no-op)."""
        if ticket is not None and self._native_sink is not None:
            self._native_sink.wait_ticket(ticket)

    def drain_post_writes(self) -> None:
        """Barrier for store_post_lines_batch(..., nowait=True) writes."""
        if self._native_sink is not None:
            self._native_sink.drain()

    def truncate_posts(self, channel: str) -> None:
        """Idempotent re-crawl support for the channel-atomic GPU path: a
        crash between a channel's JSONL write and the layer's save_state
        leaves the page 'unfetched', so resume re-processes it — truncate
        before the re-write so posts are exactly-once at channel
        granularity (the CPU path gets the same via per-message
        update_message, crawl/runner.go:1572-1633)."""
        if self.post_sink is not None:
            return  # combine-files mode writes unique temp names
        path = self._posts_path(channel)
        f = self._post_files.pop(path, None)
        if f is not None:
            f.close()
        if os.path.exists(path):
            open(path, "wb").close()

    def store_file(self, channel: str, source_path: str,
                   file_name: str) -> Tuple[str, str]:
        dst_dir = os.path.join(self._crawl_dir(), "media", channel)
        os.makedirs(dst_dir, exist_ok=True)
        dst = os.path.join(dst_dir, file_name)
        os.replace(source_path, dst)
        return dst, file_name

    # ---- page export (daprstate.go:2906-3012) ----

    export_chunk_size_bytes = 100 * 1024 * 1024  # daprstate.go:2909

    def export_pages_to_binding(self, crawl_id: str) -> List[str]:
        """Chunked JSONL export of all pages to
        analysis/channels/channel-pages-<crawlID>-partN.jsonl
        (100 MB raw per chunk; daprstate.go:2906-3012)."""
        out_dir = os.path.join(self.base_path, "analysis", "channels")
        os.makedirs(out_dir, exist_ok=True)
        paths: List[str] = []
        part = 0
        buf: List[bytes] = []
        size = 0

        def flush():
            nonlocal part, buf, size
            if not buf:
                return
            path = os.path.join(
                out_dir, f"channel-pages-{crawl_id}-part{part}.jsonl"
            )
            with open(path, "wb") as f:
                f.writelines(buf)
            paths.append(path)
            part += 1
            buf = []
            size = 0

        with self._lock:
            pages = [self.pages[i] for ids in self.layer_map.values()
                     for i in ids]
        for p in pages:
            line = (json.dumps(p.to_dict()) + "\n").encode()
            if size + len(line) > self.export_chunk_size_bytes and buf:
                flush()
            buf.append(line)
            size += len(line)
        flush()
        return paths

    # ---- media cache ----

    def has_processed_media(self, media_id: str) -> bool:
        return media_id in self.media_cache

    def mark_media_as_processed(self, media_id: str) -> None:
        self.media_cache[media_id] = {
            "id": media_id, "firstSeen": _now().isoformat(),
        }

    MEDIA_CACHE_TTL_DAYS = 30  # daprstate.go:1252-1678 shard expiry

    def _expire_media_cache(self) -> int:
        """Drop cache entries older than the TTL (the reference's
        sharded-cache cleanup that runs on Close)."""
        cutoff = _now() - _dt.timedelta(days=self.MEDIA_CACHE_TTL_DAYS)
        dropped = 0
        for k in list(self.media_cache):
            v = self.media_cache[k]
            try:
                seen = _dt.datetime.fromisoformat(v["firstSeen"])
            except (KeyError, TypeError, ValueError):
                continue  # legacy entry without a timestamp: keep
            if seen.tzinfo is None:
                seen = seen.replace(tzinfo=UTC)
            if seen < cutoff:
                del self.media_cache[k]
                dropped += 1
        return dropped

    def close(self) -> None:
        for f in self._post_files.values():
            f.close()
        self._post_files.clear()
        self._expire_media_cache()
        if self._native_sink is not None:
            self._native_sink.close()
            self._native_sink = None
            self._native_tried = False
        os.makedirs(self._crawl_dir(), exist_ok=True)
        with open(self._media_cache_path() + ".tmp", "w") as f:
            json.dump(self.media_cache, f)
        os.replace(self._media_cache_path() + ".tmp",
                   self._media_cache_path())


class RandomWalkStore:
    """In-process re-host of the random-walk / validator Postgres tables.

    Claim operations reproduce `FOR UPDATE SKIP LOCKED` + attempt_count
    semantics (state/daprstate.go:3944-4391) with a lock + status fields:
    a claim transitions rows atomically so concurrent claimers never see
    the same row. Poison detection: attempt_count >= 3 parks the batch.
    """

    MAX_ATTEMPTS = 3

    def __init__(self):
        self._lock = threading.RLock()
        self.page_buffer: Dict[str, Page] = {}       # page_id -> Page
        self._edge_list: List[EdgeRecord] = []
        # skipped-edge BLOCKS (source, sequence_id, time, names): the
        # GPU walk appends ~160k skipped edges per hop — one tuple per
        # walker here, expanded to EdgeRecords lazily on first read
        # ('append to device edge log; host spill')
        self._edge_blocks: List[tuple] = []
        self.seed_channels: Dict[str, dict] = {}     # username -> row
        self.invalid_channels: Dict[str, _dt.datetime] = {}
        self.pending_batches: Dict[str, PendingEdgeBatch] = {}
        self.pending_edges: Dict[int, PendingEdge] = {}
        self.discovered_channels: Dict[str, dict] = {}
        self.access_events: List[dict] = []
        self.source_type_stats: Dict[Tuple[str, str], int] = {}
        self._next_edge_id = 1

    # ---- page_buffer (sql/random-walk-schema.sql page_buffer) ----

    def add_page(self, page: Page) -> None:
        with self._lock:
            if not page.id:
                page.id = str(uuid.uuid4())
            self.page_buffer[page.id] = page

    def get_pages(self, limit: int) -> List[Page]:
        with self._lock:
            return list(self.page_buffer.values())[:limit]

    def delete_pages(self, page_ids: List[str]) -> None:
        with self._lock:
            for pid in page_ids:
                self.page_buffer.pop(pid, None)

    def buffer_size(self) -> int:
        with self._lock:
            return len(self.page_buffer)

    # ---- edge_records ----

    @property
    def edge_records(self) -> List[EdgeRecord]:
        with self._lock:
            self._materialize_edges()
            return self._edge_list

    @edge_records.setter
    def edge_records(self, value: List[EdgeRecord]) -> None:
        with self._lock:
            self._edge_blocks = []
            self._edge_list = value

    def _materialize_edges(self) -> None:
        if not self._edge_blocks:
            return
        for (src, seq, t, names) in self._edge_blocks:
            # names may be str lists (CPU path) or S<w> byte rows (the
            # GPU walk stores the device slice verbatim; decode here,
            # on first read — most runs only ever ask edge_count())
            self._edge_list.extend(
                EdgeRecord(n.decode() if isinstance(n, bytes) else n,
                           t, src, False, True, seq, "")
                for n in names)
        self._edge_blocks = []

    def save_skipped_edges_block(self, source: str, sequence_id: str,
                                 names, now) -> None:
        """O(1) append of one walker's skipped-edge set (str list or
        S<w> bytes array); expanded to EdgeRecord rows on first
        edge_records read."""
        with self._lock:
            if len(names):
                self._edge_blocks.append((source, sequence_id, now,
                                          names))

    def edge_count(self) -> int:
        """len(edge_records) without forcing materialization."""
        with self._lock:
            return (len(self._edge_list)
                    + sum(len(b[3]) for b in self._edge_blocks))

    def save_edge_records(self, edges: List[EdgeRecord]) -> None:
        with self._lock:
            for e in edges:
                if e.discovery_time is None:
                    e.discovery_time = _now()
                self._edge_list.append(e)

    def save_edge_records_fast(self, edges: List[EdgeRecord]) -> None:
        """Bulk append; caller guarantees discovery_time is set."""
        with self._lock:
            self._edge_list.extend(edges)

    def get_random_skipped_edge(self, exclude: set, rng=None,
                                sequence_id: Optional[str] = None,
                                source_channel: Optional[str] = None
                                ) -> Optional[EdgeRecord]:
        """Random skipped edge, optionally filtered to the same sequence +
        source (daprstate.go:3206-3277 semantics)."""
        r = rng or random
        with self._lock:
            candidates = [
                e for e in self.edge_records
                if e.skipped and e.destination_channel not in exclude
                and (sequence_id is None or e.sequence_id == sequence_id)
                and (source_channel is None
                     or e.source_channel == source_channel)
            ]
            return r.choice(candidates) if candidates else None

    def get_edge_record(self, sequence_id: str,
                        destination: str) -> Optional[EdgeRecord]:
        """Non-skipped edge arriving at destination in this chain
        (daprstate.go GetEdgeRecord semantics)."""
        with self._lock:
            for e in self.edge_records:
                if (e.sequence_id == sequence_id
                        and e.destination_channel == destination
                        and not e.skipped):
                    return e
            return None

    def delete_edge_record(self, sequence_id: str, destination: str) -> None:
        with self._lock:
            self.edge_records = [
                e for e in self.edge_records
                if not (e.sequence_id == sequence_id
                        and e.destination_channel == destination
                        and not e.skipped)
            ]

    def promote_edge(self, sequence_id: str, destination: str) -> None:
        """Flip a skipped edge to followed (daprstate.go PromoteEdge)."""
        with self._lock:
            for e in self.edge_records:
                if (e.sequence_id == sequence_id
                        and e.destination_channel == destination
                        and e.skipped):
                    e.skipped = False
                    return

    # ---- seed_channels ----

    # Fresh seed rows are stored as None and materialized on first
    # field access: bulk admission inserts ~100k names per random-walk
    # hop, and building the 4-field row dict eagerly for every name
    # cost 3.3x the dict insert itself (measured). All reads go
    # through _seed_row / the accessors below, so the sentinel never
    # escapes this class.
    def _seed_row(self, username: str) -> dict:
        row = self.seed_channels.get(username)
        if row is None:
            row = {"username": username, "chat_id": 0,
                   "last_crawled_at": None, "invalidated_at": None}
            self.seed_channels[username] = row
        return row

    def upsert_seed_channel(self, username: str, chat_id: int = 0) -> None:
        with self._lock:
            if chat_id:
                self._seed_row(username)["chat_id"] = chat_id
            elif username not in self.seed_channels:
                self.seed_channels[username] = None

    def upsert_seed_channels_bulk(self, usernames) -> None:
        """Bulk upsert without chat ids (the ~500k-row seed_channels
        table of sql/random-walk-schema.sql fills at hop granularity)."""
        with self._lock:
            sc = self.seed_channels
            for u in usernames:
                if u not in sc:
                    sc[u] = None

    def mark_channel_crawled(self, username: str, chat_id: int) -> None:
        with self._lock:
            row = self._seed_row(username)
            if chat_id:
                row["chat_id"] = chat_id
            row["last_crawled_at"] = _now()

    def get_channel_last_crawled(self, username: str) -> Optional[_dt.datetime]:
        with self._lock:
            row = self.seed_channels.get(username)
            return row["last_crawled_at"] if row else None

    def mark_seed_channel_invalid(self, username: str) -> None:
        with self._lock:
            if username in self.seed_channels:
                self._seed_row(username)["invalidated_at"] = _now()

    def get_random_seed_channel(self, rng=None,
                                ttl_days: int = 30) -> Optional[str]:
        """Random non-invalidated (or TTL-expired) seed username
        (daprstate.go:4181-4196: invalidated_at IS NULL OR older than
        30 days, ORDER BY RANDOM() LIMIT 1). Returns None when the
        table has no eligible rows."""
        r = rng or random
        cutoff = _now() - _dt.timedelta(days=ttl_days)
        with self._lock:
            elig = [u for u, row in self.seed_channels.items()
                    if row is None or row["invalidated_at"] is None
                    or row["invalidated_at"] < cutoff]
            return r.choice(elig) if elig else None

    def load_seed_channels(self, ttl_days: int = 30) -> List[dict]:
        """Rows whose invalidation is absent or older than the TTL
        (daprstate.go:3327-3429)."""
        cutoff = _now() - _dt.timedelta(days=ttl_days)
        with self._lock:
            out = []
            for u, row in self.seed_channels.items():
                if row is None:
                    out.append({"username": u, "chat_id": 0,
                                "last_crawled_at": None,
                                "invalidated_at": None})
                elif (row["invalidated_at"] is None
                      or row["invalidated_at"] < cutoff):
                    out.append(dict(row))
            return out

    # ---- invalid_channels (30-day TTL cache) ----

    def mark_invalid_channel(self, username: str) -> None:
        with self._lock:
            self.invalid_channels[username] = _now()
            # version stamp for read-side caches (the GPU walk keeps a
            # sorted snapshot keyed on this; direct dict edits in tests
            # don't go through here and don't need the snapshot)
            self.invalid_version = getattr(self, "invalid_version", 0) + 1

    def is_invalid_channel(self, username: str, ttl_days: int = 30) -> bool:
        with self._lock:
            t = self.invalid_channels.get(username)
            if t is None:
                return False
            return t > _now() - _dt.timedelta(days=ttl_days)

    # ---- pending edge batches / edges (tandem validator) ----

    def open_batch(self, crawl_id: str, source_channel: str,
                   source_page_id: str, source_depth: int,
                   sequence_id: str) -> str:
        with self._lock:
            bid = str(uuid.uuid4())
            self.pending_batches[bid] = PendingEdgeBatch(
                batch_id=bid, crawl_id=crawl_id,
                source_channel=source_channel,
                source_page_id=source_page_id, source_depth=source_depth,
                sequence_id=sequence_id, status="open",
            )
            return bid

    def insert_pending_edge(self, batch_id: str, crawl_id: str,
                            destination: str, source: str,
                            sequence_id: str, source_type: str) -> int:
        with self._lock:
            eid = self._next_edge_id
            self._next_edge_id += 1
            self.pending_edges[eid] = PendingEdge(
                pending_id=eid, batch_id=batch_id, crawl_id=crawl_id,
                destination_channel=destination, source_channel=source,
                sequence_id=sequence_id, discovery_time=_now(),
                source_type=source_type, validation_status="pending",
            )
            return eid

    def close_batch(self, batch_id: str) -> None:
        with self._lock:
            b = self.pending_batches.get(batch_id)
            if b and b.status == "open":
                b.status = "closed"

    def claim_pending_edges(self, limit: int) -> List[PendingEdge]:
        """SKIP LOCKED claim (daprstate.go:3944-3996)."""
        with self._lock:
            claimed = []
            for e in self.pending_edges.values():
                if e.validation_status == "pending":
                    e.validation_status = "validating"
                    claimed.append(dataclasses.replace(e))
                    if len(claimed) >= limit:
                        break
            return claimed

    def update_pending_edges(self, updates: List[Tuple[int, str, str]]):
        with self._lock:
            for (eid, status, reason) in updates:
                e = self.pending_edges.get(eid)
                if e:
                    e.validation_status = status
                    e.validation_reason = reason

    def claim_walkback_batch(self) -> Optional[PendingEdgeBatch]:
        """Closed batch with no pending/validating edges
        (daprstate.go:4016-4033)."""
        with self._lock:
            for b in self.pending_batches.values():
                if b.status != "closed":
                    continue
                if b.attempt_count >= self.MAX_ATTEMPTS:
                    continue
                busy = any(
                    e.batch_id == b.batch_id
                    and e.validation_status in ("pending", "validating")
                    for e in self.pending_edges.values()
                )
                if busy:
                    continue
                b.status = "processing"
                b.attempt_count += 1
                return dataclasses.replace(b)
            return None

    def complete_batch(self, batch_id: str) -> None:
        with self._lock:
            b = self.pending_batches.get(batch_id)
            if b:
                b.status = "completed"

    def count_incomplete_batches(self, crawl_id: str) -> int:
        with self._lock:
            return sum(
                1 for b in self.pending_batches.values()
                if b.crawl_id == crawl_id
                and b.status in ("open", "closed", "processing")
            )

    def edges_of_batch(self, batch_id: str) -> List[PendingEdge]:
        with self._lock:
            return [dataclasses.replace(e)
                    for e in self.pending_edges.values()
                    if e.batch_id == batch_id]

    # ---- stale / orphan recovery (daprstate.go:4264-4391) ----

    def recover_stale_claims(self) -> Tuple[int, int]:
        """Reset 'validating' edges to 'pending' and 'processing' batches to
        'closed' (attempt_count retained for poison detection)."""
        with self._lock:
            ne = nb = 0
            for e in self.pending_edges.values():
                if e.validation_status == "validating":
                    e.validation_status = "pending"
                    ne += 1
            for b in self.pending_batches.values():
                if b.status == "processing":
                    b.status = "closed"
                    nb += 1
            return ne, nb

    def delete_orphan_edges(self) -> int:
        with self._lock:
            valid_batches = set(self.pending_batches)
            orphans = [eid for eid, e in self.pending_edges.items()
                       if e.batch_id not in valid_batches]
            for eid in orphans:
                del self.pending_edges[eid]
            return len(orphans)

    # ---- discovered_channels (validator exactly-once claim CTE) ----

    def claim_discovered_channel(self, username: str, crawl_id: str) -> bool:
        """INSERT ... ON CONFLICT DO NOTHING claim (daprstate.go:4198-4225):
        True iff this call discovered the channel first."""
        with self._lock:
            if username in self.discovered_channels:
                return False
            self.discovered_channels[username] = {
                "username": username, "crawl_id": crawl_id,
                "discovered_at": _now(),
            }
            return True

    # ---- stats / access events ----

    def flush_batch_stats(self, crawl_id: str, stats: Dict[str, int]):
        with self._lock:
            for stype, cnt in stats.items():
                key = (crawl_id, stype)
                self.source_type_stats[key] = (
                    self.source_type_stats.get(key, 0) + cnt
                )

    def insert_access_event(self, event_type: str, detail: str = ""):
        with self._lock:
            self.access_events.append({
                "type": event_type, "detail": detail, "at": _now(),
            })


class StateManagerFactory:
    """state/statefactory.go:20-52 — local is the only backend here; the
    Dapr/Redis/Postgres stack is replaced by LocalStateManager +
    RandomWalkStore + the GPU-resident seen-set."""

    @staticmethod
    def create(config, base_path: Optional[str] = None) -> LocalStateManager:
        return LocalStateManager(config, base_path=base_path)
