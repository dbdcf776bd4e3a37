"""Scheduled-job mode: job payload <-> config merge and a local scheduler.

Parity (reference dapr/job.go):
- JobData payload shape and mergeConfigWithJobData semantics (:305-362):
  job fields override the CLI base config only when non-zero/non-empty;
- the Dapr Jobs API (schedule/get/delete/handle, :48-200) becomes a
  local interval scheduler: no sidecar, same lifecycle (schedule a named
  job, trigger runs the merged crawl, delete cancels).
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import json
import threading
from typing import Callable, Dict, List, Optional

from ..config import CrawlerConfig


@dataclasses.dataclass
class JobData:
    """dapr/job.go JobData payload."""

    urls: List[str] = dataclasses.field(default_factory=list)
    max_depth: int = 0
    concurrency: int = 0
    crawl_id: str = ""
    platform: str = ""
    youtube_api_key: str = ""
    sampling_method: str = ""
    min_channel_videos: int = 0
    max_posts: int = 0
    sample_size: int = 0
    min_post_date: Optional[_dt.datetime] = None
    date_between_min: Optional[_dt.datetime] = None
    date_between_max: Optional[_dt.datetime] = None
    tdlib_database_urls: List[str] = dataclasses.field(default_factory=list)
    max_pages: int = 0

    @classmethod
    def from_json(cls, payload: str) -> "JobData":
        d = json.loads(payload)

        def ts(k):
            v = d.get(k)
            return _dt.datetime.fromisoformat(v) if v else None

        return cls(
            urls=d.get("urls") or [],
            max_depth=d.get("max_depth", 0),
            concurrency=d.get("concurrency", 0),
            crawl_id=d.get("crawl_id", ""),
            platform=d.get("platform", ""),
            youtube_api_key=d.get("youtube_api_key", ""),
            sampling_method=d.get("sampling_method", ""),
            min_channel_videos=d.get("min_channel_videos", 0),
            max_posts=d.get("max_posts", 0),
            sample_size=d.get("sample_size", 0),
            min_post_date=ts("min_post_date"),
            date_between_min=ts("date_between_min"),
            date_between_max=ts("date_between_max"),
            tdlib_database_urls=d.get("tdlib_database_urls") or [],
            max_pages=d.get("max_pages", 0),
        )


def merge_config_with_job_data(base: CrawlerConfig,
                               job: JobData) -> CrawlerConfig:
    """mergeConfigWithJobData (dapr/job.go:305-362): non-zero job fields
    override the CLI base."""
    cfg = dataclasses.replace(base)
    if job.max_depth != 0:
        cfg.max_depth = job.max_depth
    if job.concurrency != 0:
        cfg.concurrency = job.concurrency
    if job.crawl_id:
        cfg.crawl_id = job.crawl_id
    if job.platform:
        cfg.platform = job.platform
    if job.youtube_api_key:
        cfg.youtube_api_key = job.youtube_api_key
    if job.sampling_method:
        cfg.sampling_method = job.sampling_method
    if job.min_channel_videos != 0:
        cfg.min_channel_videos = job.min_channel_videos
    if job.max_posts != 0:
        cfg.max_posts = job.max_posts
    if job.sample_size != 0:
        cfg.sample_size = job.sample_size
    if job.min_post_date is not None:
        cfg.min_post_date = job.min_post_date
    if job.date_between_min is not None:
        cfg.date_between_min = job.date_between_min
    if job.date_between_max is not None:
        cfg.date_between_max = job.date_between_max
    if job.tdlib_database_urls:
        cfg.tdlib_database_urls = list(job.tdlib_database_urls)
    if job.max_pages != 0:
        cfg.max_pages = job.max_pages
    return cfg


class JobScheduler:
    """Local stand-in for the Dapr Jobs API (schedule/get/delete/trigger)."""

    def __init__(self):
        self._jobs: Dict[str, dict] = {}
        self._lock = threading.Lock()

    def schedule(self, name: str, payload: str, interval_s: float,
                 handler: Callable[[JobData], None]) -> None:
        with self._lock:
            self.delete(name)
            stop = threading.Event()

            def loop():
                while not stop.wait(interval_s):
                    handler(JobData.from_json(payload))

            t = threading.Thread(target=loop, daemon=True)
            self._jobs[name] = {
                "payload": payload, "interval_s": interval_s,
                "thread": t, "stop": stop,
            }
            t.start()

    def get(self, name: str) -> Optional[dict]:
        with self._lock:
            j = self._jobs.get(name)
            if j is None:
                return None
            return {"name": name, "payload": j["payload"],
                    "interval_s": j["interval_s"]}

    def delete(self, name: str) -> bool:
        j = self._jobs.pop(name, None)
        if j is None:
            return False
        j["stop"].set()
        return True

    def trigger(self, name: str, handler: Callable[[JobData], None]) -> bool:
        with self._lock:
            j = self._jobs.get(name)
        if j is None:
            return False
        handler(JobData.from_json(j["payload"]))
        return True
