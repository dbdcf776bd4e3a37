"""Crawler configuration: CrawlerConfig, rate limits, time parsing, sampling validation.

Behavioral parity targets (reference, cited file:line):
- ``CrawlerConfig`` fields: common/utils.go:49-99
- ``TelegramRateLimitConfig`` defaults (30/6/20/60 calls/min): common/utils.go:19-46
- ``generate_crawl_id`` "YYYYMMDDHHMMSS": common/utils.go:103-111
- ``parse_time_ago`` h/d/w/m/y parser: main.go:91-142
- sampling validity matrix: common/sampling_validation.go:19-66
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import os
import re
from typing import List, Optional


@dataclasses.dataclass
class TelegramRateLimitConfig:
    """Per-connection token-bucket limits for the (synthetic) Telegram API.

    Matches common/utils.go:19-46. GetMessage is reactive: a token is only
    consumed on a TDLib-cache miss (rate_limiter.go:145-169).
    """

    get_chat_history_rate: float = 30.0      # calls/min
    search_public_chat_rate: float = 6.0
    get_supergroup_info_rate: float = 20.0
    get_chat_history_jitter_ms: int = 500
    search_public_chat_jitter_ms: int = 1500
    get_supergroup_info_jitter_ms: int = 800
    get_message_server_hit_rate: float = 60.0
    get_message_server_hit_jitter_ms: int = 300


@dataclasses.dataclass
class CrawlerConfig:
    """Unified crawl configuration (reference common/utils.go:49-99).

    Fields that were Dapr-specific in the reference (DaprPort etc.) are kept
    for CLI compatibility but drive the local runtime instead of a sidecar.
    """

    dapr_mode: bool = False
    dapr_port: int = 6481
    concurrency: int = 1
    timeout: int = 30
    user_agent: str = "Mozilla/5.0 Crawler"
    output_format: str = "json"
    storage_root: str = "/tmp/crawl"
    tdlib_database_url: str = ""
    tdlib_database_urls: List[str] = dataclasses.field(default_factory=list)
    min_post_date: Optional[_dt.datetime] = None
    post_recency: Optional[_dt.datetime] = None
    date_between_min: Optional[_dt.datetime] = None
    date_between_max: Optional[_dt.datetime] = None
    sample_size: int = 0
    dapr_job_mode: bool = False
    min_users: int = 100
    crawl_id: str = ""
    crawl_label: str = ""
    max_comments: int = -1
    max_posts: int = -1
    max_depth: int = -1
    max_pages: int = 108000
    tdlib_verbosity: int = 1
    skip_media_download: bool = False
    platform: str = "telegram"
    youtube_api_key: str = ""
    sampling_method: str = "channel"
    seed_size: int = 0
    walkback_rate: int = 15
    min_channel_videos: int = 10
    combine_files: bool = False
    combine_temp_dir: str = "/tmp/temp-files"
    combine_watch_dir: str = "/tmp/watch-files"
    combine_write_dir: str = "/tmp/combine-write"
    combine_trigger_size: int = 170   # MiB
    combine_hard_cap: int = 200       # MiB
    null_config: str = "{}"
    exit_on_complete: bool = False
    max_crawl_duration_s: float = 0.0
    rate_limit: TelegramRateLimitConfig = dataclasses.field(
        default_factory=TelegramRateLimitConfig
    )

    # Validator / tandem-crawl mode (common/utils.go:92-99)
    tandem_crawl: bool = False
    validate_only: bool = False
    validator_request_rate: float = 6.0
    validator_request_jitter_ms: int = 200
    validator_claim_batch_size: int = 10
    validator_timeout_s: float = 0.0

    # MI355X-native knobs (no reference analog)
    device_batch_posts: int = 1 << 20   # posts per GPU parse batch
    synthetic_seed: int = 1234          # RNG seed for the synthetic feed
    disable_rate_limits: bool = False   # neutralize pacing for benchmarks

    # Runtime-injected (not a flag)
    null_validator: object = None


def generate_crawl_id(now: Optional[_dt.datetime] = None) -> str:
    """Timestamp crawl ID, "YYYYMMDDHHMMSS" (common/utils.go:103-111)."""
    now = now or _dt.datetime.now()
    return now.strftime("%Y%m%d%H%M%S")


_TIME_AGO_RE = re.compile(r"^(\d+)([hdwmy])$")


def parse_time_ago(s: str, now: Optional[_dt.datetime] = None) -> _dt.datetime:
    """Parse '30d' / '6h' / '2w' / '1m' / '1y' into an absolute datetime.

    Mirrors main.go:91-142: h=hours, d=days, w=weeks, m=months (30 days),
    y=years (365 days). Raises ValueError on bad syntax.
    """
    m = _TIME_AGO_RE.match(s.strip())
    if not m:
        raise ValueError(
            f"invalid time-ago format: {s!r} (expected e.g. '30d', '6h', '2w', '1m', '1y')"
        )
    n = int(m.group(1))
    unit = m.group(2)
    delta = {
        "h": _dt.timedelta(hours=n),
        "d": _dt.timedelta(days=n),
        "w": _dt.timedelta(weeks=n),
        "m": _dt.timedelta(days=30 * n),
        "y": _dt.timedelta(days=365 * n),
    }[unit]
    now = now or _dt.datetime.now()
    return now - delta


VALID_SAMPLING_METHODS = {
    "telegram": ["channel", "snowball", "random-walk"],
    "youtube": ["channel", "random", "snowball"],
}


def validate_sampling_method(
    platform: str,
    sampling_method: str,
    url_list: Optional[List[str]] = None,
    url_file: str = "",
    url_file_url: str = "",
    mode: str = "",
    seed_size: int = 0,
    crawl_id: str = "",
) -> None:
    """Validate platform/sampling combination (common/sampling_validation.go:19-66).

    Raises ValueError with a message matching the reference's semantics:
    - unsupported platform / method-for-platform
    - random-walk: exactly one of URL sources xor seed-size; crawl-id <= 32 chars
    - random (YouTube): no URLs required
    - channel/snowball: URLs required unless dapr-job mode
    """
    supported = VALID_SAMPLING_METHODS.get(platform)
    if supported is None:
        raise ValueError(f"unsupported platform: {platform}")
    if sampling_method not in supported:
        raise ValueError(
            f"sampling method '{sampling_method}' is not supported for platform "
            f"'{platform}'. Supported methods: {supported}"
        )
    has_url_source = bool(url_list) or bool(url_file) or bool(url_file_url)
    if sampling_method == "random-walk":
        if has_url_source == (seed_size > 0):
            raise ValueError(
                "must provide either seed urls or seed size in random-walk crawl, "
                "not both or neither"
            )
        if len(crawl_id) > 32:
            raise ValueError("crawl IDs cannot exceed 32 characters")
        return
    if sampling_method == "random":
        return
    if not has_url_source and mode != "dapr-job":
        raise ValueError(
            f"{sampling_method} sampling requires URLs to be provided. "
            "Use --urls or --url-file to specify them"
        )


def calculate_date_filters(cfg, now=None):
    """Crawl date window with the reference's STRICT precedence
    (dapr/standalone.go:1092-1117 CalculateDateFilters): date-between
    (both bounds set) > post_recency > min_post_date; lower-priority
    fields are IGNORED, not combined. Returns (from_time, to_time);
    from_time may be None (the reference's zero time) and to_time
    defaults to now for the non-date-between branches."""
    now = now or _dt.datetime.now(_dt.timezone.utc)
    if cfg.date_between_min is not None and cfg.date_between_max is not None:
        return cfg.date_between_min, cfg.date_between_max
    if cfg.post_recency is not None:
        return cfg.post_recency, now
    return cfg.min_post_date, now


def config_from_env(cfg: CrawlerConfig, env=os.environ) -> CrawlerConfig:
    """Apply CRAWLER_-prefixed environment overrides (viper env parity:
    main.go:231-261 — dots/dashes map to underscores, prefix CRAWLER_)."""
    mapping = {
        "CRAWLER_CONCURRENCY": ("concurrency", int),
        "CRAWLER_STORAGE_ROOT": ("storage_root", str),
        "CRAWLER_MAX_DEPTH": ("max_depth", int),
        "CRAWLER_MAX_POSTS": ("max_posts", int),
        "CRAWLER_MAX_PAGES": ("max_pages", int),
        "CRAWLER_MAX_COMMENTS": ("max_comments", int),
        "CRAWLER_PLATFORM": ("platform", str),
        "CRAWLER_SAMPLING": ("sampling_method", str),
        "CRAWLER_CRAWL_ID": ("crawl_id", str),
        "CRAWLER_CRAWL_LABEL": ("crawl_label", str),
        "CRAWLER_MIN_USERS": ("min_users", int),
        "CRAWLER_SKIP_MEDIA": ("skip_media_download", lambda v: v.lower() in ("1", "true", "yes")),
        "CRAWLER_SEED_SIZE": ("seed_size", int),
        "CRAWLER_WALKBACK_RATE": ("walkback_rate", int),
    }
    for key, (attr, conv) in mapping.items():
        if key in env:
            setattr(cfg, attr, conv(env[key]))
    return cfg
