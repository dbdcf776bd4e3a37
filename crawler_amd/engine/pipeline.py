"""Per-channel crawl pipeline — the engine's hot path orchestration.

Parity map (reference crawl/runner.go):
- run_for_channel_with_pool: RunForChannelWithPool (runner.go:506-561)
- run_for_channel: RunForChannel (runner.go:563-817) — channel info,
  activity filter, message processing, outlink handling per sampling mode
- fetch_channel_messages: FetchChannelMessagesWithSampling
  (telegramutils.go:25-157): newest-first pagination 100/batch, min/max
  date window, max-posts cap, Fisher-Yates date-between sampling
- random-walk tail: walkback decision, page-buffer append, edge records
  (runner.go:1413-1541); tandem pending-edge batching (runner.go:1252-1306)
- FLOOD_WAIT handling: sleep <300s / retire >=300s (runner.go:1310-1383)
- 400 replacement: Handle400Replacement (runner.go:152-284)

The CPU path parses per message via ops.golden; the GPU path
(engine/gpu_pipeline.py) runs whole channel batches through the HIP
kernels. Both produce the same JSONL and discovery decisions.
"""
from __future__ import annotations

import dataclasses
import random
from typing import List, Optional, Tuple

from ..feed.client import TelegramAPIError
from ..models.post import ChannelData, EngagementData
from ..ops import golden as G
from ..utils.logging import get_logger
from . import errors as E
from .state import Page

_log = get_logger("pipeline")


@dataclasses.dataclass
class ChannelResult:
    channel: str
    status: str = "fetched"      # fetched | deadend | error
    error: str = ""
    posts_stored: int = 0
    skipped_fetched: int = 0
    parse_errors: int = 0
    discovered: List[str] = dataclasses.field(default_factory=list)
    discovered_links: List[G.DiscoveredLink] = dataclasses.field(
        default_factory=list
    )


def fetch_channel_messages(client, chat_id: int, cfg,
                           rng: Optional[random.Random] = None) -> list:
    """FetchChannelMessagesWithSampling (telegramutils.go:25-157).

    Filter precedence matches crawl/runner.go:909-912 exactly: when
    BOTH date-between bounds are set the fetch uses that window and
    min_post_date is IGNORED; otherwise min_post_date alone bounds the
    walk. post_recency NEVER filters messages — the reference uses it
    only for the channel-activity deadend check
    (isChannelActiveWithinPeriod, runner.go:628)."""
    out = []
    from_id = 0
    max_posts = cfg.max_posts if cfg.max_posts and cfg.max_posts > 0 else None
    lo = hi = min_ts = None
    if cfg.date_between_min is not None and cfg.date_between_max is not None:
        lo = cfg.date_between_min.timestamp()
        hi = cfg.date_between_max.timestamp()
    elif cfg.min_post_date is not None:
        min_ts = cfg.min_post_date.timestamp()
    while True:
        page = client.get_chat_history(chat_id, from_message_id=from_id,
                                       limit=100)
        if not page:
            break
        for m in page:
            if hi is not None and m.date > hi:
                continue
            if lo is not None and m.date < lo:
                return _maybe_sample(out, cfg, rng)
            if min_ts is not None and m.date < min_ts:
                return _maybe_sample(out, cfg, rng)
            out.append(m)
            if max_posts is not None and len(out) >= max_posts:
                return _maybe_sample(out, cfg, rng)
        from_id = page[-1].msg_id
    return _maybe_sample(out, cfg, rng)


def _maybe_sample(msgs, cfg, rng):
    """Fisher-Yates sample when date-between + sample-size are set
    (telegramutils.go:124-154)."""
    if (cfg.sample_size and cfg.sample_size > 0
            and cfg.date_between_min is not None and
            len(msgs) > cfg.sample_size):
        r = rng or random
        msgs = list(msgs)
        for i in range(len(msgs) - 1, 0, -1):
            j = r.randrange(i + 1)
            msgs[i], msgs[j] = msgs[j], msgs[i]
        return msgs[: cfg.sample_size]
    return msgs


def is_channel_active(latest_ts, member_count: int, cfg,
                      message_count: int) -> Tuple[bool, str]:
    """Deadend filter (runner.go:628-643 exactly):
    `!active || messageCount == 0 || (sampling != "random-walk" &&
    MinUsers > 0 && memberCount < MinUsers)` where active =
    latestMessageTime.After(PostRecency) over the chat's actual
    newest message (getLatestMessageTime — independent of the fetch
    window, runner.go:724-732; a tie with the cutoff is INACTIVE).
    min-users never applies in random-walk mode."""
    if cfg.post_recency is not None:
        cutoff = cfg.post_recency.timestamp()
        if latest_ts is None or latest_ts <= cutoff:
            return False, "latest message older than recency window"
    if message_count == 0:
        return False, "no messages in channel"
    if (cfg.sampling_method != "random-walk" and cfg.min_users > 0
            and member_count < cfg.min_users):
        return False, f"member count {member_count} < min_users {cfg.min_users}"
    return True, ""


def run_for_channel_with_pool(pool, page: Page, sm, cfg, rw=None,
                              seen=None, mode_hooks=None,
                              rng: Optional[random.Random] = None,
                              now=None) -> ChannelResult:
    """Checkout -> run -> release|retire (runner.go:506-561)."""
    client = pool.get_connection()
    retire = False
    recreate = False
    try:
        return run_for_channel(client, page, sm, cfg, rw=rw, seen=seen,
                               mode_hooks=mode_hooks, rng=rng, now=now)
    except E.FloodWaitRetire:
        retire = True
        raise
    except E.ConnectionDropped:
        recreate = True
        raise
    except TelegramAPIError as err:
        # transport errors escaping from mid-channel calls (comments
        # fetch etc.) also destroy+recreate the session
        if E.is_connection_error(str(err)):
            recreate = True
        raise
    finally:
        if retire:
            pool.retire_connection(client)
        elif recreate:
            pool.handle_connection_error(client)
        else:
            pool.release_connection(client)


def _classify_api_error(err: Exception):
    msg = str(err)
    if E.is_connection_error(msg):
        raise E.ConnectionDropped(msg)
    secs, is_flood = E.parse_flood_wait_secs(msg)
    if is_flood:
        if secs >= E.FLOOD_WAIT_RETIRE_THRESHOLD_SECS:
            raise E.FloodWaitRetire(msg)
        # transient ban: skip the channel but keep the client
        # (runner.go:1310-1345 sleeps <300s; the synthetic engine does not
        # actually sleep — pacing is a token-bucket concern)
        return "flood_skip"
    if E.is_tdlib_400(msg):
        raise E.TDLib400(msg)
    return "error"


def run_for_channel(client, page: Page, sm, cfg, rw=None, seen=None,
                    mode_hooks=None, rng=None, now=None) -> ChannelResult:
    """The per-channel pipeline (runner.go:563-817 + 1110-1550)."""
    rng = rng or random.Random()
    channel = page.url
    result = ChannelResult(channel=channel)

    # -- channel info (getChannelInfoWithDeps, runner.go:819-984) --
    try:
        cached_id = None
        if rw is not None:
            cid, ok = sm.get_cached_chat_id(channel)
            cached_id = cid if ok else None
        if cached_id:
            info = client.get_chat(cached_id)
        else:
            info = client.search_public_chat(channel)
        messages = fetch_channel_messages(client, info.chat_id, cfg, rng)
        sg = client.get_supergroup_info(info.chat_id)
    except TelegramAPIError as err:
        kind = _classify_api_error(err)  # raises FloodWaitRetire / TDLib400
        result.status = "error" if kind == "error" else "deadend"
        result.error = str(err)
        return result

    # -- null-validate channel data (runner.go:612) --
    if cfg.null_validator is not None:
        cd = ChannelData(
            channel_id=str(info.chat_id), channel_name=info.title,
            channel_url=f"https://t.me/c/{channel}",
            channel_engagement_data=EngagementData(
                follower_count=info.member_count,
                post_count=info.message_count,
                views_count=info.total_views,
            ),
        )
        cfg.null_validator.validate_channel_data(cd)

    # -- activity filter -> deadend (runner.go:628-643): the recency
    # check uses the chat's NEWEST message (getLatestMessageTime, one
    # GetChatHistory(limit=1) like the reference), not the filtered
    # fetch window --
    latest_ts = None
    if messages:
        latest_ts = max(m.date for m in messages)
    else:
        newest = client.get_chat_history(info.chat_id,
                                         from_message_id=0, limit=1)
        if newest:
            latest_ts = newest[0].date
    active, why = is_channel_active(latest_ts, sg["member_count"], cfg,
                                    info.message_count)
    if not active:
        result.status = "deadend"
        result.error = why
        return result

    # -- per-message processing (processAllMessagesWithProcessor) --
    # resampleMarker / addNewMessages (crawl/runner.go:1572-1697): on a
    # re-crawl, messages already "fetched" are skipped (no duplicate
    # posts); tracked messages absent from the new fetch are marked
    # "deleted"; everything else (re)processes. (The GPU engine path
    # processes a channel atomically in one batch, so its resume
    # granularity is the channel — see gpu_runner.py.)
    username = sg["active_usernames"][0] if sg["active_usernames"] else ""
    tracked = {}
    if page.messages:
        tracked = {(pm.chat_id, pm.message_id): pm for pm in page.messages}
        discovered_keys = {(m.chat_id, m.msg_id) for m in messages}
        for key, pm in tracked.items():
            if pm.status == "fetched":
                continue
            pm.status = "resample" if key in discovered_keys else "deleted"
    all_links: dict = {}
    for m in messages:
        pm = tracked.get((m.chat_id, m.msg_id))
        if pm is not None and pm.status == "fetched":
            result.skipped_fetched += 1
            continue
        if page.id and page.id in getattr(sm, "pages", {}):
            sm.update_message(page.id, m.chat_id, m.msg_id, "fetched")
        elif pm is not None:
            pm.status = "fetched"
        # Panic containment at message granularity (tdutils.go:395-405
        # recover()): one malformed message never aborts the channel.
        try:
            comments = None
            if m.reply_count > 0:
                # paginated thread walk; reply_count plays the
                # reference's commentcount role (tdutils.go:434)
                comments = client.get_message_comments(
                    m.chat_id, m.msg_id, cfg.max_comments,
                    comment_count=m.reply_count,
                )
            post = G.parse_message(
                m,
                channel_username=username or channel,
                chat_title=info.title,
                member_count=sg["member_count"],
                post_count=info.message_count,
                total_views=info.total_views,
                comments=comments,
                min_post_date=cfg.min_post_date,
                skip_media=cfg.skip_media_download,
                now=now,
            )
            if post is not None:
                sm.store_post(channel, post)
                result.posts_stored += 1
            for link in G.extract_links_with_source(m):
                if link.name not in all_links:
                    all_links[link.name] = link
        except TelegramAPIError:
            raise  # API faults keep their channel-level semantics
        except Exception as perr:
            result.parse_errors += 1
            _log.warn("message-parse-recovered", channel=channel,
                      msg_id=m.msg_id, error=str(perr))
    result.discovered_links = list(all_links.values())

    # -- outlink handling per sampling mode --
    hooks = mode_hooks or {}
    handler = hooks.get("outlinks")
    if handler is not None:
        handler(page, result, client)
    else:
        # channel/snowball default: discovered names become next-layer pages
        # (runner.go:1236-1244)
        result.discovered = [l.name for l in result.discovered_links
                             if l.name != channel]
    result.status = "fetched"
    return result
