"""Date-window / pagination / sampling math for message fetching.

Mirrors the reference's dapr/calculate_date_filters_test.go and
crawl/fetch_messages_test.go coverage: window clamps,
pagination stop conditions, Fisher-Yates date-between sampling and the
activity filter."""
import datetime as dt
import random

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine.pipeline import (fetch_channel_messages,
                                         is_channel_active)

UTC = dt.timezone.utc
BASE = dt.datetime(2026, 1, 1, tzinfo=UTC).timestamp()
DAY = 86400.0


class Msg:
    def __init__(self, msg_id, date):
        self.msg_id = msg_id
        self.date = date


class PagedClient:
    """Newest-first pages of 100, from_message_id cursor — the TDLib
    GetChatHistory contract the synthetic client also implements."""

    def __init__(self, n, newest_ts=BASE, step=DAY):
        # msg i (1-based) has date newest_ts - (n-i)*step; msg n is newest
        self.msgs = [Msg(i, newest_ts - (n - i) * step)
                     for i in range(1, n + 1)]
        self.calls = 0

    def get_chat_history(self, chat_id, from_message_id=0, limit=100):
        self.calls += 1
        ordered = sorted(self.msgs, key=lambda m: -m.msg_id)
        if from_message_id:
            ordered = [m for m in ordered if m.msg_id < from_message_id]
        return ordered[:limit]


def cfg(**kw):
    return CrawlerConfig(crawl_id="dw", storage_root="/tmp/x", **kw)


def ts2dt(ts):
    return dt.datetime.fromtimestamp(ts, UTC)


def test_pagination_walks_all_pages():
    c = PagedClient(250)
    out = fetch_channel_messages(c, 1, cfg())
    assert len(out) == 250
    assert c.calls == 4  # 100+100+50+empty
    assert out[0].msg_id == 250 and out[-1].msg_id == 1  # newest-first


def test_max_posts_stops_pagination_early():
    c = PagedClient(1000)
    out = fetch_channel_messages(c, 1, cfg(max_posts=120))
    assert len(out) == 120
    assert c.calls == 2  # stops mid-second-page


def test_min_post_date_stops_at_boundary():
    c = PagedClient(300)
    # keep only messages strictly newer than 50 days before newest
    cutoff = BASE - 50 * DAY
    out = fetch_channel_messages(c, 1, cfg(min_post_date=ts2dt(cutoff)))
    # newest-first: msg 300 has date BASE, msg i has BASE-(300-i)*DAY;
    # date >= cutoff -> 300-i <= 50 -> i >= 250 -> 51 messages
    assert len(out) == 51
    assert min(m.date for m in out) >= cutoff
    assert c.calls == 1  # stopped inside the first page, no more fetches


def test_post_recency_never_filters_messages():
    """post_recency is the ACTIVITY check's bound only
    (isChannelActiveWithinPeriod, runner.go:628); the message fetch
    ignores it — only min_post_date bounds the walk
    (runner.go:909-912)."""
    c = PagedClient(300)
    loose = BASE - 100 * DAY
    tight = BASE - 10 * DAY
    out = fetch_channel_messages(
        c, 1, cfg(min_post_date=ts2dt(loose), post_recency=ts2dt(tight)))
    assert len(out) == 101  # min_post_date governs, recency ignored
    c2 = PagedClient(300)
    out2 = fetch_channel_messages(
        c2, 1, cfg(min_post_date=ts2dt(tight), post_recency=ts2dt(loose)))
    assert len(out2) == 11


def test_date_between_overrides_min_post_date():
    """When BOTH date-between bounds are set, min_post_date is ignored
    (runner.go:909-912 picks the sampling fetch exclusively)."""
    c = PagedClient(300)
    lo = BASE - 60 * DAY
    hi = BASE - 30 * DAY
    out = fetch_channel_messages(
        c, 1, cfg(date_between_min=ts2dt(lo), date_between_max=ts2dt(hi),
                  min_post_date=ts2dt(BASE - 40 * DAY)))
    # min_post_date (40d) would cut the window short; it must NOT
    assert len(out) == 31


def test_date_between_window_clamps_both_sides():
    c = PagedClient(300)
    lo = BASE - 60 * DAY
    hi = BASE - 30 * DAY
    out = fetch_channel_messages(
        c, 1, cfg(date_between_min=ts2dt(lo), date_between_max=ts2dt(hi)))
    assert all(lo <= m.date <= hi for m in out)
    assert len(out) == 31
    # newer-than-hi messages are SKIPPED (continue), older-than-lo STOPS —
    # so pagination never reaches page 3
    assert c.calls == 1


def test_date_between_sampling_fisher_yates_deterministic():
    c = PagedClient(300)
    lo, hi = BASE - 200 * DAY, BASE
    conf = cfg(date_between_min=ts2dt(lo), date_between_max=ts2dt(hi),
               sample_size=20)
    out = fetch_channel_messages(c, 1, conf, rng=random.Random(42))
    assert len(out) == 20
    # deterministic under the same seed
    c2 = PagedClient(300)
    out2 = fetch_channel_messages(c2, 1, conf, rng=random.Random(42))
    assert [m.msg_id for m in out] == [m.msg_id for m in out2]
    # a different seed gives a different permutation
    c3 = PagedClient(300)
    out3 = fetch_channel_messages(c3, 1, conf, rng=random.Random(7))
    assert [m.msg_id for m in out] != [m.msg_id for m in out3]


def test_sampling_requires_date_between():
    """sample_size without a date window is ignored
    (telegramutils.go:124-130 gate)."""
    c = PagedClient(100)
    out = fetch_channel_messages(c, 1, cfg(sample_size=5))
    assert len(out) == 100


def test_sampling_noop_when_fewer_than_sample_size():
    c = PagedClient(10)
    out = fetch_channel_messages(
        c, 1, cfg(date_between_min=ts2dt(BASE - 400 * DAY),
                  date_between_max=ts2dt(BASE), sample_size=50))
    assert len(out) == 10


def test_is_channel_active_matrix():
    """runner.go:635 exactly: !active || messageCount==0 ||
    (sampling != random-walk && MinUsers>0 && members<MinUsers)."""
    conf = cfg(min_users=100,
               post_recency=ts2dt(BASE - 30 * DAY))
    fresh = BASE - 1 * DAY
    stale = BASE - 90 * DAY
    ok, _ = is_channel_active(fresh, 500, conf, 10)
    assert ok
    ok, why = is_channel_active(fresh, 50, conf, 10)
    assert not ok and "min_users" in why
    # min-users NEVER applies in random-walk mode (runner.go:635)
    ok, _ = is_channel_active(
        fresh, 50, cfg(min_users=100, sampling_method="random-walk",
                       post_recency=ts2dt(BASE - 30 * DAY)), 10)
    assert ok
    # min_users == 0 disables the member check entirely
    ok, _ = is_channel_active(fresh, 0, cfg(
        min_users=0, post_recency=ts2dt(BASE - 30 * DAY)), 10)
    assert ok
    ok, why = is_channel_active(stale, 500, conf, 10)
    assert not ok and "older than recency" in why
    # exact tie with the cutoff is INACTIVE (time.After is strict)
    ok, why = is_channel_active(BASE - 30 * DAY, 500, conf, 10)
    assert not ok and "recency" in why
    # channel with zero messages -> deadend even without recency
    ok, why = is_channel_active(None, 500, cfg(min_users=100), 0)
    assert not ok and "no messages" in why
    # no recency filter -> stale is fine
    ok, _ = is_channel_active(stale, 500, cfg(min_users=100), 10)
    assert ok
