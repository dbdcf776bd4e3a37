"""Rate limiter semantics (reference telegramhelper/rate_limiter.go):
per-method token buckets with jitter; REACTIVE GetMessage throttling
(tokens consumed only on TDLib-cache misses, rate_limiter.go:145-169)."""
import random

from crawler_amd.config import TelegramRateLimitConfig
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.client import (
    RateLimitedClient,
    SyntheticTelegramClient,
    TokenBucket,
)


class FakeClock:
    def __init__(self):
        self.t = 0.0
        self.sleeps = []

    def clock(self):
        return self.t

    def sleep(self, s):
        self.sleeps.append(s)
        self.t += s


def test_token_bucket_rate():
    fc = FakeClock()
    rng = random.Random(0)
    b = TokenBucket(60, jitter_ms=0, rng=rng, clock=fc.clock,
                    sleeper=fc.sleep)
    # 60/min = 1/s; capacity 1 token
    b.acquire()          # uses the initial token
    b.acquire()          # must wait ~1s
    assert b.waits == 1
    assert 0.9 <= fc.sleeps[0] <= 1.1
    fc.t += 10           # tokens refill (capped at capacity)
    b.acquire()
    assert b.waits == 1  # no extra wait


def test_token_bucket_jitter_bounded():
    fc = FakeClock()
    b = TokenBucket(60, jitter_ms=500, rng=random.Random(1),
                    clock=fc.clock, sleeper=fc.sleep)
    b.acquire()
    b.acquire()
    assert 1.0 <= fc.sleeps[0] <= 1.5  # base wait + <=500ms jitter


def test_reactive_get_message_throttle():
    """Cache hits are free; only server hits consume tokens
    (rate_limiter.go:145-169)."""
    fc = FakeClock()
    feed = SyntheticFeed(FeedConfig(seed=3, universe=50,
                                    posts_per_channel=10))
    base = SyntheticTelegramClient(feed, posts_per_channel=10)
    rl = TelegramRateLimitConfig(get_message_server_hit_rate=60,
                                 get_message_server_hit_jitter_ms=0)
    c = RateLimitedClient(base, rl, rng=random.Random(0),
                          sleeper=fc.sleep)
    # patch bucket clocks for determinism
    c.message_bucket.clock = fc.clock
    c.message_bucket.last = fc.clock()
    info = c.search_public_chat("c0000000001")
    # first get_chat_history populates the TDLib-style local cache
    msgs = c.get_chat_history(info.chat_id)
    # server hit consumed the initial token...
    c.get_message(info.chat_id, msgs[0].msg_id)
    n_waits0 = c.message_bucket.waits
    # ...but subsequent CACHE hits consume none (no waits accumulate)
    for m in msgs[:5]:
        c.get_message(info.chat_id, m.msg_id)
    assert base.last_call_latency_class == "cache"
    assert c.message_bucket.waits == n_waits0


def test_disabled_rate_limits_never_sleep():
    fc = FakeClock()
    feed = SyntheticFeed(FeedConfig(seed=3, universe=50,
                                    posts_per_channel=5))
    base = SyntheticTelegramClient(feed, posts_per_channel=5)
    c = RateLimitedClient(base, TelegramRateLimitConfig(),
                          sleeper=fc.sleep, disabled=True)
    for _ in range(10):
        c.search_public_chat("c0000000001")
    assert fc.sleeps == []
