"""Tandem crawler+validator end-to-end: the P5 pipeline of SURVEY §2.4 —
crawler streams pending edges, validator claims+validates and writes the
next hop into the page buffer, crawler continues; circuit breaker aborts
when the validator stalls (dapr/standalone.go:837-867)."""
import random
import threading

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, RandomWalkStore
from crawler_amd.engine import errors as E
from crawler_amd.engine.runner import RandomWalkRunner
from crawler_amd.engine.validator import TandemValidator
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.client import ConnectionPool
from crawler_amd.feed.tme import MockTMe


def mk_tandem(tmp_path, validator_timeout=0.0):
    cfg = CrawlerConfig(
        crawl_id="td1", storage_root=str(tmp_path), min_users=1,
        sampling_method="random-walk", tandem_crawl=True,
        disable_rate_limits=True, validator_timeout_s=validator_timeout,
        walkback_rate=15,
    )
    feed = SyntheticFeed(FeedConfig(seed=13, universe=300,
                                    posts_per_channel=40))
    pool = ConnectionPool(feed, 2, cfg.rate_limit, posts_per_channel=40,
                          disable_rate_limits=True)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    runner = RandomWalkRunner(cfg, sm, rw, pool, rng=random.Random(2),
                              poll_interval=0.005)
    tme = MockTMe(universe=300, user_permille=100, unoccupied_permille=100)
    validator = TandemValidator(cfg, sm, rw, fetcher=tme,
                                rng=random.Random(3), probe_interval=0.0)
    return cfg, sm, rw, runner, validator


def test_tandem_crawler_validator_pipeline(tmp_path):
    cfg, sm, rw, runner, validator = mk_tandem(tmp_path,
                                               validator_timeout=30.0)
    sm.add_discovered_channel("c0000000001")
    sm.add_discovered_channel("c0000000002")
    runner.seed(["c0000000001"])

    stop = threading.Event()

    def validate_loop():
        while not stop.is_set():
            n = validator.pump_edges()
            b = validator.pump_walkback()
            if not n and not b:
                stop.wait(0.005)

    vt = threading.Thread(target=validate_loop, daemon=True)
    vt.start()
    try:
        stats = runner.run(max_pages=5)
    finally:
        stop.set()
        vt.join(timeout=5)
    assert stats["pages"] == 5
    # the validator produced next-hop pages and edge records
    assert validator.stats["batches"] >= 1 or rw.edge_records


def test_tandem_circuit_breaker_aborts(tmp_path):
    cfg, sm, rw, runner, validator = mk_tandem(tmp_path,
                                               validator_timeout=0.05)
    sm.add_discovered_channel("c0000000001")
    runner.seed(["c0000000001"])
    # NO validator running: after the first channel the crawler blocks on
    # pending batches and the breaker must fire
    with pytest.raises(E.PoolExhausted, match="circuit breaker"):
        runner.run(max_pages=5)
    assert runner.stats.get("circuit_breaker") == 1
