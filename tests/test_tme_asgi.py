"""Tandem validator over uvicorn (ASGI) — the production-server hop.

Same contract as tests/test_tme_http.py but through a real ASGI server
and event loop instead of the stdlib threaded server, mirroring the
reference validator's full client->server stack
(telegramhelper/channelvalidator.go:64-103).
"""
import random

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, RandomWalkStore
from crawler_amd.engine.htmlvalidator import (
    ValidationHTTPError,
    validate_channel_http,
)
from crawler_amd.engine.validator import TandemValidator
from crawler_amd.feed.tme import MockTMe
from crawler_amd.feed.tme_server import http_fetcher

uvicorn = pytest.importorskip("uvicorn")

from crawler_amd.feed.asgi import UvicornTMeServer  # noqa: E402


@pytest.fixture
def server():
    tme = MockTMe(universe=1000)
    srv = UvicornTMeServer(tme).start()
    yield srv, tme
    srv.stop()


def test_asgi_fetch_and_classify(server):
    srv, tme = server
    fetch = http_fetcher(srv.base_url)
    res = validate_channel_http("c0000000001", fetch)
    assert res.status in ("valid", "invalid", "not_channel")
    # classification over ASGI == classification straight from the mock
    direct = tme("c0000000001")
    assert fetch("c0000000001") == direct


def test_asgi_blocked_raises(server):
    srv, tme = server
    tme.blocked = True
    fetch = http_fetcher(srv.base_url)
    with pytest.raises(ValidationHTTPError) as ei:
        validate_channel_http("c0000000001", fetch)
    assert ei.value.kind == "blocked"


def test_validator_over_asgi_end_to_end(tmp_path, server):
    srv, tme = server
    cfg = CrawlerConfig(crawl_id="a1", storage_root=str(tmp_path),
                        walkback_rate=0)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    sm.add_discovered_channel("fallbackchan")
    v = TandemValidator(cfg, sm, rw, fetcher=http_fetcher(srv.base_url),
                        rng=random.Random(1), probe_interval=0.0)
    bid = rw.open_batch("a1", "src", "p1", 0, "seq")
    for i in range(25):
        rw.insert_pending_edge(bid, "a1", "c%010d" % i, "src", "seq", "url")
    rw.close_batch(bid)
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    assert v.stats["validated"] == 25
    assert v.stats["valid"] > 0
    assert rw.get_pages(1)


def test_asgi_matches_stdlib_server_bytes(server):
    """The two server stacks serve byte-identical bodies for the same
    MockTMe state — the validator's inputs don't depend on the hop."""
    from crawler_amd.feed.tme_server import TMeServer

    srv, tme = server
    std = TMeServer(tme).start()
    try:
        fa = http_fetcher(srv.base_url)
        fs = http_fetcher(std.base_url)
        for i in (0, 1, 7, 999):
            n = "c%010d" % i
            assert fa(n) == fs(n)
    finally:
        std.stop()
