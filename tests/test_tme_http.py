"""Tandem validator over a real HTTP socket (local t.me mock server)."""
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
from crawler_amd.feed.tme_server import TMeServer, http_fetcher


@pytest.fixture
def server():
    tme = MockTMe(universe=1000)
    srv = TMeServer(tme).start()
    yield srv, tme
    srv.stop()


def test_http_fetch_and_classify(server):
    srv, tme = server
    fetch = http_fetcher(srv.base_url)
    # a valid supergroup over a real socket
    valid_name = None
    for i in range(1000):
        n = "c%010d" % i
        if tme(n)[1] == tme.__dict__["_valid"]:
            valid_name = n
            break
    res = validate_channel_http(valid_name, fetch)
    assert res.status == "valid"
    res2 = validate_channel_http("nonexistent_channel_xyz", fetch)
    assert res2.status == "invalid"


def test_http_blocked_raises(server):
    srv, tme = server
    tme.blocked = True
    fetch = http_fetcher(srv.base_url)
    with pytest.raises(ValidationHTTPError) as ei:
        validate_channel_http("c0000000001", fetch)
    assert ei.value.kind == "blocked"


def test_validator_over_http_end_to_end(tmp_path, server):
    srv, tme = server
    cfg = CrawlerConfig(crawl_id="h1", storage_root=str(tmp_path),
                        walkback_rate=0)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    sm.add_discovered_channel("fallbackchan")
    v = TandemValidator(cfg, sm, rw, fetcher=http_fetcher(srv.base_url),
                        rng=random.Random(1), probe_interval=0.0)
    bid = rw.open_batch("h1", "src", "p1", 0, "seq")
    for i in range(25):
        rw.insert_pending_edge(bid, "h1", "c%010d" % i, "src", "seq", "url")
    rw.close_batch(bid)
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    assert v.stats["validated"] == 25
    assert v.stats["valid"] > 0
    assert rw.get_pages(1)  # next hop produced


def test_http_404_invalid_and_server_classes(server):
    """Full HTTP status taxonomy over a real socket: 404 -> invalid
    not_found; unknown paths never crash the server."""
    import urllib.error
    import urllib.request

    srv, tme = server
    base = srv.base_url
    fetch = http_fetcher(base)
    # a username outside the mock's shape still classifies (not crash)
    r = validate_channel_http("zzzzz_no_such_name_zzzzz", fetch)
    assert r.status in ("invalid", "not_channel", "valid")
    # raw root path -> server answers without hanging or 5xx-crashing
    try:
        code = urllib.request.urlopen(base + "/", timeout=5).status
    except urllib.error.HTTPError as e:
        code = e.code
    assert code < 500
