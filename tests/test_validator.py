"""Validator tests: HTML rules, fixtures, blocked machine, walkback batches.

Mirror of the reference's channelvalidator_test.go + validator_test.go
coverage (fixtures parsed directly; injectable
ValidateFunc for edge outcomes)."""
import os
import random

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page, RandomWalkStore
from crawler_amd.engine.htmlvalidator import (
    ChannelValidationResult,
    ValidationHTTPError,
    parse_channel_html,
    validate_channel_http,
)
from crawler_amd.engine.validator import BLOCKED_THRESHOLD, TandemValidator
from crawler_amd.feed.tme import MockTMe

FIX = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "fixtures", "telegram-html")


def fixture(name):
    with open(os.path.join(FIX, name), "rb") as f:
        return f.read()


# ---------- HTML classification (channelvalidator.go:132-153) ----------

def test_valid_channel_fixture():
    r = parse_channel_html(fixture("valid-channel.html"))
    assert r.status == "valid"


def test_not_supergroup_fixture():
    r = parse_channel_html(fixture("not-a-supergroup.html"))
    assert r.status == "not_channel"
    assert r.reason == "not_supergroup"


def test_username_not_occupied_fixture():
    r = parse_channel_html(fixture("username-not-occupied.html"))
    assert r.status == "invalid"
    assert r.reason == "username_not_occupied"


def test_invalid_channel_fixture():
    r = parse_channel_html(fixture("invalid-channel.html"))
    assert r.status == "invalid"
    assert r.reason == "not_found"


def test_http_error_taxonomy():
    with pytest.raises(ValidationHTTPError) as ei:
        validate_channel_http("x", lambda u: (429, b""))
    assert ei.value.kind == "blocked"
    with pytest.raises(ValidationHTTPError) as ei:
        validate_channel_http("x", lambda u: (503, b""))
    assert ei.value.kind == "transient"
    r = validate_channel_http("x", lambda u: (404, b""))
    assert r.status == "invalid" and r.reason == "not_found"


# ---------- tandem validator ----------

def mk_validator(tmp_path, tme=None, **cfg_kw):
    cfg_kw.setdefault("walkback_rate", 15)
    cfg = CrawlerConfig(crawl_id="v1", storage_root=str(tmp_path), **cfg_kw)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    tme = tme or MockTMe(universe=1000)
    v = TandemValidator(cfg, sm, rw, fetcher=tme, rng=random.Random(4),
                        probe_interval=0.0)
    return cfg, sm, rw, tme, v


def seed_batch(rw, names, crawl_id="v1", source="srcchan1"):
    bid = rw.open_batch(crawl_id, source, "p1", 0, "seq1")
    for n in names:
        rw.insert_pending_edge(bid, crawl_id, n, source, "seq1", "mention")
    rw.close_batch(bid)
    return bid


def test_edge_validation_outcomes(tmp_path):
    cfg, sm, rw, tme, v = mk_validator(tmp_path)
    # choose usernames covering valid / user-account / unoccupied buckets
    names = ["c%010d" % i for i in range(40)]
    bid = seed_batch(rw, names)
    while v.pump_edges():
        pass
    edges = rw.edges_of_batch(bid)
    statuses = {e.validation_status for e in edges}
    assert "valid" in statuses
    # every edge got a definitive status
    assert all(e.validation_status != "pending" for e in edges)
    assert v.stats["valid"] > 0


def test_valid_claim_is_exactly_once(tmp_path):
    cfg, sm, rw, tme, v = mk_validator(tmp_path)
    # same destination twice in one batch -> first 'valid', second 'duplicate'
    valid_name = None
    for i in range(1000):
        n = "c%010d" % i
        if tme(n)[1] == tme.__dict__["_valid"]:
            valid_name = n
            break
    assert valid_name
    bid = seed_batch(rw, [valid_name, valid_name])
    while v.pump_edges():
        pass
    statuses = sorted(e.validation_status
                      for e in rw.edges_of_batch(bid))
    assert statuses == ["duplicate", "valid"]


def test_blocked_state_machine_and_probe(tmp_path):
    cfg, sm, rw, tme, v = mk_validator(tmp_path)
    names = ["c%010d" % i for i in range(BLOCKED_THRESHOLD + 2)]
    seed_batch(rw, names)
    tme.blocked = True
    while v.pump_edges():
        pass
    assert v.blocked
    assert any(e["type"] == "ip_blocked" for e in rw.access_events)
    # edges stay pending while blocked
    assert all(e.validation_status in ("pending", "validating")
               for e in rw.pending_edges.values())
    # access returns; probe clears the block and work resumes
    tme.blocked = False
    rw.recover_stale_claims()
    while v.pump_edges():
        pass
    assert not v.blocked
    assert any(e["type"] == "ip_unblocked" for e in rw.access_events)


def test_walkback_batch_forward_choice(tmp_path):
    cfg, sm, rw, tme, v = mk_validator(tmp_path, walkback_rate=0)
    for n in ["seedchan1", "seedchan2"]:
        sm.add_discovered_channel(n)
    names = ["c%010d" % i for i in range(30)]
    bid = seed_batch(rw, names)
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    # a next-hop page landed in the buffer with the batch's crawl id
    pages = rw.get_pages(10)
    assert len(pages) == 1
    page = pages[0]
    assert page.crawl_id == "v1"
    primary = [e for e in rw.edge_records if not e.skipped]
    assert len(primary) == 1
    assert primary[0].destination_channel == page.url
    # forward edge keeps the chain sequence id
    assert not primary[0].walkback
    assert page.sequence_id == "seq1"
    # skipped edges recorded for unchosen valid channels
    assert rw.pending_batches[bid].status == "completed"
    # batch edges were flushed
    assert not any(e.batch_id == bid for e in rw.pending_edges.values())
    assert sum(rw.source_type_stats.values()) >= 1


def test_walkback_batch_no_valid_forces_walkback(tmp_path):
    tme = MockTMe(universe=0)  # everything invalid
    cfg, sm, rw, tme, v = mk_validator(tmp_path, tme=tme)
    sm.add_discovered_channel("fallbackchan")
    bid = seed_batch(rw, ["c0000000001"])
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    primary = [e for e in rw.edge_records if not e.skipped]
    assert primary[0].walkback
    page = rw.get_pages(1)[0]
    assert page.url == "fallbackchan"
    assert page.sequence_id != "seq1"  # fresh chain after walkback


def test_recover_runs_at_startup(tmp_path):
    cfg, sm, rw, tme, v = mk_validator(tmp_path)
    bid = seed_batch(rw, ["c0000000001"])
    rw.claim_pending_edges(1)  # leave a stale 'validating' claim
    v.recover()
    assert all(e.validation_status == "pending"
               for e in rw.pending_edges.values())
