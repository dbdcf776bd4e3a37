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
    assert r.reason == "not_found"


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


# ---------- scripted-validate_fn matrix (validator.go:194-310 branches) ----

class Script:
    """validate_fn whose outcome per call is scripted; records calls."""

    def __init__(self, outcomes):
        self.outcomes = list(outcomes)
        self.calls = 0

    def __call__(self, username):
        self.calls += 1
        out = self.outcomes.pop(0) if self.outcomes else "valid"
        if out == "blocked":
            raise ValidationHTTPError("blocked", 429)
        if out == "transient":
            raise ValidationHTTPError("transient", 503)
        if out == "not_channel":
            return ChannelValidationResult("not_channel", "not_supergroup")
        if out == "invalid":
            return ChannelValidationResult("invalid", "not_found")
        return ChannelValidationResult("valid")


def mk_scripted(tmp_path, outcomes, **cfg_kw):
    cfg_kw.setdefault("walkback_rate", 0)
    cfg = CrawlerConfig(crawl_id="v1", storage_root=str(tmp_path), **cfg_kw)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    script = Script(outcomes)
    v = TandemValidator(cfg, sm, rw, validate_fn=script,
                        rng=random.Random(9), probe_interval=0.0)
    return cfg, sm, rw, v, script


def test_transient_error_leaves_edge_pending(tmp_path):
    cfg, sm, rw, v, script = mk_scripted(tmp_path, ["transient", "valid"])
    bid = seed_batch(rw, ["aaaa1", "bbbb2"])
    v.pump_edges()
    edges = {e.destination_channel: e.validation_status
             for e in rw.edges_of_batch(bid)}
    assert edges["aaaa1"] == "pending"   # left for a later claim round
    assert edges["bbbb2"] == "valid"
    assert not v.blocked and v.consecutive_blocked == 0
    # next round re-claims ONLY the pending edge
    v.pump_edges()
    assert script.calls == 3


def test_blocked_counter_resets_on_success(tmp_path):
    """4 blocked, 1 ok, 4 blocked: never crosses the threshold of 5
    CONSECUTIVE blocked outcomes (validator.go:34-38)."""
    outcomes = ["blocked"] * 4 + ["valid"] + ["blocked"] * 4
    cfg, sm, rw, v, script = mk_scripted(tmp_path, outcomes)
    seed_batch(rw, ["n%04d" % i for i in range(9)])
    while v.pump_edges():
        pass
    assert not v.blocked
    assert v.stats["blocked_events"] == 0


def test_rate_limiter_skipped_on_fast_paths(tmp_path):
    """The HTTP rate limiter is consumed only for real HTTP validations —
    cached-invalid and duplicate fast paths bypass it
    (validator rate limiter, telegramhelper/validator_rate_limiter.go)."""
    cfg, sm, rw, v, script = mk_scripted(tmp_path, ["valid"])

    class CountingLimiter:
        n = 0

        def acquire(self):
            self.n += 1

    v.rate_limiter = CountingLimiter()
    rw.mark_invalid_channel("badcha")
    sm.add_discovered_channel("dupcha")
    rw.discovered_channels["dupcha"] = {"username": "dupcha"}
    seed_batch(rw, ["badcha", "dupcha", "newcha"])
    v.pump_edges()
    assert v.rate_limiter.n == 1         # only "newcha" hit HTTP
    assert script.calls == 1


def test_probe_gating_and_unblock_event(tmp_path):
    outcomes = ["blocked"] * 5 + ["blocked", "valid"]
    cfg, sm, rw, v, script = mk_scripted(tmp_path, outcomes)
    t = {"now": 0.0}
    v.clock = lambda: t["now"]
    v.probe_interval = 300.0
    seed_batch(rw, ["n%04d" % i for i in range(5)])
    while v.pump_edges():
        pass
    assert v.blocked
    # within the probe interval: no probe, no work
    t["now"] += 100
    assert v.pump_edges() == 0
    assert script.calls == 5
    # past the interval: probe runs but is still blocked
    t["now"] += 300
    assert v.pump_edges() == 0
    assert script.calls == 6
    # next interval: probe succeeds -> unblocked, edges get re-validated
    t["now"] += 300
    v.pump_edges()
    assert not v.blocked
    assert any(e["type"] == "ip_unblocked" for e in rw.access_events)


def test_walkback_rate_forces_walkback_with_valid_channels(tmp_path):
    cfg, sm, rw, v, script = mk_scripted(tmp_path, ["valid", "valid"],
                                         walkback_rate=100)
    sm.add_discovered_channel("backstop")
    bid = seed_batch(rw, ["ccc33", "ddd44"])
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    pages = rw.get_pages(10)
    assert len(pages) == 1 and pages[0].url == "backstop"
    # both valid channels became skipped edges; walkback edge primary
    recs = [e for e in rw.edge_records if e.sequence_id == "seq1"]
    assert sum(1 for e in recs if e.walkback) == 1
    assert {e.destination_channel for e in recs if e.skipped} == \
        {"ccc33", "ddd44"}


def test_batch_stats_flushed_per_source_type(tmp_path):
    cfg, sm, rw, v, script = mk_scripted(tmp_path, ["valid"] * 3)
    sm.add_discovered_channel("backstop")
    bid = rw.open_batch("v1", "srcchan1", "p1", 0, "seqS")
    rw.insert_pending_edge(bid, "v1", "eee55", "srcchan1", "seqS",
                           "mention")
    rw.insert_pending_edge(bid, "v1", "fff66", "srcchan1", "seqS",
                           "mention")
    rw.insert_pending_edge(bid, "v1", "ggg77", "srcchan1", "seqS",
                           "forward")
    rw.close_batch(bid)
    while v.pump_edges():
        pass
    assert v.pump_walkback()
    assert rw.source_type_stats[("v1", "mention")] == 2
    assert rw.source_type_stats[("v1", "forward")] == 1


def test_walkback_exhausted_leaves_batch_for_recovery(tmp_path):
    """No discovered channel to walk back to: the claimed batch is NOT
    completed; stale recovery re-opens it (validator.go:360-487 +
    daprstate.go:4264-4354)."""
    cfg, sm, rw, v, script = mk_scripted(tmp_path, [], walkback_rate=100)
    # an empty closed batch: no valid channels -> forced walkback
    bid = rw.open_batch("v1", "srcchan1", "p1", 0, "seqE")
    rw.close_batch(bid)
    assert not v.pump_walkback()        # WalkbackExhausted inside
    b = rw.pending_batches[bid]
    assert b.status == "processing"     # stuck claim
    rw.recover_stale_claims()
    assert rw.pending_batches[bid].status == "closed"
    # once a backstop exists the retried batch completes
    sm.add_discovered_channel("backstop")
    assert v.pump_walkback()
    assert rw.pending_batches[bid].status == "completed"


# ---------- synthetic title-rule matrix (channelvalidator.go:132-153) ----

def _html(title, noindex=False):
    meta = b'<meta name="robots" content="noindex, nofollow">' \
        if noindex else b""
    return (b"<html><head><title>" + title + b"</title>" + meta +
            b"</head><body>x</body></html>")


@pytest.mark.parametrize("title,noindex,status,reason", [
    (b"Telegram: View @somechan", False, "valid", ""),
    (b"View @somechan", False, "valid", ""),
    (b"Telegram: Contact @someone", True, "invalid",
     "not_found"),
    (b"Contact @someone", True, "invalid", "not_found"),
    (b"Telegram: Contact @someone", False, "not_channel",
     "not_supergroup"),
    (b"Telegram Messenger", False, "invalid", "not_found"),
    (b"Some Random Page", False, "invalid", "unrecognized"),
    (b"", False, "invalid", "unrecognized"),
])
def test_title_rule_matrix(title, noindex, status, reason):
    r = parse_channel_html(_html(title, noindex))
    assert r.status == status
    assert r.reason == reason


def test_body_cap_64kb():
    """Only the first 64KB are parsed (channelvalidator.go:103 cap);
    a title appearing after the cap never classifies."""
    filler = b"<!-- " + b"x" * (64 * 1024) + b" -->"
    body = filler + _html(b"Telegram: View @late")
    r = parse_channel_html(body)
    assert r.status == "invalid" and r.reason == "unrecognized"


def test_missing_title_tag():
    r = parse_channel_html(b"<html><body>no title here</body></html>")
    assert r.status == "invalid" and r.reason == "unrecognized"
