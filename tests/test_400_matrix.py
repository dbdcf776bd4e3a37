"""400-replacement + walk-decision matrix.

Mirrors the reference's heaviest suites (SURVEY.md §4:
crawl/runner_400_test.go — 36 tests over seed/forward/walkback cases —
and the walkback decision of runner.go:1459-1541): every branch of
handle_400_replacement, walk_tail and validate_outlinks is pinned here
against the real in-process stores."""
import random
import uuid

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page, RandomWalkStore
from crawler_amd.engine import errors as E
from crawler_amd.engine import randomwalk as RW
from crawler_amd.engine.state import EdgeRecord
from crawler_amd.feed.client import TelegramAPIError
from crawler_amd.ops.golden import DiscoveredLink


def mk(tmp_path, **kw):
    cfg = CrawlerConfig(crawl_id="m1", storage_root=str(tmp_path),
                        disable_rate_limits=True, **kw)
    sm = LocalStateManager(cfg)
    rw = RandomWalkStore()
    return cfg, sm, rw


def seed_discovered(sm, names):
    for n in names:
        sm.add_discovered_channel(n)


def edges_for(rw, seq):
    return [e for e in rw.edge_records if e.sequence_id == seq]


def mk_page(url, seq=None, depth=3, parent="par"):
    return Page(id=str(uuid.uuid4()), url=url, depth=depth,
                parent_id=parent, sequence_id=seq or str(uuid.uuid4()),
                status="unfetched")


# ---------------------------------------------------------------- 400 matrix

def test_400_marks_both_invalid_caches(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha", "bravo"])
    rw.upsert_seed_channel("deadchan", 7)
    p = mk_page("deadchan")
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(1))
    assert rw.is_invalid_channel("deadchan")
    # seed-channel row flipped invalid (runner.go:161-171)
    assert not any(r["username"] == "deadchan"
                   for r in rw.load_seed_channels())


def test_400_no_edge_non_seed_walks_back_from_self(tmp_path):
    """No edge record + not a seed: replacement walks back from the dead
    channel itself (runner.go:219-234)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha", "bravo", "candy"])
    p = mk_page("deadchan", seq="seq-1")
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(2))
    pages = rw.get_pages(10)
    assert len(pages) == 1
    repl = pages[0]
    assert repl.url in {"alpha", "bravo", "candy"}
    assert repl.depth == p.depth          # same depth, not +1
    assert repl.sequence_id != p.sequence_id  # fresh chain
    # walkback edge recorded on the OLD sequence
    edges = edges_for(rw, "seq-1")
    assert len(edges) == 1 and edges[0].walkback
    assert edges[0].source_channel == "deadchan"


def test_400_seed_channel_replacement(tmp_path):
    """Dead channel IS a seed: replaced by a random VALID seed from
    seed_channels — fresh chain, NO edge record
    (handle400SeedReplacement, runner.go:263-284; the seed pick
    excludes invalidated rows per daprstate.go:4181-4196)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha", "bravo"])
    sm.upsert_seed_channel_chat_id("deadseed", 42)
    for u in ("deadseed", "seedx", "seedy"):
        rw.upsert_seed_channel(u)
    n_edges = rw.edge_count()
    p = mk_page("deadseed", seq="seq-s")
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(3))
    pages = rw.get_pages(10)
    # deadseed was invalidated by the handler itself, so only the
    # other seed rows are eligible — never the discovered set
    assert len(pages) == 1 and pages[0].url in {"seedx", "seedy"}
    assert pages[0].sequence_id != "seq-s"
    assert rw.edge_count() == n_edges  # no edge record written


def test_400_seed_replacement_empty_table_exhausts(tmp_path):
    """No eligible seed rows -> the error path (reference
    GetRandomSeedChannel errors on an empty table)."""
    import pytest

    from crawler_amd.engine import errors as E

    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha"])
    sm.upsert_seed_channel_chat_id("deadseed", 42)
    p = mk_page("deadseed", seq="seq-s")
    with pytest.raises(E.WalkbackExhausted):
        RW.handle_400_replacement(sm, rw, p, cfg, random.Random(3))


def test_400_walkback_edge_rewalks_from_edge_source(tmp_path):
    """Page was reached BY a walkback: pick a new walkback target from the
    edge's source channel (runner.go:240-252)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha", "bravo"])
    p = mk_page("deadchan", seq="seq-w")
    rw.save_edge_records([EdgeRecord(
        destination_channel="deadchan", source_channel="origin",
        walkback=True, skipped=False, sequence_id="seq-w")])
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(4))
    # dead edge deleted, new walkback edge from "origin"
    assert rw.get_edge_record("seq-w", "deadchan") is None
    edges = edges_for(rw, "seq-w")
    assert len(edges) == 1
    assert edges[0].source_channel == "origin" and edges[0].walkback
    assert rw.get_pages(10)[0].url in {"alpha", "bravo"}


def test_400_forward_edge_promotes_skipped_sibling(tmp_path):
    """Forward edge with skipped siblings: promote one (same sequence,
    same depth) instead of walking back (runner.go:254-283)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha"])
    p = mk_page("deadchan", seq="seq-f")
    rw.save_edge_records([
        EdgeRecord(destination_channel="deadchan", source_channel="src",
                   walkback=False, skipped=False, sequence_id="seq-f"),
        EdgeRecord(destination_channel="sib1", source_channel="src",
                   walkback=False, skipped=True, sequence_id="seq-f"),
        EdgeRecord(destination_channel="sib2", source_channel="src",
                   walkback=False, skipped=True, sequence_id="seq-f"),
    ])
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(5))
    pages = rw.get_pages(10)
    assert len(pages) == 1
    repl = pages[0]
    assert repl.url in {"sib1", "sib2"}
    assert repl.sequence_id == "seq-f"   # chain continues
    assert repl.depth == p.depth
    promoted = rw.get_edge_record("seq-f", repl.url)
    assert promoted is not None and not promoted.skipped


def test_400_forward_edge_no_siblings_walks_back_from_source(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha", "bravo"])
    p = mk_page("deadchan", seq="seq-n")
    rw.save_edge_records([EdgeRecord(
        destination_channel="deadchan", source_channel="src",
        walkback=False, skipped=False, sequence_id="seq-n")])
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(6))
    edges = edges_for(rw, "seq-n")
    assert len(edges) == 1 and edges[0].walkback
    assert edges[0].source_channel == "src"


def test_400_skipped_sibling_excludes_dead_channel(tmp_path):
    """The dead channel itself is never re-picked even if it appears as a
    skipped edge (exclude set, runner.go:256)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["alpha"])
    p = mk_page("deadchan", seq="seq-x")
    rw.save_edge_records([
        EdgeRecord(destination_channel="deadchan", source_channel="src",
                   walkback=False, skipped=False, sequence_id="seq-x"),
        EdgeRecord(destination_channel="deadchan", source_channel="src",
                   walkback=False, skipped=True, sequence_id="seq-x"),
    ])
    RW.handle_400_replacement(sm, rw, p, cfg, random.Random(7))
    # falls through to walkback (no usable sibling)
    edges = [e for e in edges_for(rw, "seq-x") if e.walkback]
    assert len(edges) == 1


def test_400_exhausted_when_no_discovered_channels(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    p = mk_page("deadchan")
    with pytest.raises(E.WalkbackExhausted):
        RW.handle_400_replacement(sm, rw, p, cfg, random.Random(8))


def test_400_walkback_excludes_dead_channel_target(tmp_path):
    """Walkback target can never be the dead channel (exclude {p.url})."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["deadchan"])  # the ONLY discovered channel
    p = mk_page("deadchan")
    with pytest.raises(E.WalkbackExhausted):
        RW.handle_400_replacement(sm, rw, p, cfg, random.Random(9))


# ------------------------------------------------------------- walkback pick

def test_pick_walkback_excludes_and_source(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["aaa", "bbb"])
    # only "aaa" (source) and "bbb" (excluded) exist -> all 10 attempts
    # fail and the pick exhausts
    with pytest.raises(E.WalkbackExhausted):
        RW.pick_walkback_channel(sm, "aaa", {"bbb"}, random.Random(10))
    seed_discovered(sm, ["ccc"])
    got = RW.pick_walkback_channel(sm, "aaa", {"bbb"}, random.Random(11))
    assert got == "ccc"


def test_pick_walkback_max_attempts(tmp_path):
    """10 attempts then WalkbackExhausted (runner.go:118-139)."""
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["only"])
    with pytest.raises(E.WalkbackExhausted):
        RW.pick_walkback_channel(sm, "only", None, random.Random(12))


# ----------------------------------------------------------- walk_tail rules

def test_walk_tail_forward_keeps_sequence_and_skips_rest(tmp_path):
    cfg, sm, rw = mk(tmp_path, walkback_rate=0)
    owner = mk_page("src", seq="chain-1", depth=2)
    nxt = RW.walk_tail(owner, {"aa1aa": True, "bb2bb": True, "cc3cc": True},
                       sm, rw, cfg, random.Random(13))
    assert nxt.sequence_id == "chain-1"       # forward: chain continues
    assert nxt.depth == 3
    edges = edges_for(rw, "chain-1")
    fwd = [e for e in edges if not e.skipped]
    skipped = [e for e in edges if e.skipped]
    assert len(fwd) == 1 and fwd[0].destination_channel == nxt.url
    assert not fwd[0].walkback
    assert {e.destination_channel for e in skipped} == \
        {"aa1aa", "bb2bb", "cc3cc"} - {nxt.url}
    assert all(e.sequence_id == "chain-1" for e in edges)


def test_walk_tail_walkback_rate_100_always_walks_back(tmp_path):
    cfg, sm, rw = mk(tmp_path, walkback_rate=100)
    seed_discovered(sm, ["backstop"])
    owner = mk_page("src", seq="chain-2")
    nxt = RW.walk_tail(owner, {"aa1aa": True}, sm, rw, cfg,
                       random.Random(14))
    assert nxt.url == "backstop"
    assert nxt.sequence_id != "chain-2"       # walkback: fresh chain
    edges = edges_for(rw, "chain-2")
    wb = [e for e in edges if e.walkback]
    # the walkback edge carries the OWNER's sequence id
    assert len(wb) == 1 and wb[0].sequence_id == "chain-2"
    # the unused candidate became a skipped edge
    assert any(e.skipped and e.destination_channel == "aa1aa"
               for e in edges)


def test_walk_tail_empty_candidates_forces_walkback(tmp_path):
    cfg, sm, rw = mk(tmp_path, walkback_rate=0)
    seed_discovered(sm, ["backstop"])
    owner = mk_page("src", seq="chain-3")
    nxt = RW.walk_tail(owner, {}, sm, rw, cfg, random.Random(15))
    assert nxt.url == "backstop" and nxt.sequence_id != "chain-3"


def test_walk_tail_walkback_excludes_candidates(tmp_path):
    """Walkback target may not be one of this page's own candidates
    (runner.go:1474 exclude=newChannels)."""
    cfg, sm, rw = mk(tmp_path, walkback_rate=100)
    seed_discovered(sm, ["aa1aa"])  # only discovered == the candidate
    owner = mk_page("src", seq="chain-4")
    with pytest.raises(E.WalkbackExhausted):
        RW.walk_tail(owner, {"aa1aa": True}, sm, rw, cfg,
                     random.Random(16))


def test_walk_tail_rate_boundary_inclusive(tmp_path):
    """rand(1..100) <= rate: rate=1 walks back only when rnd==1."""
    cfg, sm, rw = mk(tmp_path, walkback_rate=1)
    seed_discovered(sm, ["backstop"])

    class FixedRng(random.Random):
        def __init__(self, v):
            super().__init__(0)
            self.v = v

        def randint(self, a, b):
            return self.v

    owner = mk_page("src", seq="chain-5")
    nxt = RW.walk_tail(owner, {"aa1aa": True}, sm, rw, cfg, FixedRng(1))
    assert nxt.url == "backstop"               # rnd==1 <= rate==1
    owner2 = mk_page("src", seq="chain-6")
    nxt2 = RW.walk_tail(owner2, {"aa1aa": True}, sm, rw, cfg, FixedRng(2))
    assert nxt2.url == "aa1aa"                 # rnd==2 > rate==1


# ------------------------------------------------------ validate_outlinks

class ScriptedClient:
    """search_public_chat scripted per name (mock-TDLib pattern,
    crawl/mocks_test.go)."""

    def __init__(self, script):
        self.script = script
        self.calls = []

    def search_public_chat(self, name):
        self.calls.append(name)
        act = self.script.get(name)
        if isinstance(act, Exception):
            raise act

        class Info:
            chat_id = abs(hash(name)) % 10**9
        return Info()


def links(*names):
    return [DiscoveredLink(name=n, source_type="message_text")
            for n in names]


def test_outlinks_filter_matrix(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    sm.add_discovered_channel("known")
    rw.mark_invalid_channel("badcached")
    client = ScriptedClient({
        "valid": None,
        "gone400": TelegramAPIError("400 USERNAME_NOT_OCCUPIED"),
        "shortban": TelegramAPIError("[429] FLOOD_WAIT_5"),
        "weird": TelegramAPIError("Internal error"),
    })
    got = RW.validate_outlinks(
        links("selfchan", "bad!", "abc", "badcached", "known", "valid",
              "gone400", "shortban", "weird"),
        "selfchan", sm, rw, client, cfg)
    # self, regex-invalid ("bad!"), too-short ("abc"), cached-invalid all
    # dropped WITHOUT an API call; discovered fast path skips the call too
    assert set(client.calls) == {"valid", "gone400", "shortban", "weird"}
    assert got == {"known": True, "valid": True}
    assert rw.is_invalid_channel("gone400")
    assert not rw.is_invalid_channel("shortban")  # transient, not cached
    # the validated channel joined the discovered set + seed cache
    assert sm.is_discovered_channel("valid")


def test_outlinks_long_flood_retires(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    client = ScriptedClient(
        {"victim": TelegramAPIError("[429] FLOOD_WAIT_3600")})
    with pytest.raises(E.FloodWaitRetire):
        RW.validate_outlinks(links("victim"), "src", sm, rw, client, cfg)


def test_outlinks_cached_chat_id_fast_path(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    sm.upsert_seed_channel_chat_id("cachedc", 777)
    client = ScriptedClient({})
    got = RW.validate_outlinks(links("cachedc"), "src", sm, rw, client,
                               cfg)
    assert got == {"cachedc": True} and client.calls == []


def test_tandem_tail_skips_self_and_invalid(tmp_path):
    """Tandem pending-edge stream applies the same pre-filters as
    standard mode: self-links, regex-invalid names and cached-invalid
    channels never become pending edges (runner.go:1252-1306)."""
    cfg, sm, rw = mk(tmp_path)
    rw.mark_invalid_channel("badcha")
    owner = mk_page("selfchan", seq="sq-t")
    links = [DiscoveredLink(name=n, source_type="message_text")
             for n in ("selfchan", "ab", "badcha", "goodch1", "goodch2")]
    bid = RW.tandem_tail(owner, links, sm, rw, cfg, random.Random(1))
    assert bid is not None
    edges = rw.edges_of_batch(bid)
    assert {e.destination_channel for e in edges} == \
        {"goodch1", "goodch2"}
    assert rw.pending_batches[bid].status == "closed"


def test_tandem_tail_forced_walkback_when_nothing_survives(tmp_path):
    cfg, sm, rw = mk(tmp_path)
    seed_discovered(sm, ["backstop"])
    owner = mk_page("selfchan", seq="sq-u")
    links = [DiscoveredLink(name="selfchan", source_type="message_text")]
    bid = RW.tandem_tail(owner, links, sm, rw, cfg, random.Random(2))
    assert bid is None                      # no batch opened
    pages = rw.get_pages(5)
    assert len(pages) == 1 and pages[0].url == "backstop"
    wb = [e for e in edges_for(rw, "sq-u") if e.walkback]
    assert len(wb) == 1
