"""State management tests (reference: state/{base,storageproviders}_test
coverage areas; claim semantics of state/daprstate.go:3944-4391)."""
import datetime as dt

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import (
    EdgeRecord,
    LocalStateManager,
    Page,
    RandomWalkStore,
)
from crawler_amd.models.post import Post


def mk_sm(tmp_path, **kw):
    cfg = CrawlerConfig(crawl_id="testcrawl", storage_root=str(tmp_path), **kw)
    return LocalStateManager(cfg)


def pages(urls, depth=0, status="unfetched"):
    return [Page(url=u, depth=depth, status=status) for u in urls]


def test_initialize_seeds_layer0(tmp_path):
    sm = mk_sm(tmp_path)
    sm.initialize(["chan_a", "chan_b"])
    layer = sm.get_layer_by_depth(0)
    assert {p.url for p in layer} == {"chan_a", "chan_b"}
    assert all(p.status == "unfetched" for p in layer)


def test_add_layer_dedups_urls(tmp_path):
    sm = mk_sm(tmp_path)
    sm.initialize(["chan_a"])
    added = sm.add_layer(pages(["chan_a", "chan_c"], depth=1))
    assert len(added) == 1
    assert {p.url for p in sm.get_layer_by_depth(1)} == {"chan_c"}


def test_max_pages_deadend_replacement_budget(tmp_path):
    sm = mk_sm(tmp_path, max_pages=3)
    sm.initialize(["u1", "u2", "u3"])
    # at cap: new pages rejected
    assert sm.add_layer(pages(["u4"], depth=1)) == []
    # mark one deadend -> exactly one replacement allowed
    p = sm.get_layer_by_depth(0)[0]
    p.status = "deadend"
    sm.update_page(p)
    added = sm.add_layer(pages(["u5", "u6"], depth=1))
    assert len(added) == 1


def test_update_message_status(tmp_path):
    sm = mk_sm(tmp_path)
    sm.initialize(["chan_a"])
    pid = sm.get_layer_by_depth(0)[0].id
    sm.update_message(pid, 100, 5, "fetched")
    sm.update_message(pid, 100, 5, "deleted")
    page = sm.get_page(pid)
    assert len(page.messages) == 1
    assert page.messages[0].status == "deleted"


def test_save_load_roundtrip(tmp_path):
    sm = mk_sm(tmp_path)
    sm.initialize(["chan_a", "chan_b"])
    p = sm.get_layer_by_depth(0)[0]
    p.status = "fetched"
    sm.update_page(p)
    sm.add_layer(pages(["chan_c"], depth=1))
    sm.save_state()

    sm2 = mk_sm(tmp_path)
    assert sm2.load_state()
    assert sm2.get_max_depth() == 1
    statuses = {pg.url: pg.status for pg in sm2.get_layer_by_depth(0)}
    assert statuses[p.url] == "fetched"


def test_progress_json_written(tmp_path):
    import json

    sm = mk_sm(tmp_path)
    sm.initialize(["chan_a", "chan_b"])
    p = sm.get_layer_by_depth(0)[0]
    p.status = "fetched"
    sm.update_page(p)
    sm.save_state()
    with open(tmp_path / "testcrawl" / "progress.json") as f:
        prog = json.load(f)
    assert prog["crawlId"] == "testcrawl"
    assert prog["layers"][0]["total"] == 2
    assert prog["layers"][0]["completed"] == 1


def test_find_incomplete_crawl_rules(tmp_path):
    sm = mk_sm(tmp_path)
    # no state yet
    assert sm.find_incomplete_crawl("testcrawl") == ("", False)
    sm.initialize(["chan_a"])
    sm.save_state()
    exec_id, ok = sm.find_incomplete_crawl("testcrawl")
    assert ok and exec_id
    # completed crawls never resume (daprstate.go:1703-2199)
    sm.update_crawl_metadata("testcrawl", {"status": "completed"})
    sm.save_state()
    assert sm.find_incomplete_crawl("testcrawl") == ("", False)


def test_store_post_jsonl_path(tmp_path):
    sm = mk_sm(tmp_path)
    sm.store_post("mychan", Post(post_link="x", platform_name="Telegram"))
    sm.close()
    path = tmp_path / "testcrawl" / "mychan" / "posts" / "posts.jsonl"
    assert path.exists()
    assert path.read_bytes().count(b"\n") == 1


def test_media_cache_roundtrip(tmp_path):
    sm = mk_sm(tmp_path)
    assert not sm.has_processed_media("m1")
    sm.mark_media_as_processed("m1")
    assert sm.has_processed_media("m1")
    sm.close()
    sm2 = mk_sm(tmp_path)
    sm2.load_state() if (tmp_path / "testcrawl" / "state.json").exists() else None
    # media cache loads independently via load_state; write state to test
    sm3 = mk_sm(tmp_path)
    sm3.initialize(["a12345"])
    sm3.save_state()
    sm3.mark_media_as_processed("m2")
    sm3.close()
    sm4 = mk_sm(tmp_path)
    assert sm4.load_state()
    assert sm4.has_processed_media("m2")


def test_discovered_channels_random_pick(tmp_path):
    sm = mk_sm(tmp_path)
    for n in ["aaaaa", "bbbbb", "ccccc"]:
        sm.add_discovered_channel(n)
    assert sm.is_discovered_channel("aaaaa")
    assert sm.get_random_discovered_channel() in {"aaaaa", "bbbbb", "ccccc"}
    assert len(sm.discovered) == 3


# ---------- RandomWalkStore ----------

def test_page_buffer_crud():
    rw = RandomWalkStore()
    rw.add_page(Page(url="chan_a"))
    rw.add_page(Page(url="chan_b"))
    got = rw.get_pages(10)
    assert len(got) == 2
    rw.delete_pages([got[0].id])
    assert rw.buffer_size() == 1


def test_seed_channel_lifecycle():
    rw = RandomWalkStore()
    rw.upsert_seed_channel("chan_a", 0)
    assert rw.get_channel_last_crawled("chan_a") is None
    rw.mark_channel_crawled("chan_a", 999)
    assert rw.get_channel_last_crawled("chan_a") is not None
    assert len(rw.load_seed_channels()) == 1
    rw.mark_seed_channel_invalid("chan_a")
    assert rw.load_seed_channels() == []  # fresh invalidation filtered


def test_seed_channel_bulk_lazy_rows():
    """Bulk-admitted names are stored as lazy sentinels; every accessor
    sees full-row semantics (materialize-on-touch never escapes)."""
    rw = RandomWalkStore()
    rw.upsert_seed_channels_bulk(["a_chan", "b_chan", "c_chan"])
    # reads on fresh rows behave like default rows
    assert rw.get_channel_last_crawled("a_chan") is None
    rows = rw.load_seed_channels()
    assert {r["username"] for r in rows} == {"a_chan", "b_chan", "c_chan"}
    assert all(r["chat_id"] == 0 and r["last_crawled_at"] is None
               and r["invalidated_at"] is None for r in rows)
    # mutations materialize
    rw.mark_channel_crawled("a_chan", 42)
    assert rw.get_channel_last_crawled("a_chan") is not None
    rw.mark_seed_channel_invalid("b_chan")
    left = {r["username"] for r in rw.load_seed_channels()}
    assert left == {"a_chan", "c_chan"}  # fresh invalidation filtered
    # invalidating an unknown name never creates a row
    rw.mark_seed_channel_invalid("ghost_chan")
    assert "ghost_chan" not in {r["username"]
                                for r in rw.load_seed_channels()}
    # re-upsert of an existing materialized row keeps its state
    rw.upsert_seed_channels_bulk(["a_chan"])
    assert rw.get_channel_last_crawled("a_chan") is not None
    # chat_id upsert on a fresh sentinel materializes with the id
    rw.upsert_seed_channel("c_chan", 7)
    row = [r for r in rw.load_seed_channels()
           if r["username"] == "c_chan"][0]
    assert row["chat_id"] == 7


def test_invalid_channel_ttl():
    rw = RandomWalkStore()
    rw.mark_invalid_channel("badchan")
    assert rw.is_invalid_channel("badchan")
    rw.invalid_channels["badchan"] = (
        dt.datetime.now(dt.timezone.utc) - dt.timedelta(days=40)
    )
    assert not rw.is_invalid_channel("badchan")  # expired


def test_pending_edge_claim_exactly_once():
    rw = RandomWalkStore()
    bid = rw.open_batch("c1", "src", "p1", 0, "seq1")
    for i in range(5):
        rw.insert_pending_edge(bid, "c1", f"dest{i}", "src", "seq1", "mention")
    a = rw.claim_pending_edges(3)
    b = rw.claim_pending_edges(10)
    ids_a = {e.pending_id for e in a}
    ids_b = {e.pending_id for e in b}
    assert len(ids_a) == 3 and len(ids_b) == 2
    assert not (ids_a & ids_b)  # SKIP LOCKED: no double claim


def test_walkback_batch_claim_requires_closed_and_drained():
    rw = RandomWalkStore()
    bid = rw.open_batch("c1", "src", "p1", 0, "seq1")
    rw.insert_pending_edge(bid, "c1", "dest1", "src", "seq1", "url")
    assert rw.claim_walkback_batch() is None  # still open
    rw.close_batch(bid)
    assert rw.claim_walkback_batch() is None  # edge still pending
    edges = rw.claim_pending_edges(10)
    rw.update_pending_edges([(edges[0].pending_id, "valid", "")])
    got = rw.claim_walkback_batch()
    assert got is not None and got.batch_id == bid
    assert rw.claim_walkback_batch() is None  # now processing


def test_batch_poison_detection():
    rw = RandomWalkStore()
    bid = rw.open_batch("c1", "src", "p1", 0, "s")
    rw.close_batch(bid)
    for _ in range(RandomWalkStore.MAX_ATTEMPTS):
        b = rw.claim_walkback_batch()
        assert b is not None
        rw.recover_stale_claims()  # simulate crash: processing -> closed
    assert rw.claim_walkback_batch() is None  # poisoned


def test_stale_recovery_and_orphans():
    rw = RandomWalkStore()
    bid = rw.open_batch("c1", "src", "p1", 0, "s")
    rw.insert_pending_edge(bid, "c1", "dst", "src", "s", "url")
    rw.claim_pending_edges(1)
    ne, nb = rw.recover_stale_claims()
    assert ne == 1
    # orphan edge: batch deleted
    del rw.pending_batches[bid]
    assert rw.delete_orphan_edges() == 1


def test_claim_discovered_channel_exactly_once():
    rw = RandomWalkStore()
    assert rw.claim_discovered_channel("newchan", "c1")
    assert not rw.claim_discovered_channel("newchan", "c2")


def test_skipped_edge_promotion_excludes():
    rw = RandomWalkStore()
    rw.save_edge_records([
        EdgeRecord(destination_channel="d1", source_channel="s",
                   skipped=True),
        EdgeRecord(destination_channel="d2", source_channel="s",
                   skipped=True),
        EdgeRecord(destination_channel="d3", source_channel="s",
                   skipped=False),
    ])
    e = rw.get_random_skipped_edge({"d1"})
    assert e is not None and e.destination_channel == "d2"
    assert rw.get_random_skipped_edge({"d1", "d2"}) is None


def test_concurrent_edge_claims_stress():
    """Race exercise (reference concurrent_test.go pattern):
    8 threads claiming pending edges never double-claim."""
    import threading

    rw = RandomWalkStore()
    bid = rw.open_batch("c1", "src", "p1", 0, "seq")
    for i in range(400):
        rw.insert_pending_edge(bid, "c1", f"d{i}", "src", "seq", "url")
    claimed = []
    lock = threading.Lock()

    def worker():
        while True:
            got = rw.claim_pending_edges(7)
            if not got:
                return
            with lock:
                claimed.extend(e.pending_id for e in got)

    ts = [threading.Thread(target=worker) for _ in range(8)]
    [t.start() for t in ts]
    [t.join() for t in ts]
    assert len(claimed) == 400
    assert len(set(claimed)) == 400


def test_store_file_moves_into_media_layout(tmp_path):
    """StoreFile (state interface parity): move a fetched file into
    crawlID/media/channel/ (storageproviders.go paths)."""
    sm = mk_sm(tmp_path)
    src = tmp_path / "dl.tmp"
    src.write_bytes(b"blobdata")
    dst, name = sm.store_file("chanZ", str(src), "photo1.jpg")
    assert name == "photo1.jpg"
    assert not src.exists()
    assert open(dst, "rb").read() == b"blobdata"
    assert "/media/chanZ/" in dst
    sm.close()


def test_media_cache_ttl_expiry_on_close(tmp_path):
    """30-day media-cache expiry at Close (daprstate.go:1252-1678)."""
    import datetime as dt

    sm = mk_sm(tmp_path)
    sm.mark_media_as_processed("fresh1")
    old = (dt.datetime.now(dt.timezone.utc)
           - dt.timedelta(days=31)).isoformat()
    sm.media_cache["stale1"] = {"id": "stale1", "firstSeen": old}
    sm.media_cache["legacy"] = {"id": "legacy"}  # no timestamp: kept
    sm.close()
    import json
    cache = json.load(open(sm._media_cache_path()))
    sm2 = mk_sm(tmp_path)
    sm2.media_cache = cache
    assert sm2.has_processed_media("fresh1")
    assert not sm2.has_processed_media("stale1")
    assert sm2.has_processed_media("legacy")
    sm2.close()
