"""Distributed layer tests: queues, contracts, orchestrator+worker rounds,
reassignment, plus a real multi-process gloo all-gather (world_size 2).

Mirrors reference distributed/{messages,pubsub,integration}_test.go and
orchestrator/worker tests."""
import os
import threading
import time

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager
from crawler_amd.feed import FeedConfig, SyntheticFeed
from crawler_amd.feed.client import ConnectionPool
from crawler_amd.parallel import messages as M
from crawler_amd.parallel.orchestrator import Orchestrator
from crawler_amd.parallel.queue import Heartbeats, InMemoryStore, StoreQueue
from crawler_amd.parallel.worker import Worker


# ---------- contracts ----------

def test_work_item_roundtrip_and_validation():
    item = M.WorkItem(id="i1", url="chan", depth=1, crawl_id="c1",
                      trace_id=M.new_trace_id())
    item.validate()
    got = M.WorkItem.from_json(item.to_json())
    assert got == item
    with pytest.raises(ValueError):
        M.WorkItem(url="chan", crawl_id="c1").validate()
    with pytest.raises(ValueError):
        M.WorkItem(id="x", crawl_id="c1").validate()


def test_trace_ids_unique():
    assert M.new_trace_id() != M.new_trace_id()


# ---------- store queue ----------

def test_queue_fifo_and_exactly_once():
    store = InMemoryStore()
    q = StoreQueue(store, "t1")
    for i in range(5):
        q.publish(f"m{i}")
    assert q.size() == 5
    got = [q.claim() for _ in range(5)]
    assert got == [f"m{i}" for i in range(5)]
    assert q.claim(timeout_s=0.0) is None


def test_queue_concurrent_claims_no_duplicates():
    store = InMemoryStore()
    q = StoreQueue(store, "t2")
    for i in range(200):
        q.publish(str(i))
    claimed = []
    lock = threading.Lock()

    def worker():
        while True:
            v = q.claim(timeout_s=0.0)
            if v is None:
                return
            with lock:
                claimed.append(v)

    threads = [threading.Thread(target=worker) for _ in range(8)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    assert sorted(map(int, claimed)) == list(range(200))


def test_heartbeats_offline_detection():
    store = InMemoryStore()
    hb = Heartbeats(store)
    hb.register("w1")
    hb.register("w2")
    hb.beat("w1")
    hb.beat("w2")
    assert hb.offline_workers(timeout_s=10) == []
    # age w2's beat artificially
    ts, st = hb.last_seen("w2")
    store.set("hb/w2", f"{ts - 600}|active")
    assert hb.offline_workers(timeout_s=300) == ["w2"]


# ---------- orchestrator + worker integration (in-proc, like the
# reference's integration_test.go which shares a mock state manager) ----

def mk_dist_env(tmp_path, sampling="snowball", max_depth=1, n_workers=2):
    cfg = CrawlerConfig(
        crawl_id="d1", storage_root=str(tmp_path), min_users=1,
        sampling_method=sampling, max_depth=max_depth,
        disable_rate_limits=True,
    )
    feed = SyntheticFeed(FeedConfig(seed=21, universe=200,
                                    posts_per_channel=40))
    store = InMemoryStore()
    sm = LocalStateManager(cfg)
    orch = Orchestrator(cfg, sm, store, worker_timeout_s=300)
    workers = []
    for i in range(n_workers):
        pool = ConnectionPool(feed, 1, cfg.rate_limit,
                              posts_per_channel=40,
                              disable_rate_limits=True)
        wcfg = CrawlerConfig(
            crawl_id="d1", storage_root=str(tmp_path / f"w{i}"),
            min_users=1, sampling_method=sampling, max_depth=max_depth,
            disable_rate_limits=True,
        )
        workers.append(Worker(f"w{i}", wcfg, pool, store))
    return cfg, orch, workers, store


def drive(orch, workers, rounds=200):
    for _ in range(rounds):
        orch.distribute()
        for w in workers:
            w.run_once(timeout_s=0.0)
        orch.pump_results()
        if orch.done:
            break
    return orch


def test_orchestrator_worker_snowball_crawl(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path)
    orch.sm.initialize(["c0000000001"])
    drive(orch, workers)
    assert orch.done
    assert orch.stats["results"] >= 1
    # depth advanced into discovered pages
    assert orch.sm.get_max_depth() >= 1
    layer1 = orch.sm.get_layer_by_depth(1)
    assert layer1 and all(
        p.status in ("fetched", "deadend", "error") for p in layer1
    )
    # both workers processed something
    assert sum(w.processed for w in workers) == orch.stats["results"]


def test_work_split_across_workers(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path, max_depth=1)
    orch.sm.initialize(["c%010d" % i for i in range(8)])
    drive(orch, workers)
    assert orch.done
    assert all(w.processed > 0 for w in workers)


def test_reassignment_after_timeout(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path, sampling="channel",
                                            n_workers=1)
    t = {"now": 0.0}
    orch.clock = lambda: t["now"]
    orch.worker_timeout_s = 100
    orch.sm.initialize(["c0000000001"])
    orch.distribute()
    assert orch.work_q.size() == 1
    # worker claims the item but "dies" (never publishes a result)
    raw = orch.work_q.claim()
    assert raw is not None
    t["now"] = 200.0
    n = orch.check_worker_health()
    assert n == 1
    item = M.WorkItem.from_json(orch.work_q.claim())
    assert item.retry_count == 1
    assert item.priority == M.PRIORITY_HIGH


def test_error_pages_marked(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path, sampling="channel")
    orch.sm.initialize(["c0000009999"])  # outside universe -> 400
    drive(orch, workers)
    page = orch.sm.get_layer_by_depth(0)[0]
    assert page.status == "error"


def test_poison_pill_stops_workers(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path, n_workers=1)
    orch.broadcast_stop(1)
    assert workers[0].run_once(timeout_s=0.0) is False


# ---------- real multi-process gloo collective (world_size 2) ----------

def _gloo_worker(rank, world, port, results_dir):
    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    # simulate the per-step discovery exchange of bench.py: all-gather
    # fixed-size hash buffers and union
    local = torch.zeros(8, dtype=torch.int64)
    local[: 2 + rank] = torch.arange(1, 3 + rank) + rank * 100
    gathered = [torch.empty_like(local) for _ in range(world)]
    dist.all_gather(gathered, local)
    union = set()
    for g in gathered:
        union |= {int(x) for x in g[g != 0]}
    with open(os.path.join(results_dir, f"r{rank}.txt"), "w") as f:
        f.write(",".join(map(str, sorted(union))))
    dist.destroy_process_group()


def test_gloo_discovery_allgather_two_procs(tmp_path):
    import torch.multiprocessing as mp

    port = 29712
    ctx = mp.spawn(
        _gloo_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True
    )
    r0 = (tmp_path / "r0.txt").read_text()
    r1 = (tmp_path / "r1.txt").read_text()
    assert r0 == r1
    assert "101" in r0 and "1" in r0


def _names_worker(rank, world, port, results_dir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from crawler_amd.parallel import collectives as C

    mine = [f"rank{rank}chan{i}" for i in range(2 + rank)]
    merged = C.rows_to_names(C.allgather_rows(
        C.names_to_rows(mine), dist, world))
    with open(os.path.join(results_dir, f"n{rank}.txt"), "w") as f:
        f.write(",".join(sorted(merged)))
    dist.destroy_process_group()


def test_gpu_engine_name_exchange_two_procs(tmp_path):
    """The multi-rank discovered-name exchange used by the GPU snowball
    engine, run on gloo CPU (the RCCL path shares this code)."""
    import torch.multiprocessing as mp

    mp.spawn(_names_worker, args=(2, 29713, str(tmp_path)), nprocs=2,
             join=True)
    r0 = (tmp_path / "n0.txt").read_text()
    assert r0 == (tmp_path / "n1.txt").read_text()
    assert "rank0chan0" in r0 and "rank1chan2" in r0


def test_work_item_config_propagates_to_worker(tmp_path):
    """WorkItemConfig overrides the worker's base config
    (worker.go:302-381): the orchestrator's max_posts cap applies."""
    cfg, orch, workers, store = mk_dist_env(tmp_path, sampling="channel",
                                            n_workers=1)
    w = workers[0]
    item = M.WorkItem(id="wi1", url="c0000000001", depth=0, crawl_id="d1",
                      config={"max_posts": 5, "min_users": 1})
    res = w.process_item(item)
    assert res.status == M.STATUS_SUCCESS
    assert res.posts_stored == 5  # capped by the item config, not base cfg


def test_retry_results_republished_then_succeed(tmp_path):
    """A transient (FLOOD_WAIT-class) failure republishes the item at
    high priority; the page completes on the retry, not as an error."""
    cfg, orch, workers, store = mk_dist_env(tmp_path, n_workers=1,
                                            sampling="channel")
    orch.sm.initialize(["c0000000001"])
    w = workers[0]
    fail_once = {"left": 1}
    real = w.process_item

    def flaky(item):
        if fail_once["left"] > 0:
            fail_once["left"] -= 1
            return M.WorkResult(
                work_item_id=item.id, worker_id=w.worker_id,
                status=M.STATUS_RETRY, error="[429] FLOOD_WAIT_5",
                page_status="error", trace_id=item.trace_id,
            )
        return real(item)

    w.process_item = flaky
    drive(orch, workers)
    assert orch.done
    assert orch.stats["retried"] == 1
    assert orch.stats["errors"] == 0
    page = [p for p in orch.sm.pages.values()][0]
    assert page.status == "fetched"


def test_retry_exhaustion_marks_error(tmp_path):
    cfg, orch, workers, store = mk_dist_env(tmp_path, n_workers=1,
                                            sampling="channel")
    orch.sm.initialize(["c0000000001"])
    w = workers[0]

    def always_retry(item):
        return M.WorkResult(
            work_item_id=item.id, worker_id=w.worker_id,
            status=M.STATUS_RETRY, error="[429] FLOOD_WAIT_5",
            page_status="error", trace_id=item.trace_id,
        )

    w.process_item = always_retry
    drive(orch, workers)
    assert orch.done
    assert orch.stats["retried"] == orch.MAX_RETRIES
    assert orch.stats["errors"] == 1
    page = [p for p in orch.sm.pages.values()][0]
    assert page.status == "error"


def test_depth_advance_waits_for_processing_pages(tmp_path):
    """_maybe_advance must not advance or complete while any page of the
    current layer is processing (orchestrator.go:315-383 barrier)."""
    cfg, orch, workers, store = mk_dist_env(tmp_path, n_workers=1)
    orch.sm.initialize(["c0000000001", "c0000000002"])
    orch.distribute()
    # worker processes only ONE of the two queued items
    workers[0].run_once(timeout_s=0.0)
    orch.pump_results()
    assert not orch.done
    assert orch.current_depth == 0
    # finish the second
    workers[0].run_once(timeout_s=0.0)
    orch.pump_results()
    # both done; layer 0 complete -> advance (snowball found channels)
    assert orch.current_depth in (0, 1)  # advances iff a next layer exists
    if orch.current_depth == 0:
        assert orch.done


def test_work_result_count_only_roundtrip():
    """discovered_count survives the JSON hop (orchestrated path sends
    counts; names travel via the RCCL all-gather instead)."""
    r = M.WorkResult(work_item_id="w1", worker_id="r0",
                     posts_stored=5, discovered_count=123456)
    r2 = M.WorkResult.from_json(r.to_json())
    assert r2.discovered_count == 123456
    assert r2.discovered == []
    # old-style payloads (no field) still parse
    import json as _json
    d = _json.loads(r.to_json())
    del d["discovered_count"]
    r3 = M.WorkResult.from_json(_json.dumps(d))
    assert r3.discovered_count == 0


def test_queue_overclaim_ticket_stashed_and_honored():
    """A claim that races past the published tail holds its ticket and
    consumes the matching element once it is published — nothing lost,
    nothing blocked (the deadlock class fixed in StoreQueue)."""
    from crawler_amd.parallel.queue import InMemoryStore, StoreQueue

    store = InMemoryStore()
    q = StoreQueue(store, "t-oc")
    q.publish("a")
    assert q.claim(timeout_s=0.0) == "a"
    # force the race: bump the claim counter past tail manually, then
    # construct the stash through a real claim attempt
    store.add("t-oc/claim", 0)
    # drain on empty: returns None fast, takes a stash ticket internally
    # only when counters race; simulate by direct counter manipulation
    q2 = StoreQueue(store, "t-oc2")
    # claim on empty -> None, no ticket (claim counter untouched)
    assert q2.claim(timeout_s=0.0) is None
    assert store.add("t-oc2/claim", 0) == 0
    # publish-before-tail: payload visible before tail bumps
    idx = store.add("t-oc2/pub", 1) - 1
    store.set(f"t-oc2/{idx}", "x")
    # tail not yet bumped -> still invisible
    assert q2.claim(timeout_s=0.0) is None
    store.add("t-oc2/tail", 1)
    assert q2.claim(timeout_s=0.0) == "x"


def test_queue_stash_survives_slow_publish():
    """Two concurrent claimers + one element: the loser's stashed ticket
    is honored by the NEXT publish."""
    import threading

    from crawler_amd.parallel.queue import InMemoryStore, StoreQueue

    store = InMemoryStore()
    qa = StoreQueue(store, "t-st")
    qb = StoreQueue(store, "t-st")
    qa.publish("first")
    got = []

    def racer(q):
        got.append(q.claim(timeout_s=1.0))

    # drive the race artificially: qb grabs a ticket past the tail
    store.add("t-st/claim", 0)
    ta = threading.Thread(target=racer, args=(qa,))
    tb = threading.Thread(target=racer, args=(qb,))
    ta.start(); tb.start()
    import time
    time.sleep(0.1)
    qa.publish("second")  # satisfies whichever racer stashed
    ta.join(); tb.join()
    assert sorted(x for x in got if x) == ["first", "second"]


def test_offline_workers_tracked_and_reassigned(tmp_path):
    """A worker that stops heartbeating is marked offline and its
    in-flight item is republished at high priority with retry_count++
    (orchestrator.go:493-559)."""
    clock = {"t": 1000.0}
    cfg, orch, workers, store = mk_dist_env(tmp_path, n_workers=1,
                                            sampling="channel")
    orch.clock = lambda: clock["t"]
    orch.sm.initialize(["c0000000001"])
    orch.heartbeats.register("w0")
    store.set("hb/w0", f"{clock['t']}|active")
    orch.distribute()
    assert len(orch.in_flight) == 1
    # 6 minutes pass with no beat and no result
    clock["t"] += 360
    n = orch.check_worker_health()
    assert n == 1
    assert "w0" in orch.offline
    item = next(iter(orch.in_flight.values()))[0]
    assert item.retry_count == 1
    assert item.priority == M.PRIORITY_HIGH
    # the republished copy is claimable
    raw = orch.work_q.claim(timeout_s=0.0)
    assert raw is not None


def test_cli_orchestrator_worker_two_processes(tmp_path):
    """Real cross-process run of --mode orchestrator + --mode worker over
    a TCPStore on 127.0.0.1 (the reference's pod topology, one host)."""
    import os
    import subprocess
    import sys

    port = 29877
    env = {**os.environ, "MASTER_ADDR": "127.0.0.1",
           "MASTER_PORT": str(port), "CRAWLER_NUM_WORKERS": "1"}
    common = [
        "--storage-root", str(tmp_path), "--crawl-id", "dist1",
        "--synthetic-universe", "100", "--synthetic-posts", "10",
        "--disable-rate-limits", "--min-users", "1",
        "--sampling", "channel", "--urls", "c0000000001,c0000000002",
    ]
    orch = subprocess.Popen(
        [sys.executable, "-m", "crawler_amd.cli", "--mode",
         "orchestrator"] + common,
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True,
    )
    worker = subprocess.Popen(
        [sys.executable, "-m", "crawler_amd.cli", "--mode", "worker",
         "--worker-id", "wA"] + common,
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        out_o, _ = orch.communicate(timeout=90)
        out_w, _ = worker.communicate(timeout=90)
    finally:
        for p in (orch, worker):
            if p.poll() is None:
                p.kill()
    assert orch.returncode == 0, out_o
    assert worker.returncode == 0, out_w
    assert "orchestrator complete" in out_o
    # the worker's state manager stored the posts
    jsonls = list(tmp_path.rglob("posts.jsonl"))
    assert jsonls, out_w
    total = sum(p.read_bytes().count(b"\n") for p in jsonls)
    assert total == 2 * 10


def test_queue_drain_returns_fifo_and_empties():
    from crawler_amd.parallel.queue import InMemoryStore, StoreQueue

    store = InMemoryStore()
    q = StoreQueue(store, "t-dr")
    for i in range(7):
        q.publish(f"m{i}")
    assert q.size() == 7
    got = q.drain()
    assert got == [f"m{i}" for i in range(7)]
    assert q.size() == 0
    assert q.drain() == []


def test_gpu_dist_crawl_dry_run_world2(tmp_path):
    """World>1-shaped dry run of scripts/gpu_dist_crawl.py: the exact
    launcher wiring (torchrun -> TCPStore queue -> OrchestratedCrawl ->
    collectives) with --fake-engine on gloo."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29752", os.path.join(repo, "scripts",
         "gpu_dist_crawl.py"), "--fake-engine", "--seeds", "12",
         "--posts", "50", "--max-depth", "1", "--max-pages", "100",
         "--storage", str(tmp_path), "--store-port", "29773"],
        capture_output=True, text=True, timeout=120, cwd=repo,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    import json

    line = [l for l in out.stdout.splitlines()
            if l.startswith("{")][-1]
    stats = json.loads(line)
    assert stats["world"] == 2
    assert stats["layers"] == 2
    prog_path = os.path.join(str(tmp_path), "r0", "dist-crawl",
                             "progress.json")
    with open(prog_path) as f:
        assert json.load(f)["status"] == "completed"
