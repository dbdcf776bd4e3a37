"""Orchestrated multi-rank crawl logic tests (CPU):
- 2-process gloo + TCPStore run with a stub hot path, verifying dynamic
  chunk claiming, collective discovery union, deterministic layer growth;
- single-rank run with a fake dist."""
import json
import os

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager
from crawler_amd.parallel.orchestrated import OrchestratedCrawl
from crawler_amd.parallel.queue import InMemoryStore


class FakeDist:
    @staticmethod
    def barrier():
        pass

    @staticmethod
    def all_gather(out_list, t):
        out_list[0].copy_(t)


def stub_process(names):
    """Each channel chanNNN discovers chan(N+100) until 300."""
    discovered = []
    for n in names:
        k = int(n[4:])
        if k + 100 < 300:
            discovered.append(f"chan{k + 100:03d}")
    return discovered, len(names) * 10


def test_single_rank_orchestrated(tmp_path):
    cfg = CrawlerConfig(crawl_id="o1", storage_root=str(tmp_path),
                        sampling_method="snowball", max_depth=2,
                        min_users=1)
    sm = LocalStateManager(cfg)
    store = InMemoryStore()
    crawl = OrchestratedCrawl(cfg, sm, store, rank=0, world=1,
                              process_fn=stub_process, chunk_channels=2,
                              dist=FakeDist())
    stats = crawl.run(["chan000", "chan001", "chan002"])
    # depth 0: 3 chans -> discover 100..102; depth 1 -> 200..202; depth 2
    # discoveries (300+) are out of range
    assert stats["layers"] == 3
    assert stats["pages"] == 9
    assert stats["posts"] == 90
    prog = json.loads((tmp_path / "o1" / "progress.json").read_text())
    assert prog["status"] == "completed"
    assert prog["layers"][2]["total"] == 3


def _rank_main(rank, world, port, tmp_dir):
    import torch.distributed as dist
    from torch.distributed import TCPStore

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    store = TCPStore("127.0.0.1", port + 1, is_master=(rank == 0),
                     wait_for_workers=False)
    cfg = CrawlerConfig(crawl_id="o2", storage_root=f"{tmp_dir}/r{rank}",
                        sampling_method="snowball", max_depth=1,
                        min_users=1)
    sm = LocalStateManager(cfg)
    processed = []
    rows_mode = os.environ.get("ORCH_TEST_ROWS") == "1"

    def process(names):
        processed.extend(names)
        d, p = stub_process(names)
        if rows_mode:
            from crawler_amd.parallel import collectives as C

            return C.names_to_rows(d).numpy(), p
        return d, p

    crawl = OrchestratedCrawl(cfg, sm, store, rank, world,
                              process_fn=process, chunk_channels=2,
                              dist=dist)
    stats = crawl.run([f"chan{i:03d}" for i in range(8)])
    with open(f"{tmp_dir}/stats{rank}.json", "w") as f:
        json.dump({"stats": stats, "processed": sorted(processed)}, f)
    dist.destroy_process_group()


def test_two_rank_orchestrated_gloo(tmp_path):
    import torch.multiprocessing as mp

    mp.spawn(_rank_main, args=(2, 29725, str(tmp_path)), nprocs=2,
             join=True)
    _check_two_rank(tmp_path)


def test_two_rank_orchestrated_gloo_rows(tmp_path):
    """World-2 gloo with the GPU engine's rows-mode discovery payload:
    uint8[N,32] rows travel through allgather_rows and the vectorized
    admission — same split/coverage/totals as the string path."""
    import torch.multiprocessing as mp

    os.environ["ORCH_TEST_ROWS"] = "1"
    try:
        mp.spawn(_rank_main, args=(2, 29739, str(tmp_path)), nprocs=2,
                 join=True)
    finally:
        os.environ.pop("ORCH_TEST_ROWS", None)
    _check_two_rank(tmp_path)


def _check_two_rank(tmp_path):
    import pathlib

    tmp_path = pathlib.Path(tmp_path)
    s0 = json.loads((tmp_path / "stats0.json").read_text())
    s1 = json.loads((tmp_path / "stats1.json").read_text())
    # work was split dynamically, no chunk processed twice, all covered
    all_proc = s0["processed"] + s1["processed"]
    # depth 0: chans 0..7; depth 1: 100..107
    expect = sorted([f"chan{i:03d}" for i in range(8)]
                    + [f"chan{i:03d}" for i in range(100, 108)])
    assert sorted(all_proc) == expect
    assert set(s0["processed"]) & set(s1["processed"]) == set()
    # both ranks agree on totals via the collective exchange
    assert s0["stats"]["layers"] == s1["stats"]["layers"] == 2
    total_pages = s0["stats"]["pages"] + s1["stats"]["pages"]
    assert total_pages == 16
    # rank 0 persisted the completed crawl
    prog = json.loads(
        (tmp_path / "r0" / "o2" / "progress.json").read_text()
    )
    assert prog["status"] == "completed"


def test_orchestrated_marks_deadends(tmp_path):
    """deadends_fn routes engine-classified deadends into page status,
    keeping the MaxPages deadend-replacement budget live (base.go:284)
    — and the set is exchanged so all ranks would agree."""
    cfg = CrawlerConfig(crawl_id="o5", storage_root=str(tmp_path),
                        sampling_method="channel", max_depth=0,
                        min_users=1)
    sm = LocalStateManager(cfg)
    store = InMemoryStore()
    state = {"dead": set()}

    def process(names):
        state["dead"] = {n for n in names if n.startswith("dead")}
        return [], len(names)

    crawl = OrchestratedCrawl(cfg, sm, store, rank=0, world=1,
                              process_fn=process, chunk_channels=8,
                              dist=FakeDist(),
                              deadends_fn=lambda: state["dead"])
    crawl.run(["chan001", "deadbeef", "chan002"])
    statuses = {p.url: p.status for p in sm.get_layer_by_depth(0)}
    assert statuses == {"chan001": "fetched", "deadbeef": "deadend",
                        "chan002": "fetched"}


def test_single_rank_rows_mode(tmp_path):
    """process_fn returning uint8[N,32] rows (the GPU engine's
    as_arrays form): the exchange/dedup/budget path stays in rows and
    admits the same next layer as the string path."""
    import numpy as np

    from crawler_amd.parallel import collectives as C

    def rows_process(names):
        out = []
        for n in names:
            for d in (n + "x1", n + "x2", "sharedchan"):
                out.append(d)
        rows = C.names_to_rows(out).numpy()
        return rows, len(names) * 10

    def str_process(names):
        out = []
        for n in names:
            for d in (n + "x1", n + "x2", "sharedchan"):
                out.append(d)
        return out, len(names) * 10

    res = {}
    for key, fn in (("rows", rows_process), ("strs", str_process)):
        cfg = CrawlerConfig(crawl_id=f"rm-{key}",
                            storage_root=str(tmp_path / key),
                            sampling_method="snowball", max_depth=1,
                            max_pages=50, min_users=1)
        sm = LocalStateManager(cfg)
        store = InMemoryStore()
        crawl = OrchestratedCrawl(cfg, sm, store, rank=0, world=1,
                                  process_fn=fn, chunk_channels=2,
                                  dist=FakeDist())
        stats = crawl.run(["seedaa", "seedbb"])
        res[key] = (stats["pages"], stats["posts"],
                    sorted(p.url for p in sm.pages.values()))
    assert res["rows"] == res["strs"]
