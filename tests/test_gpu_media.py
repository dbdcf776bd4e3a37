"""GPU media spill test: HBM blob -> pinned ring -> host file."""
import os

import pytest
import torch

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager
from crawler_amd.engine.media import MediaEngine, synth_blob_size

pytestmark = pytest.mark.gpu


def test_gpu_spill_matches_cpu_generation(tmp_path):
    cfg = CrawlerConfig(crawl_id="gm1", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    eng = MediaEngine(sm, use_gpu=True, slot_bytes=1 << 20)
    path = eng.fetch_and_upload("chanG", "AgAD42t")
    eng.close()
    assert os.path.exists(path)
    data_gpu = open(path, "rb").read()
    assert len(data_gpu) == synth_blob_size("AgAD42t")

    sm2 = LocalStateManager(
        CrawlerConfig(crawl_id="gm2", storage_root=str(tmp_path))
    )
    eng2 = MediaEngine(sm2, use_gpu=False)
    path2 = eng2.fetch_and_upload("chanG", "AgAD42t")
    eng2.close()
    assert open(path2, "rb").read() == data_gpu


def test_gpu_spill_multi_slot(tmp_path):
    """A blob larger than one ring slot exercises the chunked spill."""
    cfg = CrawlerConfig(crawl_id="gm3", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    eng = MediaEngine(sm, use_gpu=True, slot_bytes=64 * 1024)
    # find a >256KB blob
    mid = None
    for i in range(2000):
        m = f"AgAD{i}v"
        if 256 * 1024 < synth_blob_size(m) < 4 * 1024 * 1024:
            mid = m
            break
    assert mid
    path = eng.fetch_and_upload("chanG", mid)
    eng.close()
    assert os.path.getsize(path) == synth_blob_size(mid)
