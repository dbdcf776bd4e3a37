"""Media pipeline (CPU mode), file cleaner, page export tests."""
import json
import os
import time

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page
from crawler_amd.engine.filecleaner import FileCleaner
from crawler_amd.engine.media import (
    MEDIA_SIZE_CAP_MB,
    MediaEngine,
    synth_blob_size,
)


def mk_sm(tmp_path):
    return LocalStateManager(
        CrawlerConfig(crawl_id="m1", storage_root=str(tmp_path))
    )


def test_blob_sizes_deterministic():
    assert synth_blob_size("AgAD1t") == synth_blob_size("AgAD1t")
    assert synth_blob_size("AgAD1t") != synth_blob_size("AgAD2t")


def test_media_store_and_dedup(tmp_path):
    sm = mk_sm(tmp_path)
    eng = MediaEngine(sm, use_gpu=False)
    p1 = eng.fetch_and_upload("chanA", "AgAD5t")
    eng.close()
    assert p1.endswith("AgAD5t.bin")
    assert os.path.exists(p1)
    assert os.path.getsize(p1) == synth_blob_size("AgAD5t")
    # dedup on second fetch
    eng2 = MediaEngine(sm, use_gpu=False)
    assert eng2.fetch_and_upload("chanA", "AgAD5t") == ""
    assert eng2.stats["deduped"] == 1
    eng2.close()


def test_media_skip_flag(tmp_path):
    sm = mk_sm(tmp_path)
    eng = MediaEngine(sm, use_gpu=False)
    assert eng.fetch_and_upload("c", "AgAD6t", skip_media=True) == ""
    assert eng.stats["stored"] == 0
    eng.close()


def test_media_size_cap(tmp_path):
    sm = mk_sm(tmp_path)
    eng = MediaEngine(sm, use_gpu=False)
    # find an over-cap id deterministically
    over = None
    for i in range(5000):
        mid = f"AgAD{i}v"
        if synth_blob_size(mid) / (1024 * 1024) > MEDIA_SIZE_CAP_MB:
            over = mid
            break
    assert over is not None
    assert eng.fetch_and_upload("c", over) == ""
    assert eng.stats["over_cap"] == 1
    assert sm.has_processed_media(over)  # cached so it is not refetched
    eng.close()


def test_file_cleaner_removes_old_files(tmp_path):
    d = tmp_path / "staging"
    d.mkdir()
    old = d / "old.bin"
    new = d / "new.bin"
    old.write_bytes(b"x" * 100)
    new.write_bytes(b"y" * 100)
    past = time.time() - 7200
    os.utime(old, (past, past))
    fc = FileCleaner([str(d)], max_age_s=3600)
    n = fc.sweep_once()
    assert n == 1
    assert not old.exists() and new.exists()
    assert fc.stats["bytes_freed"] == 100


def test_export_pages_chunks(tmp_path):
    sm = mk_sm(tmp_path)
    sm.initialize([f"chan{i:05d}" for i in range(50)])
    sm.export_chunk_size_bytes = 2000  # force several chunks
    paths = sm.export_pages_to_binding("m1")
    assert len(paths) > 1
    total = 0
    for p in paths:
        with open(p) as f:
            for line in f:
                obj = json.loads(line)
                assert "url" in obj and "status" in obj
                total += 1
    assert total == 50
