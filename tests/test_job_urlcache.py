"""Job-mode config merge + scheduler; cross-crawl URL dedup cache tests."""
import json
import time

import pytest

from crawler_amd.config import CrawlerConfig
from crawler_amd.engine import LocalStateManager, Page
from crawler_amd.engine.job import (
    JobData,
    JobScheduler,
    merge_config_with_job_data,
)


def test_job_merge_overrides_only_nonzero():
    base = CrawlerConfig(crawl_id="cli", max_depth=2, concurrency=3,
                         platform="telegram", max_pages=100)
    job = JobData(max_depth=5, platform="youtube",
                  sampling_method="random")
    merged = merge_config_with_job_data(base, job)
    assert merged.max_depth == 5
    assert merged.platform == "youtube"
    assert merged.sampling_method == "random"
    assert merged.concurrency == 3       # untouched
    assert merged.crawl_id == "cli"      # untouched
    assert merged.max_pages == 100
    assert base.max_depth == 2           # base unchanged


def test_job_data_from_json():
    payload = json.dumps({
        "urls": ["abcde"], "max_depth": 3, "crawl_id": "j1",
        "min_post_date": "2024-01-01T00:00:00+00:00",
    })
    jd = JobData.from_json(payload)
    assert jd.urls == ["abcde"]
    assert jd.min_post_date.year == 2024


def test_scheduler_lifecycle_and_trigger():
    sched = JobScheduler()
    runs = []
    sched.schedule("j1", json.dumps({"crawl_id": "x"}), interval_s=3600,
                   handler=lambda jd: runs.append(jd.crawl_id))
    assert sched.get("j1")["interval_s"] == 3600
    assert sched.trigger("j1", lambda jd: runs.append(jd.crawl_id))
    assert runs == ["x"]
    assert sched.delete("j1")
    assert sched.get("j1") is None
    assert not sched.trigger("j1", lambda jd: None)


def test_scheduler_interval_fires():
    sched = JobScheduler()
    runs = []
    sched.schedule("fast", "{}", interval_s=0.02,
                   handler=lambda jd: runs.append(1))
    time.sleep(0.2)
    sched.delete("fast")
    assert len(runs) >= 2


# ---------- cross-crawl URL dedup cache ----------

def test_url_dedup_cache_across_crawls(tmp_path):
    cfg_a = CrawlerConfig(crawl_id="crawlA", storage_root=str(tmp_path))
    sm_a = LocalStateManager(cfg_a)
    sm_a.initialize(["chan_one"])
    sm_a.add_layer([Page(url="chan_two", depth=1)])
    sm_a.save_state()

    cfg_b = CrawlerConfig(crawl_id="crawlB", storage_root=str(tmp_path))
    sm_b = LocalStateManager(cfg_b)
    n = sm_b.load_url_dedup_cache()
    assert n == 2
    assert sm_b.url_dedup["chan_one"].startswith("crawlA:")
    # seeds always admitted; discovered duplicates skipped
    sm_b.initialize(["chan_one"])
    assert len(sm_b.get_layer_by_depth(0)) == 1
    added = sm_b.add_layer([Page(url="chan_two", depth=1),
                            Page(url="chan_new", depth=1)])
    assert len(added) == 1
    assert sm_b.get_layer_by_depth(1)[0].url == "chan_new"


def test_url_dedup_not_loaded_when_absent(tmp_path):
    cfg = CrawlerConfig(crawl_id="solo", storage_root=str(tmp_path))
    sm = LocalStateManager(cfg)
    assert sm.load_url_dedup_cache() == 0
    sm.initialize(["abcde"])
    added = sm.add_layer([Page(url="fghij", depth=1)])
    assert len(added) == 1
