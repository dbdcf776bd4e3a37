"""CI coverage for the native C++ CPU reference (cpp/cpu_reference.cc):
builds it and asserts byte-identical output vs the Python oracle."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def cpu_ref(tmp_path_factory):
    exe = str(tmp_path_factory.mktemp("cpuref") / "cpu_ref")
    subprocess.run(
        ["g++", "-O2", "-std=c++17", "-pthread",
         os.path.join(REPO, "cpp", "cpu_reference.cc"), "-o", exe],
        check=True,
    )
    return exe


def test_cpp_reference_matches_oracle(cpu_ref, tmp_path):
    sys.path.insert(0, os.path.join(REPO, "tools"))
    from measure_cpu_baseline import NOW, dump_batch

    from crawler_amd.feed import FeedConfig, SyntheticFeed
    from crawler_amd.ops.golden_batch import encode_batch

    feed = SyntheticFeed(FeedConfig(seed=77, universe=5000))
    batch = feed.build_batch(np.arange(6), posts_per_channel=150)
    d = str(tmp_path / "dump")
    os.makedirs(d)
    dump_batch(batch, d)
    out_path = str(tmp_path / "out.jsonl")
    res = subprocess.run([cpu_ref, d, "2", "1", out_path],
                         check=True, capture_output=True)
    stats = json.loads(res.stdout)
    assert stats["posts"] == batch.n
    golden = b"".join(encode_batch(batch, now=NOW)[0])
    got = open(out_path, "rb").read()
    assert got == golden
