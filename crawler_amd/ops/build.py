"""In-tree build of the CDNA4 HIP extension (gfx950 only, no fallbacks).

Usage: python -m crawler_amd.ops.build
Called by __graft_entry__.build() — the driver's build check. hipcc
cross-compiles on CPU-only hosts; the .so travels to GPU boxes in-tree.
"""
from __future__ import annotations

import os
import subprocess
import sys

CSRC = os.path.dirname(os.path.abspath(__file__)) + "/csrc"
SOURCES = ["parse_encode.hip", "dedup.hip", "feedgen.hip", "yt_encode.hip",
           "yt_feedgen.hip", "aggregate.hip", "htmlclass.hip",
           "sample.hip"]
OUT = os.path.join(CSRC, "libcrawlhip.so")


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    out_m = os.path.getmtime(OUT)
    deps = [os.path.join(CSRC, s) for s in SOURCES]
    deps += [os.path.join(CSRC, "common.h")]
    return any(os.path.getmtime(d) > out_m for d in deps)


def build(force: bool = False) -> str:
    if not force and not needs_build():
        return OUT
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared",
    ] + [os.path.join(CSRC, s) for s in SOURCES] + ["-o", OUT]
    subprocess.run(cmd, check=True)
    return OUT


NATIVE_DIR = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "native"
)
NATIVE_SRC = os.path.join(NATIVE_DIR, "fanout_sink.cc")


def _native_out() -> str:
    import sysconfig

    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(NATIVE_DIR, "fanout_native" + suffix)


def build_native(force: bool = False) -> str:
    """Plain-C++ runtime extension (no GPU dependency): g++ + pybind11."""
    out = _native_out()
    if (not force and os.path.exists(out)
            and os.path.getmtime(out) >= os.path.getmtime(NATIVE_SRC)):
        return out
    import pybind11
    import sysconfig

    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        "g++", "-O2", "-std=c++17", "-fPIC", "-shared", "-pthread",
        f"-I{py_inc}", f"-I{pybind11.get_include()}",
        NATIVE_SRC, "-o", out,
    ]
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    path = build(force="--force" in sys.argv)
    print(path)
    print(build_native(force="--force" in sys.argv))
