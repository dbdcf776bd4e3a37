"""Package setup. The HIP extension stays IN-TREE (crawler_amd/ops/csrc/
libcrawlhip.so, built by `python -m crawler_amd.ops.build`); install with
`pip install -e .` so the .so travels with the source checkout."""
from setuptools import find_packages, setup

setup(
    name="crawler-amd",
    version="0.1.0",
    description=(
        "MI355X-native distributed crawl/ingest engine (CDNA4 HIP hot "
        "path, RCCL-over-xGMI distribution, synthetic TDLib/YouTube feeds)"
    ),
    packages=find_packages(include=["crawler_amd", "crawler_amd.*"]),
    package_data={"crawler_amd.ops": ["csrc/*.hip", "csrc/*.h",
                                      "csrc/*.so"]},
    python_requires=">=3.10",
    install_requires=["numpy", "torch"],
    entry_points={
        "console_scripts": [
            "crawler-amd=crawler_amd.cli:main",
        ],
    },
)
