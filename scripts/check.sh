#!/usr/bin/env bash
# Local build+test gate (reference build.sh:55-96 analog):
# HIP build -> CPU suite; run the gpu-marked suite on an MI355X box.
set -euo pipefail
cd "$(dirname "$0")/.."
python -m crawler_amd.ops.build --force
python -m pytest tests -q -m "not gpu"
echo "OK: build + CPU suite green. Run 'python -m pytest tests -q -m gpu' on a GPU box."
