#!/usr/bin/env bash
# Local gate: build the gfx950 extension + run the CPU suite (what the
# driver checks off-GPU).  GPU suite: python -m pytest tests -q -m gpu.
set -euo pipefail
cd "$(dirname "$0")/.."
python -m dwt_amd.kernels.build
python -m pytest tests -x -q -m "not gpu"
