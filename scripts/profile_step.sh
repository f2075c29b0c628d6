#!/usr/bin/env bash
# Clean steady-state rocprofv3 capture of the flagship step on an MI355X box.
# MIOpen's find phase pollutes a cold capture with naive-kernel time, so run
# the bench once un-profiled (populates the find DB on this box), then
# capture.  Usage: scripts/profile_step.sh [outdir] [batch]
set -euo pipefail
OUT=${1:-gpurun_out/prof_step}
BATCH=${2:-512}
REPO=$(cd "$(dirname "$0")/.." && pwd)
cd "$REPO"
python bench.py --steps 3 --warmup 2 --batch "$BATCH" > /dev/null
cd /tmp && export TMPDIR=/tmp
rocprofv3 --kernel-trace --stats -d "$REPO/$OUT" -- \
    python "$REPO/bench.py" --steps 4 --warmup 1 --batch "$BATCH"
echo "profile in $OUT"
