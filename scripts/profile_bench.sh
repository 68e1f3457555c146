#!/bin/bash
# rocprofv3 kernel-stats profile of bench.py on a GPU box.
#   bash scripts/profile_bench.sh <out-name> [bench args...]
# Writes CSVs under gpurun_out/prof/.  Counter (--pmc) collection must be a
# separate run without trace domains (pool policy).
set -e
NAME=${1:-bench}
shift || true
REPO=$(cd "$(dirname "$0")/.." && pwd)
mkdir -p "$REPO/gpurun_out/prof"
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof" \
    -o "$NAME" -- bash -c "cd '$REPO' && python bench.py --steps 5 --warmup 2 $*"
echo "profile written to gpurun_out/prof/${NAME}*"
