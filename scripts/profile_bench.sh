#!/bin/bash
# rocprofv3 kernel-stats profile of bench.py on a GPU box.
#   bash scripts/profile_bench.sh <out-name> [bench args...]
# Keeps only the aggregated *_stats.csv under gpurun_out/prof/ (the raw
# trace db of a graph-replay run exceeds the 64 MiB copy-back budget).
# Counter (--pmc) collection must be a separate run without trace domains.
set -e
NAME=${1:-bench}
shift || true
REPO=$(cd "$(dirname "$0")/.." && pwd)
PROF="$REPO/gpurun_out/prof"
mkdir -p "$PROF"
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d "$PROF" \
    -o "$NAME" -- bash -c "cd '$REPO' && python bench.py --steps 3 --warmup 2 $*"
# keep CSVs only; the .db can be hundreds of MiB; gzip the trace (the
# copy-back budget is 64 MiB total)
find "$PROF" -name "${NAME}*" ! -name "*.csv" -delete || true
gzip -f "$PROF/${NAME}_kernel_trace.csv" 2>/dev/null || true
ls -la "$PROF"
echo "profile written to gpurun_out/prof/${NAME}*"
