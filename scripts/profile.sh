#!/bin/bash
# rocprofv3 capture of the flagship bench.
# Usage: scripts/profile.sh [scale] [outdir]
set -e
SCALE=${1:-0.3}
OUT=${2:-profiles/capture}
REPO=$(cd "$(dirname "$0")/.." && pwd)     # resolve BEFORE leaving cwd
mkdir -p "$REPO/$OUT"
cd /tmp && export TMPDIR=/tmp
# kernel trace + per-kernel stats (PMC counters crash rocprofv3 on this pool;
# see profiles/r01_NOTES.md)
rocprofv3 --kernel-trace --stats --output-format csv -d "$REPO/$OUT" -o bench \
  -- python "$REPO/bench.py" --steps 3 --warmup 1 --scale "$SCALE" \
     --part-dir "$REPO/part_data_bench"
find "$REPO/$OUT" -name "*.db" -delete
echo "wrote $OUT/bench_kernel_stats.csv"
