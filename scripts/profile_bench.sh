#!/bin/bash
# Profile bench.py with rocprofv3 on a gpurun box.  Usage: profile_bench.sh <tag> [bench args...]
set -x
TAG=${1:-prof}
shift || true
export HSA_ENABLE_IPC_MODE_LEGACY=0
export TMPDIR=/tmp
OUT=/root/repo/gpurun_out/$TAG
mkdir -p "$OUT"
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$OUT" -o "$TAG" -- \
    bash -c "cd /root/repo && python bench.py $*" > "$OUT/run.log" 2>&1
tail -5 "$OUT/run.log"
for f in "$OUT"/*stats.csv; do echo "== $f"; head -30 "$f"; done
