#!/bin/bash
# rocprofv3 kernel-trace+stats wrapper for gpurun.
# Usage: bash scripts/profile_cmd.sh <tag> <command...>
# The command runs with cwd=/root/repo; profile db lands in gpurun_out/<tag>_results.db
set -x
TAG=$1
shift
export HSA_ENABLE_IPC_MODE_LEGACY=0
export TMPDIR=/tmp
export PYTHONPATH=/root/repo
mkdir -p /root/repo/gpurun_out
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out -o "$TAG" -- \
    bash -c "cd /root/repo && $*" > /root/repo/gpurun_out/"$TAG".log 2>&1
tail -3 /root/repo/gpurun_out/"$TAG".log
# summarize on the box; the raw db can exceed the gpurun copy-back limit
python /root/repo/scripts/rocpd_stats.py /root/repo/gpurun_out/"$TAG"_results.db 40 \
    > /root/repo/gpurun_out/"$TAG"_stats.txt 2>&1
rm -f /root/repo/gpurun_out/"$TAG"_results.db
cat /root/repo/gpurun_out/"$TAG"_stats.txt
