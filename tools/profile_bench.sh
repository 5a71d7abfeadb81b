#!/bin/bash
# rocprofv3 kernel-trace profile of the flagship bench, summarized on-box.
# Usage (on the GPU box): bash tools/profile_bench.sh [bench args...]
set -x
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out"
rm -rf /tmp/prof
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof -- \
  bash -c "cd '$REPO' && python bench.py ${*:---steps 2 --warmup 1}" \
  > "$REPO/gpurun_out/prof_bench.log" 2>&1
cd "$REPO"
python tools/prof_summarize.py "/tmp/prof/**/*.db" gpurun_out/r01_current_engine_profile.md
grep agent-loop gpurun_out/prof_bench.log | head -c 300
echo
head -30 gpurun_out/r01_current_engine_profile.md
