#!/bin/bash
# rocprofv3 kernel-stats profile of the flagship bench (run on the GPU box).
# Usage: bash scripts/prof_flagship.sh [extra bench args...]
set -e
cd /tmp && export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv \
  -d /root/repo/gpurun_out/prof_flagship -o flag -- \
  bash -c "cd /root/repo && python bench.py --steps 10 --warmup 2 --no-pairwise $*" \
  2>&1 | grep -E 'metric|Error' | head -3
f=$(ls /root/repo/gpurun_out/prof_flagship/*kernel_stats.csv | head -1)
echo "=== kernel stats (top by total ns) ==="
sort -t, -k4 -rn -g "$f" | head -14 | cut -c1-190
