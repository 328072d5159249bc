#!/bin/bash
# Kernel-stats profile of the 100M-row kNN headline. Run via gpurun.
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
rocprofv3 --kernel-trace --stats --output-format csv \
  -d gpurun_out/prof_knn -o knn -- \
  bash -c "cd $GRAFT_REPO_ROOT && timeout 500 python benchmarks/bench_knn.py" \
  > gpurun_out/prof_knn.log 2>&1
grep -o '{.*}' gpurun_out/prof_knn.log | tail -1
python - <<'EOF'
import csv
rows = list(csv.DictReader(open([f for f in __import__('glob').glob(
    'gpurun_out/prof_knn/*kernel_stats.csv')][0])))
rows.sort(key=lambda r: float(r['TotalDurationNs']), reverse=True)
for r in rows[:10]:
    print("%9.1f ms %5dx  %s" % (float(r['TotalDurationNs']) / 1e6,
                                 int(r['Calls']), r['Name'][:85]))
EOF
