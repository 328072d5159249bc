#!/bin/bash
# PMC counters for the kNN filter sweep (10M-row index proxy). Run via gpurun.
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
rocprofv3 --output-format csv \
  --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_BUSY_CYCLES SQ_WAVE_CYCLES \
  -d gpurun_out/pmc_knn -o knn -- \
  bash -c "cd $GRAFT_REPO_ROOT && timeout 400 python benchmarks/bench_knn.py --index-rows 10000000 --queries 8192" \
  > gpurun_out/pmc_knn.log 2>&1
grep -o '{.*}' gpurun_out/pmc_knn.log | tail -1
python - <<'EOF'
import collections, csv, glob
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for f in glob.glob("gpurun_out/pmc_knn/**/*counter_collection.csv", recursive=True):
    for row in csv.DictReader(open(f)):
        agg[row.get("Kernel_Name", "?")][row.get("Counter_Name")] += \
            float(row.get("Counter_Value", 0))
for k, c in sorted(agg.items(), key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))[:5]:
    mfma = c.get("SQ_INSTS_MFMA", 0)
    valu = c.get("SQ_INSTS_VALU", 0)
    print(k[:85])
    print("  busy %.3e mfma %.3e valu %.3e VALU/MFMA %.2f"
          % (c.get("SQ_BUSY_CYCLES", 0), mfma, valu, valu / max(mfma, 1)))
EOF
