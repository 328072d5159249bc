#!/bin/bash
# PMC counters for kmeans_update_verify (the 3.0 ms flagship block).
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
rocprofv3 --output-format csv \
  --pmc SQ_INSTS_VALU SQ_INSTS_LDS SQ_BUSY_CYCLES SQ_WAVE_CYCLES SQ_INSTS_VMEM \
  -d gpurun_out/pmc_verify -o pv -- \
  bash -c "cd $GRAFT_REPO_ROOT && timeout 250 python bench.py --steps 5 --warmup 1 --no-pairwise" \
  > gpurun_out/pmc_verify.log 2>&1
grep -o '{.*}' gpurun_out/pmc_verify.log | tail -1 | cut -c1-120
python - <<'EOF'
import collections, csv, glob
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for f in glob.glob("gpurun_out/pmc_verify/**/*counter_collection.csv", recursive=True):
    for row in csv.DictReader(open(f)):
        agg[row.get("Kernel_Name", "?")][row.get("Counter_Name")] += \
            float(row.get("Counter_Value", 0))
for k, c in agg.items():
    if "update_verify" in k or "l2nn_2d_kernel" in k:
        print(k[:70])
        for n in ("SQ_INSTS_VALU", "SQ_INSTS_LDS", "SQ_INSTS_VMEM",
                  "SQ_BUSY_CYCLES", "SQ_WAVE_CYCLES"):
            print("  %-18s %.3e" % (n, c.get(n, 0)))
        if c.get("SQ_BUSY_CYCLES"):
            print("  VALU/busy %.3f  LDS/busy %.3f  wave/busy %.2f"
                  % (c["SQ_INSTS_VALU"] / c["SQ_BUSY_CYCLES"],
                     c.get("SQ_INSTS_LDS", 0) / c["SQ_BUSY_CYCLES"],
                     c["SQ_WAVE_CYCLES"] / c["SQ_BUSY_CYCLES"]))
EOF
