#!/bin/bash
# PMC comparison of the 256^2 vs 2D fused L2-NN engines (run via gpurun).
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
M=${M:-1000000}; N=${N:-1024}; D=${D:-256}
for eng in 1 0; do
  RAFT_AMD_L2NN_256=$eng rocprofv3 --output-format csv \
    --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT SQ_WAVE_CYCLES SQ_BUSY_CYCLES \
    -d gpurun_out/prof256_$eng -o p$eng -- \
    timeout 200 python benchmarks/ab_l2nn_256.py --mode one \
    --m $M --n $N --d $D --steps 5 > gpurun_out/prof256_run_$eng.log 2>&1
  grep -o '{.*}' gpurun_out/prof256_run_$eng.log | tail -1
done
python scripts/prof_l2nn_256_parse.py
