#!/bin/bash
# A/B the fused L2-NN K-loop variants (RAFT_AMD_L2NN_DB=0/1) across engines.
cd "$(dirname "$0")/.."
for db in 0 1; do
  for mode in bf16x2 bf16x2v bf16x3; do
    out=$(RAFT_AMD_L2NN_DB=$db timeout 150 python bench.py --steps 3 --warmup 1 --fp32-mode $mode 2>/dev/null)
    ms=$(echo "$out" | python -c "import json,sys; print(json.load(sys.stdin)['ms_per_step'])")
    echo "DB=$db $mode: $ms ms/step"
  done
done
