#!/bin/bash
# PMC counters for the 1-slice (bf16x1v) fused L2-NN kernel at the flagship
# shape (10M x 256, k=1024, GT8 default). Run via gpurun.
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
rocprofv3 --output-format csv \
  --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT SQ_WAVE_CYCLES SQ_BUSY_CYCLES \
  -d gpurun_out/pmc_x1v -o x1v -- \
  timeout 300 python -c "
import torch
from raft_amd._ext import require_ext
ext = require_ext()
m, d, k = 10_000_000, 256, 1024
torch.manual_seed(0)
x = torch.randn(m, d, device='cuda')
c = torch.randn(k, d, device='cuda')
xs = [x.bfloat16().contiguous()]; cs = [c.bfloat16().contiguous()]
xn = (x*x).sum(dim=1); cn = (c*c).sum(dim=1)
for _ in range(5):
    ext.fused_l2nn_split(xs, cs, xn, cn)
torch.cuda.synchronize()
print('PMC_RUN_DONE')
" > gpurun_out/pmc_x1v_run.log 2>&1
tail -2 gpurun_out/pmc_x1v_run.log
python - <<'EOF'
import collections, csv, glob
files = glob.glob("gpurun_out/pmc_x1v/**/*counter_collection.csv", recursive=True)
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for f in files:
    for row in csv.DictReader(open(f)):
        agg[row.get("Kernel_Name", "?")][row.get("Counter_Name")] += \
            float(row.get("Counter_Value", 0))
for kname, c in agg.items():
    if "l2nn" in kname:
        print(kname[:90])
        for n, v in sorted(c.items()):
            print(f"  {n:24s} {v:.4e}")
        if c.get("SQ_BUSY_CYCLES"):
            # 4 SIMDs/CU issue MFMA; each mfma_16x16x32 keeps the MFMA pipe
            # 4 issue-cycles busy per wave inst (bf16 2-pass) — report raw
            # inst/cycle ratios for the record instead of a derived percent
            print(f"  MFMA insts / busy-cycle  {c['SQ_INSTS_MFMA']/c['SQ_BUSY_CYCLES']:.4f}")
            print(f"  VALU insts / busy-cycle  {c['SQ_INSTS_VALU']/c['SQ_BUSY_CYCLES']:.4f}")
            print(f"  LDS conflicts / MFMA     {c['SQ_LDS_BANK_CONFLICT']/max(c['SQ_INSTS_MFMA'],1):.4f}")
EOF
