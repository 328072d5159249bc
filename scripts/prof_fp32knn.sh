cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof_fknn -o fk -- \
  bash -c "cd $GRAFT_REPO_ROOT && timeout 400 python -c \"
import torch
from raft_amd.neighbors import brute_force_build
torch.manual_seed(0)
x = torch.randn(50_000_000, 128, device='cuda')
q = torch.randn(16384, 128, device='cuda')
idx = brute_force_build(x)
idx.search(q, 64); torch.cuda.synchronize()
print('DONE')\"" > gpurun_out/prof_fknn.log 2>&1
tail -1 gpurun_out/prof_fknn.log
python - <<'PYEOF'
import csv, glob
f = glob.glob('gpurun_out/prof_fknn/*kernel_stats.csv')[0]
rows = sorted(csv.DictReader(open(f)), key=lambda r: -float(r['TotalDurationNs']))
for r in rows[:8]:
    print("%8.1f ms %4dx %s" % (float(r['TotalDurationNs'])/1e6, int(r['Calls']), r['Name'][:80]))
PYEOF
