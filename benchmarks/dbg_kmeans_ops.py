import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.profiler import profile, ProfilerActivity
from raft_amd.random import make_blobs, RngState
from raft_amd.cluster.kmeans import kmeans_iterate

x, _, c = make_blobs(10_000_000, 256, n_clusters=1024, cluster_std=1.0,
                     state=RngState(seed=1), device="cuda")
c = c + 0.1
kmeans_iterate(x, c.clone(), 1, fp32_mode="bf16x2v")  # warmup
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    kmeans_iterate(x, c.clone(), 2, fp32_mode="bf16x2v")
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=22))
