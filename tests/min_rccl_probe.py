"""Minimal 2-rank RCCL probe: init + one allreduce. Used to isolate RCCL
bring-up failures from the full comms suite."""
import os
import sys
import traceback

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    try:
        ndev = torch.cuda.device_count()
        print(f"[rank {rank}] cuda devices: {ndev}", flush=True)
        torch.cuda.set_device(rank % max(ndev, 1))
        dist.init_process_group("nccl")
        t = torch.ones(4, device="cuda") * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        print(f"[rank {rank}] allreduce -> {t.tolist()} PROBE_OK", flush=True)
        dist.destroy_process_group()
    except Exception:
        traceback.print_exc()
        sys.exit(1)


if __name__ == "__main__":
    main()
