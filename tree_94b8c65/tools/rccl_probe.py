"""Probe: can this RCCL build run 2 ranks of one communicator on one GPU?
Prints the exact init/all_reduce error if not. Evidence for the 2-rank
RCCL test's skip path (tests/test_rccl_2rank.py)."""
import os
import sys

import torch


def worker(rank):
    import torch.distributed as dist

    os.environ.update(RANK=str(rank), WORLD_SIZE="2",
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29771")
    torch.cuda.set_device(0)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=2)
        t = torch.ones(1024, device="cuda:0") * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        print(f"[rank {rank}] all_reduce OK: {t[0].item()} (expect 3.0)", flush=True)
        dist.destroy_process_group()
    except Exception as e:
        print(f"[rank {rank}] FAILED: {type(e).__name__}: {e}", flush=True)
        sys.exit(1)


if __name__ == "__main__":
    import torch.multiprocessing as mp

    mp.start_processes(worker, nprocs=2, start_method="spawn", join=True)
    print("probe done")
