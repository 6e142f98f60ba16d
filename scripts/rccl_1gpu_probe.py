"""Probe: 2-rank RCCL communicator with BOTH ranks on one GPU.

If RCCL accepts same-device communicators, the gang/reducer nccl code
path (never exercised on a 1-GPU lease) can be tested before a real
multi-GPU run: all_reduce, then a 2-rank gang JobRunner step through
GradReducer, then a broadcast_state grow.
"""
import os
import sys

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def worker(rank, world):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29501"
    torch.cuda.set_device(0)  # BOTH ranks on cuda:0
    dist.init_process_group("nccl", rank=rank, world_size=world)
    try:
        t = torch.full((1024,), float(rank + 1), device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert t[0].item() == 3.0, t[0].item()
        print(f"rank {rank}: all_reduce OK", flush=True)

        from gpuschedule_amd.exec.trainer import JobRunner
        r = JobRunner("g", "llama3-tiny", 2, "cuda:0", torch.bfloat16,
                      dist.group.WORLD, rank, world)
        r.step(3)
        print(f"rank {rank}: gang step OK loss={r.last_loss:.4f}",
              flush=True)

        from gpuschedule_amd.parallel.ddp import broadcast_state
        broadcast_state(r.model, r.optim, dist.group.WORLD, 0)
        torch.cuda.synchronize()
        print(f"rank {rank}: broadcast_state OK", flush=True)
        dist.barrier()
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    mp.spawn(worker, args=(2,), nprocs=2, join=True)
    print("RCCL 1-GPU 2-rank probe: ALL OK", flush=True)
