"""GradReducer correctness: bucketed async all-reduce == manual mean."""
import os

import torch


def _worker(rank, world, port, out_dir):
    import torch.distributed as dist
    from gpuschedule_amd.parallel.ddp import GradReducer
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(42)  # identical replicas
        m = torch.nn.Sequential(torch.nn.Linear(64, 128),
                                torch.nn.ReLU(),
                                torch.nn.Linear(128, 8))
        red = GradReducer(m, None if world == 1 else dist.group.WORLD,
                          bucket_mb=1)
        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(16, 64)
        y = m(x).square().mean()
        y.backward()
        local = [p.grad.clone() for p in m.parameters()]
        red.finalize()
        # reference: manual average of local grads
        for lg, p in zip(local, m.parameters()):
            ref = lg.clone()
            dist.all_reduce(ref)
            ref /= world
            assert torch.allclose(p.grad, ref, atol=1e-6), "reducer wrong"
        if rank == 0:
            open(os.path.join(out_dir, "ok"), "w").write("ok")
    finally:
        dist.destroy_process_group()


def test_grad_reducer_world2(tmp_path):
    import torch.multiprocessing as mp
    mp.spawn(_worker, args=(2, 29350, str(tmp_path)), nprocs=2, join=True)
    assert (tmp_path / "ok").exists()
