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


class _PartialUse(torch.nn.Module):
    """Two branches; forward uses only branch a — branch b's params get
    no grads this pass (the reducer must still reduce every READY
    bucket and leave untouched params' grads None on all ranks)."""

    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(32, 32)
        self.b = torch.nn.Linear(32, 32)

    def forward(self, x):
        return self.a(x)


def _unused_worker(rank, world, port, out_dir):
    import torch.distributed as dist
    from gpuschedule_amd.parallel.ddp import GradReducer
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(7)
        m = _PartialUse()
        red = GradReducer(m, dist.group.WORLD, bucket_mb=1)
        torch.manual_seed(200 + rank)
        x = torch.randn(8, 32)
        m(x).square().mean().backward()
        local_a = [p.grad.clone() for p in m.a.parameters()]
        red.finalize()
        for lg, p in zip(local_a, m.a.parameters()):
            ref = lg.clone()
            dist.all_reduce(ref)
            ref /= world
            assert torch.allclose(p.grad, ref, atol=1e-6), \
                "used-branch grads not averaged"
        assert all(p.grad is None for p in m.b.parameters()), \
            "unused-branch grads must stay None"
        # second pass must work identically (ready-counters reset)
        m(torch.randn(8, 32)).square().mean().backward()
        red.finalize()
        if rank == 0:
            open(os.path.join(out_dir, "ok2"), "w").write("ok")
    finally:
        dist.destroy_process_group()


def test_grad_reducer_unused_params_world2(tmp_path):
    import torch.multiprocessing as mp
    mp.spawn(_unused_worker, args=(2, 29361, str(tmp_path)), nprocs=2,
             join=True)
    assert (tmp_path / "ok2").exists()
