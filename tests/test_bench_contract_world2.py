"""Driver-contract check: bench.py launched exactly the way the driver
launches it for N>1 (torch.distributed.run, one rank per slot, gloo on
CPU here) must print one valid JSON line from rank 0 with the
contract fields.  Tiny shape so it stays CPU-affordable; completion of
the full trace is exercised separately (test_cli_configs CPU bench +
the GPU runs in profiles/)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_world2_torchrun_contract():
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", "29561",
           "bench.py", "--gpus", "2", "--steps", "4", "--warmup", "1",
           "--jobs", "3", "--iters-per-tick", "1", "--device", "cpu"]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["steps"] == 4
    assert d["scaling"] == "strong"
    assert d["dtype"] == "bf16"
    assert d["higher_is_better"] is False
    assert d["unit"] == "s"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["trace_jobs"] == 3
    # exactly ONE rank printed the JSON line
    assert sum(ln.startswith("{") for ln in out.stdout.splitlines()) == 1
