#!/usr/bin/env python3
"""End-to-end iteration time, hand-written MFMA GEMM vs hipBLASLt
(GSAMD_GEMM_LIB=1) for every model family that uses MFMALinear.
The mode is baked into the runner's hipGraph at capture, so each mode
gets a fresh runner."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch  # noqa: E402

from gpuschedule_amd.exec.trainer import JobRunner  # noqa: E402


def measure(model, batch, iters=10):
    r = JobRunner(f"m-{model}", model, batch, "cuda:0", torch.bfloat16,
                  None, 0, 1)
    r.step(4)  # warmup + capture
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n, _ = r.step(iters)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / n * 1000
    loss = r.last_loss
    del r
    torch.cuda.empty_cache()
    return round(ms, 2), round(float(loss), 3)


def main():
    out = {}
    for model, batch, iters in (("bert-base", 32, 20),
                                ("llama3-8b", 1, 6)):
        row = {}
        for mode in ("0", "1"):
            os.environ["GSAMD_GEMM_LIB"] = mode
            ms, loss = measure(model, batch, iters)
            row["mfma" if mode == "0" else "lib"] = {"ms_per_iter": ms,
                                                     "loss": loss}
        row["lib_speedup"] = round(row["mfma"]["ms_per_iter"]
                                   / row["lib"]["ms_per_iter"], 3)
        out[model] = row
        print(json.dumps({model: row}), flush=True)
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
