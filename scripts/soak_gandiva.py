"""64-job Gandiva soak on 1 GPU: the real Philly-format trace
(clamp_gpus=1 so every job fits a single-GPU cluster), time-compressed."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch  # noqa: E402

from gpuschedule_amd import Cluster, Scheduler, make_policy  # noqa: E402
from gpuschedule_amd.exec import TorchBackend  # noqa: E402
from gpuschedule_amd.trace.philly_log import load_philly_log  # noqa: E402

# time_scale 1/6000: sim_iter_time calibration (catalog.py) shrank the
# per-iteration times ~5x, which grows the derived total_iters ~5x for
# the same scaled duration — compress 5x harder so the soak's wall time
# stays ~the round-2 run (profiles/soak_r2.jsonl, taken at 1/1200
# pre-calibration; scheduler churn per job is equivalent).
specs = load_philly_log("data/philly_sample.json", time_scale=1 / 6000,
                        clamp_gpus=1, seed=4)
sched = Scheduler(Cluster(1, 1), make_policy("gandiva", quantum=2.0))
sched.submit(specs)
be = TorchBackend(sched, rank=0, world=1,
                  device_type="cuda" if torch.cuda.is_available() else "cpu",
                  iters_per_tick=2)
log = be.run(max_seconds=float(sys.argv[1]) if len(sys.argv) > 1 else 240)
s = log.summary()
print({k: round(v, 3) for k, v in s.items()})
log.save("gpurun_out/soak_r2.jsonl")
