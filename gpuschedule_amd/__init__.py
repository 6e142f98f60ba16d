"""gpuschedule_amd — an MI355X-native DL-job GPU cluster scheduler.

A from-scratch framework with the capabilities of GPUSchedule
(matthewygf/GPUSchedule; reference mount empty this build — see
SURVEY.md for provenance): Philly-style trace replay through a policy
suite (FIFO / SRTF / Tiresias-LAS / Gandiva time-slicing / elastic
resize) over either a discrete-event simulator or a real execution
backend where every job step is a PyTorch-ROCm training iteration whose
GEMM / attention / LayerNorm run in hand-written gfx950 HIP kernels and
gangs all-reduce over RCCL/xGMI.
"""

__version__ = "0.1.0"

from .trace.job import Job, JobSpec, JobState, MI355X_HBM_BYTES
from .trace.philly import load_trace, save_trace
from .trace.synth import make_trace, make_elastic_trace
from .sched.cluster import Cluster
from .sched.events import EventLog
from .sched.scheduler import Scheduler, Plan, Placement
from .sched.policies import (POLICIES, make_policy, FIFOPolicy, SRTFPolicy,
                             LASPolicy, GandivaPolicy, ElasticPolicy)
from .sim.simulator import Simulator, simulate

__all__ = [
    "Job", "JobSpec", "JobState", "MI355X_HBM_BYTES",
    "load_trace", "save_trace", "make_trace", "make_elastic_trace",
    "Cluster", "EventLog", "Scheduler", "Plan", "Placement",
    "POLICIES", "make_policy", "FIFOPolicy", "SRTFPolicy", "LASPolicy",
    "GandivaPolicy", "ElasticPolicy", "Simulator", "simulate",
]
