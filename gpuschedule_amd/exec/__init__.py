from .backend import TorchBackend, init_distributed
from .checkpoint import CheckpointStore
from .gangs import GangGroups
from .trainer import JobRunner

__all__ = ["TorchBackend", "init_distributed", "CheckpointStore",
           "GangGroups", "JobRunner"]
