from .base import Policy
from .fifo import FIFOPolicy
from .srtf import SRTFPolicy
from .las import LASPolicy
from .gandiva import GandivaPolicy
from .elastic import ElasticPolicy

POLICIES = {
    "fifo": FIFOPolicy,
    "srtf": SRTFPolicy,
    "las": LASPolicy,
    "gandiva": GandivaPolicy,
    "elastic": ElasticPolicy,
}


def make_policy(name: str, **kw) -> Policy:
    try:
        cls = POLICIES[name]
    except KeyError:
        raise ValueError(f"unknown policy {name!r}; have {sorted(POLICIES)}")
    return cls(**kw)


__all__ = ["Policy", "FIFOPolicy", "SRTFPolicy", "LASPolicy",
           "GandivaPolicy", "ElasticPolicy", "POLICIES", "make_policy"]
