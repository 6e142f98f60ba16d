from ._ext import get_ext, have_ext
from .norms import FusedLayerNorm, FusedRMSNorm
from .linear import MFMALinear
from .optim import FusedAdamW, FusedSGD
from .bn import FusedBN2d
from .ce import fused_cross_entropy
from .embedding import FusedEmbedding

__all__ = ["get_ext", "have_ext", "FusedLayerNorm", "FusedRMSNorm",
           "MFMALinear", "FusedAdamW", "FusedSGD", "FusedBN2d", "fused_cross_entropy", "FusedEmbedding"]
