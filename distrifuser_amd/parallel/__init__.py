from .state import ParallelState
from .patch_ops import PatchConv2d, PatchGroupNorm, PatchSelfAttention, CachedCrossAttention

__all__ = [
    "ParallelState",
    "PatchConv2d",
    "PatchGroupNorm",
    "PatchSelfAttention",
    "CachedCrossAttention",
]
