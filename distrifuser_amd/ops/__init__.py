"""Device ops: hand-written gfx950 HIP kernels with eager CPU references.

Dispatch rule: on a ROCm GPU the HIP extension is REQUIRED — a missing
extension raises instead of silently falling back to eager PyTorch (so GPU
runs always exercise the native kernels). On CPU the eager reference path
runs, which is what the non-GPU test suite checks numerics against.
Set DFA_FORCE_EAGER=1 to force the eager path on GPU (A/B debugging only).
"""

from .conv import NativeConv2d, conv3x3_halo, pack_conv3x3_weight
from .dispatch import (
    add_layer_norm,
    flash_attention,
    flash_attention_chunked,
    geglu,
    group_norm_apply,
    group_norm_silu,
    group_norm_stats,
    hip_ext,
    hip_ext_available,
    layer_norm,
    vae_attention,
)

__all__ = [
    "NativeConv2d",
    "add_layer_norm",
    "layer_norm",
    "conv3x3_halo",
    "flash_attention",
    "flash_attention_chunked",
    "geglu",
    "group_norm_apply",
    "group_norm_silu",
    "group_norm_stats",
    "hip_ext",
    "hip_ext_available",
    "pack_conv3x3_weight",
    "vae_attention",
]
