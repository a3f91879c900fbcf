"""HIP-extension loading and CPU/GPU dispatch for the hot ops."""

from __future__ import annotations

import os

import torch

from . import eager

_EXT = None
_EXT_ERR: Exception | None = None


def _force_eager() -> bool:
    return os.environ.get("DFA_FORCE_EAGER", "0") == "1"


def hip_ext():
    """Load the in-tree HIP extension (built by setup.py / __graft_entry__.build)."""
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        if _EXT is None:
            raise RuntimeError(f"distrifuser_amd HIP extension failed to load: {_EXT_ERR}")
        return _EXT
    try:
        import importlib

        _EXT = importlib.import_module("distrifuser_amd._C")
    except Exception as exc:  # noqa: BLE001
        _EXT_ERR = exc
        raise RuntimeError(
            "distrifuser_amd HIP extension (_C) is not built. Run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {exc}"
        ) from exc
    return _EXT


def hip_ext_available() -> bool:
    try:
        hip_ext()
        return True
    except RuntimeError:
        return False


def _use_hip(x: torch.Tensor) -> bool:
    return x.is_cuda and not _force_eager()


# -- public ops --------------------------------------------------------------


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Attention over (possibly stale, strided) full-sequence KV.

    q: [B, H, Lq, D]; k, v: [B, H, Lkv, D] (arbitrary stride in the Lkv dim
    so the stale-KV flat comm buffer can be consumed without a torch.cat).
    """
    if _use_hip(q):
        ext = hip_ext()
        if q.dtype == torch.bfloat16 and q.shape[-1] in (64,) and hasattr(ext, "flash_attention"):
            return ext.flash_attention(q, k, v)
        # Non-bf16 / odd head-dims ride PyTorch's SDPA (rocm AOTriton path).
        return eager.flash_attention(q, k, v)
    return eager.flash_attention(q, k, v)


def group_norm_stats(x: torch.Tensor, num_groups: int) -> torch.Tensor:
    if _use_hip(x):
        return hip_ext().group_norm_stats(x, num_groups)
    return eager.group_norm_stats(x, num_groups)


def group_norm_apply(x, mean, meansq, weight, bias, eps, silu=False):
    if _use_hip(x):
        n = x.shape[0]
        g = mean.reshape(n, -1).shape[1]
        return hip_ext().group_norm_apply(
            x,
            mean.reshape(n, g).contiguous(),
            meansq.reshape(n, g).contiguous(),
            weight,
            bias,
            eps,
            silu,
        )
    return eager.group_norm_apply(x, mean, meansq, weight, bias, eps, silu)


def group_norm_silu(x, num_groups, weight, bias, eps, silu=True):
    if _use_hip(x):
        return hip_ext().group_norm_silu(x, num_groups, weight, bias, eps, silu)
    return eager.group_norm_silu(x, num_groups, weight, bias, eps, silu)


def geglu(hidden: torch.Tensor) -> torch.Tensor:
    if _use_hip(hidden):
        return hip_ext().geglu(hidden)
    return eager.geglu(hidden)
