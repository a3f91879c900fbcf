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


_FLASH_HEAD_DIMS = (40, 64, 80, 96, 128, 160)  # SD-family head dims


def _flash_ok(q: torch.Tensor, *kv: torch.Tensor) -> bool:
    if q.dtype != torch.bfloat16 or q.shape[-1] not in _FLASH_HEAD_DIMS:
        return False
    for t in (q, *kv):
        if t.stride(-1) != 1:
            return False
        if any(s % 8 != 0 for s in t.stride()[:-1]):
            return False
        if t.dim() == 5 and t.shape[2] > 1 and t.shape[3] % 8 != 0:
            return False  # chunked KV: 8-token windows must not straddle chunks
    return True


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Attention over (possibly strided) full-sequence KV.

    q: [B, H, Lq, D]; k, v: [B, H, Lkv, D] (arbitrary stride in the Lkv dim).
    """
    if _use_hip(q):
        if _flash_ok(q, k, v):
            return hip_ext().flash_attention(q, k, v)
        # Non-bf16 / odd head-dims ride PyTorch's SDPA (rocm AOTriton path).
        return eager.flash_attention(q, k, v)
    return eager.flash_attention(q, k, v)


def flash_attention_chunked(
    q: torch.Tensor, kv_chunks: torch.Tensor, heads: int, dim_head: int
) -> torch.Tensor:
    """Displaced-patch attention: q [B, Lq, inner]; kv_chunks is the flat comm
    buffer viewed as [n_peers, B, L_local, 2*inner] (strided — peer rows live
    in the flat buffer). On GPU the gfx950 kernel walks the chunks in place;
    the eager path materializes the concatenated KV.
    """
    b, lq, inner = q.shape
    n, _, l, _ = kv_chunks.shape
    if _use_hip(q):
        q4 = q.view(b, lq, heads, dim_head).permute(0, 2, 1, 3)
        k = kv_chunks[..., :inner].unflatten(-1, (heads, dim_head)).permute(1, 3, 0, 2, 4)
        v = kv_chunks[..., inner:].unflatten(-1, (heads, dim_head)).permute(1, 3, 0, 2, 4)
        if _flash_ok(q4, k, v):
            o = hip_ext().flash_attention(q4, k, v)  # [B,H,Lq,D] (view of B,Lq,H,D)
            return o.transpose(1, 2).reshape(b, lq, inner)
    full_kv = kv_chunks.permute(1, 0, 2, 3).reshape(b, n * l, 2 * inner)
    k, v = full_kv.split(inner, dim=-1)
    q4 = q.view(b, lq, heads, dim_head).transpose(1, 2)
    k = k.view(b, n * l, heads, dim_head).transpose(1, 2)
    v = v.view(b, n * l, heads, dim_head).transpose(1, 2)
    out = flash_attention(q4, k.contiguous(), v.contiguous())
    return out.transpose(1, 2).reshape(b, lq, inner)


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


def layer_norm(x, weight, bias, eps):
    if _use_hip(x) and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0 and x.shape[-1] <= 2048:
        return hip_ext().layer_norm(x, weight, bias, eps)
    return eager.layer_norm(x, weight, bias, eps)


def add_layer_norm(x, res, weight, bias, eps):
    """(x + res, LN(x + res)) in one kernel on GPU."""
    if _use_hip(x) and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0 and x.shape[-1] <= 2048:
        s, y = hip_ext().add_layer_norm(x, res, weight, bias, eps)
        return s, y
    return eager.add_layer_norm(x, res, weight, bias, eps)


def vae_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Single-head d=512 attention, q/k/v [B, L, 512] (VAE mid block)."""
    if _use_hip(q) and q.dtype == torch.bfloat16 and q.shape[-1] == 512:
        return hip_ext().vae_attention(q, k, v)
    return eager.vae_attention(q, k, v)


def geglu(hidden: torch.Tensor) -> torch.Tensor:
    if _use_hip(hidden):
        return hip_ext().geglu(hidden)
    return eager.geglu(hidden)
