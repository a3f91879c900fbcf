"""Eager PyTorch reference implementations of every HIP kernel.

These define the NUMERICS contract: each gfx950 kernel in ops/hip/ is
unit-tested against the fp32 run of the matching function here
(tests/test_ops_gpu.py). They also serve as the CPU execution path for the
no-GPU test suite.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """softmax(q k^T / sqrt(d)) v with rectangular Lq x Lkv.

    q: [B, H, Lq, D]; k, v: [B, H, Lkv, D]. Returns [B, H, Lq, D].
    """
    return F.scaled_dot_product_attention(q, k, v, dropout_p=0.0, is_causal=False)


def group_norm_stats(x: torch.Tensor, num_groups: int) -> torch.Tensor:
    """Per-(sample, group) first/second moments over (group_size, H, W).

    x: [N, C, H, W] -> [2, N, G, 1, 1, 1] stacked (E[x], E[x^2]), computed in
    fp32 and cast back to x.dtype (the stale-stats comm slot dtype).
    """
    n, c, h, w = x.shape
    xg = x.reshape(n, num_groups, -1).float()
    mean = xg.mean(dim=-1)
    meansq = (xg * xg).mean(dim=-1)
    return torch.stack([mean, meansq], dim=0).reshape(2, n, num_groups, 1, 1, 1).to(x.dtype)


def group_norm_apply(
    x: torch.Tensor,
    mean: torch.Tensor,
    meansq: torch.Tensor,
    weight: torch.Tensor | None,
    bias: torch.Tensor | None,
    eps: float,
    silu: bool = False,
) -> torch.Tensor:
    """Normalize x with externally supplied group stats; optionally fuse SiLU.

    x: [N, C, H, W]; mean/meansq: broadcastable to [N, G] (e.g. the
    [N, G, 1, 1, 1] comm-slot views). var = E[x^2] - E[x]^2 (population
    variance, matching F.group_norm; we deliberately drop the reference's
    n/(n-1) Bessel factor — see reference pp/groupnorm.py:65-66 — so that
    full_sync output is bit-identical to the single-GPU oracle).
    """
    n, c, h, w = x.shape
    g = mean.reshape(n, -1).shape[1]
    mean = mean.reshape(n, g, 1).float()
    meansq = meansq.reshape(n, g, 1).float()
    var = (meansq - mean * mean).clamp_min_(0.0)
    inv_std = torch.rsqrt(var + eps)
    out = (x.reshape(n, g, -1).float() - mean) * inv_std
    out = out.reshape(n, c, h, w)
    if weight is not None:
        out = out * weight.float().view(1, -1, 1, 1)
    if bias is not None:
        out = out + bias.float().view(1, -1, 1, 1)
    if silu:
        out = F.silu(out)
    return out.to(x.dtype)


def group_norm_silu(
    x: torch.Tensor,
    num_groups: int,
    weight: torch.Tensor | None,
    bias: torch.Tensor | None,
    eps: float,
    silu: bool = True,
) -> torch.Tensor:
    """Plain (single-device) GroupNorm with fused SiLU epilogue."""
    out = F.group_norm(x.float(), num_groups, None if weight is None else weight.float(),
                       None if bias is None else bias.float(), eps)
    if silu:
        out = F.silu(out)
    return out.to(x.dtype)


def geglu(hidden: torch.Tensor) -> torch.Tensor:
    """GEGLU gate: split last dim in half, return a * gelu(b) (tanh=false)."""
    a, b = hidden.chunk(2, dim=-1)
    return a * F.gelu(b)


def layer_norm(x, weight, bias, eps):
    import torch.nn.functional as _F

    return _F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def add_layer_norm(x, res, weight, bias, eps):
    """Fused residual add + LayerNorm: returns (x + res, LN(x + res))."""
    import torch.nn.functional as _F

    s = x + res
    return s, _F.layer_norm(s, (s.shape[-1],), weight, bias, eps)


def vae_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Single-head attention over [B, L, C] (fp32 softmax)."""
    scale = q.shape[-1] ** -0.5
    scores = torch.einsum("bqc,bkc->bqk", q.float() * scale, k.float())
    probs = scores.softmax(dim=-1)
    return torch.einsum("bqk,bkc->bqc", probs, v.float()).to(q.dtype)


def conv3x3_halo(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor | None,
    stride: int = 1,
    top: torch.Tensor | None = None,
    bot: torch.Tensor | None = None,
) -> torch.Tensor:
    """3x3 pad-1 conv whose top/bottom halo rows come from separate tensors.

    x: [B, Cin, H, W]; top/bot: [B, Cin, 1, W] neighbour rows (None => zero
    pad at that border). Semantics of the HIP conv3x3 kernel: the reference
    materializes cat([top, x, bot]) per conv
    (/root/reference/distrifuser/modules/pp/conv2d.py:72-88); the kernel —
    and this oracle — treat the halos as extra input rows in place.
    """
    b, c, h, w = x.shape
    parts = []
    pad_top = 1 if top is None else 0
    pad_bot = 1 if bot is None else 0
    if top is not None:
        parts.append(top.reshape(b, c, 1, w))
    parts.append(x)
    if bot is not None:
        parts.append(bot.reshape(b, c, 1, w))
    full = torch.cat(parts, dim=2) if len(parts) > 1 else x
    full = F.pad(full, [1, 1, pad_top, pad_bot])
    out = F.conv2d(full, weight, bias, stride=stride)
    # pad-1 stride-s output has ceil(H/s) rows regardless of halo presence
    ho = (h - 1) // stride + 1
    return out[:, :, :ho, :]
