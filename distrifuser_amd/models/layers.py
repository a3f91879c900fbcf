"""Layer factory: picks the parallelism-aware implementation of each op.

The reference rewired a diffusers U-Net after the fact via ``setattr``
(reference models/distri_sdxl_unet_pp.py:19-40). We own the model, so each
building block asks this factory for the right variant at CONSTRUCTION time:

    parallelism="patch"        -> Patch* modules (degrade to plain at n=1)
    parallelism="tensor"       -> TP* modules (Megatron sharding)
    parallelism="naive_patch"  -> plain modules (slicing happens at the
                                  model level, with zero cross-patch comm)
"""

from __future__ import annotations

import torch
from torch import nn

from .. import ops
from ..ops import NativeConv2d
from ..parallel.patch_ops import (
    CachedCrossAttention,
    PatchConv2d,
    PatchGroupNorm,
    PatchSelfAttention,
)
from ..parallel.state import ParallelState
from ..parallel.tensor_ops import TPAttention, TPConv2d, TPFeedForward


class PlainGroupNorm(nn.GroupNorm):
    """GroupNorm with optionally fused SiLU, routed through ops (HIP kernel)."""

    def __init__(self, num_groups, num_channels, eps=1e-5, affine=True, fuse_silu=False):
        super().__init__(num_groups, num_channels, eps=eps, affine=affine)
        self.fuse_silu = fuse_silu

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.group_norm_silu(
            x, self.num_groups, self.weight, self.bias, self.eps, silu=self.fuse_silu
        )


class PlainSelfAttention(nn.Module):
    def __init__(self, query_dim: int, heads: int, dim_head: int, out_bias: bool = True):
        super().__init__()
        inner = heads * dim_head
        self.heads, self.dim_head = heads, dim_head
        self.to_q = nn.Linear(query_dim, inner, bias=False)
        self.to_kv = nn.Linear(query_dim, 2 * inner, bias=False)
        self.to_out = nn.Linear(inner, query_dim, bias=out_bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, l, _ = x.shape
        inner = self.heads * self.dim_head
        q = self.to_q(x).view(b, l, self.heads, self.dim_head).transpose(1, 2)
        k, v = self.to_kv(x).split(inner, dim=-1)
        k = k.view(b, l, self.heads, self.dim_head).transpose(1, 2)
        v = v.view(b, l, self.heads, self.dim_head).transpose(1, 2)
        out = ops.flash_attention(q, k, v).transpose(1, 2).reshape(b, l, inner)
        return self.to_out(out)


class PlainCrossAttention(nn.Module):
    def __init__(self, query_dim: int, cross_dim: int, heads: int, dim_head: int, out_bias=True):
        super().__init__()
        inner = heads * dim_head
        self.heads, self.dim_head = heads, dim_head
        self.to_q = nn.Linear(query_dim, inner, bias=False)
        self.to_kv = nn.Linear(cross_dim, 2 * inner, bias=False)
        self.to_out = nn.Linear(inner, query_dim, bias=out_bias)

    def forward(self, x: torch.Tensor, encoder_hidden_states: torch.Tensor) -> torch.Tensor:
        b, lq, _ = x.shape
        inner = self.heads * self.dim_head
        lkv = encoder_hidden_states.shape[1]
        q = self.to_q(x).view(b, lq, self.heads, self.dim_head).transpose(1, 2)
        k, v = self.to_kv(encoder_hidden_states).split(inner, dim=-1)
        k = k.view(b, lkv, self.heads, self.dim_head).transpose(1, 2)
        v = v.view(b, lkv, self.heads, self.dim_head).transpose(1, 2)
        out = ops.flash_attention(q, k, v).transpose(1, 2).reshape(b, lq, inner)
        return self.to_out(out)


class FeedForward(nn.Module):
    """GEGLU MLP: Linear(dim -> 2*mult*dim), a*gelu(gate), Linear(mult*dim -> dim)."""

    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = dim * mult
        self.proj_in = nn.Linear(dim, inner * 2)
        self.proj_out = nn.Linear(inner, dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.proj_out(ops.geglu(self.proj_in(x)))


class LayerFactory:
    def __init__(self, state: ParallelState):
        self.state = state
        self.parallelism = state.config.parallelism

    # Spatial ops ----------------------------------------------------------

    def conv2d(
        self,
        in_ch: int,
        out_ch: int,
        kernel: int = 3,
        stride: int = 1,
        padding: int = 1,
        bias: bool = True,
        *,
        is_first_layer: bool = False,
        tp_shard: bool = False,
    ) -> nn.Module:
        """tp_shard: under tensor parallelism, input-channel-shard this conv
        (samplers + conv_out; reference distri_sdxl_unet_tp.py:27-38)."""
        if self.parallelism == "patch":
            return PatchConv2d(
                in_ch, out_ch, kernel, stride, padding, bias,
                state=self.state, is_first_layer=is_first_layer,
            )
        if (
            self.parallelism == "tensor"
            and tp_shard
            and self.state.config.n_device_per_batch > 1
            and in_ch % self.state.config.n_device_per_batch == 0
        ):
            return TPConv2d(in_ch, out_ch, kernel, stride, padding, bias, state=self.state)
        return NativeConv2d(in_ch, out_ch, kernel, stride=stride, padding=padding, bias=bias)

    def group_norm(self, groups: int, ch: int, eps: float = 1e-5, fuse_silu: bool = False):
        if self.parallelism == "patch":
            return PatchGroupNorm(groups, ch, eps=eps, state=self.state, fuse_silu=fuse_silu)
        return PlainGroupNorm(groups, ch, eps=eps, fuse_silu=fuse_silu)

    # Sequence ops ---------------------------------------------------------

    def self_attention(self, dim: int, heads: int, dim_head: int) -> nn.Module:
        if self.parallelism == "patch":
            return PatchSelfAttention(dim, heads, dim_head, state=self.state)
        if self.parallelism == "tensor" and self.state.config.n_device_per_batch > 1:
            return TPAttention(dim, heads, dim_head, state=self.state)
        return PlainSelfAttention(dim, heads, dim_head)

    def cross_attention(self, dim: int, cross_dim: int, heads: int, dim_head: int) -> nn.Module:
        if self.parallelism == "patch":
            return CachedCrossAttention(dim, cross_dim, heads, dim_head, state=self.state)
        if self.parallelism == "tensor" and self.state.config.n_device_per_batch > 1:
            return TPAttention(dim, heads, dim_head, cross_dim=cross_dim, state=self.state)
        return PlainCrossAttention(dim, cross_dim, heads, dim_head)

    def feed_forward(self, dim: int, mult: int = 4) -> nn.Module:
        if self.parallelism == "tensor" and self.state.config.n_device_per_batch > 1:
            return TPFeedForward(dim, mult, state=self.state)
        return FeedForward(dim, mult)
