"""Spatial transformer (Transformer2DModel equivalent), native."""

from __future__ import annotations

import torch
from torch import nn

from .. import ops

from .layers import LayerFactory


class BasicTransformerBlock(nn.Module):
    def __init__(
        self,
        dim: int,
        heads: int,
        dim_head: int,
        cross_dim: int,
        *,
        factory: LayerFactory,
    ):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim)
        self.attn1 = factory.self_attention(dim, heads, dim_head)
        self.norm2 = nn.LayerNorm(dim)
        self.attn2 = factory.cross_attention(dim, cross_dim, heads, dim_head)
        self.norm3 = nn.LayerNorm(dim)
        self.ff = factory.feed_forward(dim)

    def forward(self, x: torch.Tensor, encoder_hidden_states: torch.Tensor,
                pending=None):
        # residual adds fused into the next LayerNorm (one kernel emits
        # both the running sum and the normalized view — csrc/layernorm.hip).
        # `pending` is the PREVIOUS block's un-added ff output, folded into
        # this block's norm1 the same way; the pair (x, ff_out) is returned
        # so the chain never materializes a standalone add.
        if pending is None:
            n1 = ops.layer_norm(x, self.norm1.weight, self.norm1.bias, self.norm1.eps)
        else:
            x, n1 = ops.add_layer_norm(x, pending, self.norm1.weight, self.norm1.bias,
                                       self.norm1.eps)
        x, h = ops.add_layer_norm(x, self.attn1(n1), self.norm2.weight, self.norm2.bias,
                                  self.norm2.eps)
        x, n3 = ops.add_layer_norm(x, self.attn2(h, encoder_hidden_states),
                                   self.norm3.weight, self.norm3.bias, self.norm3.eps)
        return x, self.ff(n3)


class Transformer2DModel(nn.Module):
    """GN -> proj_in -> N transformer blocks -> proj_out -> +residual.

    ``use_linear_projection`` (SDXL) projects after flattening with a Linear;
    SD1.5 uses 1x1 convs. Both are spatially local ops, so they compose with
    row patching untouched.
    """

    def __init__(
        self,
        channels: int,
        heads: int,
        dim_head: int,
        layers: int,
        cross_dim: int,
        *,
        factory: LayerFactory,
        groups: int = 32,
        use_linear_projection: bool = True,
    ):
        super().__init__()
        inner = heads * dim_head
        self.use_linear_projection = use_linear_projection
        self.norm = factory.group_norm(groups, channels, eps=1e-6, fuse_silu=False)
        if use_linear_projection:
            self.proj_in = nn.Linear(channels, inner)
            self.proj_out = nn.Linear(inner, channels)
        else:
            self.proj_in = nn.Conv2d(channels, inner, 1)
            self.proj_out = nn.Conv2d(inner, channels, 1)
        self.transformer_blocks = nn.ModuleList(
            [
                BasicTransformerBlock(inner, heads, dim_head, cross_dim, factory=factory)
                for _ in range(layers)
            ]
        )

    def forward(self, x: torch.Tensor, encoder_hidden_states: torch.Tensor) -> torch.Tensor:
        b, c, h, w = x.shape
        residual = x
        x = self.norm(x)
        if self.use_linear_projection:
            x = x.permute(0, 2, 3, 1).reshape(b, h * w, c)
            x = self.proj_in(x)
        else:
            x = self.proj_in(x)
            inner = x.shape[1]
            x = x.permute(0, 2, 3, 1).reshape(b, h * w, inner)
        import os
        if os.environ.get("DFA_NO_PENDING_CHAIN", "0") == "1":
            for block in self.transformer_blocks:
                x, p = block(x, encoder_hidden_states, None)
                x = x + p
        else:
            pending = None
            for block in self.transformer_blocks:
                x, pending = block(x, encoder_hidden_states, pending)
            x = x + pending
        if self.use_linear_projection:
            x = self.proj_out(x)
            x = x.reshape(b, h, w, c).permute(0, 3, 1, 2)
        else:
            inner = x.shape[-1]
            x = x.reshape(b, h, w, inner).permute(0, 3, 1, 2)
            x = self.proj_out(x)
        return x + residual
