"""ResNet blocks and up/down sampling (native; structural semantics match
the diffusers blocks the reference leaned on, numerics defined by our ops).

GroupNorm+SiLU pairs are fused (one kernel) — in SD-family U-Nets every GN
inside a ResBlock is followed by SiLU, so the fused epilogue halves the
HBM traffic of the norm (MI355X is bandwidth-bound at 8 TB/s).
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch import nn

from ..ops import NativeConv2d
from .layers import LayerFactory


class ResnetBlock2D(nn.Module):
    """norm1+SiLU -> conv1 -> +temb -> norm2+SiLU -> conv2 -> +shortcut.

    Under tensor parallelism uses the Megatron pairing (reference
    tp/resnet.py:43-104): conv1 out-channel shard + temb-proj shard +
    group-sharded norm2 + conv2 in-channel shard, ONE all-reduce per block
    with the bias applied after the reduce.
    """

    def __init__(
        self,
        in_channels: int,
        out_channels: int,
        temb_channels: int,
        *,
        factory: LayerFactory,
        groups: int = 32,
        eps: float = 1e-5,
    ):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        state = factory.state
        n = state.config.n_device_per_batch
        self.tp_pair = (
            factory.parallelism == "tensor"
            and n > 1
            and out_channels % n == 0
            and groups % n == 0
        )
        self.state = state

        self.norm1 = factory.group_norm(groups, in_channels, eps=eps, fuse_silu=True)
        if self.tp_pair:
            rank = state.config.split_idx()
            self.c_local = out_channels // n
            self.c_start = rank * self.c_local
            self.conv1 = NativeConv2d(in_channels, self.c_local, 3, padding=1, bias=True)
            self.time_emb_proj = nn.Linear(temb_channels, self.c_local)
            self.norm2 = nn.GroupNorm(groups // n, self.c_local, eps=eps)
            self.conv2 = NativeConv2d(self.c_local, out_channels, 3, padding=1, bias=False)
            self.conv2_bias = nn.Parameter(torch.zeros(out_channels))
        else:
            self.conv1 = factory.conv2d(in_channels, out_channels, 3, 1, 1)
            self.time_emb_proj = nn.Linear(temb_channels, out_channels)
            self.norm2 = factory.group_norm(groups, out_channels, eps=eps, fuse_silu=True)
            self.conv2 = factory.conv2d(out_channels, out_channels, 3, 1, 1)
        if in_channels != out_channels:
            self.conv_shortcut = NativeConv2d(in_channels, out_channels, 1)
        else:
            self.conv_shortcut = None

    def copy_from_full_tp(self, sd: dict, prefix: str) -> None:
        """Load a full (unsharded) diffusers-layout state dict into TP shards."""
        assert self.tp_pair
        s, e = self.c_start, self.c_start + self.c_local
        self.conv1.weight.data.copy_(sd[f"{prefix}conv1.weight"][s:e])
        self.conv1.bias.data.copy_(sd[f"{prefix}conv1.bias"][s:e])
        self.time_emb_proj.weight.data.copy_(sd[f"{prefix}time_emb_proj.weight"][s:e])
        self.time_emb_proj.bias.data.copy_(sd[f"{prefix}time_emb_proj.bias"][s:e])
        self.norm2.weight.data.copy_(sd[f"{prefix}norm2.weight"][s:e])
        self.norm2.bias.data.copy_(sd[f"{prefix}norm2.bias"][s:e])
        self.conv2.weight.data.copy_(sd[f"{prefix}conv2.weight"][:, s:e])
        self.conv2_bias.data.copy_(sd[f"{prefix}conv2.bias"])

    def forward(self, x: torch.Tensor, temb: torch.Tensor) -> torch.Tensor:
        # time-embedding add fused into conv1's epilogue (one scalar per
        # output row, like the conv bias — csrc/conv.hip bias2)
        import os
        t2 = self.time_emb_proj(F.silu(temb))
        if os.environ.get("DFA_NO_TEMB_FUSE", "0") == "1":
            h = self.conv1(self.norm1(x)) + t2[:, :, None, None]
        else:
            h = self.conv1(self.norm1(x), bias2=t2)
        if self.tp_pair:
            h = F.silu(self.norm2(h))
            h = self.conv2(h)
            if dist.is_initialized():
                dist.all_reduce(h, op=dist.ReduceOp.SUM, group=self.state.config.batch_group)
            h = h + self.conv2_bias.view(1, -1, 1, 1)
        else:
            # shortcut add fused into the conv2 epilogue (HIP kernel
            # residual pointer; eager paths fall back to out + residual)
            res = self.conv_shortcut(x) if self.conv_shortcut is not None else x
            return self.conv2(self.norm2(h), residual=res)
        if self.conv_shortcut is not None:
            x = self.conv_shortcut(x)
        return x + h


class Downsample2D(nn.Module):
    def __init__(self, channels: int, *, factory: LayerFactory):
        super().__init__()
        self.conv = factory.conv2d(channels, channels, 3, stride=2, padding=1, tp_shard=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.conv(x)


class Upsample2D(nn.Module):
    def __init__(self, channels: int, *, factory: LayerFactory):
        super().__init__()
        self.conv = factory.conv2d(channels, channels, 3, stride=1, padding=1, tp_shard=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # Nearest-neighbour 2x is purely row-local, so it composes with row
        # patching with no extra communication.
        x = F.interpolate(x, scale_factor=2.0, mode="nearest")
        return self.conv(x)
