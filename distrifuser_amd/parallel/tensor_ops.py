"""Tensor-parallel (Megatron-style) layers for ``parallelism="tensor"``.

Semantics parity with the reference TP modules
(/root/reference/distrifuser/modules/tp/*.py): heads / channels are sharded
across the ``n_device_per_batch`` ranks of one CFG branch, partial outputs
are summed with ONE all-reduce per layer (or per ResNet block via the
Megatron conv1-out-shard / conv2-in-shard pairing), and biases are applied
once AFTER the reduce. On MI355X the small per-block all-reduces are
latency-bound over xGMI, so fusing the block's two convs into a single
reduce matters more than on NVSwitch.

All shards are materialized directly (no full-weight intermediate); the
weight loader slices full checkpoints via each module's ``copy_from_full``.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch import nn

from .. import ops
from .state import ParallelState


def _tp_degree(state: ParallelState) -> int:
    if state.config.parallelism != "tensor":
        return 1
    return state.config.n_device_per_batch


def head_shard(heads: int, n: int, rank: int) -> tuple[int, int]:
    """(start, count) of this rank's heads; first ``heads % n`` ranks get +1."""
    base, rem = divmod(heads, n)
    count = base + (1 if rank < rem else 0)
    start = rank * base + min(rank, rem)
    return start, count


class TPAttention(nn.Module):
    """Head-sharded attention (self or cross).

    Full-sequence SDPA on the local heads, output projection WITHOUT bias,
    all-reduce(SUM) of the partial outputs over the batch group, bias added
    once after the reduce (reference tp/attention.py:150-161).
    """

    def __init__(
        self,
        query_dim: int,
        heads: int,
        dim_head: int,
        cross_dim: int | None = None,
        out_bias: bool = True,
        *,
        state: ParallelState,
    ):
        super().__init__()
        self.state = state
        self.heads = heads
        self.dim_head = dim_head
        n = _tp_degree(state)
        rank_in_group = state.config.split_idx()
        self.h_start, self.h_count = head_shard(heads, n, rank_in_group)
        local_inner = self.h_count * dim_head
        kv_dim = cross_dim if cross_dim is not None else query_dim
        # Zero-head ranks keep 1-element dummies out of the graph: they just
        # contribute zeros to the reduce (reference tp/attention.py:153-158).
        self.to_q = nn.Linear(query_dim, max(local_inner, 0), bias=False) if local_inner else None
        self.to_k = nn.Linear(kv_dim, local_inner, bias=False) if local_inner else None
        self.to_v = nn.Linear(kv_dim, local_inner, bias=False) if local_inner else None
        self.to_out_weight = (
            nn.Parameter(torch.empty(query_dim, local_inner)) if local_inner else None
        )
        if self.to_out_weight is not None:
            nn.init.kaiming_uniform_(self.to_out_weight, a=5**0.5)
        self.to_out_bias = nn.Parameter(torch.zeros(query_dim)) if out_bias else None
        self.query_dim = query_dim

    def copy_from_full(self, wq, wk, wv, wo, bo=None) -> None:
        """Slice full [inner, dim] projection weights into this rank's shard."""
        s = self.h_start * self.dim_head
        e = (self.h_start + self.h_count) * self.dim_head
        if self.h_count:
            self.to_q.weight.data.copy_(wq[s:e])
            self.to_k.weight.data.copy_(wk[s:e])
            self.to_v.weight.data.copy_(wv[s:e])
            self.to_out_weight.data.copy_(wo[:, s:e])
        if bo is not None and self.to_out_bias is not None:
            self.to_out_bias.data.copy_(bo)

    def forward(self, x: torch.Tensor, encoder_hidden_states: torch.Tensor | None = None):
        b, l, _ = x.shape
        ctx = x if encoder_hidden_states is None else encoder_hidden_states
        if self.h_count:
            from .. import ops

            q = self.to_q(x).view(b, l, self.h_count, self.dim_head).transpose(1, 2)
            k = self.to_k(ctx).view(b, ctx.shape[1], self.h_count, self.dim_head).transpose(1, 2)
            v = self.to_v(ctx).view(b, ctx.shape[1], self.h_count, self.dim_head).transpose(1, 2)
            out = ops.flash_attention(q, k, v)
            out = out.transpose(1, 2).reshape(b, l, self.h_count * self.dim_head)
            out = F.linear(out, self.to_out_weight)  # bias deferred past the reduce
        else:
            out = x.new_zeros(b, l, self.query_dim)
        if _tp_degree(self.state) > 1 and dist.is_initialized():
            dist.all_reduce(out, op=dist.ReduceOp.SUM, group=self.state.config.batch_group)
        if self.to_out_bias is not None:
            out = out + self.to_out_bias
        return out


class TPFeedForward(nn.Module):
    """Column-sharded GEGLU fc1 + row-sharded fc2, bias after the reduce.

    Reference tp/feed_forward.py:27-79; both the value and gate halves of the
    GEGLU projection are sharded so the local GEGLU is self-contained.
    """

    def __init__(self, dim: int, mult: int = 4, *, state: ParallelState):
        super().__init__()
        self.state = state
        self.dim = dim
        inner = dim * mult
        n = _tp_degree(state)
        rank = state.config.split_idx()
        assert inner % n == 0, f"ff inner dim {inner} must divide TP degree {n}"
        self.inner_local = inner // n
        self.shard_start = rank * self.inner_local
        self.fc1 = nn.Linear(dim, 2 * self.inner_local, bias=True)
        self.fc2_weight = nn.Parameter(torch.empty(dim, self.inner_local))
        nn.init.kaiming_uniform_(self.fc2_weight, a=5**0.5)
        self.fc2_bias = nn.Parameter(torch.zeros(dim))

    def copy_from_full(self, w1, b1, w2, b2) -> None:
        """w1: [2*inner, dim] with [value; gate] halves; w2: [dim, inner]."""
        inner = w1.shape[0] // 2
        s, e = self.shard_start, self.shard_start + self.inner_local
        self.fc1.weight.data[: self.inner_local].copy_(w1[s:e])
        self.fc1.weight.data[self.inner_local :].copy_(w1[inner + s : inner + e])
        self.fc1.bias.data[: self.inner_local].copy_(b1[s:e])
        self.fc1.bias.data[self.inner_local :].copy_(b1[inner + s : inner + e])
        self.fc2_weight.data.copy_(w2[:, s:e])
        self.fc2_bias.data.copy_(b2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.fc1(x)
        a, gate = h.chunk(2, dim=-1)
        h = a * F.gelu(gate)
        out = F.linear(h, self.fc2_weight)
        if _tp_degree(self.state) > 1 and dist.is_initialized():
            dist.all_reduce(out, op=dist.ReduceOp.SUM, group=self.state.config.batch_group)
        return out + self.fc2_bias


class TPConv2d(nn.Module):
    """Input-channel-sharded conv: bias-free local conv + all-reduce + bias.

    Reference tp/conv2d.py:15-55. Used for conv_out and the down/up-sampler
    convs (conv_in keeps 4 input channels and stays replicated).
    """

    def __init__(
        self,
        in_channels: int,
        out_channels: int,
        kernel_size: int = 3,
        stride: int = 1,
        padding: int = 1,
        bias: bool = True,
        *,
        state: ParallelState,
    ):
        super().__init__()
        self.state = state
        n = _tp_degree(state)
        rank = state.config.split_idx()
        assert in_channels % n == 0, f"conv in_channels {in_channels} must divide TP degree {n}"
        self.c_local = in_channels // n
        self.c_start = rank * self.c_local
        self.conv = ops.NativeConv2d(
            self.c_local, out_channels, kernel_size, stride=stride, padding=padding, bias=False
        )
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None

    def copy_from_full(self, w, b=None) -> None:
        self.conv.weight.data.copy_(w[:, self.c_start : self.c_start + self.c_local])
        if b is not None and self.bias is not None:
            self.bias.data.copy_(b)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.conv(x[:, self.c_start : self.c_start + self.c_local])
        if _tp_degree(self.state) > 1 and dist.is_initialized():
            dist.all_reduce(out, op=dist.ReduceOp.SUM, group=self.state.config.batch_group)
        if self.bias is not None:
            out = out + self.bias.view(1, -1, 1, 1)
        return out
