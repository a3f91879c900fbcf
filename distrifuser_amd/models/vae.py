"""Native VAE (AutoencoderKL) decoder.

The reference delegated VAE decode to diffusers; here it is owned. Every
rank decodes the identical full latent (parity with the reference's
replicated decode — the denoise loop is the parallel part). The mid-block
attention (single 512-dim head over up to 480x480 = 230k tokens at
3840x3840) runs on the split-D gfx950 HIP kernel (csrc/vae_attn.hip) on
GPU; the CPU path uses a query-chunked softmax so no full score matrix is
ever materialized. Convolutions ride the implicit-GEMM conv3x3 kernel
(NativeConv2d).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F
from torch import nn

from ..ops import NativeConv2d

from .. import ops


@dataclass(frozen=True)
class VAEDecoderConfig:
    latent_channels: int = 4
    out_channels: int = 3
    block_out_channels: tuple = (128, 256, 512, 512)
    layers_per_block: int = 2
    norm_num_groups: int = 32
    scaling_factor: float = 0.13025  # SDXL


SDXL_VAE = VAEDecoderConfig()
SD_VAE = VAEDecoderConfig(scaling_factor=0.18215)
# 4 up blocks = 8x upscale, same ratio as the real VAE
TINY_VAE = VAEDecoderConfig(
    latent_channels=4, block_out_channels=(16, 16, 32, 32), layers_per_block=1,
    norm_num_groups=8, scaling_factor=0.18215,
)


class VAEResnetBlock(nn.Module):
    """ResNet block without time embedding (VAE variant)."""

    def __init__(self, in_ch: int, out_ch: int, groups: int):
        super().__init__()
        self.norm1 = nn.GroupNorm(groups, in_ch, eps=1e-6)
        self.conv1 = NativeConv2d(in_ch, out_ch, 3, padding=1)
        self.norm2 = nn.GroupNorm(groups, out_ch, eps=1e-6)
        self.conv2 = NativeConv2d(out_ch, out_ch, 3, padding=1)
        self.conv_shortcut = NativeConv2d(in_ch, out_ch, 1) if in_ch != out_ch else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.conv1(ops.group_norm_silu(x, self.norm1.num_groups, self.norm1.weight,
                                           self.norm1.bias, self.norm1.eps))
        res = self.conv_shortcut(x) if self.conv_shortcut is not None else x
        return self.conv2(ops.group_norm_silu(h, self.norm2.num_groups, self.norm2.weight,
                                              self.norm2.bias, self.norm2.eps),
                          residual=res)


def _chunked_single_head_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, chunk: int = 4096
) -> torch.Tensor:
    """[B, L, C] single-head attention with O(chunk*L) memory.

    At 3840^2 the VAE mid block attends over 480*480 = 230k tokens with a
    single 512-dim head; the score slab per chunk is chunk*L fp32
    (4096*230k*4B ~= 3.8 GB) — bounded regardless of resolution.
    """
    scale = q.shape[-1] ** -0.5
    # GPU: bf16 GEMMs (hipBLASLt) with fp32 softmax — at 230k tokens the
    # score GEMM is ~5e16 FLOP and fp32 (vector-ALU only on CDNA4, no fp32
    # MFMA) would take minutes; CPU keeps fp32 for the test oracle.
    mm_dtype = torch.bfloat16 if q.is_cuda else torch.float32
    qs = (q.float() * scale).to(mm_dtype)
    ks = k.to(mm_dtype)
    vs = v.to(mm_dtype)
    outs = []
    for s in range(0, q.shape[1], chunk):
        scores = torch.einsum("bqc,bkc->bqk", qs[:, s : s + chunk], ks)
        probs = scores.float().softmax(dim=-1).to(mm_dtype)
        outs.append(torch.einsum("bqk,bkc->bqc", probs, vs))
    return torch.cat(outs, dim=1).to(q.dtype)


class VAEAttention(nn.Module):
    """Single-head spatial self-attention of the VAE mid block."""

    def __init__(self, channels: int, groups: int):
        super().__init__()
        self.group_norm = nn.GroupNorm(groups, channels, eps=1e-6)
        self.to_q = nn.Linear(channels, channels)
        self.to_k = nn.Linear(channels, channels)
        self.to_v = nn.Linear(channels, channels)
        self.to_out = nn.Linear(channels, channels)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, c, h, w = x.shape
        residual = x
        x = ops.group_norm_silu(x, self.group_norm.num_groups, self.group_norm.weight,
                                self.group_norm.bias, self.group_norm.eps, silu=False)
        x = x.permute(0, 2, 3, 1).reshape(b, h * w, c)
        q, k, v = self.to_q(x), self.to_k(x), self.to_v(x)
        if x.is_cuda and c == 512 and q.dtype == torch.bfloat16:
            out = ops.vae_attention(q, k, v)  # split-D HIP kernel
        elif x.is_cuda and c <= 256:
            out = ops.flash_attention(q[:, None], k[:, None], v[:, None])[:, 0]
        else:
            out = _chunked_single_head_attention(q, k, v)
        out = self.to_out(out)
        return out.reshape(b, h, w, c).permute(0, 3, 1, 2) + residual


class UpDecoderBlock2D(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, layers: int, add_upsample: bool, groups: int):
        super().__init__()
        self.resnets = nn.ModuleList(
            [VAEResnetBlock(in_ch if i == 0 else out_ch, out_ch, groups) for i in range(layers)]
        )
        self.upsampler = NativeConv2d(out_ch, out_ch, 3, padding=1) if add_upsample else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for r in self.resnets:
            x = r(x)
        if self.upsampler is not None:
            x = F.interpolate(x, scale_factor=2.0, mode="nearest")
            x = self.upsampler(x)
        return x


class VAEDecoder(nn.Module):
    def __init__(self, config: VAEDecoderConfig):
        super().__init__()
        self.config = config
        ch = config.block_out_channels
        groups = config.norm_num_groups
        top = ch[-1]
        self.post_quant_conv = NativeConv2d(config.latent_channels, config.latent_channels, 1)
        self.conv_in = NativeConv2d(config.latent_channels, top, 3, padding=1)
        self.mid_resnet_1 = VAEResnetBlock(top, top, groups)
        self.mid_attn = VAEAttention(top, groups)
        self.mid_resnet_2 = VAEResnetBlock(top, top, groups)
        rev = list(reversed(ch))
        blocks = []
        prev = top
        for i, out_ch in enumerate(rev):
            blocks.append(
                UpDecoderBlock2D(
                    prev, out_ch, config.layers_per_block + 1,
                    add_upsample=i < len(rev) - 1, groups=groups,
                )
            )
            prev = out_ch
        self.up_blocks = nn.ModuleList(blocks)
        self.conv_norm_out = nn.GroupNorm(groups, ch[0], eps=1e-6)
        self.conv_out = NativeConv2d(ch[0], config.out_channels, 3, padding=1)

    # -- tiled decode (diffusers AutoencoderKL.enable_tiling parity) --------
    # The mid-block attention is O(L^2) in latent tokens: a full 3840^2
    # decode attends over 230k tokens (~109 TFLOP in that one layer). Tiling
    # decodes overlapping latent tiles independently and linearly blends the
    # seams — the standard high-resolution VAE path.
    use_tiling: bool = False
    tile_latent_size: int = 96
    tile_overlap: int = 16  # latent pixels blended between neighbouring tiles

    def enable_tiling(self, tile_latent_size: int = 96, tile_overlap: int = 16) -> None:
        self.use_tiling = True
        self.tile_latent_size = tile_latent_size
        self.tile_overlap = tile_overlap

    def disable_tiling(self) -> None:
        self.use_tiling = False

    @torch.no_grad()
    def _decode_one(self, z: torch.Tensor) -> torch.Tensor:
        x = self.conv_in(z)
        x = self.mid_resnet_1(x)
        x = self.mid_attn(x)
        x = self.mid_resnet_2(x)
        for block in self.up_blocks:
            x = block(x)
        x = ops.group_norm_silu(x, self.conv_norm_out.num_groups, self.conv_norm_out.weight,
                                self.conv_norm_out.bias, self.conv_norm_out.eps)
        return self.conv_out(x)

    @torch.no_grad()
    def _decode_tiled(self, z: torch.Tensor) -> torch.Tensor:
        ts, ov = self.tile_latent_size, self.tile_overlap
        stride = ts - ov
        b, _, h, w = z.shape
        sf = 8  # spatial upscale of the decoder
        ys = list(range(0, max(h - ov, 1), stride))
        xs = list(range(0, max(w - ov, 1), stride))
        # Full-size tiles decode as ONE batched pass (288 GB HBM3E holds all
        # activations comfortably); ragged edge tiles decode individually.
        tiles = {}
        full = [(y0, x0) for y0 in ys for x0 in xs
                if y0 + ts <= h and x0 + ts <= w]
        if len(full) > 1 and b == 1:
            batch = torch.cat([z[:, :, y0 : y0 + ts, x0 : x0 + ts]
                               for (y0, x0) in full])
            dec = self._decode_one(batch)
            for i, key in enumerate(full):
                tiles[key] = dec[i : i + 1]
        for y0 in ys:
            for x0 in xs:
                if (y0, x0) not in tiles:
                    tiles[(y0, x0)] = self._decode_one(
                        z[:, :, y0 : y0 + ts, x0 : x0 + ts])
        rows = [[tiles[(y0, x0)] for x0 in xs] for y0 in ys]

        def blend_v(a, bt, k):
            k = min(k, a.shape[2], bt.shape[2])
            w_ = torch.linspace(0, 1, k, device=a.device, dtype=a.dtype).view(1, 1, k, 1)
            bt[:, :, :k] = a[:, :, a.shape[2] - k :] * (1 - w_) + bt[:, :, :k] * w_
            return bt

        def blend_h(a, bt, k):
            k = min(k, a.shape[3], bt.shape[3])
            w_ = torch.linspace(0, 1, k, device=a.device, dtype=a.dtype).view(1, 1, 1, k)
            bt[:, :, :, :k] = a[:, :, :, a.shape[3] - k :] * (1 - w_) + bt[:, :, :, :k] * w_
            return bt

        ov_px = ov * sf
        st_px = stride * sf
        out_rows = []
        for i, row in enumerate(rows):
            parts = []
            for j, tile in enumerate(row):
                if i > 0:
                    tile = blend_v(rows[i - 1][j], tile, ov_px)
                if j > 0:
                    tile = blend_h(row[j - 1], tile, ov_px)
                keep_w = st_px if j < len(row) - 1 else tile.shape[3]
                parts.append(tile[:, :, :st_px, :keep_w] if i < len(rows) - 1
                             else tile[:, :, :, :keep_w])
            out_rows.append(torch.cat(parts, dim=3))
        out = torch.cat(out_rows, dim=2)
        return out[:, :, : h * sf, : w * sf]

    @torch.no_grad()
    def decode(self, latents: torch.Tensor) -> torch.Tensor:
        """latents (scaled) -> images in [-1, 1]."""
        z = latents / self.config.scaling_factor
        z = self.post_quant_conv(z)
        if self.use_tiling and (z.shape[2] > self.tile_latent_size
                                or z.shape[3] > self.tile_latent_size):
            return self._decode_tiled(z)
        return self._decode_one(z)

    forward = decode
