"""Native UNet2DCondition for the SD family (SDXL / SD1.5 / tiny-test).

The reference reused diffusers' ``UNet2DConditionModel`` wholesale
(reference pipelines.py:26-28); this is our own implementation, built from
the parallelism-aware layer factory so displaced patch parallelism, tensor
parallelism, and plain execution are construction-time choices, not
monkey-patches. State-dict naming follows the diffusers layout so real HF
checkpoints load 1:1 (models/weights.py).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
from torch import nn

from ..parallel.state import ParallelState
from .embeddings import TimestepEmbedding, sinusoidal_embedding
from .layers import LayerFactory
from .resnet import Downsample2D, ResnetBlock2D, Upsample2D
from .transformer import Transformer2DModel


@dataclass(frozen=True)
class UNetConfig:
    in_channels: int = 4
    out_channels: int = 4
    block_out_channels: tuple = (320, 640, 1280)
    down_block_types: tuple = ("DownBlock2D", "CrossAttnDownBlock2D", "CrossAttnDownBlock2D")
    layers_per_block: int = 2
    transformer_layers_per_block: tuple = (1, 2, 10)
    num_attention_heads: tuple = (5, 10, 20)
    cross_attention_dim: int = 2048
    norm_num_groups: int = 32
    use_linear_projection: bool = True
    addition_embed_type: str | None = "text_time"
    addition_time_embed_dim: int = 256
    projection_class_embeddings_input_dim: int = 2816
    sample_size: int = 128

    @property
    def time_embed_dim(self) -> int:
        return self.block_out_channels[0] * 4

    @property
    def up_block_types(self) -> tuple:
        mapping = {"DownBlock2D": "UpBlock2D", "CrossAttnDownBlock2D": "CrossAttnUpBlock2D"}
        return tuple(mapping[t] for t in reversed(self.down_block_types))


# stabilityai/stable-diffusion-xl-base-1.0 unet/config.json shapes
SDXL_UNET = UNetConfig()

# runwayml/stable-diffusion-v1-5 unet/config.json shapes
SD15_UNET = UNetConfig(
    block_out_channels=(320, 640, 1280, 1280),
    down_block_types=(
        "CrossAttnDownBlock2D",
        "CrossAttnDownBlock2D",
        "CrossAttnDownBlock2D",
        "DownBlock2D",
    ),
    transformer_layers_per_block=(1, 1, 1, 1),
    num_attention_heads=(8, 8, 8, 8),
    cross_attention_dim=768,
    use_linear_projection=False,
    addition_embed_type=None,
    sample_size=64,
)

# stabilityai/stable-diffusion-2-1 unet/config.json shapes
SD21_UNET = UNetConfig(
    block_out_channels=(320, 640, 1280, 1280),
    down_block_types=(
        "CrossAttnDownBlock2D",
        "CrossAttnDownBlock2D",
        "CrossAttnDownBlock2D",
        "DownBlock2D",
    ),
    transformer_layers_per_block=(1, 1, 1, 1),
    num_attention_heads=(5, 10, 20, 20),  # head_dim 64 per block
    cross_attention_dim=1024,
    use_linear_projection=True,
    addition_embed_type=None,
    sample_size=96,
)

# Tiny config for CPU tests (structure-preserving, 64x smaller)
TINY_UNET = UNetConfig(
    block_out_channels=(32, 64),
    down_block_types=("DownBlock2D", "CrossAttnDownBlock2D"),
    layers_per_block=1,
    transformer_layers_per_block=(1, 1),
    num_attention_heads=(2, 4),
    cross_attention_dim=32,
    norm_num_groups=8,
    use_linear_projection=True,
    addition_embed_type="text_time",
    addition_time_embed_dim=8,
    projection_class_embeddings_input_dim=8 * 6 + 16,
    sample_size=16,
)


class DownBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, temb_ch, layers, add_downsample, *, factory, groups):
        super().__init__()
        self.resnets = nn.ModuleList(
            [
                ResnetBlock2D(
                    in_ch if i == 0 else out_ch, out_ch, temb_ch, factory=factory, groups=groups
                )
                for i in range(layers)
            ]
        )
        self.downsamplers = (
            nn.ModuleList([Downsample2D(out_ch, factory=factory)]) if add_downsample else None
        )

    def forward(self, x, temb, encoder_hidden_states=None):
        states = []
        for resnet in self.resnets:
            x = resnet(x, temb)
            states.append(x)
        if self.downsamplers is not None:
            x = self.downsamplers[0](x)
            states.append(x)
        return x, states


class CrossAttnDownBlock2D(nn.Module):
    def __init__(
        self, in_ch, out_ch, temb_ch, layers, tlayers, heads, cross_dim, add_downsample,
        *, factory, groups, use_linear_projection,
    ):
        super().__init__()
        dim_head = out_ch // heads
        self.resnets = nn.ModuleList(
            [
                ResnetBlock2D(
                    in_ch if i == 0 else out_ch, out_ch, temb_ch, factory=factory, groups=groups
                )
                for i in range(layers)
            ]
        )
        self.attentions = nn.ModuleList(
            [
                Transformer2DModel(
                    out_ch, heads, dim_head, tlayers, cross_dim,
                    factory=factory, groups=groups, use_linear_projection=use_linear_projection,
                )
                for _ in range(layers)
            ]
        )
        self.downsamplers = (
            nn.ModuleList([Downsample2D(out_ch, factory=factory)]) if add_downsample else None
        )

    def forward(self, x, temb, encoder_hidden_states):
        states = []
        for resnet, attn in zip(self.resnets, self.attentions):
            x = resnet(x, temb)
            x = attn(x, encoder_hidden_states)
            states.append(x)
        if self.downsamplers is not None:
            x = self.downsamplers[0](x)
            states.append(x)
        return x, states


class UpBlock2D(nn.Module):
    def __init__(self, in_ch, prev_ch, out_ch, temb_ch, layers, add_upsample, *, factory, groups):
        super().__init__()
        resnets = []
        for i in range(layers):
            skip_ch = in_ch if i == layers - 1 else out_ch
            res_in = prev_ch if i == 0 else out_ch
            resnets.append(
                ResnetBlock2D(res_in + skip_ch, out_ch, temb_ch, factory=factory, groups=groups)
            )
        self.resnets = nn.ModuleList(resnets)
        self.upsamplers = (
            nn.ModuleList([Upsample2D(out_ch, factory=factory)]) if add_upsample else None
        )

    def forward(self, x, skips, temb, encoder_hidden_states=None):
        for resnet in self.resnets:
            x = torch.cat([x, skips.pop()], dim=1)
            x = resnet(x, temb)
        if self.upsamplers is not None:
            x = self.upsamplers[0](x)
        return x


class CrossAttnUpBlock2D(nn.Module):
    def __init__(
        self, in_ch, prev_ch, out_ch, temb_ch, layers, tlayers, heads, cross_dim, add_upsample,
        *, factory, groups, use_linear_projection,
    ):
        super().__init__()
        dim_head = out_ch // heads
        resnets, attentions = [], []
        for i in range(layers):
            skip_ch = in_ch if i == layers - 1 else out_ch
            res_in = prev_ch if i == 0 else out_ch
            resnets.append(
                ResnetBlock2D(res_in + skip_ch, out_ch, temb_ch, factory=factory, groups=groups)
            )
            attentions.append(
                Transformer2DModel(
                    out_ch, heads, dim_head, tlayers, cross_dim,
                    factory=factory, groups=groups, use_linear_projection=use_linear_projection,
                )
            )
        self.resnets = nn.ModuleList(resnets)
        self.attentions = nn.ModuleList(attentions)
        self.upsamplers = (
            nn.ModuleList([Upsample2D(out_ch, factory=factory)]) if add_upsample else None
        )

    def forward(self, x, skips, temb, encoder_hidden_states):
        for resnet, attn in zip(self.resnets, self.attentions):
            x = torch.cat([x, skips.pop()], dim=1)
            x = resnet(x, temb)
            x = attn(x, encoder_hidden_states)
        if self.upsamplers is not None:
            x = self.upsamplers[0](x)
        return x


class UNetMidBlock2DCrossAttn(nn.Module):
    def __init__(
        self, ch, temb_ch, tlayers, heads, cross_dim, *, factory, groups, use_linear_projection
    ):
        super().__init__()
        dim_head = ch // heads
        self.resnets = nn.ModuleList(
            [
                ResnetBlock2D(ch, ch, temb_ch, factory=factory, groups=groups),
                ResnetBlock2D(ch, ch, temb_ch, factory=factory, groups=groups),
            ]
        )
        self.attentions = nn.ModuleList(
            [
                Transformer2DModel(
                    ch, heads, dim_head, tlayers, cross_dim,
                    factory=factory, groups=groups, use_linear_projection=use_linear_projection,
                )
            ]
        )

    def forward(self, x, temb, encoder_hidden_states):
        x = self.resnets[0](x, temb)
        x = self.attentions[0](x, encoder_hidden_states)
        x = self.resnets[1](x, temb)
        return x


class UNet2DConditionNative(nn.Module):
    def __init__(self, config: UNetConfig, state: ParallelState):
        super().__init__()
        self.config = config
        self.state = state
        factory = LayerFactory(state)
        cfg = config
        ch0 = cfg.block_out_channels[0]
        temb_ch = cfg.time_embed_dim
        groups = cfg.norm_num_groups

        self.conv_in = factory.conv2d(cfg.in_channels, ch0, 3, 1, 1, is_first_layer=True)
        self.time_embedding = TimestepEmbedding(ch0, temb_ch)
        if cfg.addition_embed_type == "text_time":
            self.add_embedding = TimestepEmbedding(
                cfg.projection_class_embeddings_input_dim, temb_ch
            )
        else:
            self.add_embedding = None

        # down
        down_blocks = []
        out_ch = ch0
        for i, btype in enumerate(cfg.down_block_types):
            in_ch = out_ch
            out_ch = cfg.block_out_channels[i]
            is_final = i == len(cfg.block_out_channels) - 1
            if btype == "DownBlock2D":
                down_blocks.append(
                    DownBlock2D(
                        in_ch, out_ch, temb_ch, cfg.layers_per_block, not is_final,
                        factory=factory, groups=groups,
                    )
                )
            else:
                down_blocks.append(
                    CrossAttnDownBlock2D(
                        in_ch, out_ch, temb_ch, cfg.layers_per_block,
                        cfg.transformer_layers_per_block[i], cfg.num_attention_heads[i],
                        cfg.cross_attention_dim, not is_final,
                        factory=factory, groups=groups,
                        use_linear_projection=cfg.use_linear_projection,
                    )
                )
        self.down_blocks = nn.ModuleList(down_blocks)

        # mid
        mid_ch = cfg.block_out_channels[-1]
        self.mid_block = UNetMidBlock2DCrossAttn(
            mid_ch, temb_ch, cfg.transformer_layers_per_block[-1], cfg.num_attention_heads[-1],
            cfg.cross_attention_dim, factory=factory, groups=groups,
            use_linear_projection=cfg.use_linear_projection,
        )

        # up
        up_blocks = []
        rev_ch = list(reversed(cfg.block_out_channels))
        rev_tlayers = list(reversed(cfg.transformer_layers_per_block))
        rev_heads = list(reversed(cfg.num_attention_heads))
        out_ch = rev_ch[0]
        for i, btype in enumerate(cfg.up_block_types):
            prev_ch = out_ch
            out_ch = rev_ch[i]
            in_ch = rev_ch[min(i + 1, len(rev_ch) - 1)]
            is_final = i == len(rev_ch) - 1
            if btype == "UpBlock2D":
                up_blocks.append(
                    UpBlock2D(
                        in_ch, prev_ch, out_ch, temb_ch, cfg.layers_per_block + 1, not is_final,
                        factory=factory, groups=groups,
                    )
                )
            else:
                up_blocks.append(
                    CrossAttnUpBlock2D(
                        in_ch, prev_ch, out_ch, temb_ch, cfg.layers_per_block + 1,
                        rev_tlayers[i], rev_heads[i], cfg.cross_attention_dim, not is_final,
                        factory=factory, groups=groups,
                        use_linear_projection=cfg.use_linear_projection,
                    )
                )
        self.up_blocks = nn.ModuleList(up_blocks)

        self.conv_norm_out = factory.group_norm(groups, ch0, eps=1e-5, fuse_silu=True)
        # conv_out is input-channel-sharded under TP (reference
        # distri_sdxl_unet_tp.py:27-38); conv_in stays replicated (4 inputs)
        self.conv_out = factory.conv2d(ch0, cfg.out_channels, 3, 1, 1, tp_shard=True)

    def forward(
        self,
        sample: torch.Tensor,
        timestep: torch.Tensor,
        encoder_hidden_states: torch.Tensor,
        added_cond_kwargs: dict | None = None,
    ) -> torch.Tensor:
        b = sample.shape[0]
        if not torch.is_tensor(timestep):
            timestep = torch.tensor([timestep], device=sample.device)
        if timestep.ndim == 0:
            timestep = timestep[None]
        timestep = timestep.expand(b)

        cfg = self.config
        temb = sinusoidal_embedding(timestep, cfg.block_out_channels[0]).to(sample.dtype)
        temb = self.time_embedding(temb)

        if self.add_embedding is not None:
            assert added_cond_kwargs is not None, "SDXL needs text_embeds + time_ids"
            text_embeds = added_cond_kwargs["text_embeds"]
            time_ids = added_cond_kwargs["time_ids"]
            time_embeds = sinusoidal_embedding(
                time_ids.flatten(), cfg.addition_time_embed_dim
            ).reshape(b, -1)
            aug = torch.cat([text_embeds, time_embeds.to(text_embeds.dtype)], dim=-1)
            temb = temb + self.add_embedding(aug.to(sample.dtype))

        x = self.conv_in(sample)
        skips = [x]
        for block in self.down_blocks:
            x, states = block(x, temb, encoder_hidden_states)
            skips.extend(states)
        x = self.mid_block(x, temb, encoder_hidden_states)
        for block in self.up_blocks:
            x = block(x, skips, temb, encoder_hidden_states)
        x = self.conv_norm_out(x)
        return self.conv_out(x)
