"""Native CLIP text encoders (SD1.5's ViT-L and SDXL's ViT-L + OpenCLIP bigG).

The reference used transformers' CLIPTextModel via the diffusers pipeline;
here the causal text transformer is owned. ``hidden_state_index=-2``
reproduces SDXL's penultimate-layer conditioning; the pooled embedding is the
EOT-token hidden state through the text projection (OpenCLIP convention).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F
from torch import nn


@dataclass(frozen=True)
class CLIPTextConfig:
    vocab_size: int = 49408
    hidden_size: int = 768
    intermediate_size: int = 3072
    num_layers: int = 12
    num_heads: int = 12
    max_position_embeddings: int = 77
    projection_dim: int | None = None  # text projection (OpenCLIP bigG: 1280)
    act: str = "quick_gelu"  # ViT-L uses quick_gelu; bigG uses gelu


CLIP_VIT_L = CLIPTextConfig()
OPEN_CLIP_BIG_G = CLIPTextConfig(
    hidden_size=1280, intermediate_size=5120, num_layers=32, num_heads=20,
    projection_dim=1280, act="gelu",
)
# SD2.x text encoder (OpenCLIP ViT-H text tower, truncated to 23 layers as in
# the stabilityai/stable-diffusion-2-1 checkpoint)
OPEN_CLIP_VIT_H = CLIPTextConfig(
    hidden_size=1024, intermediate_size=4096, num_layers=23, num_heads=16,
    act="gelu",
)
TINY_CLIP = CLIPTextConfig(
    vocab_size=1000, hidden_size=16, intermediate_size=32, num_layers=2, num_heads=2,
    projection_dim=16, act="gelu",
)


class CLIPMLP(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size)
        self.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        self.act = cfg.act

    def forward(self, x):
        x = self.fc1(x)
        x = x * torch.sigmoid(1.702 * x) if self.act == "quick_gelu" else F.gelu(x)
        return self.fc2(x)


class CLIPAttention(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.heads = cfg.num_heads
        self.head_dim = cfg.hidden_size // cfg.num_heads
        self.q_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.k_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.v_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.out_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, x):
        b, l, c = x.shape
        q = self.q_proj(x).view(b, l, self.heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(x).view(b, l, self.heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(x).view(b, l, self.heads, self.head_dim).transpose(1, 2)
        out = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        return self.out_proj(out.transpose(1, 2).reshape(b, l, c))


class CLIPLayer(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.layer_norm1 = nn.LayerNorm(cfg.hidden_size)
        self.self_attn = CLIPAttention(cfg)
        self.layer_norm2 = nn.LayerNorm(cfg.hidden_size)
        self.mlp = CLIPMLP(cfg)

    def forward(self, x):
        x = x + self.self_attn(self.layer_norm1(x))
        x = x + self.mlp(self.layer_norm2(x))
        return x


class CLIPTextEncoder(nn.Module):
    def __init__(self, config: CLIPTextConfig):
        super().__init__()
        self.config = config
        self.token_embedding = nn.Embedding(config.vocab_size, config.hidden_size)
        self.position_embedding = nn.Embedding(
            config.max_position_embeddings, config.hidden_size
        )
        self.layers = nn.ModuleList([CLIPLayer(config) for _ in range(config.num_layers)])
        self.final_layer_norm = nn.LayerNorm(config.hidden_size)
        if config.projection_dim is not None:
            self.text_projection = nn.Linear(config.hidden_size, config.projection_dim, bias=False)
        else:
            self.text_projection = None

    def forward(
        self, input_ids: torch.Tensor, hidden_state_index: int = -1
    ) -> tuple[torch.Tensor, torch.Tensor | None]:
        """Returns (hidden_states[hidden_state_index], pooled_projection).

        hidden_state_index=-1: final-LN output (SD1.5 conditioning);
        hidden_state_index=-2: penultimate layer, pre-final-LN (SDXL).
        """
        b, l = input_ids.shape
        pos = torch.arange(l, device=input_ids.device)
        x = self.token_embedding(input_ids) + self.position_embedding(pos)[None]
        hidden_states = []
        for layer in self.layers:
            x = layer(x)
            hidden_states.append(x)
        final = self.final_layer_norm(x)

        pooled = None
        if self.text_projection is not None:
            # EOT token = argmax(ids) under the CLIP tokenizer (eos has the
            # highest id); pooled through the projection.
            eot = input_ids.argmax(dim=-1)
            pooled = self.text_projection(final[torch.arange(b, device=x.device), eot])

        if hidden_state_index == -1:
            return final, pooled
        # e.g. -2 -> output of the penultimate layer, pre-final-LN (SDXL).
        return hidden_states[hidden_state_index], pooled
