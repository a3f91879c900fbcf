"""safetensors checkpoint loading: HF diffusers layout <-> native modules.

The reference loaded weights via diffusers' ``from_pretrained``
(reference pipelines.py:26-41); we own the mapping. The translation is
MODULE-AWARE (built by walking the model), not string-guessing: for every
native parameter the diffusers key is derived from the module tree, so real
checkpoints load 1:1 and unexpected gaps are reported.

Deliberate layout deviations handled here:
* fused KV: diffusers ``to_k`` + ``to_v``  ->  our ``to_kv`` (cat on dim 0)
* ``to_out.0``                             ->  ``to_out`` (dropout dropped)
* GEGLU ``ff.net.0.proj`` / ``ff.net.2``   ->  ``ff.proj_in`` / ``ff.proj_out``
* PatchConv2d wraps an inner nn.Conv2d (adds a ``.conv`` path segment)
* VAE decoder: our flat names <-> diffusers ``decoder.*`` structure
* CLIP: our flat names <-> HF ``text_model.*`` structure
"""

from __future__ import annotations

import os
import re

import torch
from torch import nn


def _unet_key_map(model: nn.Module) -> dict[str, str]:
    """native key -> diffusers key for U-Net-shaped models (to_kv keys are
    mapped with ``to_kv`` kept; the load/export functions split them)."""
    from ..parallel.patch_ops import PatchConv2d

    wrap_prefixes = [
        name + ".conv."
        for name, m in model.named_modules()
        if isinstance(m, PatchConv2d)
    ]
    out = {}
    for key in model.state_dict():
        k = key
        for w in wrap_prefixes:
            if k.startswith(w):
                k = w[: -len("conv.")] + k[len(w):]
                break
        k = re.sub(r"\.to_out\.(weight|bias)$", r".to_out.0.\1", k)
        k = k.replace(".ff.proj_in.", ".ff.net.0.proj.").replace(".ff.proj_out.", ".ff.net.2.")
        out[key] = k
    return out


_VAE_RENAMES = [
    (re.compile(r"^conv_in\."), "decoder.conv_in."),
    (re.compile(r"^mid_resnet_1\."), "decoder.mid_block.resnets.0."),
    (re.compile(r"^mid_attn\."), "decoder.mid_block.attentions.0."),
    (re.compile(r"^mid_resnet_2\."), "decoder.mid_block.resnets.1."),
    (re.compile(r"^up_blocks\.(\d+)\.upsampler\."), r"decoder.up_blocks.\1.upsamplers.0.conv."),
    (re.compile(r"^up_blocks\."), "decoder.up_blocks."),
    (re.compile(r"^conv_norm_out\."), "decoder.conv_norm_out."),
    (re.compile(r"^conv_out\."), "decoder.conv_out."),
]

_CLIP_RENAMES = [
    (re.compile(r"^token_embedding\."), "text_model.embeddings.token_embedding."),
    (re.compile(r"^position_embedding\."), "text_model.embeddings.position_embedding."),
    (re.compile(r"^layers\."), "text_model.encoder.layers."),
    (re.compile(r"^final_layer_norm\."), "text_model.final_layer_norm."),
    (re.compile(r"^text_projection\."), "text_projection."),
]


def _renamed_key_map(model: nn.Module, renames) -> dict[str, str]:
    out = {}
    for key in model.state_dict():
        k = key
        for pat, repl in renames:
            k2 = pat.sub(repl, k)
            if k2 != k:
                k = k2
                break
        k = re.sub(r"\.to_out\.(weight|bias)$", r".to_out.0.\1", k)
        out[key] = k
    return out


def key_map_for(model: nn.Module) -> dict[str, str]:
    from .clip import CLIPTextEncoder
    from .vae import VAEDecoder

    if isinstance(model, VAEDecoder):
        return _renamed_key_map(model, _VAE_RENAMES)
    if isinstance(model, CLIPTextEncoder):
        return _renamed_key_map(model, _CLIP_RENAMES)
    return _unet_key_map(model)


def state_dict_to_native(model: nn.Module, sd: dict) -> tuple[dict, list[str]]:
    """Translate a diffusers-layout state dict into this model's layout.

    Returns (native_state_dict, missing_keys)."""
    kmap = key_map_for(model)
    native = {}
    missing = []
    for key, dk in kmap.items():
        if ".to_kv." in dk:
            kk = dk.replace(".to_kv.", ".to_k.")
            vk = dk.replace(".to_kv.", ".to_v.")
            if kk in sd and vk in sd:
                native[key] = torch.cat([sd[kk], sd[vk]], dim=0)
            else:
                missing.append(key)
            continue
        if dk in sd:
            native[key] = sd[dk]
        else:
            missing.append(key)
    return native, missing


def export_diffusers_state_dict(model: nn.Module) -> dict:
    """Inverse of ``state_dict_to_native`` (used by round-trip tests)."""
    kmap = key_map_for(model)
    state = model.state_dict()
    out = {}
    for key, dk in kmap.items():
        value = state[key]
        if ".to_kv." in dk:
            half = value.shape[0] // 2
            out[dk.replace(".to_kv.", ".to_k.")] = value[:half].clone()
            out[dk.replace(".to_kv.", ".to_v.")] = value[half:].clone()
        else:
            out[dk] = value.clone()
    return out


def load_safetensors(path: str) -> dict:
    from safetensors.torch import load_file

    return load_file(path)


def load_into(model: nn.Module, sd: dict, strict_coverage: float = 0.999) -> None:
    """Load a diffusers-layout dict; raises if too many params are missing."""
    native, missing = state_dict_to_native(model, sd)
    model.load_state_dict(native, strict=False)
    total = len(model.state_dict())
    if missing and (total - len(missing)) / total < strict_coverage:
        raise RuntimeError(
            f"checkpoint covers only {total - len(missing)}/{total} params; "
            f"first missing: {missing[:8]}"
        )


def load_unet_checkpoint(model: nn.Module, sd: dict) -> None:
    """Load a full diffusers-layout U-Net checkpoint, sharding on the fly for
    tensor-parallel modules (the reference loaded full weights then re-sliced
    at wrap time, tp/attention.py:33-91; we shard directly from the dict)."""
    from ..parallel.tensor_ops import TPAttention, TPConv2d, TPFeedForward
    from .resnet import ResnetBlock2D

    handled: list[str] = []
    for name, m in model.named_modules():
        prefix = f"{name}." if name else ""
        if isinstance(m, TPAttention):
            m.copy_from_full(
                sd[f"{prefix}to_q.weight"],
                sd[f"{prefix}to_k.weight"],
                sd[f"{prefix}to_v.weight"],
                sd[f"{prefix}to_out.0.weight"],
                sd.get(f"{prefix}to_out.0.bias"),
            )
            handled.append(prefix)
        elif isinstance(m, TPFeedForward):
            m.copy_from_full(
                sd[f"{prefix}net.0.proj.weight"],
                sd[f"{prefix}net.0.proj.bias"],
                sd[f"{prefix}net.2.weight"],
                sd[f"{prefix}net.2.bias"],
            )
            handled.append(prefix)
        elif isinstance(m, TPConv2d):
            # module path IS the diffusers conv name (e.g. downsamplers.0.conv)
            m.copy_from_full(sd[f"{name}.weight"], sd.get(f"{name}.bias"))
            handled.append(prefix)
        elif isinstance(m, ResnetBlock2D) and getattr(m, "tp_pair", False):
            m.copy_from_full_tp(sd, prefix)
            # only the sharded members are handled; norm1/conv_shortcut load
            # through the standard mapping below
            for sub in ("conv1.", "conv2.", "conv2_bias", "norm2.", "time_emb_proj."):
                handled.append(prefix + sub)

    if not handled:
        load_into(model, sd)
        return
    native, missing = state_dict_to_native(model, sd)
    missing = [k for k in missing if not any(k.startswith(h) for h in handled)]
    native = {k: v for k, v in native.items() if not any(k.startswith(h) for h in handled)}
    model.load_state_dict(native, strict=False)
    if missing:
        raise RuntimeError(f"unmapped non-TP params ({len(missing)}): {missing[:8]}")


def find_component_weights(model_dir: str, component: str) -> str | None:
    """Locate <model_dir>/<component>/*.safetensors (diffusers repo layout)."""
    comp_dir = os.path.join(model_dir, component)
    if not os.path.isdir(comp_dir):
        return None
    for name in sorted(os.listdir(comp_dir)):
        if name.endswith(".safetensors"):
            return os.path.join(comp_dir, name)
    return None
