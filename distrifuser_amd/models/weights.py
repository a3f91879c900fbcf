"""safetensors checkpoint loading: HF diffusers layout -> native modules.

The reference loaded weights via diffusers' ``from_pretrained``
(reference pipelines.py:26-41); we own the mapping. Our module tree mirrors
the diffusers naming, with these deliberate deviations:

* fused KV: diffusers ``to_k`` + ``to_v``  ->  our ``to_kv`` (cat on dim 0)
* ``to_out.0``                             ->  ``to_out`` (dropout dropped)
* GEGLU ``ff.net.0.proj`` / ``ff.net.2``   ->  ``ff.proj_in`` / ``ff.proj_out``
* PatchConv2d / samplers wrap an inner nn.Conv2d, adding a ``.conv`` path
  segment (``downsamplers.0.conv.conv.weight`` etc.)
* VAE decoder flattens ``decoder.`` / ``mid_block`` naming (see _VAE_RULES)

``state_dict_to_native`` is model-key-driven: for every native parameter it
derives the diffusers key, so unexpected checkpoint keys are reported rather
than silently dropped.
"""

from __future__ import annotations

import json
import os
import re

import torch
from torch import nn


def _diffusers_key_candidates(native_key: str) -> list[str]:
    """Possible diffusers names for a native parameter key (sans to_kv)."""
    cands = [native_key]
    # strip wrapper ".conv" segments: a.conv.weight -> a.weight (possibly 2 deep)
    k = native_key
    while ".conv.conv." in k or re.search(r"\.conv\.(weight|bias)$", k):
        k2 = re.sub(r"\.conv\.(weight|bias)$", r".\1", k, count=1)
        if k2 == k:
            break
        k = k2
        cands.append(k)
    out = []
    for c in cands:
        out.append(c)
        c2 = c.replace(".to_out.", ".to_out.0.")
        if c2 != c:
            out.append(c2)
        c3 = c.replace(".ff.proj_in.", ".ff.net.0.proj.").replace(".ff.proj_out.", ".ff.net.2.")
        if c3 != c:
            out.append(c3)
            out.append(c3.replace(".to_out.", ".to_out.0."))
    return out


def state_dict_to_native(model: nn.Module, sd: dict) -> tuple[dict, list[str]]:
    """Translate a diffusers-layout state dict into this model's layout.

    Returns (native_state_dict, missing_keys)."""
    native = {}
    missing = []
    for key, param in model.state_dict().items():
        if ".to_kv." in key:
            kk = None
            for cand in _diffusers_key_candidates(key.replace(".to_kv.", ".to_k.")):
                if cand in sd:
                    kk = cand
                    break
            if kk is None:
                missing.append(key)
                continue
            vk = kk.replace(".to_k.", ".to_v.")
            native[key] = torch.cat([sd[kk], sd[vk]], dim=0)
            continue
        found = None
        for cand in _diffusers_key_candidates(key):
            if cand in sd:
                found = cand
                break
        if found is None:
            missing.append(key)
        else:
            native[key] = sd[found]
    return native, missing


def export_diffusers_state_dict(model: nn.Module) -> dict:
    """Inverse of ``state_dict_to_native`` (used by round-trip tests)."""
    out = {}
    for key, value in model.state_dict().items():
        if ".to_kv." in key:
            half = value.shape[0] // 2
            base = _diffusers_key_candidates(key.replace(".to_kv.", ".to_k."))[-1]
            out[base] = value[:half].clone()
            out[base.replace(".to_k.", ".to_v.")] = value[half:].clone()
            continue
        out[_diffusers_key_candidates(key)[-1]] = value.clone()
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


def find_component_weights(model_dir: str, component: str) -> str | None:
    """Locate <model_dir>/<component>/*.safetensors (diffusers repo layout)."""
    comp_dir = os.path.join(model_dir, component)
    if not os.path.isdir(comp_dir):
        return None
    for name in sorted(os.listdir(comp_dir)):
        if name.endswith(".safetensors"):
            return os.path.join(comp_dir, name)
    return None
