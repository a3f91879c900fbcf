"""U-Net MACs at a given resolution (parity with the reference's
profile_macs.py, natively — forward hooks over our own modules instead of
torchprofile)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse

import torch
from torch import nn

from distrifuser_amd import DistriConfig
from distrifuser_amd.models import DistriUNet
from distrifuser_amd.models.unet import SD15_UNET, SDXL_UNET


def count_macs(unet, sample, ehs, added):
    macs = [0]
    hooks = []

    def conv_hook(m, inp, out):
        kh, kw = m.kernel_size
        macs[0] += out.numel() * m.in_channels * kh * kw // m.groups

    def linear_hook(m, inp, out):
        macs[0] += out.numel() * m.in_features

    for m in unet.modules():
        if isinstance(m, nn.Conv2d):
            hooks.append(m.register_forward_hook(conv_hook))
        elif isinstance(m, nn.Linear):
            hooks.append(m.register_forward_hook(linear_hook))

    # attention score/value MACs via module-level hooks on SDPA wrappers
    from distrifuser_amd.models.layers import PlainSelfAttention, PlainCrossAttention
    from distrifuser_amd.parallel.patch_ops import CachedCrossAttention, PatchSelfAttention

    def attn_hook(m, inp, out):
        x = inp[0]
        b, l, _ = x.shape
        lkv = l if isinstance(m, (PlainSelfAttention, PatchSelfAttention)) else 77
        macs[0] += 2 * b * m.heads * l * lkv * m.dim_head

    for m in unet.modules():
        if isinstance(m, (PlainSelfAttention, PlainCrossAttention, PatchSelfAttention,
                          CachedCrossAttention)):
            hooks.append(m.register_forward_hook(attn_hook))

    with torch.no_grad():
        unet.set_counter(0)
        unet(sample, 500.0, ehs, added)
    for h in hooks:
        h.remove()
    return macs[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--image_size", type=int, default=1024)
    ap.add_argument("--model", type=str, default="sdxl", choices=["sdxl", "sd15"])
    args = ap.parse_args()

    cfg = DistriConfig(height=args.image_size, width=args.image_size,
                       do_classifier_free_guidance=False, use_cuda_graph=False,
                       device="cpu")
    ucfg = SDXL_UNET if args.model == "sdxl" else SD15_UNET
    torch.manual_seed(0)
    unet = DistriUNet(ucfg, cfg).eval()
    h = args.image_size // 8
    sample = torch.randn(1, 4, h, h)
    ehs = torch.randn(1, 77, ucfg.cross_attention_dim)
    added = None
    if ucfg.addition_embed_type == "text_time":
        added = {
            "text_embeds": torch.randn(1, 1280),
            "time_ids": torch.tensor([[args.image_size, args.image_size, 0, 0,
                                       args.image_size, args.image_size]],
                                     dtype=torch.float32),
        }
    macs = count_macs(unet, sample, ehs, added)
    params = sum(p.numel() for p in unet.parameters())
    print(f"{args.model} @ {args.image_size}^2: {macs / 1e9:.1f} GMACs/step, "
          f"{params / 1e6:.1f} M params")


if __name__ == "__main__":
    main()
