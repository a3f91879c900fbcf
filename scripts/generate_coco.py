"""Quality benchmark: generate images from COCO-style captions with
deterministic per-index seeds (parity with the reference's generate_coco.py;
the HuggingFace COCO download is replaced by a local caption file since this
environment has no network — `--caption_file` takes a JSON list of strings or
a prompts.json produced by scripts/dump_coco.py; without it a deterministic
synthetic caption set is used)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import json
import os

import numpy as np
import torch

from distrifuser_amd import DistriConfig, DistriSDXLPipeline

SYNTHETIC_CAPTIONS = [
    "a {} {} on a {} in the {}".format(a, b, c, d)
    for a in ("red", "small", "wooden", "shiny", "old")
    for b in ("bicycle", "dog", "teapot", "airplane", "boat")
    for c in ("table", "street", "beach", "mountain")
    for d in ("morning", "rain", "snow", "sunset", "fog")
]


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--caption_file", type=str, default=None)
    p.add_argument("--n_images", type=int, default=5000)
    p.add_argument("--image_size", type=int, default=1024)
    p.add_argument("--num_inference_steps", type=int, default=50)
    p.add_argument("--guidance_scale", type=float, default=5.0)
    p.add_argument("--scheduler", type=str, default="ddim")
    p.add_argument("--sync_mode", type=str, default="corrected_async_gn")
    p.add_argument("--parallelism", type=str, default="patch")
    p.add_argument("--warmup_steps", type=int, default=4)
    p.add_argument("--no_cuda_graph", action="store_true")
    p.add_argument("--no_split_batch", action="store_true")
    p.add_argument("--pretrained", type=str, default=None)
    p.add_argument("--output_root", type=str, default="results/coco")
    p.add_argument("--split", type=int, nargs=2, default=None, metavar=("IDX", "N"),
                   help="process only chunk IDX of N (coarse resumability)")
    return p.parse_args()


def main():
    args = get_args()
    captions = SYNTHETIC_CAPTIONS
    if args.caption_file:
        with open(args.caption_file) as f:
            data = json.load(f)
        captions = data if isinstance(data, list) else list(data.values())
    captions = (captions * ((args.n_images // len(captions)) + 1))[: args.n_images]

    cfg = DistriConfig(
        height=args.image_size, width=args.image_size,
        do_classifier_free_guidance=args.guidance_scale > 1,
        split_batch=not args.no_split_batch,
        warmup_steps=args.warmup_steps, mode=args.sync_mode,
        parallelism=args.parallelism,
        use_cuda_graph=not args.no_cuda_graph and torch.cuda.is_available(),
    )
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    pipe = DistriSDXLPipeline.from_pretrained(
        cfg, torch_dtype=dtype, scheduler=args.scheduler,
        pretrained_model_name_or_path=args.pretrained,
    )
    out_dir = os.path.join(
        args.output_root,
        f"{args.scheduler}-{args.num_inference_steps}",
        f"gpus{cfg.world_size}-warmup{args.warmup_steps}-{args.sync_mode}",
    )
    if cfg.rank == 0:
        os.makedirs(out_dir, exist_ok=True)

    indices = range(len(captions))
    if args.split is not None:
        i, n = args.split
        per = (len(captions) + n - 1) // n
        indices = range(i * per, min((i + 1) * per, len(captions)))

    for idx in indices:
        path = os.path.join(out_dir, f"{idx:05d}.npy")
        if os.path.exists(path):
            continue
        g = torch.Generator().manual_seed(idx)  # seed = image index (reproducible)
        img = pipe(prompt=captions[idx], generator=g,
                   num_inference_steps=args.num_inference_steps,
                   guidance_scale=args.guidance_scale, output_type="np")
        if cfg.rank == 0:
            np.save(path, img[0])
            try:
                from PIL import Image

                Image.fromarray(img[0]).save(path.replace(".npy", ".png"))
            except ImportError:
                pass
    if cfg.rank == 0:
        print(f"done -> {out_dir}")


if __name__ == "__main__":
    main()
