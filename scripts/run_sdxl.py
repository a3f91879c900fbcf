"""Generation + latency benchmark CLI (parity with the reference's
scripts/run_sdxl.py flag surface; launched under torchrun for N>1):

  torchrun --nproc_per_node=N scripts/run_sdxl.py --mode benchmark ...
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import time

import numpy as np
import torch

from distrifuser_amd import DistriConfig, DistriSDXLPipeline


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--mode", type=str, default="generation",
                   choices=["generation", "benchmark"])
    # Diffuser-level
    p.add_argument("--prompt", type=str,
                   default="Astronaut in a jungle, cold color palette, detailed, 8k")
    p.add_argument("--output_path", type=str, default="astronaut.png")
    p.add_argument("--num_inference_steps", type=int, default=50)
    p.add_argument("--image_size", type=int, nargs="*", default=[1024, 1024],
                   help="height [width]")
    p.add_argument("--guidance_scale", type=float, default=5.0)
    p.add_argument("--scheduler", type=str, default="ddim",
                   choices=["ddim", "euler", "dpm-solver"])
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--pretrained", type=str, default=None,
                   help="local diffusers-layout model dir (random init if omitted)")
    # Parallelism
    p.add_argument("--no_split_batch", action="store_true")
    p.add_argument("--warmup_steps", type=int, default=4)
    p.add_argument("--tile_decode", action="store_true",
                   help="force tiled VAE decode (auto-enabled >= 2048^2)")
    p.add_argument("--sync_mode", type=str, default="corrected_async_gn",
                   choices=["separate_gn", "stale_gn", "corrected_async_gn", "sync_gn",
                            "full_sync", "no_sync"])
    p.add_argument("--parallelism", type=str, default="patch",
                   choices=["patch", "tensor", "naive_patch"])
    p.add_argument("--no_cuda_graph", action="store_true")
    p.add_argument("--split_scheme", type=str, default="row",
                   choices=["row", "col", "alternate"])
    # Benchmark
    p.add_argument("--output_type", type=str, default="pil", choices=["latent", "pil"])
    p.add_argument("--warmup_times", type=int, default=5)
    p.add_argument("--test_times", type=int, default=20)
    p.add_argument("--ignore_ratio", type=float, default=0.2)
    p.add_argument("--preset", type=str, default="sdxl", choices=["sdxl", "tiny"],
                   help="tiny = fast CPU smoke of the CLI")
    return p.parse_args()


def main():
    args = get_args()
    size = args.image_size if len(args.image_size) == 2 else args.image_size * 2
    cfg = DistriConfig(
        height=size[0],
        width=size[1],
        do_classifier_free_guidance=args.guidance_scale > 1,
        split_batch=not args.no_split_batch,
        warmup_steps=args.warmup_steps,
        mode=args.sync_mode,
        parallelism=args.parallelism,
        split_scheme=args.split_scheme,
        use_cuda_graph=not args.no_cuda_graph and torch.cuda.is_available(),
    )
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    pipe = DistriSDXLPipeline.from_pretrained(
        cfg, torch_dtype=dtype, scheduler=args.scheduler,
        pretrained_model_name_or_path=args.pretrained, preset=args.preset,
    )
    if args.tile_decode:
        pipe.vae.enable_tiling()

    def run(output_type):
        g = torch.Generator().manual_seed(args.seed)
        return pipe(
            prompt=args.prompt,
            generator=g,
            num_inference_steps=args.num_inference_steps,
            guidance_scale=args.guidance_scale,
            output_type=output_type,
        )

    if args.mode == "generation":
        out = run(args.output_type)
        if cfg.rank == 0 and args.output_type == "pil":
            img = out[0] if isinstance(out, list) else out[0]
            try:
                img.save(args.output_path)
            except AttributeError:  # numpy fallback (no pillow in image)
                np.save(args.output_path + ".npy", img)
            print(f"saved {args.output_path}")
    else:
        for _ in range(args.warmup_times):
            run("latent")
            if torch.cuda.is_available():
                torch.cuda.synchronize()
        latencies = []
        for _ in range(args.test_times):
            t0 = time.perf_counter()
            run("latent")
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            latencies.append(time.perf_counter() - t0)
        latencies.sort()
        drop = int(len(latencies) * args.ignore_ratio / 2)
        kept = latencies[drop:len(latencies) - drop] if drop else latencies
        if cfg.rank == 0:
            print(f"latency: {sum(kept) / len(kept):.3f} s "
                  f"(trimmed mean of {len(kept)}/{len(latencies)} runs, "
                  f"{cfg.world_size} GPUs, {args.parallelism}/{args.sync_mode})")


if __name__ == "__main__":
    main()
