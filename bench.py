"""Flagship benchmark: SDXL 50-step-DDIM single-image latency (BASELINE.json
metric) under displaced patch parallelism on MI355X.

Contract (driver):
    python bench.py --gpus N --steps K --warmup W
    (N>1 launched via torch.distributed.run, one rank per GPU over RCCL)

Protocol: random-init SDXL weights (no network), synthetic prompt embeddings,
guidance_scale 5.0 (CFG batch split), DDIM, bf16, default 3840x3840 (the
reference's headline config, README.md:30). W untimed denoise steps (covers
the sync-comm warmup phase), then EXACTLY K timed steps bracketed by a
barrier + torch.cuda.synchronize on both sides; per-step time is the MAX over
ranks; value = ms_per_step * 50 / 1000 = end-to-end 50-step latency
(output_type=latent protocol: VAE/text-encode excluded, as in the
reference's run_sdxl.py benchmark mode).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--height", type=int, default=3840)
    ap.add_argument("--width", type=int, default=3840)
    ap.add_argument("--guidance-scale", type=float, default=5.0)
    ap.add_argument("--scheduler", type=str, default="ddim")
    ap.add_argument("--mode", type=str, default="corrected_async_gn")
    ap.add_argument("--parallelism", type=str, default="patch")
    ap.add_argument("--no-cuda-graph", action="store_true")
    ap.add_argument("--force-cuda-graph", action="store_true",
                    help="capture hipGraphs also at world_size>1 (RCCL-in-graph)")
    ap.add_argument("--no-split-batch", action="store_true")
    ap.add_argument("--warmup-steps", type=int, default=4, help="sync-comm denoise steps")
    ap.add_argument("--preset", type=str, default="sdxl", choices=["sdxl", "tiny"],
                    help="tiny = CPU plumbing test of the distributed bench path")
    args = ap.parse_args()

    from distrifuser_amd import DistriConfig, DistriSDXLPipeline

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        # let MIOpen search for the best conv algorithm per shape (the
        # default immediate mode picked an im2col path for some 3840^2 convs
        # — profiles/rocprof_3840_r01.md)
        torch.backends.cudnn.benchmark = True
    world = int(os.environ.get("WORLD_SIZE", "1"))
    # hipGraph capture is the verified fast path at world_size 1; at 3840^2
    # the step is kernel-bound and graphs measured neutral, so multi-rank
    # runs stay eager by default (RCCL-in-graph capture is opt-in via
    # --force-cuda-graph until verified on an 8-GPU node).
    use_graphs = use_cuda and not args.no_cuda_graph and (
        world == 1 or args.force_cuda_graph
    )
    if args.force_cuda_graph and world > 1:
        print("WARNING: RCCL collectives inside hipGraph capture HANG on this "
              "stack (docs/DESIGN.md §5) — --force-cuda-graph at world_size>1 "
              "is a debug probe, not a fast path", flush=True)
    cfg = DistriConfig(
        height=args.height,
        width=args.width,
        do_classifier_free_guidance=True,
        split_batch=not args.no_split_batch,
        warmup_steps=args.warmup_steps,
        mode=args.mode,
        parallelism=args.parallelism,
        use_cuda_graph=use_graphs,
    )
    if cfg.world_size > 1:
        assert cfg.world_size == args.gpus, (
            f"WORLD_SIZE={cfg.world_size} != --gpus {args.gpus}"
        )
    dtype = torch.bfloat16 if use_cuda else torch.float32

    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(
        cfg, torch_dtype=dtype, scheduler=args.scheduler, preset=args.preset
    )

    # ---- manual denoise loop so we can time exactly K steps ----
    total_steps = args.warmup + args.steps
    sched = pipe.scheduler
    sched.set_timesteps(max(total_steps, 50))
    embeds, pooled = pipe.encode_prompt(
        "a photo of a supersonic aircraft over mountains", "", True
    )
    embeds = embeds.to(dtype)
    added = pipe._added_cond(embeds.shape[0], pooled.to(dtype))
    latents = pipe._prepare_latents(1, torch.Generator().manual_seed(0))
    pipe.unet.set_counter(0)

    def one_step(i: int, latents: torch.Tensor) -> torch.Tensor:
        t = sched.timesteps[i]
        latent_in = torch.cat([latents] * 2)
        latent_in = sched.scale_model_input(latent_in, t)
        noise = pipe.unet(latent_in, t.to(cfg.device) if torch.is_tensor(t) else t, embeds, added)
        if hasattr(sched, "guided_step"):
            return sched.guided_step(noise, t, latents, args.guidance_scale)
        nu, nc = noise.chunk(2)
        noise = nu + args.guidance_scale * (nc - nu)
        return sched.step(noise, t, latents)

    def sync():
        if dist.is_initialized():
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    with torch.no_grad():
        for i in range(args.warmup):
            latents = one_step(i, latents)
        sync()
        t0 = time.perf_counter()
        for i in range(args.warmup, total_steps):
            latents = one_step(i, latents)
        sync()
        elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=cfg.device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    latency_50 = ms_per_step * 50 / 1000.0

    if cfg.rank == 0:
        result = {
            "metric": "end-to-end latency (s) per image, SDXL 50-step DDIM",
            "value": latency_50,
            "unit": "s/image",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic prompts, random-init SDXL weights (no network)",
            "config": {
                "model": ("sdxl-base (random init)" if args.preset == "sdxl" else f"{args.preset} preset (random init)"),
                "global_batch": 1,
                "image": f"{args.height}x{args.width}",
                "seq_len": (args.height // 8) * (args.width // 8),
                "scheduler": args.scheduler,
                "guidance_scale": args.guidance_scale,
                "steps_per_image": 50,
                "parallelism": (
                    f"cfg2x{cfg.n_device_per_batch}patch" if cfg.split_batch
                    else f"{args.parallelism}{cfg.n_device_per_batch}"
                ),
                "mode": args.mode,
                "cuda_graph": cfg.use_cuda_graph,
            },
        }
        if pipe.comm_manager is not None and pipe.comm_manager.stats["gathers"]:
            s = pipe.comm_manager.stats
            result["config"]["comm"] = {
                "async_gathers": s["gathers"],
                "bytes_per_rank": s["bytes"],
            }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
