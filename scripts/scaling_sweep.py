"""Run the flagship bench at N = 1, 2, 4, 8 on one node and print the
speedup table (the reference's headline format, README.md:30). For use on a
multi-GPU MI355X node:

    python scripts/scaling_sweep.py --height 3840 --width 3840 --steps 6
"""

import argparse
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_bench(n: int, args) -> dict:
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
        "--master-port", str(free_port()),
        os.path.join(REPO, "bench.py"),
        "--gpus", str(n), "--steps", str(args.steps), "--warmup", str(args.warmup),
        "--height", str(args.height), "--width", str(args.width),
    ] if n > 1 else [
        sys.executable, os.path.join(REPO, "bench.py"),
        "--gpus", "1", "--steps", str(args.steps), "--warmup", str(args.warmup),
        "--height", str(args.height), "--width", str(args.width),
    ]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    out = subprocess.run(cmd, capture_output=True, text=True, env=env, timeout=3600)
    if out.returncode != 0:
        raise SystemExit(f"N={n} failed:\n{out.stdout[-1500:]}\n{out.stderr[-1500:]}")
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--height", type=int, default=3840)
    ap.add_argument("--width", type=int, default=3840)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--gpus", type=int, nargs="*", default=None,
                    help="GPU counts to sweep (default: 1,2,4,.. up to available)")
    args = ap.parse_args()

    import torch

    avail = torch.cuda.device_count()
    counts = args.gpus or [n for n in (1, 2, 4, 8) if n <= avail]
    results = {}
    for n in counts:
        results[n] = run_bench(n, args)
        print(f"N={n}: {results[n]['value']:.2f} s/image "
              f"({results[n]['ms_per_step']:.1f} ms/step)", flush=True)

    base = results[counts[0]]["value"]
    print(f"\nSDXL {args.height}x{args.width} 50-step DDIM — speedup vs {counts[0]} GPU "
          "(reference A100 headline: 1.8x/3.4x/6.1x at 2/4/8):")
    for n in counts:
        print(f"  {n} GPU: {base / results[n]['value']:.2f}x "
              f"({results[n]['value']:.2f} s/image)")
    with open("scaling_sweep.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
