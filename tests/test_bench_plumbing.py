"""The driver launches bench.py via torch.distributed.run for N>1 — exercise
that exact path on CPU (gloo, tiny preset) so the unattended multi-GPU run
has no untested plumbing."""

import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_bench(nproc: int):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
        os.path.join(REPO, "bench.py"),
        "--gpus", str(nproc), "--steps", "2", "--warmup", "1",
        "--height", "128", "--width", "128", "--preset", "tiny",
    ]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def test_bench_single_process():
    cmd = [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "1",
           "--steps", "2", "--warmup", "1", "--height", "128", "--width", "128",
           "--preset", "tiny"]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert rec["metric"].startswith("end-to-end latency")
    assert rec["n_gpus"] == 1
    assert rec["value"] > 0
    assert rec["higher_is_better"] is False


def test_bench_torchrun_two_ranks():
    rec = _run_bench(2)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "cfg2x1patch"
    assert rec["value"] > 0
