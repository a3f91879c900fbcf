import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run with -m gpu on an MI355X box)")


def _worker(rank, world_size, init_file, fn, args, result_queue):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.pop("MASTER_PORT", None)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group(
        backend="gloo", init_method=f"file://{init_file}", rank=rank, world_size=world_size
    )
    torch.manual_seed(0)
    try:
        out = fn(rank, world_size, *args)
        # Serialize by value: shared-memory tensor handles die with the worker.
        import io

        buf = io.BytesIO()
        torch.save(out, buf)
        result_queue.put((rank, "ok", buf.getvalue()))
    except Exception as exc:  # noqa: BLE001
        import traceback

        result_queue.put((rank, "err", f"{exc}\n{traceback.format_exc()}"))
        raise
    finally:
        dist.destroy_process_group()


def run_distributed(world_size, fn, args=()):
    """Run fn(rank, world_size, *args) in world_size gloo processes.

    Returns {rank: return_value}. fn must be a module-level (picklable)
    function.
    """
    ctx = mp.get_context("spawn")
    result_queue = ctx.Queue()
    with tempfile.NamedTemporaryFile(delete=False) as f:
        init_file = f.name
    os.unlink(init_file)
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, init_file, fn, args, result_queue))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, status, payload = result_queue.get(timeout=300)
        if status == "err":
            for p in procs:
                p.terminate()
            raise AssertionError(f"rank {rank} failed:\n{payload}")
        import io

        results[rank] = torch.load(io.BytesIO(payload), weights_only=False)
    for p in procs:
        p.join(timeout=60)
    return results


@pytest.fixture
def distributed_runner():
    return run_distributed
