"""CommManager: registration, batched async all-gather, staleness lifecycle."""

import torch

from distrifuser_amd import DistriConfig, PatchParallelismCommManager

from conftest import run_distributed


def _roundtrip(rank, world_size):
    cfg = DistriConfig(do_classifier_free_guidance=False, device="cpu")
    comm = PatchParallelismCommManager(cfg)
    i0 = comm.register_tensor((2, 3), torch.float32, layer_type="attn")
    i1 = comm.register_tensor((4,), torch.float32, layer_type="gn")
    comm.create_buffer()
    # slot starts and the row stride are 16B-aligned (8-element grid):
    # (2,3) at [0,6), (4,) at [8,12), row stride padded to 16
    assert comm.starts == [0, 8]
    assert comm.numels == 16

    t0 = torch.full((2, 3), float(rank))
    t1 = torch.arange(4, dtype=torch.float32) + rank * 10
    comm.enqueue(i0, t0)
    comm.enqueue(i1, t1)
    comm.communicate()
    comm.wait(i0)
    comm.wait(i1)

    bl0 = comm.get_buffer_list(i0)
    bl1 = comm.get_buffer_list(i1)
    for p in range(world_size):
        assert torch.equal(bl0[p], torch.full((2, 3), float(p)))
        assert torch.equal(bl1[p], torch.arange(4, dtype=torch.float32) + p * 10)
    return True


def test_comm_roundtrip_ws2():
    out = run_distributed(2, _roundtrip)
    assert all(out.values())


def _wrap_flush(rank, world_size):
    """Re-enqueueing idx 0 must flush the previous step's batch first."""
    cfg = DistriConfig(do_classifier_free_guidance=False, device="cpu")
    comm = PatchParallelismCommManager(cfg)
    i0 = comm.register_tensor((2,), torch.float32)
    i1 = comm.register_tensor((2,), torch.float32)
    comm.create_buffer()

    comm.enqueue(i0, torch.tensor([1.0 + rank, 0.0]))
    comm.enqueue(i1, torch.tensor([2.0 + rank, 0.0]))
    # new step: wrapping to idx 0 flushes the queue above
    comm.enqueue(i0, torch.tensor([3.0 + rank, 0.0]))
    assert comm.handles[i0] is not None
    comm.wait(i0)
    comm.wait(i1)
    bl = comm.get_buffer_list(i1)
    for p in range(world_size):
        assert bl[p][0].item() == 2.0 + p
    comm.clear()
    bl0 = comm.get_buffer_list(i0)
    for p in range(world_size):
        assert bl0[p][0].item() == 3.0 + p
    return True


def test_comm_wrap_flush_ws2():
    out = run_distributed(2, _wrap_flush)
    assert all(out.values())


def _checkpoint_flush(rank, world_size):
    cfg = DistriConfig(do_classifier_free_guidance=False, device="cpu")
    cfg.comm_checkpoint = 3
    comm = PatchParallelismCommManager(cfg)
    idxs = [comm.register_tensor((1,), torch.float32) for _ in range(5)]
    comm.create_buffer()
    for j, i in enumerate(idxs):
        comm.enqueue(i, torch.tensor([float(rank * 100 + j)]))
        if j < 2:
            assert comm.handles[i] is None  # still queued
    # after the 3rd enqueue the first batch flushed
    assert comm.handles[idxs[0]] is not None
    comm.clear()
    for j, i in enumerate(idxs):
        bl = comm.get_buffer_list(i)
        for p in range(world_size):
            assert bl[p].item() == p * 100 + j
    return True


def test_comm_checkpoint_flush_ws2():
    out = run_distributed(2, _checkpoint_flush)
    assert all(out.values())


def _stats_worker(rank, world_size):
    cfg = DistriConfig(do_classifier_free_guidance=False, device="cpu")
    comm = PatchParallelismCommManager(cfg)
    i0 = comm.register_tensor((8,), torch.float32)
    comm.create_buffer()
    comm.enqueue(i0, torch.ones(8))
    comm.communicate()
    comm.clear()
    return dict(comm.stats)


def test_comm_stats_accounting():
    out = run_distributed(2, _stats_worker)
    for r in (0, 1):
        assert out[r]["gathers"] == 1
        assert out[r]["bytes"] == 8 * 4
