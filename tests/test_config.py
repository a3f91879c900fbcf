"""DistriConfig topology tests (single- and multi-process gloo)."""

import pytest
import torch

from distrifuser_amd import DistriConfig
from distrifuser_amd.utils.config import is_power_of_2

from conftest import run_distributed


def test_is_power_of_2():
    assert [n for n in range(1, 17) if is_power_of_2(n)] == [1, 2, 4, 8, 16]
    assert not is_power_of_2(0)


def test_single_process_defaults():
    cfg = DistriConfig(height=512, width=512, device="cpu")
    assert cfg.world_size == 1
    assert cfg.n_device_per_batch == 1
    assert not cfg.split_batch  # degrades at world_size 1
    assert cfg.batch_idx() == 0
    assert cfg.split_idx() == 0
    assert cfg.mode == "corrected_async_gn"
    assert cfg.warmup_steps == 4
    assert cfg.comm_checkpoint == 60


def test_mode_validation():
    with pytest.raises(AssertionError):
        DistriConfig(mode="bogus", device="cpu")
    with pytest.raises(AssertionError):
        DistriConfig(parallelism="bogus", device="cpu")
    with pytest.raises(AssertionError):
        DistriConfig(split_scheme="bogus", device="cpu")


def _topology(rank, world_size):
    cfg = DistriConfig(device="cpu")
    return {
        "batch_idx": cfg.batch_idx(),
        "split_idx": cfg.split_idx(),
        "n_device_per_batch": cfg.n_device_per_batch,
        "split_batch": cfg.split_batch,
    }


def test_topology_ws4():
    out = run_distributed(4, _topology)
    # CFG split: first half batch 0, second half batch 1; patch idx = rank % 2
    assert [out[r]["batch_idx"] for r in range(4)] == [0, 0, 1, 1]
    assert [out[r]["split_idx"] for r in range(4)] == [0, 1, 0, 1]
    assert all(out[r]["n_device_per_batch"] == 2 for r in range(4))


def _topology_no_cfg(rank, world_size):
    cfg = DistriConfig(do_classifier_free_guidance=False, device="cpu")
    return cfg.n_device_per_batch, cfg.split_idx()


def test_topology_ws2_no_cfg():
    out = run_distributed(2, _topology_no_cfg)
    assert out[0] == (2, 0)
    assert out[1] == (2, 1)
