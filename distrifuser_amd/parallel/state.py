"""Shared per-model parallel state.

The reference threads a per-module ``counter`` plus ``set_counter`` /
``set_comm_manager`` propagation through every wrapped layer
(reference modules/base_module.py:6-29, models/base_model.py:27-37). Our
modules are parallelism-aware natively, so they simply share ONE state
object: the denoising-step counter, the comm manager, and the config.
"""

from __future__ import annotations

from ..utils.comm import PatchParallelismCommManager
from ..utils.config import DistriConfig


class ParallelState:
    def __init__(self, config: DistriConfig):
        self.config = config
        self.comm_manager: PatchParallelismCommManager | None = None
        self.counter = 0
        # True only during the pipeline's registration / pre-run passes
        # (reference `record=True` forward): modules may register comm slots
        # and capture static state.
        self.recording = False

    @property
    def in_warmup(self) -> bool:
        """Sync-communication phase: the first ``warmup_steps`` denoise steps."""
        return self.counter <= self.config.warmup_steps

    @property
    def use_sync_comm(self) -> bool:
        return self.config.mode == "full_sync" or self.in_warmup

    def set_counter(self, value: int = 0) -> None:
        self.counter = value

    def next_step(self) -> None:
        self.counter += 1
