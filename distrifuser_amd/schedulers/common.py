"""Shared noise-schedule math for the SD-family schedulers.

SD/SDXL training schedule: 1000 steps, scaled_linear betas in
[0.00085, 0.012], epsilon prediction, leading timestep spacing with
steps_offset=1 (the reference inherited these from the diffusers scheduler
configs shipped with the checkpoints).
"""

from __future__ import annotations

import torch


class SchedulerBase:
    order = 1

    def __init__(
        self,
        num_train_timesteps: int = 1000,
        beta_start: float = 0.00085,
        beta_end: float = 0.012,
        steps_offset: int = 1,
    ):
        self.num_train_timesteps = num_train_timesteps
        self.steps_offset = steps_offset
        betas = (
            torch.linspace(beta_start**0.5, beta_end**0.5, num_train_timesteps, dtype=torch.float64)
            ** 2
        )
        alphas = 1.0 - betas
        self.alphas_cumprod = torch.cumprod(alphas, dim=0).float()
        self.timesteps: torch.Tensor | None = None
        self.num_inference_steps: int | None = None
        self.init_noise_sigma = 1.0

    def _leading_timesteps(self, num_inference_steps: int) -> torch.Tensor:
        step_ratio = self.num_train_timesteps // num_inference_steps
        t = (torch.arange(num_inference_steps) * step_ratio).flip(0) + self.steps_offset
        return t.long()

    def scale_model_input(self, sample: torch.Tensor, timestep=None) -> torch.Tensor:
        return sample

    def set_timesteps(self, num_inference_steps: int, device=None) -> None:
        raise NotImplementedError

    def step(self, model_output: torch.Tensor, timestep, sample: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError
