"""DDIM (eta=0, epsilon-prediction) — the benchmark-protocol sampler."""

from __future__ import annotations

import torch

from .common import SchedulerBase


class DDIMScheduler(SchedulerBase):
    def set_timesteps(self, num_inference_steps: int, device=None) -> None:
        self.num_inference_steps = num_inference_steps
        self.timesteps = self._leading_timesteps(num_inference_steps)
        if device is not None:
            self.timesteps = self.timesteps.to(device)

    def step(self, model_output: torch.Tensor, timestep, sample: torch.Tensor) -> torch.Tensor:
        t = int(timestep)
        prev_t = t - self.num_train_timesteps // self.num_inference_steps
        acp = self.alphas_cumprod
        alpha_t = acp[t]
        alpha_prev = acp[prev_t] if prev_t >= 0 else acp[0]

        x = sample.float()
        eps = model_output.float()
        pred_x0 = (x - (1 - alpha_t).sqrt() * eps) / alpha_t.sqrt()
        direction = (1 - alpha_prev).sqrt() * eps
        prev = alpha_prev.sqrt() * pred_x0 + direction
        return prev.to(sample.dtype)
