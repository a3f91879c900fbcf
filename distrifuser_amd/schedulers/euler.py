"""Euler discrete (epsilon-prediction, leading spacing)."""

from __future__ import annotations

import numpy as np
import torch

from .common import SchedulerBase


class EulerDiscreteScheduler(SchedulerBase):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        acp = self.alphas_cumprod.double()
        self._sigmas_train = ((1 - acp) / acp).sqrt().numpy()
        self.sigmas: torch.Tensor | None = None
        self._step_index = 0

    def set_timesteps(self, num_inference_steps: int, device=None) -> None:
        self.num_inference_steps = num_inference_steps
        t = self._leading_timesteps(num_inference_steps).numpy().astype(np.float64)
        sigmas = np.interp(t, np.arange(self.num_train_timesteps), self._sigmas_train)
        self.sigmas = torch.tensor(np.concatenate([sigmas, [0.0]]), dtype=torch.float32)
        self.timesteps = torch.tensor(t, dtype=torch.float32)
        if device is not None:
            self.timesteps = self.timesteps.to(device)
        self.init_noise_sigma = float((self.sigmas.max() ** 2 + 1) ** 0.5)
        self._step_index = 0

    def scale_model_input(self, sample: torch.Tensor, timestep=None) -> torch.Tensor:
        sigma = self.sigmas[self._step_index]
        return sample / float((sigma**2 + 1) ** 0.5)

    def step(self, model_output: torch.Tensor, timestep, sample: torch.Tensor) -> torch.Tensor:
        sigma = self.sigmas[self._step_index].to(sample.device)
        sigma_next = self.sigmas[self._step_index + 1].to(sample.device)
        x = sample.float()
        eps = model_output.float()
        # epsilon prediction: derivative d = eps; x_{t+1} = x + d * (s_next - s)
        prev = x + eps * (sigma_next - sigma)
        self._step_index += 1
        return prev.to(sample.dtype)

    def guided_step(self, noise: torch.Tensor, timestep, sample: torch.Tensor,
                    guidance_scale: float) -> torch.Tensor:
        # CFG combine + Euler update in ONE fused kernel on GPU (the update is
        # affine in (x, eps): ca = 1, cb = sigma_next - sigma)
        import os

        if noise.is_cuda and os.environ.get("DFA_FORCE_EAGER", "0") != "1":
            from ..ops.dispatch import hip_ext

            cb = float(self.sigmas[self._step_index + 1] - self.sigmas[self._step_index])
            self._step_index += 1
            return hip_ext().cfg_affine_step(noise, sample, guidance_scale, 1.0, cb)
        nu, nc = noise.float().chunk(2)
        eps = nu + guidance_scale * (nc - nu)
        return self.step(eps, timestep, sample)
