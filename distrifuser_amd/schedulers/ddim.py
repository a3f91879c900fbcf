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

    def guided_step(self, noise: torch.Tensor, timestep, sample: torch.Tensor,
                    guidance_scale: float) -> torch.Tensor:
        # CFG combine + DDIM update; noise is the [uncond; cond] pair. On GPU
        # this is ONE fused kernel (csrc/scheduler.hip) - the step math runs
        # outside the captured graphs, so its torch-op launch overhead is
        # otherwise exposed every denoise step.
        import os

        t = int(timestep)
        prev_t = t - self.num_train_timesteps // self.num_inference_steps
        acp = self.alphas_cumprod
        alpha_t = float(acp[t])
        alpha_prev = float(acp[prev_t] if prev_t >= 0 else acp[0])
        if noise.is_cuda and os.environ.get("DFA_FORCE_EAGER", "0") != "1":
            from ..ops.dispatch import hip_ext

            # x' = sqrt(a_p) * (x - sqrt(1-a_t) eps)/sqrt(a_t) + sqrt(1-a_p) eps
            #    = ca * x + cb * eps
            ca = (alpha_prev / alpha_t) ** 0.5
            cb = (1 - alpha_prev) ** 0.5 - ca * (1 - alpha_t) ** 0.5
            return hip_ext().cfg_affine_step(noise, sample, guidance_scale, ca, cb)
        nu, nc = noise.float().chunk(2)
        eps = nu + guidance_scale * (nc - nu)
        return self.step(eps, timestep, sample)
