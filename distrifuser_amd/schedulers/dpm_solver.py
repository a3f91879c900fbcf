"""DPM-Solver++ (2M, multistep, epsilon-prediction)."""

from __future__ import annotations

import torch

from .common import SchedulerBase


class DPMSolverMultistepScheduler(SchedulerBase):
    def __init__(self, *args, solver_order: int = 2, **kwargs):
        super().__init__(*args, **kwargs)
        self.solver_order = solver_order
        acp = self.alphas_cumprod
        self.alpha_t = acp.sqrt()
        self.sigma_t = (1 - acp).sqrt()
        self.lambda_t = torch.log(self.alpha_t) - torch.log(self.sigma_t)
        self._x0_prev: torch.Tensor | None = None
        self._t_prev: int | None = None
        self._step_index = 0

    def set_timesteps(self, num_inference_steps: int, device=None) -> None:
        self.num_inference_steps = num_inference_steps
        self.timesteps = self._leading_timesteps(num_inference_steps)
        if device is not None:
            self.timesteps = self.timesteps.to(device)
        self._x0_prev = None
        self._t_prev = None
        self._step_index = 0

    def _prev_timestep(self, t: int) -> int:
        idx = int((self.timesteps == t).nonzero()[0])
        return int(self.timesteps[idx + 1]) if idx + 1 < len(self.timesteps) else 0

    def step(self, model_output: torch.Tensor, timestep, sample: torch.Tensor) -> torch.Tensor:
        t = int(timestep)
        t_prev = self._prev_timestep(t)
        x = sample.float()
        eps = model_output.float()

        a_t, s_t, l_t = self.alpha_t[t], self.sigma_t[t], self.lambda_t[t]
        a_p, s_p, l_p = self.alpha_t[t_prev], self.sigma_t[t_prev], self.lambda_t[t_prev]
        x0 = (x - s_t * eps) / a_t

        h = l_p - l_t
        if self._x0_prev is None or self.solver_order == 1 or self._step_index == len(self.timesteps) - 1:
            # first-order (DPM-Solver++ 1S) update
            prev = (s_p / s_t) * x - a_p * (torch.exp(-h) - 1.0) * x0
        else:
            l_pp = self.lambda_t[self._t_prev]
            h_0 = l_t - l_pp
            r0 = h_0 / h
            d0 = x0
            d1 = (x0 - self._x0_prev) / r0
            prev = (
                (s_p / s_t) * x
                - a_p * (torch.exp(-h) - 1.0) * d0
                - 0.5 * a_p * (torch.exp(-h) - 1.0) * d1
            )

        self._x0_prev = x0
        self._t_prev = t
        self._step_index += 1
        return prev.to(sample.dtype)
