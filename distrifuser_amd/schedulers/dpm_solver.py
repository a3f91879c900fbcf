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

    def guided_step(self, noise: torch.Tensor, timestep, sample: torch.Tensor,
                    guidance_scale: float) -> torch.Tensor:
        """CFG combine + DPM-Solver++(2M) update in ONE fused kernel on GPU.

        The 2M update is affine in (x, eps, x0_prev):
            prev = ca*x + cb*eps + cc*x0_prev,   x0 = cx*x + ce*eps
        with C = a_p*(exp(-h)-1), C' = C*(1 + 0.5/r0) (order 2) or C (order 1):
            ca = s_p/s_t - C'/a_t,  cb = C'*s_t/a_t,  cc = 0.5*C/r0.
        """
        import os

        if not (noise.is_cuda and os.environ.get("DFA_FORCE_EAGER", "0") != "1"):
            nu, nc = noise.float().chunk(2)
            eps = nu + guidance_scale * (nc - nu)
            return self.step(eps, timestep, sample)

        from ..ops.dispatch import hip_ext

        t = int(timestep)
        t_prev = self._prev_timestep(t)
        a_t, s_t, l_t = (float(self.alpha_t[t]), float(self.sigma_t[t]),
                         float(self.lambda_t[t]))
        a_p, s_p, l_p = (float(self.alpha_t[t_prev]), float(self.sigma_t[t_prev]),
                         float(self.lambda_t[t_prev]))
        import math

        h = l_p - l_t
        C = a_p * (math.exp(-h) - 1.0)
        first = (self._x0_prev is None or self.solver_order == 1
                 or self._step_index == len(self.timesteps) - 1)
        if first:
            cprime, cc = C, 0.0
        else:
            l_pp = float(self.lambda_t[self._t_prev])
            r0 = (l_t - l_pp) / h
            cprime = C * (1.0 + 0.5 / r0)
            cc = 0.5 * C / r0
        ca = s_p / s_t - cprime / a_t
        cb = cprime * s_t / a_t
        cx = 1.0 / a_t
        ce = -s_t / a_t
        prev, x0 = hip_ext().cfg_dpm_step(
            noise, sample, None if first else self._x0_prev, guidance_scale,
            ca, cb, cc, cx, ce)
        self._x0_prev = x0
        self._t_prev = t
        self._step_index += 1
        return prev
