"""Scheduler math: with an oracle eps-model, each sampler must recover x0.

If sample_t = sqrt(acp_t) * x0 + sqrt(1-acp_t) * eps (DDIM/DPM convention)
or sample = x0 + sigma * eps (Euler convention) and the "model" returns the
exact eps, the deterministic samplers reconstruct x0 (up to the t=0 floor of
the schedule)."""

import torch

from distrifuser_amd.schedulers import (
    DDIMScheduler,
    DPMSolverMultistepScheduler,
    EulerDiscreteScheduler,
    get_scheduler,
)


def test_get_scheduler():
    assert isinstance(get_scheduler("ddim"), DDIMScheduler)
    assert isinstance(get_scheduler("euler"), EulerDiscreteScheduler)
    assert isinstance(get_scheduler("dpm-solver"), DPMSolverMultistepScheduler)


def test_leading_timesteps():
    s = DDIMScheduler()
    s.set_timesteps(50)
    assert len(s.timesteps) == 50
    assert int(s.timesteps[0]) == 981  # 49*20 + 1
    assert int(s.timesteps[-1]) == 1


def _run_ddim_like(scheduler, steps=50):
    torch.manual_seed(0)
    x0 = torch.randn(1, 4, 8, 8)
    eps = torch.randn(1, 4, 8, 8)
    scheduler.set_timesteps(steps)
    t0 = int(scheduler.timesteps[0])
    acp = scheduler.alphas_cumprod
    sample = acp[t0].sqrt() * x0 + (1 - acp[t0]).sqrt() * eps
    for t in scheduler.timesteps:
        sample = scheduler.step(eps, t, sample)
    return x0, sample


def test_ddim_recovers_x0():
    x0, final = _run_ddim_like(DDIMScheduler())
    # final = sqrt(acp_0) x0 + sqrt(1-acp_0) eps, acp_0 ~ 0.99915
    assert (final - x0).abs().max() < 0.15
    assert torch.allclose(final, x0, atol=0.15)


def test_dpm_recovers_x0():
    x0, final = _run_ddim_like(DPMSolverMultistepScheduler())
    assert (final - x0).abs().max() < 0.15


def test_euler_recovers_x0():
    torch.manual_seed(0)
    x0 = torch.randn(1, 4, 8, 8)
    eps = torch.randn(1, 4, 8, 8)
    s = EulerDiscreteScheduler()
    s.set_timesteps(50)
    sample = x0 + s.sigmas[0] * eps
    for t in s.timesteps:
        sample = s.step(eps, t, sample)
    # last sigma is exactly 0 -> exact recovery
    assert torch.allclose(sample, x0, atol=1e-4)


def test_euler_scale_model_input():
    s = EulerDiscreteScheduler()
    s.set_timesteps(10)
    x = torch.ones(2, 2)
    scaled = s.scale_model_input(x, s.timesteps[0])
    sigma = float(s.sigmas[0])
    assert torch.allclose(scaled, x / (sigma**2 + 1) ** 0.5)
    assert abs(s.init_noise_sigma - (sigma**2 + 1) ** 0.5) < 1e-6


def test_ddim_guided_step_matches_manual():
    torch.manual_seed(0)
    s = DDIMScheduler()
    s.set_timesteps(10)
    noise = torch.randn(2, 4, 8, 8)
    x = torch.randn(1, 4, 8, 8)
    g = 5.0
    t = s.timesteps[3]
    fused = s.guided_step(noise, t, x, g)
    nu, nc = noise.chunk(2)
    eps = nu + g * (nc - nu)
    ref = s.step(eps, t, x)
    assert torch.allclose(fused, ref, atol=1e-6)


def test_euler_guided_step_matches_manual():
    torch.manual_seed(0)
    s = EulerDiscreteScheduler()
    s.set_timesteps(10)
    noise = torch.randn(2, 4, 8, 8)
    x = torch.randn(1, 4, 8, 8)
    t = s.timesteps[0]
    fused = s.guided_step(noise, t, x, 7.5)
    s2 = EulerDiscreteScheduler()
    s2.set_timesteps(10)
    nu, nc = noise.chunk(2)
    eps = nu + 7.5 * (nc - nu)
    ref = s2.step(eps, t, x)
    assert torch.allclose(fused, ref, atol=1e-6)


def test_ddim_monotone_denoise():
    """Variance of the sample should shrink toward the data scale."""
    torch.manual_seed(0)
    s = DDIMScheduler()
    s.set_timesteps(10)
    x0 = torch.zeros(1, 4, 8, 8)
    eps = torch.randn(1, 4, 8, 8)
    t0 = int(s.timesteps[0])
    sample = s.alphas_cumprod[t0].sqrt() * x0 + (1 - s.alphas_cumprod[t0]).sqrt() * eps
    norms = [sample.norm().item()]
    for t in s.timesteps:
        sample = s.step(eps, t, sample)
        norms.append(sample.norm().item())
    assert norms[-1] < norms[0]
