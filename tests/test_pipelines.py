"""End-to-end pipeline tests on CPU (tiny presets; BASELINE config #1's
world_size=1 CPU plumbing check runs the real SD1.5 shapes in test_sd15_*)."""

import numpy as np
import torch

from distrifuser_amd import DistriConfig, DistriSDPipeline, DistriSDXLPipeline

from conftest import run_distributed


def _tiny_sdxl(device="cpu", **cfg_kwargs):
    cfg = DistriConfig(
        height=128, width=128, use_cuda_graph=False, device=device,
        **cfg_kwargs,
    )
    torch.manual_seed(0)
    return DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)


def test_sdxl_tiny_pipeline_latent():
    pipe = _tiny_sdxl()
    out = pipe("a photo of an astronaut", num_inference_steps=3, output_type="latent")
    assert out.shape == (1, 4, 16, 16)
    assert torch.isfinite(out).all()


def test_sdxl_tiny_pipeline_np_and_determinism():
    pipe = _tiny_sdxl()
    g = torch.Generator().manual_seed(7)
    img1 = pipe("hello", num_inference_steps=2, output_type="np", generator=g)
    g = torch.Generator().manual_seed(7)
    img2 = pipe("hello", num_inference_steps=2, output_type="np", generator=g)
    assert isinstance(img1, np.ndarray)
    assert img1.shape == (1, 128, 128, 3)
    assert img1.dtype == np.uint8
    assert (img1 == img2).all()


def test_sdxl_tiny_no_cfg():
    pipe = _tiny_sdxl(do_classifier_free_guidance=False)
    out = pipe("x", num_inference_steps=2, guidance_scale=1, output_type="latent")
    assert out.shape == (1, 4, 16, 16)


def test_sd_tiny_pipeline():
    cfg = DistriConfig(height=64, width=64, use_cuda_graph=False, device="cpu")
    torch.manual_seed(0)
    pipe = DistriSDPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)
    out = pipe("a cat", num_inference_steps=2, output_type="latent")
    assert out.shape == (1, 4, 8, 8)
    assert torch.isfinite(out).all()


def test_sd_tiny_euler_and_dpm():
    for sched in ("euler", "dpm-solver"):
        cfg = DistriConfig(height=64, width=64, use_cuda_graph=False, device="cpu")
        torch.manual_seed(0)
        pipe = DistriSDPipeline.from_pretrained(
            cfg, preset="tiny", torch_dtype=torch.float32, scheduler=sched
        )
        out = pipe("a cat", num_inference_steps=4, output_type="latent")
        assert torch.isfinite(out).all(), sched


def _pipeline_worker(rank, world_size, mode):
    cfg = DistriConfig(
        height=128, width=128, mode=mode, warmup_steps=2,
        use_cuda_graph=False, device="cpu",
    )
    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)
    g = torch.Generator().manual_seed(3)
    out = pipe("a scenic mountain", num_inference_steps=6, output_type="latent", generator=g)
    return out.clone()


def test_sdxl_pipeline_parallel_matches_single_ws2():
    """ws=2 + CFG split (pure batch parallel) must equal the 1-proc run."""
    torch.manual_seed(0)
    cfg = DistriConfig(height=128, width=128, use_cuda_graph=False, device="cpu", warmup_steps=2)
    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)
    g = torch.Generator().manual_seed(3)
    ref = pipe("a scenic mountain", num_inference_steps=6, output_type="latent", generator=g)

    out = run_distributed(2, _pipeline_worker, ("corrected_async_gn",))
    for r in (0, 1):
        assert torch.allclose(out[r], ref, atol=1e-3), (
            f"rank {r}: max err {(out[r] - ref).abs().max()}"
        )


def _psnr(a: torch.Tensor, b: torch.Tensor) -> float:
    mse = float(((a - b) ** 2).mean())
    if mse == 0:
        return float("inf")
    rng = float(b.max() - b.min())
    import math

    return 10 * math.log10(rng * rng / mse)


def test_mode_ladder_quality_vs_full_sync_ws2():
    """Quantified staleness-mode ladder (the reference's PSNR workflow,
    README.md:121-144, as an automated assertion): each displaced mode's
    latents stay within a PSNR floor of full_sync on a real multi-step
    denoise, and no_sync (zero comm) is the quality baseline."""
    torch.manual_seed(0)
    cfg = DistriConfig(height=128, width=128, use_cuda_graph=False, device="cpu", warmup_steps=2)
    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)
    g = torch.Generator().manual_seed(3)
    ref = pipe("a scenic mountain", num_inference_steps=6, output_type="latent", generator=g)

    psnrs = {}
    for mode in ("corrected_async_gn", "stale_gn", "no_sync"):
        out = run_distributed(2, _pipeline_worker, (mode,))
        psnrs[mode] = _psnr(out[0], ref)
    # displaced modes must track the serial output closely at this scale
    assert psnrs["corrected_async_gn"] > 25, psnrs
    assert psnrs["stale_gn"] > 25, psnrs
    # no cross-patch comm at all is allowed to drift further but stays sane
    assert psnrs["no_sync"] > 10, psnrs


def test_sdxl_pipeline_parallel_patch_ws4():
    """ws=4: CFG split x 2 displaced patches; ranks agree, full_sync == 1-proc."""
    torch.manual_seed(0)
    cfg = DistriConfig(height=128, width=128, use_cuda_graph=False, device="cpu", warmup_steps=2)
    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.float32)
    g = torch.Generator().manual_seed(3)
    ref = pipe("a scenic mountain", num_inference_steps=6, output_type="latent", generator=g)

    out = run_distributed(4, _pipeline_worker, ("full_sync",))
    for r in range(4):
        assert torch.allclose(out[r], ref, atol=1e-3), (
            f"rank {r}: max err {(out[r] - ref).abs().max()}"
        )

    # displaced mode: same image up to staleness error, tiny on 6 steps
    out_async = run_distributed(4, _pipeline_worker, ("corrected_async_gn",))
    err = (out_async[0] - ref).abs().max()
    assert torch.isfinite(out_async[0]).all()
    assert err < 0.5, f"displaced-patch drift too large: {err}"


def test_vae_tiled_decode_matches_full():
    """Tiled decode (diffusers enable_tiling parity) matches the full decode
    away from tile seams and reproduces the exact output shape."""
    import torch

    from distrifuser_amd.models.vae import TINY_VAE, VAEDecoder

    torch.manual_seed(0)
    v = VAEDecoder(TINY_VAE).eval()
    z = torch.randn(1, TINY_VAE.latent_channels, 24, 40)
    full = v(z)
    v.enable_tiling(tile_latent_size=16, tile_overlap=4)
    tiled = v(z)
    assert tiled.shape == full.shape
    diff = (full - tiled).abs()
    # tiled decode is approximate near tile boundaries (each tile's convs
    # see zero padding instead of neighbour context — diffusers' tiled
    # decode has the same property); the aggregate deviation stays small
    assert float(diff.mean()) < 0.1
    assert float(diff.median()) < 0.05
