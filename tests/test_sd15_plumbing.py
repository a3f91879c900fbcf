"""BASELINE.json config #1: SD1.5 512x512 10-step DDIM, world_size=1 on CPU
(DistriSDPipeline plumbing with the REAL SD1.5 architecture, random init)."""

import torch

from distrifuser_amd import DistriConfig, DistriSDPipeline


def test_sd15_512_10step_ddim_cpu():
    cfg = DistriConfig(
        height=512, width=512, do_classifier_free_guidance=False,
        use_cuda_graph=False, device="cpu",
    )
    torch.manual_seed(0)
    pipe = DistriSDPipeline.from_pretrained(cfg, torch_dtype=torch.float32, scheduler="ddim")
    # real SD1.5 shapes
    assert sum(p.numel() for p in pipe.unet.parameters()) > 8.5e8
    assert pipe.text_encoder.config.hidden_size == 768
    out = pipe(
        "a photograph of an astronaut riding a horse",
        num_inference_steps=10,
        guidance_scale=1,
        generator=torch.Generator().manual_seed(42),
        output_type="latent",
    )
    assert out.shape == (1, 4, 64, 64)
    assert torch.isfinite(out).all()
