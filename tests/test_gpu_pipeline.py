"""GPU end-to-end: tiny + real-shape SDXL pipeline on one MI355X, including
hipGraph capture and the displaced-patch single-rank degenerate path."""

import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs ROCm GPU")


@requires_gpu
def test_tiny_sdxl_gpu_with_graphs():
    from distrifuser_amd import DistriConfig, DistriSDXLPipeline

    cfg = DistriConfig(height=128, width=128, use_cuda_graph=True, device="cuda:0")
    torch.manual_seed(0)
    pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.bfloat16)
    out = pipe("a photo", num_inference_steps=4, output_type="latent")
    assert out.shape == (1, 4, 16, 16)
    assert torch.isfinite(out.float()).all()


@requires_gpu
def test_tiny_sdxl_gpu_graph_vs_eager():
    from distrifuser_amd import DistriConfig, DistriSDXLPipeline

    outs = {}
    for use_graph in (False, True):
        cfg = DistriConfig(height=128, width=128, use_cuda_graph=use_graph, device="cuda:0")
        torch.manual_seed(0)
        pipe = DistriSDXLPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.bfloat16)
        g = torch.Generator().manual_seed(5)
        outs[use_graph] = pipe(
            "graph parity", num_inference_steps=4, output_type="latent", generator=g
        ).float()
    err = (outs[True] - outs[False]).abs().max().item()
    assert err < 0.05, f"graph-replay output drifted from eager: {err}"


@requires_gpu
def test_sd_tiny_gpu():
    from distrifuser_amd import DistriConfig, DistriSDPipeline

    cfg = DistriConfig(height=64, width=64, use_cuda_graph=False, device="cuda:0")
    torch.manual_seed(0)
    pipe = DistriSDPipeline.from_pretrained(cfg, preset="tiny", torch_dtype=torch.bfloat16)
    out = pipe("a cat", num_inference_steps=3, output_type="np")
    assert out.shape == (1, 64, 64, 3)


@requires_gpu
def test_hip_vs_eager_unet_integration(monkeypatch):
    """Whole-U-Net output with HIP kernels vs the eager composition
    (DFA_FORCE_EAGER=1) — catches kernel-integration drift the per-op tests
    can't see (layout/stride handling through real module wiring)."""
    import torch

    from distrifuser_amd import DistriConfig
    from distrifuser_amd.models import DistriUNet
    from distrifuser_amd.models.unet import UNetConfig

    # d64 heads so the flash kernel path is exercised
    cfg_model = UNetConfig(
        block_out_channels=(64, 128),
        down_block_types=("DownBlock2D", "CrossAttnDownBlock2D"),
        layers_per_block=1,
        transformer_layers_per_block=(1, 1),
        num_attention_heads=(1, 2),
        cross_attention_dim=64,
        norm_num_groups=8,
        use_linear_projection=True,
        addition_embed_type=None,
        sample_size=16,
    )
    outs = {}
    for eager in (False, True):
        if eager:
            monkeypatch.setenv("DFA_FORCE_EAGER", "1")
        else:
            monkeypatch.delenv("DFA_FORCE_EAGER", raising=False)
        cfg = DistriConfig(height=128, width=128, do_classifier_free_guidance=False,
                           use_cuda_graph=False, device="cuda:0")
        torch.manual_seed(0)
        unet = DistriUNet(cfg_model, cfg).to(device="cuda:0", dtype=torch.bfloat16).eval()
        x = torch.randn(1, 4, 16, 16, device="cuda:0", dtype=torch.bfloat16,
                        generator=torch.Generator("cuda:0").manual_seed(1))
        ehs = torch.randn(1, 7, 64, device="cuda:0", dtype=torch.bfloat16,
                          generator=torch.Generator("cuda:0").manual_seed(2))
        with torch.no_grad():
            unet.set_counter(0)
            outs[eager] = unet(x, 500.0, ehs, None).float()
    err = (outs[True] - outs[False]).abs().max().item()
    assert err < 0.1, f"HIP vs eager U-Net drift: {err}"


@requires_gpu
def test_sdxl_real_unet_one_step():
    """Full-size SDXL U-Net, one denoise step at 1024^2 on cuda:0 (bf16)."""
    from distrifuser_amd import DistriConfig
    from distrifuser_amd.models import DistriUNet
    from distrifuser_amd.models.unet import SDXL_UNET

    cfg = DistriConfig(height=1024, width=1024, do_classifier_free_guidance=False,
                       use_cuda_graph=False, device="cuda:0")
    torch.manual_seed(0)
    unet = DistriUNet(SDXL_UNET, cfg).to(device=cfg.device, dtype=torch.bfloat16).eval()
    lat = torch.randn(1, 4, 128, 128, device=cfg.device, dtype=torch.bfloat16)
    ehs = torch.randn(1, 77, 2048, device=cfg.device, dtype=torch.bfloat16)
    added = {
        "text_embeds": torch.randn(1, 1280, device=cfg.device, dtype=torch.bfloat16),
        "time_ids": torch.tensor([[1024, 1024, 0, 0, 1024, 1024]], device=cfg.device,
                                 dtype=torch.bfloat16),
    }
    with torch.no_grad():
        unet.set_counter(0)
        out = unet(lat, 500.0, ehs, added)
    assert out.shape == (1, 4, 128, 128)
    assert torch.isfinite(out.float()).all()


@requires_gpu
def test_comm_manager_single_rank_rccl(monkeypatch):
    """Displaced-path comm engine under a real (1-rank) RCCL group: register
    -> buffer -> enqueue -> async gather -> wait, with per-gather timing and
    the side-stream mode on, plus RCCL-inside-hipGraph capture (the capture
    path skips events/side-stream, matching multi-rank behavior)."""
    import os

    import torch.distributed as dist

    from distrifuser_amd import DistriConfig
    from distrifuser_amd.utils.comm import PatchParallelismCommManager

    monkeypatch.setenv("DFA_COMM_TIMING", "1")
    monkeypatch.setenv("DFA_COMM_SIDE_STREAM", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29617")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    cfg = DistriConfig(height=64, width=64, device="cuda:0")
    comm = PatchParallelismCommManager(cfg)
    i0 = comm.register_tensor((1, 8, 16), torch.bfloat16, layer_type="attn")
    i1 = comm.register_tensor((2, 4, 4), torch.bfloat16, layer_type="conv2d")
    comm.create_buffer()
    t0 = torch.randn(1, 8, 16, device="cuda", dtype=torch.bfloat16)
    t1 = torch.randn(2, 4, 4, device="cuda", dtype=torch.bfloat16)
    comm.enqueue(i0, t0)
    comm.enqueue(i1, t1)
    comm.communicate()
    comm.wait(i0)
    comm.wait(i1)
    got = comm.get_buffer_list(i0)[0].view(1, 8, 16)
    assert torch.equal(got, t0)
    comm.clear()
    assert comm.stats["gathers"] >= 1 and len(comm.stats["gather_ms"]) >= 1

    # RCCL collective captured inside a hipGraph (reference feature:
    # NCCL-inside-CUDA-graph, /root/reference/distrifuser/pipelines.py:147-165).
    # MEASURED ON THIS STACK: the capture attempt HANGS (ROCm 7.2 / RCCL,
    # 2026-09) — run killed at the 400 s timeout. This is the go/no-go
    # evidence for keeping multi-rank hipGraphs OFF by default (a hang
    # cannot be caught by the capture-consensus fallback). Re-enable with
    # DFA_TEST_RCCL_GRAPH=1 to re-probe on newer stacks.
    if os.environ.get("DFA_TEST_RCCL_GRAPH", "0") != "1":
        return
    g = torch.cuda.CUDAGraph()
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        comm.enqueue(i0, t0)
        comm.enqueue(i1, t1)
        comm.communicate()
        comm.wait(i0)
        comm.wait(i1)
    comm.handles = [None] * len(comm.handles)
    comm.idx_queue = []
    t0.fill_(2.0)
    g.replay()
    torch.cuda.synchronize()
    got = comm.get_buffer_list(i0)[0].view(1, 8, 16)
    assert torch.equal(got, t0)
