"""SD2.1 model family (reference README claims SD1.4/2 support via
DistriSDPipeline; here it is the sd21 preset)."""

import torch

from distrifuser_amd import DistriConfig
from distrifuser_amd.models import DistriUNet
from distrifuser_amd.models.clip import OPEN_CLIP_VIT_H, CLIPTextEncoder
from distrifuser_amd.models.unet import SD21_UNET


def test_sd21_unet_forward_cpu():
    cfg = DistriConfig(height=256, width=256, do_classifier_free_guidance=False,
                       use_cuda_graph=False, device="cpu")
    torch.manual_seed(0)
    unet = DistriUNet(SD21_UNET, cfg).eval()
    n_params = sum(p.numel() for p in unet.parameters())
    assert 8.0e8 < n_params < 9.5e8, n_params  # ~865M like the checkpoint
    x = torch.randn(1, 4, 32, 32)
    ehs = torch.randn(1, 77, 1024)
    with torch.no_grad():
        unet.set_counter(0)
        out = unet(x, 500.0, ehs, None)
    assert out.shape == (1, 4, 32, 32)
    assert torch.isfinite(out).all()


def test_openclip_vith_shapes():
    torch.manual_seed(0)
    enc = CLIPTextEncoder(OPEN_CLIP_VIT_H)
    ids = torch.randint(0, 1000, (2, 77))
    with torch.no_grad():
        hidden, pooled = enc(ids, hidden_state_index=-1)
    assert hidden.shape == (2, 77, 1024)
    assert pooled is None  # no projection head in the SD2 text encoder
