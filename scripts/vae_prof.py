import sys

import torch

sys.path.insert(0, "/root/repo")
from distrifuser_amd.models.vae import SDXL_VAE, VAEDecoder

vae = VAEDecoder(SDXL_VAE).to("cuda", torch.bfloat16).eval()
vae.enable_tiling()
with torch.no_grad():
    z = torch.randn(1, 4, 256, 256, device="cuda", dtype=torch.bfloat16)
    vae(z)
    torch.cuda.synchronize()
print("done")
