"""Minimal SD1.5 usage example (parity with the reference's sd_example.py)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import torch

from distrifuser_amd import DistriConfig, DistriSDPipeline

distri_config = DistriConfig(height=512, width=512, mode="stale_gn")
pipeline = DistriSDPipeline.from_pretrained(
    distri_config,
    torch_dtype=torch.bfloat16 if torch.cuda.is_available() else torch.float32,
)

pipeline.set_progress_bar_config(disable=distri_config.rank != 0)
image = pipeline(
    prompt="A kitten sitting in a teacup, studio lighting",
    generator=torch.Generator().manual_seed(233),
    output_type="pil",
)
if distri_config.rank == 0:
    img = image[0]
    try:
        img.save("kitten.png")
    except AttributeError:
        import numpy as np

        np.save("kitten.npy", img)
    print("saved kitten image")
