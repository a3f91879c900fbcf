"""Minimal usage example (parity with the reference's sdxl_example.py):

  torchrun --nproc_per_node=2 scripts/sdxl_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import torch

from distrifuser_amd import DistriConfig, DistriSDXLPipeline

distri_config = DistriConfig(height=1024, width=1024)
pipeline = DistriSDXLPipeline.from_pretrained(
    distri_config,
    torch_dtype=torch.bfloat16 if torch.cuda.is_available() else torch.float32,
)

pipeline.set_progress_bar_config(disable=distri_config.rank != 0)
image = pipeline(
    prompt="Astronaut in a jungle, cold color palette, muted colors, detailed, 8k",
    generator=torch.Generator().manual_seed(233),
    output_type="pil",
)
if distri_config.rank == 0:
    img = image[0]
    try:
        img.save("astronaut.png")
    except AttributeError:
        import numpy as np

        np.save("astronaut.npy", img)
    print("saved astronaut image")
