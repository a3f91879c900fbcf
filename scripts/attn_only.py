"""Run only the flash-attention kernel at the hot shapes (for rocprofv3 PMC)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from distrifuser_amd import ops


def main():
    dev = "cuda:0"
    h, l = 10, 57600
    q = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
    k = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
    v = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
    for _ in range(int(os.environ.get("ATTN_ITERS", "5"))):
        ops.hip_ext().flash_attention(q, k, v)
    torch.cuda.synchronize()
    print("done")


if __name__ == "__main__":
    main()
