"""rocprofv3 PMC target: flash attention at one shape in a loop."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from distrifuser_amd import ops

h, l = (int(a) for a in sys.argv[1:3]) if len(sys.argv) > 2 else (10, 57600)
q = torch.randn(1, h, l, 64, device="cuda", dtype=torch.bfloat16)
k = torch.randn(1, h, l, 64, device="cuda", dtype=torch.bfloat16)
v = torch.randn(1, h, l, 64, device="cuda", dtype=torch.bfloat16)
for _ in range(8):
    ops.hip_ext().flash_attention(q, k, v)
torch.cuda.synchronize()
print("done")
