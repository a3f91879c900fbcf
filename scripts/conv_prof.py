"""rocprofv3 PMC target: conv3x3 kernel in a loop at one hot shape."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from distrifuser_amd import ops

cin, cout, h, w, s = (int(a) for a in sys.argv[1:6]) if len(sys.argv) > 5 else (1280, 1280, 120, 120, 1)
x = torch.randn(2, cin, h, w, device="cuda", dtype=torch.bfloat16) * 0.5
wt = torch.randn(cout, cin, 3, 3, device="cuda", dtype=torch.bfloat16) * (cin * 9) ** -0.5
pk = ops.pack_conv3x3_weight(wt)
for _ in range(30):
    ops.conv3x3_halo(x, wt, None, s, packed=pk)
torch.cuda.synchronize()
print("done")
