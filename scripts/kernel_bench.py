"""Quick on-box kernel A/B: our gfx950 kernels vs the PyTorch-ROCm baselines.

Run on an MI355X: python scripts/kernel_bench.py > gpurun_out/kernel_bench.json
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import json
import sys
import time

import torch
import torch.nn.functional as F


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000  # ms


def main():
    assert torch.cuda.is_available()
    from distrifuser_amd import ops

    dev = "cuda:0"
    results = []

    # ---- flash attention at SDXL shapes (B=1 CFG-split branch) ----
    #   (heads, L): 1024^2 -> (10, 4096) and (20, 1024)
    #   3840^2 -> (10, 57600) and (20, 14400)
    for h, l in [(20, 1024), (10, 4096), (20, 14400), (10, 57600)]:
        q = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
        k = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
        v = torch.randn(1, h, l, 64, device=dev, dtype=torch.bfloat16)
        ms_ours = timeit(lambda: ops.hip_ext().flash_attention(q, k, v))
        ms_sdpa = timeit(lambda: F.scaled_dot_product_attention(q, k, v))
        flops = 4.0 * l * l * 64 * h
        results.append({
            "op": "flash_attention", "heads": h, "L": l,
            "ms_ours": ms_ours, "ms_sdpa": ms_sdpa,
            "tflops_ours": flops / ms_ours / 1e9,
            "tflops_sdpa": flops / ms_sdpa / 1e9,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- displaced-patch shape: local Q x full stale KV (8 chunks) ----
    for h, lq, n in [(10, 7200, 8), (20, 3600, 4)]:
        lkv = lq * n
        inner = h * 64
        q = torch.randn(1, h, lq, 64, device=dev, dtype=torch.bfloat16)
        slot = lq * 2 * inner
        buf = torch.randn(n, slot + 64, device=dev, dtype=torch.bfloat16)
        kv = buf[:, :slot].view(n, 1, lq, 2 * inner)
        k5 = kv[..., :inner].unflatten(-1, (h, 64)).permute(1, 3, 0, 2, 4)
        v5 = kv[..., inner:].unflatten(-1, (h, 64)).permute(1, 3, 0, 2, 4)
        ms_ours = timeit(lambda: ops.hip_ext().flash_attention(q, k5, v5))
        kc = kv.permute(1, 0, 2, 3).reshape(1, lkv, 2 * inner)
        kcat = kc[..., :inner].view(1, lkv, h, 64).transpose(1, 2).contiguous()
        vcat = kc[..., inner:].view(1, lkv, h, 64).transpose(1, 2).contiguous()
        ms_sdpa = timeit(lambda: F.scaled_dot_product_attention(q, kcat, vcat))
        flops = 4.0 * lq * lkv * 64 * h
        results.append({
            "op": "flash_attention_stale_chunked", "heads": h, "Lq": lq, "Lkv": lkv,
            "n_chunks": n, "ms_ours": ms_ours, "ms_sdpa_precat": ms_sdpa,
            "tflops_ours": flops / ms_ours / 1e9,
            "tflops_sdpa_precat": flops / ms_sdpa / 1e9,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- implicit-GEMM conv3x3 at SDXL 3840^2 shapes (2-sample CFG batch) ----
    torch.backends.cudnn.benchmark = True
    conv_shapes = [
        (4, 320, 480, 480, 1),      # conv_in
        (320, 320, 480, 480, 1),    # down0 ResBlock
        (320, 320, 240, 240, 2),    # downsample
        (640, 640, 240, 240, 1),
        (1280, 1280, 120, 120, 1),  # mid/up ResBlocks
        (2560, 1280, 120, 120, 1),  # up-block concat conv
        (960, 320, 480, 480, 1),
    ]
    for cin, cout, h, w, s in conv_shapes:
        x = torch.randn(2, cin, h, w, device=dev, dtype=torch.bfloat16) * 0.5
        wt = torch.randn(cout, cin, 3, 3, device=dev, dtype=torch.bfloat16) * (cin * 9) ** -0.5
        bias = torch.randn(cout, device=dev, dtype=torch.bfloat16)
        packed = ops.pack_conv3x3_weight(wt)
        ms_ours = timeit(lambda: ops.conv3x3_halo(x, wt, bias, s, packed=packed))
        ms_ref = timeit(lambda: F.conv2d(x, wt, bias, stride=s, padding=1))
        ho, wo = (h - 1) // s + 1, (w - 1) // s + 1
        flops = 2.0 * 2 * cout * ho * wo * cin * 9
        results.append({
            "op": "conv3x3", "cin": cin, "cout": cout, "h": h, "w": w, "stride": s,
            "ms_ours": ms_ours, "ms_miopen": ms_ref,
            "tflops_ours": flops / ms_ours / 1e9,
            "tflops_miopen": flops / ms_ref / 1e9,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- fused GN+SiLU at SDXL shapes ----
    for c, hw in [(320, 480), (640, 240), (1280, 120), (320, 128), (640, 64)]:
        x = torch.randn(2, c, hw, hw, device=dev, dtype=torch.bfloat16)
        w = torch.randn(c, device=dev, dtype=torch.bfloat16)
        b = torch.randn(c, device=dev, dtype=torch.bfloat16)
        ms_ours = timeit(lambda: ops.group_norm_silu(x, 32, w, b, 1e-5, silu=True))
        ms_ref = timeit(lambda: F.silu(F.group_norm(x, 32, w, b, 1e-5)))
        gb = x.numel() * 2 * 2 / 1e9  # read + write
        results.append({
            "op": "gn_silu", "C": c, "HW": hw,
            "ms_ours": ms_ours, "ms_torch": ms_ref,
            "tbps_ours": gb / ms_ours, "tbps_torch": gb / ms_ref,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- GEGLU ----
    for rows, inner in [(4096, 2560), (14400, 5120), (57600, 2560)]:
        x = torch.randn(1, rows, 2 * inner, device=dev, dtype=torch.bfloat16)
        ms_ours = timeit(lambda: ops.geglu(x))
        def ref():
            a, g = x.chunk(2, dim=-1)
            return a * F.gelu(g)
        ms_ref = timeit(ref)
        gb = x.numel() * 2 * 1.5 / 1e9
        results.append({
            "op": "geglu", "rows": rows, "inner": inner,
            "ms_ours": ms_ours, "ms_torch": ms_ref,
            "tbps_ours": gb / ms_ours, "tbps_torch": gb / ms_ref,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- fused (residual+)LayerNorm at SDXL transformer shapes ----
    for rows, c in [(57600, 640), (14400, 1280), (4096, 640)]:
        x = torch.randn(1, rows, c, device=dev, dtype=torch.bfloat16)
        r = torch.randn(1, rows, c, device=dev, dtype=torch.bfloat16)
        w = torch.randn(c, device=dev, dtype=torch.bfloat16)
        b = torch.randn(c, device=dev, dtype=torch.bfloat16)
        ms_ours = timeit(lambda: ops.hip_ext().add_layer_norm(x, r, w, b, 1e-5))
        def ref():
            s2 = x + r
            return s2, F.layer_norm(s2, (c,), w, b, 1e-5)
        ms_ref = timeit(ref)
        gb = x.numel() * 2 * 4 / 1e9  # 2 reads + 2 writes
        results.append({
            "op": "add_layer_norm", "rows": rows, "C": c,
            "ms_ours": ms_ours, "ms_torch": ms_ref,
            "tbps_ours": gb / ms_ours, "tbps_torch": gb / ms_ref,
        })
        print(json.dumps(results[-1]), flush=True)

    # ---- VAE 512-dim single-head mid attention ----
    for l in (4096, 16384):
        q = torch.randn(1, l, 512, device=dev, dtype=torch.bfloat16)
        k = torch.randn(1, l, 512, device=dev, dtype=torch.bfloat16)
        v = torch.randn(1, l, 512, device=dev, dtype=torch.bfloat16)
        ms_ours = timeit(lambda: ops.hip_ext().vae_attention(q, k, v), iters=5, warmup=2)
        flops = 4.0 * l * l * 512
        results.append({
            "op": "vae_attention", "L": l, "ms_ours": ms_ours,
            "tflops_ours": flops / ms_ours / 1e9,
        })
        print(json.dumps(results[-1]), flush=True)

    with open("gpurun_out/kernel_bench.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    sys.exit(main())
