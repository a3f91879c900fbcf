"""Eager op reference implementations vs plain PyTorch composites."""

import torch
import torch.nn.functional as F

from distrifuser_amd.ops import eager


def test_group_norm_stats_and_apply_match_group_norm():
    torch.manual_seed(0)
    x = torch.randn(2, 16, 9, 7)
    g = 4
    w = torch.randn(16)
    b = torch.randn(16)
    stats = eager.group_norm_stats(x, g)
    assert stats.shape == (2, 2, g, 1, 1, 1)
    out = eager.group_norm_apply(x, stats[0], stats[1], w, b, eps=1e-5)
    ref = F.group_norm(x, g, w, b, eps=1e-5)
    assert torch.allclose(out, ref, atol=1e-5)


def test_group_norm_silu_fused():
    torch.manual_seed(0)
    x = torch.randn(1, 8, 4, 4)
    w, b = torch.randn(8), torch.randn(8)
    out = eager.group_norm_silu(x, 2, w, b, 1e-6, silu=True)
    ref = F.silu(F.group_norm(x, 2, w, b, 1e-6))
    assert torch.allclose(out, ref, atol=1e-5)


def test_group_norm_apply_silu_matches():
    torch.manual_seed(1)
    x = torch.randn(2, 8, 5, 5)
    stats = eager.group_norm_stats(x, 4)
    out = eager.group_norm_apply(x, stats[0], stats[1], None, None, 1e-5, silu=True)
    ref = F.silu(F.group_norm(x, 4, eps=1e-5))
    assert torch.allclose(out, ref, atol=1e-5)


def test_geglu():
    torch.manual_seed(0)
    h = torch.randn(2, 5, 12)
    a, b = h.chunk(2, dim=-1)
    assert torch.allclose(eager.geglu(h), a * F.gelu(b))


def test_flash_attention_matches_manual_softmax():
    torch.manual_seed(0)
    q = torch.randn(1, 2, 6, 8)
    k = torch.randn(1, 2, 10, 8)
    v = torch.randn(1, 2, 10, 8)
    out = eager.flash_attention(q, k, v)
    scores = (q @ k.transpose(-1, -2)) / (8**0.5)
    ref = scores.softmax(dim=-1) @ v
    assert torch.allclose(out, ref, atol=1e-5)


def test_group_norm_stats_distributed_average_equals_full():
    """Averaging per-patch moments over an H-split == full-tensor moments."""
    torch.manual_seed(0)
    x = torch.randn(1, 8, 8, 6)
    g = 2
    full = eager.group_norm_stats(x, g)
    top = eager.group_norm_stats(x[:, :, :4], g)
    bot = eager.group_norm_stats(x[:, :, 4:], g)
    avg = (top + bot) / 2
    assert torch.allclose(avg, full, atol=1e-6)


def test_conv3x3_halo_oracle_matches_full_conv():
    """conv3x3_halo with neighbour halo rows == conv over the full tensor."""
    import torch.nn.functional as F

    from distrifuser_amd.ops import eager

    torch.manual_seed(0)
    full = torch.randn(2, 16, 12, 20)
    w = torch.randn(24, 16, 3, 3) * 0.1
    b = torch.randn(24)
    for stride in (1, 2):
        ref = F.conv2d(full, w, b, stride=stride, padding=1)
        # split rows 4..8 as the "local band"; halos from the neighbours
        x = full[:, :, 4:8]
        top = full[:, :, 3:4]
        bot = full[:, :, 8:9]
        got = eager.conv3x3_halo(x, w, b, stride, top, bot if stride == 1 else None)
        band = ref[:, :, 4 // stride : 8 // stride]
        assert torch.allclose(got, band, atol=1e-4), (stride, (got - band).abs().max())
    # border band: no top halo -> zero padding must match the full conv edge
    x0 = full[:, :, :4]
    got0 = eager.conv3x3_halo(x0, w, b, 1, None, full[:, :, 4:5])
    assert torch.allclose(got0, F.conv2d(full, w, b, padding=1)[:, :, :4], atol=1e-4)


def test_transformer_pending_residual_chain():
    """The fused add+LN residual threading equals the naive block math."""
    import torch

    from distrifuser_amd.models.layers import LayerFactory
    from distrifuser_amd.models.transformer import BasicTransformerBlock
    from distrifuser_amd.parallel.state import ParallelState
    from distrifuser_amd.utils.config import DistriConfig

    torch.manual_seed(0)
    cfg = DistriConfig(height=64, width=64, use_cuda_graph=False, device="cpu")
    fac = LayerFactory(ParallelState(cfg))
    blocks = [BasicTransformerBlock(32, 2, 16, 16, factory=fac).eval() for _ in range(3)]
    x = torch.randn(1, 24, 32)
    ehs = torch.randn(1, 7, 16)

    with torch.no_grad():
        # naive composition
        def naive(b, x):
            import torch.nn.functional as F

            x = x + b.attn1(F.layer_norm(x, (32,), b.norm1.weight, b.norm1.bias, b.norm1.eps))
            h = F.layer_norm(x, (32,), b.norm2.weight, b.norm2.bias, b.norm2.eps)
            x = x + b.attn2(h, ehs)
            return x + b.ff(F.layer_norm(x, (32,), b.norm3.weight, b.norm3.bias, b.norm3.eps))

        ref = x
        for b in blocks:
            ref = naive(b, ref)

        got, pending = x, None
        for b in blocks:
            got, pending = b(got, ehs, pending)
        got = got + pending
    assert torch.allclose(got, ref, atol=1e-5)


def test_native_conv2d_bias2_residual_cpu():
    import torch

    from distrifuser_amd.ops import NativeConv2d

    torch.manual_seed(1)
    m = NativeConv2d(8, 12, 3, padding=1).eval()
    x = torch.randn(2, 8, 6, 10)
    b2 = torch.randn(2, 12)
    res = torch.randn(2, 12, 6, 10)
    with torch.no_grad():
        got = m(x, bias2=b2, residual=res)
        ref = torch.nn.functional.conv2d(x, m.weight, m.bias, padding=1) \
            + b2[:, :, None, None] + res
    assert torch.allclose(got, ref, atol=1e-5)
