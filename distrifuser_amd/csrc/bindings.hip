// Python bindings for the distrifuser_amd gfx950 kernels (module _C).
// Pure HIP + ATen/hip — no CUDA-compat layer; compiled by hipcc via
// torch.utils.cpp_extension with PYTORCH_ROCM_ARCH=gfx950.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include "kernels.h"

namespace {

int dtype_of(const at::Tensor& t) {
    switch (t.scalar_type()) {
        case at::kBFloat16: return DFA_BF16;
        case at::kHalf: return DFA_F16;
        case at::kFloat: return DFA_F32;
        default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
    }
}

hipStream_t cur_stream() { return at::hip::getCurrentHIPStream().stream(); }

at::Tensor group_norm_stats(const at::Tensor& x_, int64_t num_groups) {
    TORCH_CHECK(x_.is_cuda() && x_.dim() == 4, "x must be CUDA [N,C,H,W]");
    auto x = x_.contiguous();
    const int64_t n = x.size(0), c = x.size(1), hw = x.size(2) * x.size(3);
    TORCH_CHECK(c % num_groups == 0);
    const int64_t group_len = (c / num_groups) * hw;
    const int ngt = (int)(n * num_groups);
    auto partial = at::zeros({2, (long)ngt}, x.options().dtype(at::kFloat));
    launch_gn_stats_partial(x.data_ptr(), partial.data_ptr<float>(), group_len, ngt,
                            dtype_of(x), cur_stream());
    auto out = at::empty({2, n, num_groups, 1, 1, 1}, x.options());
    launch_gn_finalize(partial.data_ptr<float>(), out.data_ptr(), 1.0f / (float)group_len, ngt,
                       dtype_of(x), cur_stream());
    return out;
}

at::Tensor group_norm_apply(const at::Tensor& x_, const at::Tensor& mean_,
                            const at::Tensor& meansq_, const c10::optional<at::Tensor>& w_,
                            const c10::optional<at::Tensor>& b_, double eps, bool silu) {
    TORCH_CHECK(x_.is_cuda() && x_.dim() == 4);
    auto x = x_.contiguous();
    const int64_t n = x.size(0), c = x.size(1), hw = x.size(2) * x.size(3);
    auto mean = mean_.to(at::kFloat).contiguous().view({-1});
    auto meansq = meansq_.to(at::kFloat).contiguous().view({-1});
    TORCH_CHECK(mean.numel() % n == 0);
    const int g = (int)(mean.numel() / n);
    TORCH_CHECK(c % g == 0, "channels ", c, " not divisible by groups ", g);
    at::Tensor w, b;
    const void* wp = nullptr;
    const void* bp = nullptr;
    if (w_.has_value()) {
        w = w_->to(x.scalar_type()).contiguous();
        wp = w.data_ptr();
    }
    if (b_.has_value()) {
        b = b_->to(x.scalar_type()).contiguous();
        bp = b.data_ptr();
    }
    auto y = at::empty_like(x);
    launch_gn_apply(x.data_ptr(), y.data_ptr(), mean.data_ptr<float>(), meansq.data_ptr<float>(),
                    wp, bp, (float)eps, hw, (int)c, g, (int)n, silu, dtype_of(x), cur_stream());
    return y;
}

at::Tensor group_norm_silu(const at::Tensor& x_, int64_t num_groups,
                           const c10::optional<at::Tensor>& w_,
                           const c10::optional<at::Tensor>& b_, double eps, bool silu) {
    TORCH_CHECK(x_.is_cuda() && x_.dim() == 4);
    auto x = x_.contiguous();
    const int64_t n = x.size(0), c = x.size(1), hw = x.size(2) * x.size(3);
    TORCH_CHECK(c % num_groups == 0);
    const int64_t group_len = (c / num_groups) * hw;
    const int ngt = (int)(n * num_groups);
    auto partial = at::zeros({2, (long)ngt}, x.options().dtype(at::kFloat));
    launch_gn_stats_partial(x.data_ptr(), partial.data_ptr<float>(), group_len, ngt,
                            dtype_of(x), cur_stream());
    // finalize in fp32 (full precision straight into the apply)
    auto moments = at::empty({2, (long)ngt}, x.options().dtype(at::kFloat));
    launch_gn_finalize(partial.data_ptr<float>(), moments.data_ptr(), 1.0f / (float)group_len,
                       ngt, DFA_F32, cur_stream());
    at::Tensor w, b;
    const void* wp = nullptr;
    const void* bp = nullptr;
    if (w_.has_value()) {
        w = w_->to(x.scalar_type()).contiguous();
        wp = w.data_ptr();
    }
    if (b_.has_value()) {
        b = b_->to(x.scalar_type()).contiguous();
        bp = b.data_ptr();
    }
    auto y = at::empty_like(x);
    launch_gn_apply(x.data_ptr(), y.data_ptr(), moments.data_ptr<float>(),
                    moments.data_ptr<float>() + ngt, wp, bp, (float)eps, hw, (int)c,
                    (int)num_groups, (int)n, silu, dtype_of(x), cur_stream());
    return y;
}

at::Tensor gn_merge_stats(const at::Tensor& buffer, int64_t slot_off, int64_t own,
                          const at::Tensor& fresh_, bool corrected) {
    // buffer: [n_peers, row_stride] (the flat comm buffer); fresh: [2, N, G,...]
    TORCH_CHECK(buffer.is_cuda() && buffer.dim() == 2 && buffer.stride(1) == 1);
    auto fresh = fresh_.contiguous();
    const int ng = (int)(fresh.numel() / 2);
    TORCH_CHECK(fresh.scalar_type() == buffer.scalar_type());
    auto out = at::empty({2, (long)ng}, buffer.options().dtype(at::kFloat));
    launch_gn_merge_stats(buffer.data_ptr(), buffer.stride(0), slot_off,
                          (int)buffer.size(0), (int)own, fresh.data_ptr(),
                          out.data_ptr<float>(), ng, corrected, dtype_of(buffer),
                          cur_stream());
    return out;
}

at::Tensor geglu(const at::Tensor& h_) {
    TORCH_CHECK(h_.is_cuda());
    auto h = h_.contiguous();
    const int64_t inner = h.size(-1) / 2;
    TORCH_CHECK(h.size(-1) % 2 == 0);
    auto sizes = h.sizes().vec();
    sizes.back() = inner;
    auto out = at::empty(sizes, h.options());
    launch_geglu(h.data_ptr(), out.data_ptr(), h.numel() / (2 * inner), inner, dtype_of(h),
                 cur_stream());
    return out;
}

at::Tensor cfg_affine_step(const at::Tensor& noise, const at::Tensor& x, double g, double ca,
                           double cb) {
    // noise: [2, C, H, W] = [uncond; cond]; x: [1, C, H, W]; out = ca*x + cb*eps
    TORCH_CHECK(noise.is_cuda() && noise.size(0) == 2);
    auto n = noise.contiguous();
    auto xc = x.contiguous();
    TORCH_CHECK(xc.numel() * 2 == n.numel());
    auto out = at::empty_like(xc);
    const int64_t total = xc.numel();
    launch_cfg_affine_step(
        n.data_ptr(),
        reinterpret_cast<const char*>(n.data_ptr()) + total * n.element_size(), xc.data_ptr(),
        out.data_ptr(), (float)g, (float)ca, (float)cb, total, dtype_of(xc), cur_stream());
    return out;
}

at::Tensor flash_attention(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v) {
    // q: [B,H,Lq,64]; k/v: [B,H,Lkv,64] or [B,H,NC,LC,64] (stale-KV chunks)
    TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16, "flash_attention: bf16 only");
    const int Dh = (int)q.size(-1);
    TORCH_CHECK(q.dim() == 4 && Dh % 8 == 0 && Dh <= 160,
                "head_dim must be a multiple of 8 and <= 160");
    TORCH_CHECK(q.stride(-1) == 1 && k.stride(-1) == 1 && v.stride(-1) == 1,
                "innermost dim must be contiguous");
    const int B = (int)q.size(0), H = (int)q.size(1);
    const int64_t Lq = q.size(2);

    FlashAttnParams p{};
    p.q = reinterpret_cast<const uint16_t*>(q.data_ptr());
    p.B = B;
    p.H = H;
    p.Lq = Lq;
    p.q_sb = q.stride(0);
    p.q_sh = q.stride(1);
    p.q_sl = q.stride(2);
    p.Dh = Dh;
    p.scale = 1.0f / std::sqrt((float)Dh);

    auto set_kv = [&](const at::Tensor& t, const uint16_t*& ptr, int64_t& sb, int64_t& sh,
                      int64_t& sc, int64_t& sl, int64_t& NC, int64_t& LC) {
        TORCH_CHECK(t.scalar_type() == at::kBFloat16);
        TORCH_CHECK((int)t.size(-1) == Dh, "k/v head_dim mismatch");
        ptr = reinterpret_cast<const uint16_t*>(t.data_ptr());
        sb = t.stride(0);
        sh = t.stride(1);
        if (t.dim() == 4) {
            NC = 1;
            LC = t.size(2);
            sc = 0;
            sl = t.stride(2);
        } else {
            TORCH_CHECK(t.dim() == 5);
            NC = t.size(2);
            LC = t.size(3);
            sc = t.stride(2);
            sl = t.stride(3);
            // V staging reads 8-token windows assuming they never straddle a
            // chunk boundary
            TORCH_CHECK(NC == 1 || LC % 8 == 0, "chunk length must be a multiple of 8");
        }
        TORCH_CHECK(sl % 8 == 0 && sc % 8 == 0 && sh % 8 == 0 && sb % 8 == 0,
                    "KV strides must be 16B-aligned (multiples of 8 elements)");
    };
    int64_t nc2, lc2;
    set_kv(k, p.k, p.k_sb, p.k_sh, p.k_sc, p.k_sl, p.NC, p.LC);
    set_kv(v, p.v, p.v_sb, p.v_sh, p.v_sc, p.v_sl, nc2, lc2);
    TORCH_CHECK(nc2 == p.NC && lc2 == p.LC, "k/v shape mismatch");
    TORCH_CHECK(p.q_sl % 8 == 0 && p.q_sb % 8 == 0 && p.q_sh % 8 == 0);

    auto o = at::empty({(long)B, (long)Lq, (long)H, (long)Dh}, q.options());
    p.o = reinterpret_cast<uint16_t*>(o.data_ptr());
    launch_flash_attention_d64(p, cur_stream());
    return o.permute({0, 2, 1, 3});  // logical [B,H,Lq,64]
}

at::Tensor conv3x3(const at::Tensor& x, const at::Tensor& wp,
                   const c10::optional<at::Tensor>& bias, int64_t cout, int64_t stride,
                   const c10::optional<at::Tensor>& top, const c10::optional<at::Tensor>& bot,
                   const c10::optional<at::Tensor>& residual,
                   const c10::optional<at::Tensor>& bias2) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kBFloat16,
                "x must be CUDA bf16 [B,Cin,H,W]");
    TORCH_CHECK(x.stride(3) == 1 && x.stride(2) == x.size(3), "x rows must be contiguous");
    TORCH_CHECK(wp.is_cuda() && wp.dim() == 5 && wp.is_contiguous() &&
                wp.scalar_type() == at::kBFloat16 && wp.size(0) == 9 &&
                wp.size(3) == 64 && wp.size(4) == 8,
                "wp must be packed [9][KS][CT][64][8] bf16");
    TORCH_CHECK(stride == 1 || stride == 2);
    const int B = (int)x.size(0), Cin = (int)x.size(1), H = (int)x.size(2), W = (int)x.size(3);
    const int Ho = (H - 1) / (int)stride + 1, Wo = (W - 1) / (int)stride + 1;

    Conv3x3Params p{};
    p.x = reinterpret_cast<const uint16_t*>(x.data_ptr());
    p.wp = reinterpret_cast<const uint16_t*>(wp.data_ptr());
    p.B = B;
    p.Cin = Cin;
    p.Cout = (int)cout;
    p.H = H;
    p.W = W;
    p.Ho = Ho;
    p.Wo = Wo;
    p.KS = (int)wp.size(1);
    p.CT = (int)wp.size(2);
    TORCH_CHECK(p.KS * 16 >= Cin && p.CT * 32 >= cout, "packed weight too small");
    p.x_sb = x.stride(0);
    p.x_sc = x.stride(1);
    if (const char* dbg = getenv("DFA_CONV_DEBUG")) p.debug = atoi(dbg);
    at::Tensor bias_c;
    if (bias.has_value()) {
        bias_c = bias->to(at::kBFloat16).contiguous();
        TORCH_CHECK(bias_c.numel() == cout);
        p.bias = reinterpret_cast<const uint16_t*>(bias_c.data_ptr());
    }
    auto set_halo = [&](const c10::optional<at::Tensor>& h, const uint16_t*& ptr, int64_t& sb,
                        int64_t& sc) {
        if (!h.has_value()) return;
        const at::Tensor& t = *h;
        TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 && t.stride(-1) == 1 &&
                    t.size(-1) == W && t.size(1) == Cin,
                    "halo must be bf16 [B,Cin,(1,)W] with contiguous rows");
        ptr = reinterpret_cast<const uint16_t*>(t.data_ptr());
        sb = t.stride(0);
        sc = t.stride(1);
    };
    set_halo(top, p.top, p.t_sb, p.t_sc);
    set_halo(bot, p.bot, p.b_sb, p.b_sc);
    at::Tensor b2_c;
    if (bias2.has_value()) {
        b2_c = bias2->to(at::kBFloat16).contiguous();
        TORCH_CHECK(b2_c.numel() == (int64_t)B * cout, "bias2 must be [B, Cout]");
        p.bias2 = reinterpret_cast<const uint16_t*>(b2_c.data_ptr());
    }
    at::Tensor res_c;
    if (residual.has_value()) {
        res_c = residual->contiguous();
        TORCH_CHECK(res_c.scalar_type() == at::kBFloat16 &&
                    res_c.sizes() == (at::IntArrayRef{(long)B, (long)cout, (long)Ho, (long)Wo}),
                    "residual must be bf16 [B,Cout,Ho,Wo]");
        p.residual = reinterpret_cast<const uint16_t*>(res_c.data_ptr());
    }

    auto o = at::empty({(long)B, (long)cout, (long)Ho, (long)Wo}, x.options());
    p.o = reinterpret_cast<uint16_t*>(o.data_ptr());
    launch_conv3x3(p, (int)stride, cur_stream());
    return o;
}

std::vector<at::Tensor> cfg_dpm_step(const at::Tensor& noise, const at::Tensor& x,
                                     const c10::optional<at::Tensor>& x0_prev, double g,
                                     double ca, double cb, double cc, double cx, double ce) {
    TORCH_CHECK(noise.is_cuda() && noise.size(0) == 2);
    auto n = noise.contiguous();
    auto xc = x.contiguous();
    TORCH_CHECK(xc.numel() * 2 == n.numel());
    const void* pp = nullptr;
    at::Tensor p;
    if (x0_prev.has_value()) {
        p = x0_prev->contiguous();
        TORCH_CHECK(p.numel() == xc.numel() && p.scalar_type() == at::kFloat,
                    "x0_prev must be the fp32 state from the previous cfg_dpm_step");
        pp = p.data_ptr();
    }
    auto out = at::empty_like(xc);
    auto x0 = at::empty_like(xc, xc.options().dtype(at::kFloat));
    const int64_t total = xc.numel();
    launch_cfg_dpm_step(n.data_ptr(),
                        reinterpret_cast<const char*>(n.data_ptr()) + total * n.element_size(),
                        xc.data_ptr(), pp, out.data_ptr(), x0.data_ptr(), (float)g, (float)ca,
                        (float)cb, (float)cc, (float)cx, (float)ce, total, dtype_of(xc),
                        cur_stream());
    return {out, x0};
}

at::Tensor layer_norm(const at::Tensor& x_, const at::Tensor& w_, const at::Tensor& b_,
                      double eps) {
    TORCH_CHECK(x_.is_cuda() && x_.scalar_type() == at::kBFloat16);
    auto x = x_.contiguous();
    const int C = (int)x.size(-1);
    TORCH_CHECK(C % 8 == 0 && C <= 2048, "layer_norm kernel: C % 8 == 0 and C <= 2048");
    auto w = w_.to(at::kBFloat16).contiguous();
    auto b = b_.to(at::kBFloat16).contiguous();
    auto y = at::empty_like(x);
    launch_layer_norm(x.data_ptr(), nullptr, y.data_ptr(), nullptr, w.data_ptr(), b.data_ptr(),
                      (float)eps, x.numel() / C, C, cur_stream());
    return y;
}

std::vector<at::Tensor> add_layer_norm(const at::Tensor& x_, const at::Tensor& res_,
                                       const at::Tensor& w_, const at::Tensor& b_, double eps) {
    TORCH_CHECK(x_.is_cuda() && x_.scalar_type() == at::kBFloat16);
    auto x = x_.contiguous();
    auto res = res_.contiguous();
    TORCH_CHECK(res.sizes() == x.sizes() && res.scalar_type() == at::kBFloat16);
    const int C = (int)x.size(-1);
    TORCH_CHECK(C % 8 == 0 && C <= 2048, "add_layer_norm kernel: C % 8 == 0 and C <= 2048");
    auto w = w_.to(at::kBFloat16).contiguous();
    auto b = b_.to(at::kBFloat16).contiguous();
    auto y = at::empty_like(x);
    auto sum = at::empty_like(x);
    launch_layer_norm(x.data_ptr(), res.data_ptr(), y.data_ptr(), sum.data_ptr(), w.data_ptr(),
                      b.data_ptr(), (float)eps, x.numel() / C, C, cur_stream());
    return {sum, y};
}

at::Tensor vae_attention(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v) {
    TORCH_CHECK(q.is_cuda() && q.dim() == 3 && q.size(-1) == 512 &&
                q.scalar_type() == at::kBFloat16,
                "vae_attention: bf16 [B, L, 512] only");
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    TORCH_CHECK(kc.sizes() == qc.sizes() && vc.sizes() == qc.sizes());
    VaeAttnParams p{};
    p.q = reinterpret_cast<const uint16_t*>(qc.data_ptr());
    p.k = reinterpret_cast<const uint16_t*>(kc.data_ptr());
    p.v = reinterpret_cast<const uint16_t*>(vc.data_ptr());
    p.B = (int)qc.size(0);
    p.L = qc.size(1);
    p.sb = qc.stride(0);
    p.scale = 1.0f / std::sqrt(512.0f);
    auto o = at::empty_like(qc);
    p.o = reinterpret_cast<uint16_t*>(o.data_ptr());
    launch_vae_attention(p, cur_stream());
    return o;
}

std::vector<at::Tensor> mfma_probe(const at::Tensor& a, const at::Tensor& b);

}  // namespace

// defined in attention.hip
void launch_mfma_probe(const float* a, const float* b, float* d, hipStream_t stream);
void launch_mfma_probe32(const float* a, const float* b, float* d, hipStream_t stream);

namespace {
std::vector<at::Tensor> mfma_probe(const at::Tensor& a, const at::Tensor& b) {
    TORCH_CHECK(a.is_cuda() && a.sizes() == (at::IntArrayRef{64, 8}));
    auto af = a.to(at::kFloat).contiguous(), bf = b.to(at::kFloat).contiguous();
    auto d = at::zeros({64, 4}, af.options());
    launch_mfma_probe(af.data_ptr<float>(), bf.data_ptr<float>(), d.data_ptr<float>(),
                      cur_stream());
    return {d};
}

std::vector<at::Tensor> mfma_probe32(const at::Tensor& a, const at::Tensor& b) {
    TORCH_CHECK(a.is_cuda() && a.sizes() == (at::IntArrayRef{64, 8}));
    auto af = a.to(at::kFloat).contiguous(), bf = b.to(at::kFloat).contiguous();
    auto d = at::zeros({64, 16}, af.options());
    launch_mfma_probe32(af.data_ptr<float>(), bf.data_ptr<float>(), d.data_ptr<float>(),
                        cur_stream());
    return {d};
}
}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("group_norm_stats", &group_norm_stats, "per-group moments [2,N,G,1,1,1]");
    m.def("group_norm_apply", &group_norm_apply, "normalize+affine(+SiLU) from given moments");
    m.def("group_norm_silu", &group_norm_silu, "fused GroupNorm(+SiLU)");
    m.def("geglu", &geglu, "a * gelu(gate) over last-dim halves");
    m.def("gn_merge_stats", &gn_merge_stats,
          "merge stale peer GN moments with fresh local ones (one launch)");
    m.def("cfg_affine_step", &cfg_affine_step, "fused CFG combine + affine scheduler update");
    m.def("flash_attention", &flash_attention, "bf16 d64 flash attention (chunked stale KV)");
    m.def("cfg_dpm_step", &cfg_dpm_step, "fused CFG + DPM-Solver++(2M) update -> (prev, x0)");
    m.def("layer_norm", &layer_norm, "bf16 fused LayerNorm");
    m.def("add_layer_norm", &add_layer_norm, "bf16 fused residual-add + LayerNorm -> (sum, y)");
    m.def("vae_attention", &vae_attention, "bf16 single-head d=512 VAE mid attention");
    m.def("conv3x3", &conv3x3,
          "bf16 implicit-GEMM 3x3 conv, stride 1/2, in-place halo rows");
    m.def("mfma_probe", &mfma_probe, "dump mfma_f32_16x16x32_bf16 fragment mapping");
    m.def("mfma_probe32", &mfma_probe32, "dump mfma_f32_32x32x16_bf16 fragment mapping");
}
