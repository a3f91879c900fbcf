// Host-side launcher declarations (implemented in the .hip translation units).
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

enum DfaDtype : int { DFA_BF16 = 0, DFA_F16 = 1, DFA_F32 = 2 };

// ---- GroupNorm family ------------------------------------------------------
// partial: fp32 [2, N*G] zero-initialized; accumulates sum and sumsq.
void launch_gn_stats_partial(const void* x, float* partial, int64_t group_len,
                             int n_groups_total, int dtype, hipStream_t stream);
// out: [2, N*G] in `dtype`; mean = partial/count.
void launch_gn_finalize(const float* partial, void* out, float inv_count,
                        int n_groups_total, int dtype, hipStream_t stream);
// normalize+affine(+SiLU) with externally supplied fp32 moments [N*G] each.
void launch_gn_apply(const void* x, void* y, const float* mean, const float* meansq,
                     const void* weight, const void* bias, float eps, int64_t hw,
                     int C, int G, int N, bool silu, int dtype, hipStream_t stream);

// ---- displaced-GN stat merge ------------------------------------------------
// Merge per-peer stale GroupNorm moments with this rank's fresh ones in ONE
// launch (replaces a ~10-kernel torch composition per GN layer):
//   corrected: full = mean(stale) + (fresh - stale_own)   [+ neg-var guard]
//   stale:     full = mean(stale with own slot := fresh)
// Also stages `fresh` into the rank's buffer slot (the enqueue needs it).
// buffer: bf16, n_peers rows of row_stride elements, moment block at
// buffer[row * row_stride + slot_off .. + 2*ng); fresh: bf16 [2*ng];
// out: fp32 [2*ng].
void launch_gn_merge_stats(void* buffer, int64_t row_stride, int64_t slot_off,
                           int n_peers, int own, const void* fresh, float* out,
                           int ng, bool corrected, int dtype, hipStream_t stream);

// ---- GEGLU ------------------------------------------------------------------
// in: [rows, 2*inner]; out: [rows, inner]; out = a * gelu(gate).
void launch_geglu(const void* in, void* out, int64_t rows, int64_t inner, int dtype,
                  hipStream_t stream);

// ---- fused CFG + scheduler step (affine: out = ca*x + cb*eps) ---------------
void launch_cfg_affine_step(const void* noise_u, const void* noise_c, const void* x, void* out,
                            float g, float ca, float cb, int64_t total, int dtype,
                            hipStream_t stream);

// ---- Implicit-GEMM 3x3 conv (bf16, NCHW, stride 1/2, pad 1) -----------------
// x: interior [B][Cin][H][W] (row-contiguous; x_sc = channel stride).
// top/bot: optional single halo rows [B][Cin][1][W] (e.g. comm-buffer views);
// null => zero padding at that border. Input row y_in=-1 reads top, y_in=H
// reads bot. o: [B][Cout][Ho][Wo] contiguous output.
// wp: weights prepacked in per-lane A-fragment order [9][KS][CT][64][8]
// (cout = ct*32 + (lane&31), cin = ks*16 + (lane>>5)*8 + j), KS*16 and CT*32
// zero-padded to multiples of 64/32. bias: bf16 [Cout] or null.
struct Conv3x3Params {
    const uint16_t* x;
    const uint16_t* top;
    const uint16_t* bot;
    const uint16_t* wp;
    const uint16_t* bias;
    const uint16_t* residual;  // optional [B][Cout][Ho][Wo]: o += residual
    const uint16_t* bias2;     // optional [B][Cout]: o += bias2 (time-emb add)
    uint16_t* o;
    int B, Cin, Cout, H, W, Ho, Wo;
    int KS, CT;
    int debug;  // 1 = skip input staging, 2 = skip MFMA (phase-cost probes)
    int64_t x_sb, x_sc;
    int64_t t_sb, t_sc;
    int64_t b_sb, b_sc;
};
void launch_conv3x3(const Conv3x3Params& p, int stride, hipStream_t stream);

// ---- fused CFG + DPM-Solver++(2M) step --------------------------------------
// out = ca*x + cb*eps + cc*x0_prev (x0_prev may be null); x0_out = cx*x + ce*eps
void launch_cfg_dpm_step(const void* nu, const void* nc, const void* x, const void* x0_prev,
                         void* out, void* x0_out, float g, float ca, float cb, float cc,
                         float cx, float ce, int64_t total, int dtype, hipStream_t stream);

// ---- fused (residual +) LayerNorm (bf16, C <= 2048, C % 8 == 0) -------------
// y = LN(x [+ res]); when res != null and sum_out != null, x+res is also
// written to sum_out (the transformer residual add fused away).
void launch_layer_norm(const void* x, const void* res, void* y, void* sum_out,
                       const void* w, const void* b, float eps, int64_t rows, int C,
                       hipStream_t stream);

// ---- VAE mid-block attention (bf16, single head, head_dim 512) --------------
// q/k/v/o: [B][L][512] row-contiguous (sb = batch stride in elements).
struct VaeAttnParams {
    const uint16_t* q;
    const uint16_t* k;
    const uint16_t* v;
    uint16_t* o;
    int B;
    int64_t L, sb;
    float scale;
};
void launch_vae_attention(const VaeAttnParams& p, hipStream_t stream);

// ---- Flash attention (bf16, SD-family head dims) ----------------------------
// q: logical [B, H, Lq, 64]; k/v: logical [B, H, NC, LC, 64] (NC stale-KV
// chunks of LC tokens; NC=1 for plain attention). All strides in ELEMENTS,
// innermost head_dim contiguous. o: [B, Lq, H, 64] contiguous output.
struct FlashAttnParams {
    const uint16_t* q;
    const uint16_t* k;
    const uint16_t* v;
    uint16_t* o;
    int B, H;
    int Dh;  // real head_dim (kernel pads to a multiple of 32)
    int64_t Lq, NC, LC;  // Lkv = NC * LC
    int64_t q_sb, q_sh, q_sl;
    int64_t k_sb, k_sh, k_sc, k_sl;
    int64_t v_sb, v_sh, v_sc, v_sl;
    float scale;  // 1/sqrt(64)
};
void launch_flash_attention_d64(const FlashAttnParams& p, hipStream_t stream);
