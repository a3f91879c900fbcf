// Fused (residual-add +) LayerNorm, gfx950, bf16 (SURVEY hotlist item:
// at::native vectorized_layer_norm + the transformer residual adds were
// ~4-5% of the 3840^2 step, profiles/rocprof_3840_r01_v2.md rows 6/9/10).
//
// One wave per row (C up to 2048 held in registers: 32 bf16/lane), fp32
// stats via wave reduction, optional fused residual: y = LN(x + res) with
// the sum also written out — so the transformer block's `x = x + attn(...)`
// add never runs as a separate elementwise kernel.

#include "common.h"
#include "kernels.h"

namespace {

constexpr int WPB = 4;  // waves (rows) per block

template <int VPT>  // 8-element vectors per lane
__global__ __launch_bounds__(WPB* WAVE_SIZE) void ln_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
    uint16_t* __restrict__ y, uint16_t* __restrict__ sum_out,
    const uint16_t* __restrict__ w, const uint16_t* __restrict__ b, float eps,
    int64_t rows, int C) {
    const int wave = threadIdx.x / WAVE_SIZE;
    const int lane = threadIdx.x % WAVE_SIZE;
    const int64_t row = (int64_t)blockIdx.x * WPB + wave;
    if (row >= rows) return;

    const uint16_t* xp = x + row * C;
    const uint16_t* rp = res ? res + row * C : nullptr;
    float v[VPT][8];
    float s = 0.f, ss = 0.f;
#pragma unroll
    for (int k = 0; k < VPT; ++k) {
        const int d0 = (k * WAVE_SIZE + lane) * 8;
        if (VPT * WAVE_SIZE * 8 == 0 || d0 < C) {
            const short8 xv = *reinterpret_cast<const short8*>(xp + d0);
            short8 rv{};
            if (rp) rv = *reinterpret_cast<const short8*>(rp + d0);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float f = to_f32(__builtin_bit_cast(bf16_t, (short)xv[j]));
                if (rp) f += to_f32(__builtin_bit_cast(bf16_t, (short)rv[j]));
                v[k][j] = f;
                s += f;
                ss += f * f;
            }
            if (sum_out) {
                uint16_t sv[8];
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    sv[j] = __builtin_bit_cast(uint16_t, __float2bfloat16(v[k][j]));
                *reinterpret_cast<uint4*>(sum_out + row * C + d0) =
                    *reinterpret_cast<const uint4*>(sv);
            }
        } else {
#pragma unroll
            for (int j = 0; j < 8; ++j) v[k][j] = 0.f;
        }
    }
    s = wave_all_reduce_sum(s);
    ss = wave_all_reduce_sum(ss);
    const float mean = s / (float)C;
    const float var = fmaxf(ss / (float)C - mean * mean, 0.f);
    const float inv = __frsqrt_rn(var + eps);

    uint16_t* yp = y + row * C;
#pragma unroll
    for (int k = 0; k < VPT; ++k) {
        const int d0 = (k * WAVE_SIZE + lane) * 8;
        if (d0 < C) {
            const short8 wv = *reinterpret_cast<const short8*>(w + d0);
            const short8 bv = *reinterpret_cast<const short8*>(b + d0);
            uint16_t out[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float wf = to_f32(__builtin_bit_cast(bf16_t, (short)wv[j]));
                const float bf = to_f32(__builtin_bit_cast(bf16_t, (short)bv[j]));
                out[j] = __builtin_bit_cast(
                    uint16_t, __float2bfloat16((v[k][j] - mean) * inv * wf + bf));
            }
            *reinterpret_cast<uint4*>(yp + d0) = *reinterpret_cast<const uint4*>(out);
        }
    }
}

}  // namespace

void launch_layer_norm(const void* x, const void* res, void* y, void* sum_out,
                       const void* w, const void* b, float eps, int64_t rows, int C,
                       hipStream_t stream) {
    dim3 grid((unsigned)((rows + WPB - 1) / WPB));
    dim3 block(WPB * WAVE_SIZE);
    const auto* xp = reinterpret_cast<const uint16_t*>(x);
    const auto* rp = reinterpret_cast<const uint16_t*>(res);
    auto* yp = reinterpret_cast<uint16_t*>(y);
    auto* sp = reinterpret_cast<uint16_t*>(sum_out);
    const auto* wp = reinterpret_cast<const uint16_t*>(w);
    const auto* bp = reinterpret_cast<const uint16_t*>(b);
    if (C <= 512)
        ln_kernel<1><<<grid, block, 0, stream>>>(xp, rp, yp, sp, wp, bp, eps, rows, C);
    else if (C <= 1024)
        ln_kernel<2><<<grid, block, 0, stream>>>(xp, rp, yp, sp, wp, bp, eps, rows, C);
    else if (C <= 1536)
        ln_kernel<3><<<grid, block, 0, stream>>>(xp, rp, yp, sp, wp, bp, eps, rows, C);
    else
        ln_kernel<4><<<grid, block, 0, stream>>>(xp, rp, yp, sp, wp, bp, eps, rows, C);
}
