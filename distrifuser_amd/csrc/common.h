// Shared device helpers for the distrifuser_amd gfx950 kernels.
// CDNA4: wave64, 32-bank LDS, MFMA via __builtin_amdgcn_mfma_*.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

using bf16_t = __hip_bfloat16;
using f16_t = __half;

typedef float float4v __attribute__((ext_vector_type(4)));
typedef short short8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float to_f32(bf16_t v) { return __bfloat162float(v); }
__device__ __forceinline__ float to_f32(f16_t v) { return __half2float(v); }
__device__ __forceinline__ float to_f32(float v) { return v; }

template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ bf16_t from_f32<bf16_t>(float v) { return __float2bfloat16(v); }
template <>
__device__ __forceinline__ f16_t from_f32<f16_t>(float v) { return __float2half(v); }
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }

// elements per 16-byte vector
template <typename T>
struct VecN {
    static constexpr int value = 16 / sizeof(T);
};

// Full-wave sum; every lane returns the total.
__device__ __forceinline__ float wave_all_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}

// Block-level reduce-sum of (s, ss) pairs across all waves; result on every
// thread of wave 0.
__device__ __forceinline__ void block_reduce_sum2(float& s, float& ss, float* lds /* [2*maxwaves] */) {
    const int wave = threadIdx.x / WAVE_SIZE;
    const int lane = threadIdx.x % WAVE_SIZE;
    const int nwaves = blockDim.x / WAVE_SIZE;
    s = wave_all_reduce_sum(s);
    ss = wave_all_reduce_sum(ss);
    if (lane == 0) {
        lds[wave * 2] = s;
        lds[wave * 2 + 1] = ss;
    }
    __syncthreads();
    if (wave == 0 && lane == 0) {
        s = 0.f;
        ss = 0.f;
        for (int w = 0; w < nwaves; ++w) {
            s += lds[w * 2];
            ss += lds[w * 2 + 1];
        }
    }
}

__device__ __forceinline__ float sigmoidf_fast(float x) { return 1.f / (1.f + __expf(-x)); }
__device__ __forceinline__ float siluf(float x) { return x * sigmoidf_fast(x); }
__device__ __forceinline__ float geluf(float x) {
    // exact erf gelu (PyTorch F.gelu default)
    return 0.5f * x * (1.f + erff(x * 0.70710678118654752440f));
}

// In-register 8x8 bf16 transpose across an 8-lane group (butterfly): group
// member r = (lane&3)|((lane&8)>>1) holds row r (8 elements = one uint4);
// afterwards member r holds column r. Lane-mask set {1,2,8} is chosen so
// every exchange is a VALU DPP op (quad_perm for ^1/^2, row_ror:8 for ^8)
// — __shfl_xor compiles to ds_bpermute, which rides the SAME LDS pipe the
// MFMA fragment reads need (v3 PMC: 2.9 LDS instrs/MFMA, 19% MFMA util).
// Stage (lane-bit m, elem-bit e): where bit(lane,m) != bit(elem,e):
// new[c] = partner(lane^m)'s e[c^e]; composition over the three stages is
// the full transpose (verified by simulation).
__device__ __forceinline__ uint32_t dpp_xor1(uint32_t v) {
    return __builtin_amdgcn_mov_dpp(v, 0xB1, 0xf, 0xf, true);  // quad_perm [1,0,3,2]
}
__device__ __forceinline__ uint32_t dpp_xor2(uint32_t v) {
    return __builtin_amdgcn_mov_dpp(v, 0x4E, 0xf, 0xf, true);  // quad_perm [2,3,0,1]
}
__device__ __forceinline__ uint32_t dpp_xor8(uint32_t v) {
    return __builtin_amdgcn_mov_dpp(v, 0x128, 0xf, 0xf, true);  // row_ror:8
}

__device__ __forceinline__ uint4 transpose8x8_bf16(uint4 v, int lane) {
    uint32_t d[4] = {v.x, v.y, v.z, v.w};
    // stage (m=1, e=1): bf16 halves within dwords
    {
        const bool hi = (lane & 1) != 0;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const uint32_t sw = (d[i] >> 16) | (d[i] << 16);  // e[c^1]
            const uint32_t recv = dpp_xor1(sw);
            d[i] = hi ? (d[i] & 0xffff0000u) | (recv & 0x0000ffffu)
                      : (d[i] & 0x0000ffffu) | (recv & 0xffff0000u);
        }
    }
    // stage (m=2, e=2): dword pairs (0<->1, 2<->3)
    {
        const bool hi = (lane & 2) != 0;
        const uint32_t r0 = dpp_xor2(d[1]);
        const uint32_t r1 = dpp_xor2(d[0]);
        const uint32_t r2 = dpp_xor2(d[3]);
        const uint32_t r3 = dpp_xor2(d[2]);
        if (hi) {
            d[0] = r0;
            d[2] = r2;
        } else {
            d[1] = r1;
            d[3] = r3;
        }
    }
    // stage (m=8, e=4): dword pairs (0<->2, 1<->3)
    {
        const bool hi = (lane & 8) != 0;
        const uint32_t r0 = dpp_xor8(d[2]);
        const uint32_t r1 = dpp_xor8(d[3]);
        const uint32_t r2 = dpp_xor8(d[0]);
        const uint32_t r3 = dpp_xor8(d[1]);
        if (hi) {
            d[0] = r0;
            d[1] = r1;
        } else {
            d[2] = r2;
            d[3] = r3;
        }
    }
    return uint4{d[0], d[1], d[2], d[3]};
}


#define DFA_HIP_CHECK(expr)                                                     \
    do {                                                                        \
        hipError_t _e = (expr);                                                 \
        if (_e != hipSuccess) {                                                 \
            printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, __LINE__); \
        }                                                                       \
    } while (0)
