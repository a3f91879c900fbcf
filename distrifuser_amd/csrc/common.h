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

#define DFA_HIP_CHECK(expr)                                                     \
    do {                                                                        \
        hipError_t _e = (expr);                                                 \
        if (_e != hipSuccess) {                                                 \
            printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, __LINE__); \
        }                                                                       \
    } while (0)
