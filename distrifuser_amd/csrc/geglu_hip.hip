#include "hip/hip_runtime.h"
// Fused GEGLU gate (gfx950): out[r, i] = in[r, i] * gelu(in[r, inner + i]).
// Memory-bound; 16 B/lane vector loads on both halves, erf-exact gelu
// (matches torch F.gelu default). Reference numerics: ops/eager.py geglu.

#include "common.h"
#include "kernels.h"
#include <algorithm>

namespace {

template <typename T, bool VEC>
__global__ void geglu_kernel(const T* __restrict__ in, T* __restrict__ out, int64_t rows,
                             int64_t inner) {
    constexpr int V = VEC ? VecN<T>::value : 1;
    const int64_t total_v = rows * inner / V;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t vid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; vid < total_v;
         vid += stride) {
        const int64_t i = vid * V;
        const int64_t r = i / inner;
        const int64_t col = i % inner;
        const T* a_ptr = in + r * 2 * inner + col;
        const T* g_ptr = a_ptr + inner;
        if (VEC) {
            uint4 araw = *reinterpret_cast<const uint4*>(a_ptr);
            uint4 graw = *reinterpret_cast<const uint4*>(g_ptr);
            const T* a = reinterpret_cast<const T*>(&araw);
            const T* g = reinterpret_cast<const T*>(&graw);
            uint4 oraw;
            T* o = reinterpret_cast<T*>(&oraw);
#pragma unroll
            for (int j = 0; j < V; ++j) o[j] = from_f32<T>(to_f32(a[j]) * geluf(to_f32(g[j])));
            *reinterpret_cast<uint4*>(out + i) = oraw;
        } else {
            out[i] = from_f32<T>(to_f32(*a_ptr) * geluf(to_f32(*g_ptr)));
        }
    }
}

template <typename T>
void geglu_t(const void* in, void* out, int64_t rows, int64_t inner, hipStream_t stream) {
    const bool vec = (inner % VecN<T>::value) == 0;
    const int block = 256;
    const int64_t work = rows * inner / (vec ? VecN<T>::value : 1);
    const int grid = (int)std::min<int64_t>((work + block - 1) / block, 4096);
    if (vec)
       hipLaunchKernelGGL(( geglu_kernel<T, true>), dim3(grid), dim3(block), 0, stream, (const T*)in, (T*)out, rows, inner);
    else
       hipLaunchKernelGGL(( geglu_kernel<T, false>), dim3(grid), dim3(block), 0, stream, (const T*)in, (T*)out, rows, inner);
}

}  // namespace

void launch_geglu(const void* in, void* out, int64_t rows, int64_t inner, int dtype,
                  hipStream_t stream) {
    switch (dtype) {
        case DFA_BF16: geglu_t<bf16_t>(in, out, rows, inner, stream); break;
        case DFA_F16: geglu_t<f16_t>(in, out, rows, inner, stream); break;
        default: geglu_t<float>(in, out, rows, inner, stream); break;
    }
}
