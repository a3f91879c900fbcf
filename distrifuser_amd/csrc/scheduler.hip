// Fused CFG-combine + DDIM update (gfx950): one elementwise pass replaces
// chunk/lerp/sqrt-chain torch ops between denoise steps (they run OUTSIDE
// the captured graphs, so their launch overhead is exposed every step).
//
//   eps   = nu + g * (nc - nu)           (classifier-free guidance)
//   x0    = (x - sqrt(1-a_t) * eps) / sqrt(a_t)
//   x'    = sqrt(a_prev) * x0 + sqrt(1-a_prev) * eps   (DDIM, eta = 0)
//
// Reference numerics: schedulers/ddim.py step + pipelines._denoise combine.

#include "common.h"
#include "kernels.h"

namespace {

template <typename T, bool VEC>
__global__ void ddim_cfg_step_kernel(const T* __restrict__ noise_u, const T* __restrict__ noise_c,
                                     const T* __restrict__ x, T* __restrict__ out, float g,
                                     float sqrt_at, float sqrt_1mat, float sqrt_ap,
                                     float sqrt_1map, int64_t total) {
    constexpr int V = VEC ? VecN<T>::value : 1;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const float inv_sqrt_at = 1.f / sqrt_at;
    for (int64_t vid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; vid * V < total;
         vid += stride) {
        const int64_t i = vid * V;
        if (VEC) {
            uint4 ur = *reinterpret_cast<const uint4*>(noise_u + i);
            uint4 cr = *reinterpret_cast<const uint4*>(noise_c + i);
            uint4 xr = *reinterpret_cast<const uint4*>(x + i);
            const T* u = reinterpret_cast<const T*>(&ur);
            const T* c = reinterpret_cast<const T*>(&cr);
            const T* xi = reinterpret_cast<const T*>(&xr);
            uint4 orw;
            T* o = reinterpret_cast<T*>(&orw);
#pragma unroll
            for (int j = 0; j < V; ++j) {
                const float nu = to_f32(u[j]);
                const float eps = nu + g * (to_f32(c[j]) - nu);
                const float x0 = (to_f32(xi[j]) - sqrt_1mat * eps) * inv_sqrt_at;
                o[j] = from_f32<T>(sqrt_ap * x0 + sqrt_1map * eps);
            }
            *reinterpret_cast<uint4*>(out + i) = orw;
        } else {
            const float nu = to_f32(noise_u[i]);
            const float eps = nu + g * (to_f32(noise_c[i]) - nu);
            const float x0 = (to_f32(x[i]) - sqrt_1mat * eps) * inv_sqrt_at;
            out[i] = from_f32<T>(sqrt_ap * x0 + sqrt_1map * eps);
        }
    }
}

template <typename T>
void ddim_t(const void* nu, const void* nc, const void* x, void* out, float g, float sat,
            float s1mat, float sap, float s1map, int64_t total, hipStream_t stream) {
    const bool vec = (total % VecN<T>::value) == 0;
    const int block = 256;
    const int64_t work = total / (vec ? VecN<T>::value : 1);
    const int grid = (int)((work + block - 1) / block < 4096 ? (work + block - 1) / block : 4096);
    if (vec)
        ddim_cfg_step_kernel<T, true><<<grid, block, 0, stream>>>(
            (const T*)nu, (const T*)nc, (const T*)x, (T*)out, g, sat, s1mat, sap, s1map, total);
    else
        ddim_cfg_step_kernel<T, false><<<grid, block, 0, stream>>>(
            (const T*)nu, (const T*)nc, (const T*)x, (T*)out, g, sat, s1mat, sap, s1map, total);
}

}  // namespace

void launch_ddim_cfg_step(const void* noise_u, const void* noise_c, const void* x, void* out,
                          float g, float sqrt_at, float sqrt_1mat, float sqrt_ap,
                          float sqrt_1map, int64_t total, int dtype, hipStream_t stream) {
    switch (dtype) {
        case DFA_BF16:
            ddim_t<bf16_t>(noise_u, noise_c, x, out, g, sqrt_at, sqrt_1mat, sqrt_ap, sqrt_1map,
                           total, stream);
            break;
        case DFA_F16:
            ddim_t<f16_t>(noise_u, noise_c, x, out, g, sqrt_at, sqrt_1mat, sqrt_ap, sqrt_1map,
                          total, stream);
            break;
        default:
            ddim_t<float>(noise_u, noise_c, x, out, g, sqrt_at, sqrt_1mat, sqrt_ap, sqrt_1map,
                          total, stream);
            break;
    }
}
