// Fused CFG-combine + scheduler update (gfx950): one elementwise pass
// replaces the chunk/lerp/axpy torch ops between denoise steps (they run
// OUTSIDE the captured graphs, so their launch overhead is exposed every
// step). Deterministic eta=0 updates are affine in (x, eps):
//
//   eps = nu + g * (nc - nu)              (classifier-free guidance)
//   x'  = ca * x + cb * eps
//
// DDIM:  ca = sqrt(a_prev/a_t), cb = sqrt(1-a_prev) - ca * sqrt(1-a_t)
// Euler: ca = 1,                cb = sigma_next - sigma
// Reference numerics: schedulers/{ddim,euler}.py + pipelines._denoise.

#include "common.h"
#include "kernels.h"

namespace {

template <typename T, bool VEC>
__global__ void cfg_affine_step_kernel(const T* __restrict__ noise_u,
                                       const T* __restrict__ noise_c, const T* __restrict__ x,
                                       T* __restrict__ out, float g, float ca, float cb,
                                       int64_t total) {
    constexpr int V = VEC ? VecN<T>::value : 1;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
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
                o[j] = from_f32<T>(ca * to_f32(xi[j]) + cb * eps);
            }
            *reinterpret_cast<uint4*>(out + i) = orw;
        } else {
            const float nu = to_f32(noise_u[i]);
            const float eps = nu + g * (to_f32(noise_c[i]) - nu);
            out[i] = from_f32<T>(ca * to_f32(x[i]) + cb * eps);
        }
    }
}

template <typename T>
void affine_t(const void* nu, const void* nc, const void* x, void* out, float g, float ca,
              float cb, int64_t total, hipStream_t stream) {
    const bool vec = (total % VecN<T>::value) == 0;
    const int block = 256;
    const int64_t work = total / (vec ? VecN<T>::value : 1);
    const int grid = (int)((work + block - 1) / block < 4096 ? (work + block - 1) / block : 4096);
    if (vec)
        cfg_affine_step_kernel<T, true><<<grid, block, 0, stream>>>(
            (const T*)nu, (const T*)nc, (const T*)x, (T*)out, g, ca, cb, total);
    else
        cfg_affine_step_kernel<T, false><<<grid, block, 0, stream>>>(
            (const T*)nu, (const T*)nc, (const T*)x, (T*)out, g, ca, cb, total);
}

}  // namespace

void launch_cfg_affine_step(const void* noise_u, const void* noise_c, const void* x, void* out,
                            float g, float ca, float cb, int64_t total, int dtype,
                            hipStream_t stream) {
    switch (dtype) {
        case DFA_BF16:
            affine_t<bf16_t>(noise_u, noise_c, x, out, g, ca, cb, total, stream);
            break;
        case DFA_F16:
            affine_t<f16_t>(noise_u, noise_c, x, out, g, ca, cb, total, stream);
            break;
        default:
            affine_t<float>(noise_u, noise_c, x, out, g, ca, cb, total, stream);
            break;
    }
}

namespace {

// DPM-Solver++(2M): x' = ca*x + cb*eps + cc*x0_prev, also emits
// x0 = cx*x + ce*eps for the next step (x0_prev null on the first step).
// x0 state kept in fp32: it reaches ~1/alpha_t * x magnitudes early in the
// trajectory, where bf16 granularity visibly perturbs the 2M correction.
template <typename T>
__global__ void cfg_dpm_step_kernel(const T* __restrict__ noise_u, const T* __restrict__ noise_c,
                                    const T* __restrict__ x, const float* __restrict__ x0_prev,
                                    T* __restrict__ out, float* __restrict__ x0_out, float g,
                                    float ca, float cb, float cc, float cx, float ce,
                                    int64_t total) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
        const float nu = to_f32(noise_u[i]);
        const float eps = nu + g * (to_f32(noise_c[i]) - nu);
        const float xf = to_f32(x[i]);
        float prev = ca * xf + cb * eps;
        if (x0_prev) prev += cc * x0_prev[i];
        out[i] = from_f32<T>(prev);
        x0_out[i] = cx * xf + ce * eps;
    }
}

}  // namespace

void launch_cfg_dpm_step(const void* nu, const void* nc, const void* x, const void* x0_prev,
                         void* out, void* x0_out, float g, float ca, float cb, float cc,
                         float cx, float ce, int64_t total, int dtype, hipStream_t stream) {
    const int block = 256;
    const int grid = (int)((total + block - 1) / block < 4096 ? (total + block - 1) / block : 4096);
    switch (dtype) {
        case DFA_BF16:
            cfg_dpm_step_kernel<bf16_t><<<grid, block, 0, stream>>>(
                (const bf16_t*)nu, (const bf16_t*)nc, (const bf16_t*)x, (const float*)x0_prev,
                (bf16_t*)out, (float*)x0_out, g, ca, cb, cc, cx, ce, total);
            break;
        case DFA_F16:
            cfg_dpm_step_kernel<f16_t><<<grid, block, 0, stream>>>(
                (const f16_t*)nu, (const f16_t*)nc, (const f16_t*)x, (const float*)x0_prev,
                (f16_t*)out, (float*)x0_out, g, ca, cb, cc, cx, ce, total);
            break;
        default:
            cfg_dpm_step_kernel<float><<<grid, block, 0, stream>>>(
                (const float*)nu, (const float*)nc, (const float*)x, (const float*)x0_prev,
                (float*)out, (float*)x0_out, g, ca, cb, cc, cx, ce, total);
    }
}
