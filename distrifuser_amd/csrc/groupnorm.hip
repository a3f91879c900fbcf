// Fused GroupNorm kernels (gfx950).
//
// Reference semantics: ops/eager.py group_norm_{stats,apply,silu}. The op is
// HBM-bandwidth-bound on MI355X (8 TB/s), so everything is vectorized to
// 16 B/lane and SiLU is fused into the normalization epilogue (every GN in an
// SD ResBlock is followed by SiLU — fusing halves the traffic of a separate
// activation pass). Stats use a grid-strided partial-reduction kernel with
// one atomicAdd pair per block (device-scope atomics are XCD-safe, guide §6
// G12); each (sample, group) slab is contiguous in NCHW.

#include "common.h"
#include "kernels.h"
#include <algorithm>

namespace {

template <typename T, bool VEC>
__global__ void gn_stats_partial_kernel(const T* __restrict__ x, float* __restrict__ partial,
                                        int64_t group_len, int chunks, int ngt) {
    constexpr int V = VEC ? VecN<T>::value : 1;
    const int ng = blockIdx.x / chunks;
    const int chunk = blockIdx.x % chunks;
    // per-chunk range, V-aligned (VEC requires group_len % V == 0, host-checked)
    const int64_t nvec = group_len / V;
    const int64_t per = (nvec + chunks - 1) / chunks;
    const int64_t v0 = chunk * per;
    const int64_t v1 = min(v0 + per, nvec);
    const T* base = x + (int64_t)ng * group_len;

    float s = 0.f, ss = 0.f;
    for (int64_t iv = v0 + threadIdx.x; iv < v1; iv += blockDim.x) {
        if (VEC) {
            uint4 raw = *reinterpret_cast<const uint4*>(base + iv * V);
            const T* e = reinterpret_cast<const T*>(&raw);
#pragma unroll
            for (int j = 0; j < V; ++j) {
                float f = to_f32(e[j]);
                s += f;
                ss += f * f;
            }
        } else {
            float f = to_f32(base[iv]);
            s += f;
            ss += f * f;
        }
    }
    __shared__ float lds[2 * 16];
    block_reduce_sum2(s, ss, lds);
    if (threadIdx.x == 0) {
        atomicAdd(&partial[ng], s);
        atomicAdd(&partial[ngt + ng], ss);
    }
}

template <typename T>
__global__ void gn_finalize_kernel(const float* __restrict__ partial, T* __restrict__ out,
                                   float inv_count, int total) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < 2 * total) out[i] = from_f32<T>(partial[i] * inv_count);
}

template <typename T, bool SILU, bool VEC>
__global__ void gn_apply_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const float* __restrict__ mean, const float* __restrict__ meansq,
                                const T* __restrict__ w, const T* __restrict__ b, float eps,
                                int64_t hw, int C, int G, int64_t total) {
    constexpr int V = VEC ? VecN<T>::value : 1;
    const int gs = C / G;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t vid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; vid * V < total;
         vid += stride) {
        const int64_t i = vid * V;
        const int64_t c = (i / hw) % C;
        const int64_t n = i / (hw * (int64_t)C);
        const int g = (int)(c / gs);
        const float m = mean[n * G + g];
        float var = meansq[n * G + g] - m * m;
        var = var < 0.f ? 0.f : var;
        const float inv = rsqrtf(var + eps);
        const float sc = (w ? to_f32(w[c]) : 1.f) * inv;
        const float sh = (b ? to_f32(b[c]) : 0.f) - m * sc;
        if (VEC) {
            uint4 raw = *reinterpret_cast<const uint4*>(x + i);
            const T* e = reinterpret_cast<const T*>(&raw);
            uint4 outv;
            T* o = reinterpret_cast<T*>(&outv);
#pragma unroll
            for (int j = 0; j < V; ++j) {
                float f = to_f32(e[j]) * sc + sh;
                if (SILU) f = siluf(f);
                o[j] = from_f32<T>(f);
            }
            *reinterpret_cast<uint4*>(y + i) = outv;
        } else {
            float f = to_f32(x[i]) * sc + sh;
            if (SILU) f = siluf(f);
            y[i] = from_f32<T>(f);
        }
    }
}

inline int grid_for(int64_t work, int block) {
    int64_t g = (work + block - 1) / block;
    return (int)std::min<int64_t>(g, 4096);
}

template <typename T>
void gn_stats_partial_t(const void* x, float* partial, int64_t group_len, int ngt,
                        hipStream_t stream) {
    // size chunks so the chip is filled: >= 2048 blocks total
    const bool vec = (group_len % VecN<T>::value) == 0;
    const int V = vec ? VecN<T>::value : 1;
    int chunks = (int)std::max<int64_t>(1, (2048 + ngt - 1) / ngt);
    // but never more chunks than work
    int64_t nvec = std::max<int64_t>(group_len / V, 1);
    chunks = (int)std::min<int64_t>(chunks, (nvec + 255) / 256);
    chunks = std::max(chunks, 1);
    dim3 grid(ngt * chunks);
    if (vec)
        gn_stats_partial_kernel<T, true><<<grid, 256, 0, stream>>>(
            reinterpret_cast<const T*>(x), partial, group_len, chunks, ngt);
    else
        gn_stats_partial_kernel<T, false><<<grid, 256, 0, stream>>>(
            reinterpret_cast<const T*>(x), partial, group_len, chunks, ngt);
}

template <typename T>
void gn_apply_t(const void* x, void* y, const float* mean, const float* meansq, const void* w,
                const void* b, float eps, int64_t hw, int C, int G, int N, bool silu,
                hipStream_t stream) {
    const int64_t total = (int64_t)N * C * hw;
    const bool vec = (hw % VecN<T>::value) == 0;
    const int block = 256;
    if (vec) {
        int grid = grid_for(total / VecN<T>::value, block);
        if (silu)
            gn_apply_kernel<T, true, true><<<grid, block, 0, stream>>>(
                (const T*)x, (T*)y, mean, meansq, (const T*)w, (const T*)b, eps, hw, C, G, total);
        else
            gn_apply_kernel<T, false, true><<<grid, block, 0, stream>>>(
                (const T*)x, (T*)y, mean, meansq, (const T*)w, (const T*)b, eps, hw, C, G, total);
    } else {
        int grid = grid_for(total, block);
        if (silu)
            gn_apply_kernel<T, true, false><<<grid, block, 0, stream>>>(
                (const T*)x, (T*)y, mean, meansq, (const T*)w, (const T*)b, eps, hw, C, G, total);
        else
            gn_apply_kernel<T, false, false><<<grid, block, 0, stream>>>(
                (const T*)x, (T*)y, mean, meansq, (const T*)w, (const T*)b, eps, hw, C, G, total);
    }
}

template <typename T, bool CORRECTED>
__global__ void gn_merge_stats_kernel(T* __restrict__ buffer, int64_t row_stride,
                                      int64_t slot_off, int n_peers, int own,
                                      const T* __restrict__ fresh, float* __restrict__ out,
                                      int ng) {
    const int g = blockIdx.x * blockDim.x + threadIdx.x;
    if (g >= ng) return;
    // moments live as [mean[0..ng), meansq[ng..2ng)] within each peer slot
    float m_sum = 0.f, q_sum = 0.f;
    float m_own = 0.f, q_own = 0.f;
    for (int p = 0; p < n_peers; ++p) {
        const T* slot = buffer + p * row_stride + slot_off;
        float m = to_f32(slot[g]);
        float q = to_f32(slot[ng + g]);
        if (p == own) {
            m_own = m;
            q_own = q;
            if (!CORRECTED) {  // stale_gn substitutes the fresh slot
                m = to_f32(fresh[g]);
                q = to_f32(fresh[ng + g]);
            }
        }
        m_sum += m;
        q_sum += q;
    }
    float mean = m_sum / n_peers;
    float msq = q_sum / n_peers;
    const float f_m = to_f32(fresh[g]);
    const float f_q = to_f32(fresh[ng + g]);
    if (CORRECTED) {
        mean += f_m - m_own;
        msq += f_q - q_own;
        // negative-variance guard: fall back to the local slice variance
        // (reference pp/groupnorm.py:60-63)
        const float var = msq - mean * mean;
        if (var < 0.f) msq = mean * mean + (f_q - f_m * f_m);
    }
    out[g] = mean;
    out[ng + g] = msq;
    // stage fresh into our slot for the async all-gather
    T* own_slot = buffer + own * row_stride + slot_off;
    own_slot[g] = fresh[g];
    own_slot[ng + g] = fresh[ng + g];
}

}  // namespace

void launch_gn_merge_stats(void* buffer, int64_t row_stride, int64_t slot_off, int n_peers,
                           int own, const void* fresh, float* out, int ng, bool corrected,
                           int dtype, hipStream_t stream) {
    const int block = 256;
    const int grid = (ng + block - 1) / block;
    switch (dtype) {
        case DFA_BF16:
            if (corrected)
                gn_merge_stats_kernel<bf16_t, true><<<grid, block, 0, stream>>>(
                    (bf16_t*)buffer, row_stride, slot_off, n_peers, own, (const bf16_t*)fresh,
                    out, ng);
            else
                gn_merge_stats_kernel<bf16_t, false><<<grid, block, 0, stream>>>(
                    (bf16_t*)buffer, row_stride, slot_off, n_peers, own, (const bf16_t*)fresh,
                    out, ng);
            break;
        case DFA_F16:
            if (corrected)
                gn_merge_stats_kernel<f16_t, true><<<grid, block, 0, stream>>>(
                    (f16_t*)buffer, row_stride, slot_off, n_peers, own, (const f16_t*)fresh,
                    out, ng);
            else
                gn_merge_stats_kernel<f16_t, false><<<grid, block, 0, stream>>>(
                    (f16_t*)buffer, row_stride, slot_off, n_peers, own, (const f16_t*)fresh,
                    out, ng);
            break;
        default:
            if (corrected)
                gn_merge_stats_kernel<float, true><<<grid, block, 0, stream>>>(
                    (float*)buffer, row_stride, slot_off, n_peers, own, (const float*)fresh,
                    out, ng);
            else
                gn_merge_stats_kernel<float, false><<<grid, block, 0, stream>>>(
                    (float*)buffer, row_stride, slot_off, n_peers, own, (const float*)fresh,
                    out, ng);
            break;
    }
}

void launch_gn_stats_partial(const void* x, float* partial, int64_t group_len, int ngt, int dtype,
                             hipStream_t stream) {
    switch (dtype) {
        case DFA_BF16: gn_stats_partial_t<bf16_t>(x, partial, group_len, ngt, stream); break;
        case DFA_F16: gn_stats_partial_t<f16_t>(x, partial, group_len, ngt, stream); break;
        default: gn_stats_partial_t<float>(x, partial, group_len, ngt, stream); break;
    }
}

void launch_gn_finalize(const float* partial, void* out, float inv_count, int ngt, int dtype,
                        hipStream_t stream) {
    int total = 2 * ngt;
    int block = 256;
    int grid = (total + block - 1) / block;
    switch (dtype) {
        case DFA_BF16:
            gn_finalize_kernel<bf16_t><<<grid, block, 0, stream>>>(partial, (bf16_t*)out, inv_count, ngt);
            break;
        case DFA_F16:
            gn_finalize_kernel<f16_t><<<grid, block, 0, stream>>>(partial, (f16_t*)out, inv_count, ngt);
            break;
        default:
            gn_finalize_kernel<float><<<grid, block, 0, stream>>>(partial, (float*)out, inv_count, ngt);
            break;
    }
}

void launch_gn_apply(const void* x, void* y, const float* mean, const float* meansq,
                     const void* weight, const void* bias, float eps, int64_t hw, int C, int G,
                     int N, bool silu, int dtype, hipStream_t stream) {
    switch (dtype) {
        case DFA_BF16:
            gn_apply_t<bf16_t>(x, y, mean, meansq, weight, bias, eps, hw, C, G, N, silu, stream);
            break;
        case DFA_F16:
            gn_apply_t<f16_t>(x, y, mean, meansq, weight, bias, eps, hw, C, G, N, silu, stream);
            break;
        default:
            gn_apply_t<float>(x, y, mean, meansq, weight, bias, eps, hw, C, G, N, silu, stream);
            break;
    }
}
