// VAE mid-block attention, gfx950, bf16: SINGLE head with head_dim 512
// (SURVEY §2.4a K12 — the SD VAE decoder's 512-channel attention).
//
// d=512 cannot ride the d64 flash kernel: per-lane O accumulators would be
// 16 MFMA tiles (256 VGPRs) and Q fragments 32 k-slices (256 VGPRs).
// Split-D design instead:
// * one 64-query block; 8 waves each own a 64-wide d-slice (Q fragments
//   2x32 VGPRs, O accumulators 2x32 per wave for the two 32-query tiles);
// * per 32-token KV tile, each wave computes its PARTIAL S^T (its d-slice's
//   QK contribution, swapped mfma_f32_32x32x16_bf16 like the d64 kernel)
//   and ds_add_f32-accumulates it into one shared S tile; the reduction
//   pass applies the shared online-softmax state (m, l per query, in LDS)
//   and materializes P[q][t] bf16;
// * each wave then rescales its O^T accumulators by the shared correction
//   and accumulates PV for its own d-slice;
// * the epilogue transposes O through LDS (aliasing the K/V buffers) so
//   the global stores are row-contiguous ([L][512] layout).
// LDS rows are 8 B-padded (stride 1032/72 B: gcd(dwords,32)=2 -> 2-way
// conflicts) and fragment reads are b64 pairs, as in the conv kernel
// (profiles/conv_ladder_r02.md v5).
//
// Replaces the chunked bf16 einsum fallback in models/vae.py, which
// materialized fp32 score slabs. v1 of this kernel
// used 4 waves x 32 queries: 64 TF at L=16k, bound by KV re-stream; this
// 64-query version doubles the compute per staged KV byte.

#include "common.h"
#include "kernels.h"

namespace {

typedef float float16v __attribute__((ext_vector_type(16)));

constexpr int NW = 8;        // waves = d-slices
constexpr int QT = 64;       // queries per block (two 32-row MFMA tiles)
constexpr int NQT = QT / 32;
constexpr int KVB = 32;      // kv tokens per tile
constexpr int C = 512;       // channels (head_dim)
constexpr int DSL = C / NW;  // d-slice per wave (64)
constexpr int KS = DSL / 16;
constexpr int DT = DSL / 32;

constexpr int K_ROW = C * 2 + 8;     // [t][d] row bytes
constexpr int VT_ROW = KVB * 2 + 8;  // [d][t] row bytes
constexpr int P_ROW = KVB * 2 + 8;   // [q][t] row bytes
constexpr int O_ROW = C * 2 + 8;     // [q][d] row bytes (aliases k+vt)

constexpr int K_SZ = KVB * K_ROW;
constexpr int VT_SZ = C * VT_ROW;
constexpr int P_SZ = QT * P_ROW;

__device__ __forceinline__ short8 lds_frag_b64x2(const char* addr) {
    const uint2 a = *reinterpret_cast<const uint2*>(addr);
    const uint2 b = *reinterpret_cast<const uint2*>(addr + 8);
    const uint4 v{a.x, a.y, b.x, b.y};
    return __builtin_bit_cast(short8, v);
}

__global__ __launch_bounds__(NW * WAVE_SIZE) void vae_attn_kernel(VaeAttnParams p) {
    __shared__ char smem[K_SZ + VT_SZ + P_SZ];  // k | vt | p; o aliases k+vt
    __shared__ float s_red[KVB][QT];
    __shared__ float m_lds[QT], l_lds[QT], corr_lds[QT];
    char* k_lds = smem;
    char* vt_lds = smem + K_SZ;
    char* p_lds = smem + K_SZ + VT_SZ;
    static_assert(QT * O_ROW <= K_SZ + VT_SZ);

    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lo = lane & 31;
    const int hi = lane >> 5;
    const int gr = (lane & 3) | ((lane & 8) >> 1);  // butterfly group member
    const int gw = ((lane >> 2) & 1) | (((lane >> 4) & 3) << 1);

    const int64_t q0 = (int64_t)blockIdx.x * QT;
    const int b = blockIdx.y;
    const uint16_t* qb = p.q + (int64_t)b * p.sb;
    const uint16_t* kb = p.k + (int64_t)b * p.sb;
    const uint16_t* vb = p.v + (int64_t)b * p.sb;
    const float scale2 = p.scale * 1.44269504088896340736f;

    // ---- Q fragments for this wave's d-slice (held in registers) ----------
    short8 qf[NQT][KS];
#pragma unroll
    for (int qt = 0; qt < NQT; ++qt) {
        const int64_t qrow = q0 + qt * 32 + lo < p.L ? q0 + qt * 32 + lo : p.L - 1;
        const uint16_t* qp = qb + qrow * C + wave * DSL;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks)
            qf[qt][ks] = *reinterpret_cast<const short8*>(qp + ks * 16 + hi * 8);
    }
    if (tid < QT) {
        m_lds[tid] = -1e30f;
        l_lds[tid] = 0.f;
    }
    float16v ot[NQT][DT] = {};

    const int n_tiles = (int)((p.L + KVB - 1) / KVB);
    for (int tile = 0; tile < n_tiles; ++tile) {
        const int64_t t0 = (int64_t)tile * KVB;
        // zero the split-K tile while staging
        for (int c = tid; c < KVB * QT; c += NW * WAVE_SIZE)
            reinterpret_cast<float*>(s_red)[c] = 0.f;
        // ---- stage K [t][d] and V^T [d][t] --------------------------------
        {
            constexpr int KITEMS = KVB * (C / 8) / (NW * WAVE_SIZE);
#pragma unroll
            for (int it = 0; it < KITEMS; ++it) {
                const int c = it * NW * WAVE_SIZE + tid;
                const int t = c / (C / 8);
                const int d8 = c % (C / 8);
                uint4 raw = {0, 0, 0, 0};
                if (t0 + t < p.L)
                    raw = *reinterpret_cast<const uint4*>(kb + (t0 + t) * C + d8 * 8);
                char* dst = &k_lds[t * K_ROW + d8 * 16];
                *reinterpret_cast<uint2*>(dst) = uint2{raw.x, raw.y};
                *reinterpret_cast<uint2*>(dst + 8) = uint2{raw.z, raw.w};
            }
            // V^T via the DPP butterfly: slots = (token-window, 64-d block)
#pragma unroll
            for (int sl = 0; sl < (KVB / 8) * (C / 64) / NW; ++sl) {
                const int slot = sl * NW + wave;
                const int tw = slot % (KVB / 8);
                const int db = slot / (KVB / 8);
                const int64_t tg = t0 + tw * 8 + gr;
                const int d0 = (db * 8 + gw) * 8;
                uint4 raw = {0, 0, 0, 0};
                if (tg < p.L)
                    raw = *reinterpret_cast<const uint4*>(vb + tg * C + d0);
                const uint4 tr = transpose8x8_bf16(raw, lane);
                const int d = (db * 8 + gw) * 8 + gr;
                char* dst = &vt_lds[d * VT_ROW + (tw * 8) * 2];
                *reinterpret_cast<uint2*>(dst) = uint2{tr.x, tr.y};
                *reinterpret_cast<uint2*>(dst + 8) = uint2{tr.z, tr.w};
            }
        }
        __syncthreads();

        // ---- partial S^T for this wave's d-slice --------------------------
#pragma unroll
        for (int qt = 0; qt < NQT; ++qt) {
            float16v s = {};
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int ks = 0; ks < KS; ++ks) {
                const short8 kf = lds_frag_b64x2(
                    &k_lds[lo * K_ROW + (wave * DSL + ks * 16 + hi * 8) * 2]);
                s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[qt][ks], s, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
            // lane holds S^T rows t = (r&3)+8*(r>>2)+4*hi for q col qt*32+lo
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int t = (r & 3) + 8 * (r >> 2) + 4 * hi;
                atomicAdd(&s_red[t][qt * 32 + lo], s[r]);
            }
        }
        __syncthreads();

        // ---- reduce + shared online softmax -------------------------------
        // thread -> (q = tid/8, 4 tokens); 8 consecutive threads per q.
        {
            const int q = tid / 8;
            const int t0l = (tid % 8) * 4;
            float sv[4];
            float tmax = -1e30f;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int t = t0l + j;
                float acc = s_red[t][q];
                if (t0 + t >= p.L) acc = -1e30f;
                sv[j] = acc;
                tmax = fmaxf(tmax, acc);
            }
            tmax = fmaxf(tmax, __shfl_xor(tmax, 1, WAVE_SIZE));
            tmax = fmaxf(tmax, __shfl_xor(tmax, 2, WAVE_SIZE));
            tmax = fmaxf(tmax, __shfl_xor(tmax, 4, WAVE_SIZE));
            const float m_old = m_lds[q];
            const float m_new = fmaxf(m_old, tmax);
            const float msc = m_new * scale2;
            float tsum = 0.f;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const float pe = __builtin_amdgcn_exp2f(__builtin_fmaf(sv[j], scale2, -msc));
                tsum += pe;
                reinterpret_cast<bf16_t*>(&p_lds[q * P_ROW])[t0l + j] = __float2bfloat16(pe);
            }
            tsum += __shfl_xor(tsum, 1, WAVE_SIZE);
            tsum += __shfl_xor(tsum, 2, WAVE_SIZE);
            tsum += __shfl_xor(tsum, 4, WAVE_SIZE);
            if ((tid % 8) == 0) {
                const float corr =
                    m_old == -1e30f ? 1.f
                                    : __builtin_amdgcn_exp2f((m_old - m_new) * scale2);
                corr_lds[q] = corr;
                l_lds[q] = l_lds[q] * corr + tsum;
                m_lds[q] = m_new;
            }
        }
        __syncthreads();

        // ---- rescale + PV for this wave's d-slice -------------------------
#pragma unroll
        for (int qt = 0; qt < NQT; ++qt) {
            const float corr = corr_lds[qt * 32 + lo];
            if (corr != 1.f) {
#pragma unroll
                for (int dt = 0; dt < DT; ++dt)
#pragma unroll
                    for (int r = 0; r < 16; ++r) ot[qt][dt][r] *= corr;
            }
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
                for (int kt = 0; kt < 2; ++kt) {
                    const int d = wave * DSL + dt * 32 + lo;
                    const short8 vf =
                        lds_frag_b64x2(&vt_lds[d * VT_ROW + (kt * 16 + hi * 8) * 2]);
                    const short8 pf = lds_frag_b64x2(
                        &p_lds[(qt * 32 + lo) * P_ROW + (kt * 16 + hi * 8) * 2]);
                    // A = V^T rows d, B = P^T cols q -> O^T[d][q]
                    ot[qt][dt] =
                        __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pf, ot[qt][dt], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
    }

    // ---- epilogue: O^T -> LDS [q][d] -> coalesced [L][512] stores ---------
    char* o_lds = smem;  // aliases k|vt (QT * O_ROW = 66.6 KB <= 70 KB)
    {
#pragma unroll
        for (int qt = 0; qt < NQT; ++qt) {
            const float inv = l_lds[qt * 32 + lo] > 0.f ? 1.f / l_lds[qt * 32 + lo] : 0.f;
#pragma unroll
            for (int dt = 0; dt < DT; ++dt)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int d = wave * DSL + dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                    reinterpret_cast<bf16_t*>(&o_lds[(qt * 32 + lo) * O_ROW])[d] =
                        __float2bfloat16(ot[qt][dt][r] * inv);
                }
        }
    }
    __syncthreads();
    {
        constexpr int OITEMS = QT * (C / 8) / (NW * WAVE_SIZE);
#pragma unroll
        for (int it = 0; it < OITEMS; ++it) {
            const int c = it * NW * WAVE_SIZE + tid;
            const int qq = c / (C / 8);
            const int d8 = c % (C / 8);
            if (q0 + qq < p.L) {
                const char* src = &o_lds[qq * O_ROW + d8 * 16];
                const uint2 a = *reinterpret_cast<const uint2*>(src);
                const uint2 bwd = *reinterpret_cast<const uint2*>(src + 8);
                uint4 v{a.x, a.y, bwd.x, bwd.y};
                *reinterpret_cast<uint4*>(p.o + (q0 + qq) * C + (int64_t)b * p.sb + d8 * 8) = v;
            }
        }
    }
}

}  // namespace

void launch_vae_attention(const VaeAttnParams& p, hipStream_t stream) {
    dim3 grid((unsigned)((p.L + QT - 1) / QT), (unsigned)p.B);
    vae_attn_kernel<<<grid, dim3(NW * WAVE_SIZE), 0, stream>>>(p);
}
