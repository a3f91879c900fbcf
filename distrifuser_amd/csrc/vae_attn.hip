// VAE mid-block attention, gfx950, bf16: SINGLE head with head_dim 512
// (SURVEY §2.4a K12 — the SD VAE decoder's 512-channel attention).
//
// d=512 cannot ride the d64 flash kernel: per-lane O accumulators would be
// 16 MFMA tiles (256 VGPRs) and Q fragments 32 k-slices (256 VGPRs).
// Split-D design instead:
// * one 32-query tile per block; 4 waves each own a 128-wide d-slice
//   (Q fragments 64 VGPRs, O accumulators 64 VGPRs per wave);
// * per 32-token KV tile, each wave computes its PARTIAL S^T (its d-slice's
//   QK contribution, swapped mfma_f32_32x32x16_bf16 like the d64 kernel),
//   written to a per-wave LDS slab; a split-K reduction sums the slabs,
//   updates the shared online-softmax state (m, l per query, in LDS), and
//   materializes P[q][t] bf16;
// * each wave then rescales its O^T accumulators by the shared correction
//   and accumulates PV for its own d-slice;
// * the epilogue transposes O through LDS so the global stores are
//   row-contiguous ([L][512] layout).
// LDS rows are 8 B-padded (stride 1032/72 B: gcd(dwords,32)=2 -> 2-way
// conflicts) and fragment reads are b64 pairs, as in the conv kernel
// (profiles/conv_ladder_r02.md v5).
//
// Replaces the chunked bf16 einsum fallback in models/vae.py, which
// materialized fp32 score slabs.

#include "common.h"
#include "kernels.h"

namespace {

typedef float float16v __attribute__((ext_vector_type(16)));

constexpr int NW = 4;        // waves = d-slices
constexpr int QT = 32;       // queries per block
constexpr int KVB = 32;      // kv tokens per tile
constexpr int C = 512;       // channels (head_dim)
constexpr int DSL = C / NW;  // d-slice per wave (128)
constexpr int KS = DSL / 16;
constexpr int DT = DSL / 32;

constexpr int K_ROW = C * 2 + 8;    // k_lds row bytes [t][d]
constexpr int VT_ROW = KVB * 2 + 8; // vt_lds row bytes [d][t]
constexpr int P_ROW = KVB * 2 + 8;  // p_lds row bytes [q][t]
constexpr int O_ROW = C * 2 + 8;    // o_lds row bytes [q][d] (aliases k_lds)

__device__ __forceinline__ short8 lds_frag_b64x2(const char* addr) {
    const uint2 a = *reinterpret_cast<const uint2*>(addr);
    const uint2 b = *reinterpret_cast<const uint2*>(addr + 8);
    const uint4 v{a.x, a.y, b.x, b.y};
    return __builtin_bit_cast(short8, v);
}

__global__ __launch_bounds__(NW * WAVE_SIZE) void vae_attn_kernel(VaeAttnParams p) {
    __shared__ char k_lds[KVB * K_ROW];          // [t][d]; reused as o_lds
    __shared__ char vt_lds[C * VT_ROW];          // [d][t]
    __shared__ char p_lds[QT * P_ROW];           // [q][t] bf16
    // split-K: waves ds_add_f32 their partials into ONE tile (4 KB; the
    // per-wave-slab version cost 16 KB and dropped occupancy to 1 block/CU)
    __shared__ float s_red[KVB][QT];
    __shared__ float m_lds[QT], l_lds[QT], corr_lds[QT];

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
    short8 qf[KS];
    {
        const int64_t qrow = q0 + lo < p.L ? q0 + lo : p.L - 1;
        const uint16_t* qp = qb + qrow * C + wave * DSL;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks)
            qf[ks] = *reinterpret_cast<const short8*>(qp + ks * 16 + hi * 8);
    }
    if (tid < QT) {
        m_lds[tid] = -1e30f;
        l_lds[tid] = 0.f;
    }
    float16v ot[DT] = {};

    const int n_tiles = (int)((p.L + KVB - 1) / KVB);
    for (int tile = 0; tile < n_tiles; ++tile) {
        const int64_t t0 = (int64_t)tile * KVB;
        // zero the split-K tile while staging
        for (int c = tid; c < KVB * QT; c += NW * WAVE_SIZE)
            reinterpret_cast<float*>(s_red)[c] = 0.f;
        // ---- stage K [t][d] and V^T [d][t] --------------------------------
        {
            // K: 32 rows x 64 x 16 B chunks = 2048 items / 256 threads
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
            // V^T via the DPP butterfly: each pass covers 8 tokens x 64 d
            // per wave; 4 waves x VPASS passes x 8 d-blocks... iterate
            // (token-window, d-block) pairs: windows = KVB/8 = 4,
            // d-blocks = C/64 = 8 -> 32 slots / 4 waves = 8 per wave.
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
        float16v s = {};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
            const short8 kf = lds_frag_b64x2(
                &k_lds[lo * K_ROW + (wave * DSL + ks * 16 + hi * 8) * 2]);
            s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ks], s, 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
        // lane holds S^T rows t = (r&3)+8*(r>>2)+4*hi for q col lo
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            const int t = (r & 3) + 8 * (r >> 2) + 4 * hi;
            atomicAdd(&s_red[t][lo], s[r]);
        }
        __syncthreads();

        // ---- split-K reduce + shared online softmax -----------------------
        // thread -> (q = tid/8, 4 tokens); groups of 8 consecutive threads
        // share one q.
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
            // per-q max across the 8 threads (3 xor-shuffles)
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
                const float corr = __builtin_amdgcn_exp2f((m_old - m_new) * scale2);
                corr_lds[q] = m_old == -1e30f ? 1.f : corr;
                l_lds[q] = l_lds[q] * (m_old == -1e30f ? 1.f : corr) + tsum;
                m_lds[q] = m_new;
            }
        }
        __syncthreads();

        // ---- rescale + PV for this wave's d-slice -------------------------
        {
            const float corr = corr_lds[lo];  // lane's q column
            if (corr != 1.f) {
#pragma unroll
                for (int dt = 0; dt < DT; ++dt)
#pragma unroll
                    for (int r = 0; r < 16; ++r) ot[dt][r] *= corr;
            }
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
                for (int kt = 0; kt < 2; ++kt) {
                    const int d = wave * DSL + dt * 32 + lo;
                    const short8 vf = lds_frag_b64x2(
                        &vt_lds[d * VT_ROW + (kt * 16 + hi * 8) * 2]);
                    const short8 pf = lds_frag_b64x2(
                        &p_lds[lo * P_ROW + (kt * 16 + hi * 8) * 2]);
                    // A = V^T rows d, B = P^T cols q -> O^T[d][q]
                    ot[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pf, ot[dt], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
    }

    // ---- epilogue: O^T -> LDS [q][d] -> coalesced [L][512] stores ---------
    char* o_lds = k_lds;  // reuse (32 x O_ROW = 33 KB fits the K buffer)
    {
#pragma unroll
        for (int dt = 0; dt < DT; ++dt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int d = wave * DSL + dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                const float inv = l_lds[lo] > 0.f ? 1.f / l_lds[lo] : 0.f;
                reinterpret_cast<bf16_t*>(&o_lds[lo * O_ROW])[d] =
                    __float2bfloat16(ot[dt][r] * inv);
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
