#include "hip/hip_runtime.h"
// Flash-attention forward, gfx950, bf16, head_dim 64 (K1/K2/K8 of SURVEY
// §2.4a): rectangular local-query x global-(stale)-KV attention.
//
// Design (cdna_hip_programming.md §B "fused attention prefill" ladder):
// * one 4-wave workgroup per (batch, head, 64-row Q tile); each wave owns a
//   16-row Q sub-tile, Q held in registers (2 MFMA A-fragments per lane).
// * KV streamed in 64-token LDS tiles shared by the 4 waves; K row-major
//   [64][64], V stored TRANSPOSED [64 d][64 t] so both QK^T and PV read
//   contiguous 16B B-fragments (ds_read_b128).
// * LDS XOR swizzle byte ^= ((row & 7) << 4) on every row-strided buffer —
//   row-major [.][64] bf16 tiles are a 16-way bank conflict otherwise
//   (guide §6 Guideline 4).
// * online softmax entirely wave-parallel: the MFMA C-layout puts one kv
//   column per lane (col = lane&15), so row max/sum are __shfl_xor(1,2,4,8)
//   reductions — no serial-lane softmax (guide common-mistake #6).
// * stale-KV chunking: KV tokens come from NC flat-comm-buffer chunks of LC
//   tokens (k_sc/v_sc chunk strides), so the displaced-patch KV is consumed
//   in place with zero torch.cat.
//
// MFMA v_mfma_f32_16x16x32_bf16 fragment maps (A/B assumed per CDNA ISA, C/D
// verified in the guide §3): A: [row=l&15][k=(l>>4)*8+j]; B: [k=(l>>4)*8+j]
// [col=l&15]; C/D: [row=(l>>4)*4+r][col=l&15]. tests/test_ops_gpu.py checks
// the whole kernel against fp32 SDPA.

#include "common.h"
#include "kernels.h"

namespace {

constexpr int QB = 16;     // q rows per wave
constexpr int WAVES = 4;   // waves per block
constexpr int QBLK = QB * WAVES;
constexpr int KVB = 64;    // kv tokens per LDS tile
constexpr int D = 64;

__device__ __forceinline__ int swz(int row, int byte_off) {
    return byte_off ^ ((row & 7) << 4);
}

typedef float float4v_ __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(WAVES * WAVE_SIZE) void flash_attn_d64_kernel(FlashAttnParams p) {
    __shared__ char k_lds[KVB * D * 2];       // [t][d] bf16, swizzled rows
    __shared__ char vt_lds[D * KVB * 2];      // [d][t] bf16, swizzled rows
    __shared__ char p_lds[WAVES][QB * KVB * 2];  // per-wave P tile [q][t]

    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lo = lane & 15;
    const int hi = lane >> 4;

    const int bh = blockIdx.y;  // b * H + h
    const int b = bh / p.H;
    const int h = bh % p.H;
    const int64_t q0 = (int64_t)blockIdx.x * QBLK;  // first q row of the block

    const int64_t Lkv = p.NC * p.LC;
    const uint16_t* qbase = p.q + b * p.q_sb + h * p.q_sh;
    const uint16_t* kbase = p.k + b * p.k_sb + h * p.k_sh;
    const uint16_t* vbase = p.v + b * p.v_sb + h * p.v_sh;

    // ---- load Q fragments (row = q0 + wave*16 + lo; d = hi*8 + 32*ks) ----
    short8 qfrag[2];
    const int64_t qrow = q0 + wave * QB + lo;
    const bool qvalid = qrow < p.Lq;
    {
        const uint16_t* qp = qbase + (qvalid ? qrow : (p.Lq - 1)) * p.q_sl;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
            qfrag[ks] = *reinterpret_cast<const short8*>(qp + hi * 8 + 32 * ks);
    }

    // online-softmax state: this lane participates in rows hi*4 + r
    float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
    float l_run[4] = {0.f, 0.f, 0.f, 0.f};
    float4v_ oacc[4];  // [dblk] -> D[row=(hi*4+r)][dcol = dblk*16 + lo]
#pragma unroll
    for (int i = 0; i < 4; ++i) oacc[i] = {0.f, 0.f, 0.f, 0.f};

    const int n_tiles = (int)((Lkv + KVB - 1) / KVB);
    for (int tile = 0; tile < n_tiles; ++tile) {
        const int64_t t0 = (int64_t)tile * KVB;
        // ---- stage K [t][d] and V^T [d][t] (256 threads cooperative) ----
        {
            const int tl = tid / 8;        // 0..31: local token row pair index
            const int d8 = tid % 8;        // 16B column
#pragma unroll
            for (int rep = 0; rep < 2; ++rep) {
                const int t_local = tl + rep * 32;
                const int64_t t_glob = t0 + t_local;
                uint4 kraw = {0, 0, 0, 0}, vraw = {0, 0, 0, 0};
                if (t_glob < Lkv) {
                    const int64_t chunk = t_glob / p.LC;
                    const int64_t tin = t_glob % p.LC;
                    kraw = *reinterpret_cast<const uint4*>(
                        kbase + chunk * p.k_sc + tin * p.k_sl + d8 * 8);
                    vraw = *reinterpret_cast<const uint4*>(
                        vbase + chunk * p.v_sc + tin * p.v_sl + d8 * 8);
                }
                *reinterpret_cast<uint4*>(&k_lds[t_local * 128 + swz(t_local, d8 * 16)]) = kraw;
                const uint16_t* ve = reinterpret_cast<const uint16_t*>(&vraw);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const int d = d8 * 8 + j;
                    *reinterpret_cast<uint16_t*>(&vt_lds[d * 128 + swz(d, t_local * 2)]) = ve[j];
                }
            }
        }
        __syncthreads();

        // ---- S = scale * Q K^T over 4 16-col blocks ----
        // A = Q tile: lane reads Q[row=l&15][d=(l>>4)*8+j] (qfrag).
        // B = K^T:    lane reads K^T[d=(l>>4)*8+j][t=l&15] = K[t][d] (kfrag;
        //             same lane arithmetic as A, t from the 16-col block).
        // D[row=q][col=t] lands at [row=(l>>4)*4+r][col=l&15].
        float4v_ s[4];
#pragma unroll
        for (int blk = 0; blk < 4; ++blk) {
            float4v_ acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                const int t = blk * 16 + lo;
                short8 kfrag = *reinterpret_cast<const short8*>(
                    &k_lds[t * 128 + swz(t, (hi * 8 + 32 * ks) * 2)]);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], kfrag, acc, 0, 0, 0);
            }
            s[blk] = acc;
        }

        // ---- masking + online softmax (rows hi*4+r, col lo per blk) ----
        float pmax[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) pmax[r] = -1e30f;
#pragma unroll
        for (int blk = 0; blk < 4; ++blk) {
            const int64_t col = t0 + blk * 16 + lo;
            const bool valid = col < Lkv;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float v = valid ? s[blk][r] * p.scale : -1e30f;
                s[blk][r] = v;
                pmax[r] = fmaxf(pmax[r], v);
            }
        }
        // row reduce across the 16 lanes of the row group
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
#pragma unroll
            for (int r = 0; r < 4; ++r) pmax[r] = fmaxf(pmax[r], __shfl_xor(pmax[r], off, 64));

        float lsum[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const float m_new = fmaxf(m_run[r], pmax[r]);
            const float corr = __expf(m_run[r] - m_new);
            m_run[r] = m_new;
            l_run[r] *= corr;
#pragma unroll
            for (int dblk = 0; dblk < 4; ++dblk) oacc[dblk][r] *= corr;
            lsum[r] = 0.f;
        }
        // P = exp(S - m); write bf16 P tile to this wave's LDS
#pragma unroll
        for (int blk = 0; blk < 4; ++blk) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const float e = __expf(s[blk][r] - m_run[r]);
                lsum[r] += e;
                const int prow = hi * 4 + r;
                const int pcol = blk * 16 + lo;
                *reinterpret_cast<uint16_t*>(
                    &p_lds[wave][prow * 128 + swz(prow, pcol * 2)]) =
                    __builtin_bit_cast(uint16_t, __float2bfloat16(e));
            }
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
#pragma unroll
            for (int r = 0; r < 4; ++r) lsum[r] += __shfl_xor(lsum[r], off, 64);
#pragma unroll
        for (int r = 0; r < 4; ++r) l_run[r] += lsum[r];

        // ---- O += P V : A = P[q=lo][t=hi*8+32ks+j], B = V^T[d=lo+16dblk][t] ----
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                const int prow = lo;
                short8 pfrag = *reinterpret_cast<const short8*>(
                    &p_lds[wave][prow * 128 + swz(prow, (hi * 8 + 32 * ks) * 2)]);
                const int vrow = dblk * 16 + lo;
                short8 vfrag = *reinterpret_cast<const short8*>(
                    &vt_lds[vrow * 128 + swz(vrow, (hi * 8 + 32 * ks) * 2)]);
                oacc[dblk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, oacc[dblk], 0, 0, 0);
            }
        }
        __syncthreads();  // K/V LDS reused next tile
    }

    // ---- epilogue: divide by l, store o[b, qrow, h, d] ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int64_t row = q0 + wave * QB + hi * 4 + r;
        if (row >= p.Lq) continue;
        const float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk) {
            const int64_t off =
                ((int64_t)b * p.Lq + row) * (p.H * D) + (int64_t)h * D + dblk * 16 + lo;
            p.o[off] = __builtin_bit_cast(uint16_t, __float2bfloat16(oacc[dblk][r] * inv));
        }
    }
}

}  // namespace

void launch_flash_attention_d64(const FlashAttnParams& p, hipStream_t stream) {
    dim3 grid((unsigned)((p.Lq + QBLK - 1) / QBLK), (unsigned)(p.B * p.H));
    dim3 block(WAVES * WAVE_SIZE);
   hipLaunchKernelGGL(( flash_attn_d64_kernel), dim3(grid), dim3(block), 0, stream, p);
}

// ---- fragment-layout probe (tests/test_ops_gpu.py verifies the A/B maps) ---
namespace {
__global__ void mfma_probe_kernel(const float* __restrict__ a, const float* __restrict__ b,
                                  float* __restrict__ d) {
    const int lane = threadIdx.x;  // single wave
    short8 af, bf;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        af[j] = __builtin_bit_cast(short, __float2bfloat16(a[lane * 8 + j]));
        bf[j] = __builtin_bit_cast(short, __float2bfloat16(b[lane * 8 + j]));
    }
    float4v_ acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) d[lane * 4 + r] = acc[r];
}
}  // namespace

void launch_mfma_probe(const float* a, const float* b, float* d, hipStream_t stream) {
   hipLaunchKernelGGL(( mfma_probe_kernel), dim3(1), dim3(WAVE_SIZE), 0, stream, a, b, d);
}
