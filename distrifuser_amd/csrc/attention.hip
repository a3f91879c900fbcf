// Flash-attention forward, gfx950, bf16, head_dim 64 (K1/K2/K8 of SURVEY
// §2.4a): rectangular local-query x global-(stale)-KV attention.
//
// Structure (cdna_hip_programming.md §B "8-warp 32x32 ladder"; evolution and
// per-step measurements in profiles/attention_ladder_r01.md):
// * 8-wave workgroup per (batch, head, 256-row Q tile); each wave owns 32 Q
//   rows held in registers (4 k-slice A/B fragments). Q-block size governs
//   KV re-read traffic (each KV tile is re-read Lq/QBLK times).
// * KV streamed in 64-token LDS tiles shared by the waves; K row-major
//   [64][64], V TRANSPOSED [d][t]; XOR swizzle byte^=((row&7)<<4) breaks the
//   16-way bank conflict of row-major [.][64] bf16 tiles (guide §6 G4).
// * SWAPPED QK^T: S^T = K_tile x Q^T via mfma_f32_32x32x16_bf16, so each
//   lane holds 16 score rows of ONE q column (q = lane&31) — the online-
//   softmax max/sum are 15 in-register ops + ONE shfl_xor(32) with the
//   partner lane, instead of 30+ cross-lane shuffles per tile (guide
//   common-mistake #6 / §B "swapped QK^T").
// * P stays in registers: v_cvt_pk_bf16_f32 packs + shfl_xor(32) partner
//   exchange build the PV B-fragments directly (guide T12), no P LDS
//   round-trip.
// * PV is also swapped: O^T = V^T x P^T accumulates in 2x16 AGPRs per lane.
// * stale-KV chunking: KV tokens come from NC flat-comm-buffer chunks of LC
//   tokens (k_sc/v_sc strides) — the displaced-patch KV is consumed in place
//   with zero torch.cat.
//
// Fragment maps (verified on MI355X by tests/test_ops_gpu.py probes):
//   16x16x32: A[row=l&15][k=(l>>4)*8+j]  B[k][col=l&15]  D[row=(l>>4)*4+r][col=l&15]
//   32x32x16: A[row=l&31][k=(l>>5)*8+j]  B[k][col=l&31]
//             D[row=(r&3)+8*(r>>2)+4*(l>>5)][col=l&31]

#include "common.h"
#include "kernels.h"

namespace {

constexpr int QW = 32;     // q rows per wave
// waves per block is a template parameter: bigger Q blocks amortize the KV
// stream (each KV tile is re-read Lq/QBLK times), smaller blocks keep small
// Lq shapes filled. Measured: 4w->8w at L=57.6k was +53% (439->673 TF).
// KVB (kv tokens per LDS tile) and DPAD (head_dim padded to a multiple of
// 32) are template parameters: KVB=128 amortizes staging barriers, KVB=64
// keeps Lkv % 128 == 64 shapes (SDXL's 14400-token stage) mask-free; DPAD
// covers the SD-family head dims (SDXL/SD2: 64; SD1.5: 40->64, 80->96,
// 160). The real head_dim rides in FlashAttnParams.Dh; padding lanes carry
// zeros (exact for QK^T scores and O columns < Dh).

typedef float float4v_ __attribute__((ext_vector_type(4)));
typedef float float16v __attribute__((ext_vector_type(16)));

template <int ROW_BYTES>
__device__ __forceinline__ int vt_swz(int row, int byte_off) {
    // XOR over all 16 B windows of a power-of-two LDS row: spreads same-
    // column reads across banks (16 windows at 256 B rows is a free 2-way
    // conflict per guide m136; 8 windows at 128 B rows is the 4-way floor)
    static_assert((ROW_BYTES & (ROW_BYTES - 1)) == 0);
    return byte_off ^ ((row & (ROW_BYTES / 16 - 1)) << 4);
}

__device__ __forceinline__ uint32_t cvt_pk_bf16(float a, float b) {
    uint32_t r;
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
    return r;
}

template <int NW, int DPAD, int KVB, bool MASK, bool DEFER>
__global__ __launch_bounds__(NW * WAVE_SIZE) void flash_attn_kernel(FlashAttnParams p) {
    constexpr int QBLK = QW * NW;
    constexpr int VT_ROW = KVB * 2;         // V^T LDS row bytes [d][t]
    // K rows padded by 8 B: row stride DPAD*2+8 gives gcd(stride/4, 32) = 2,
    // so the 32-consecutive-row column reads are 2-way (free) instead of the
    // 4-way floor of XOR-swizzled 128 B rows — same trick as the conv
    // kernel's staged tile (profiles/conv_ladder_r02.md v5). Reads/writes
    // are b64 pairs (rows are 8 B- but not 16 B-aligned).
    constexpr int K_ROW = DPAD * 2 + 8;     // padded K LDS row bytes [t][d]
    constexpr int KS = DPAD / 16;           // QK^T k-slices
    constexpr int DT = DPAD / 32;           // PV / O^T d-tiles
    __shared__ char k_lds[KVB * K_ROW];     // [t][d] bf16, padded rows
    __shared__ char vt_lds[DPAD * VT_ROW];  // [d][t] bf16, swizzled rows

    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lo = lane & 31;   // q column of this lane
    const int hi = lane >> 5;   // partner-half index

    const int bh = blockIdx.y;
    const int b = bh / p.H;
    const int h = bh % p.H;
    const int64_t q0 = (int64_t)blockIdx.x * QBLK;

    const int64_t Lkv = p.NC * p.LC;
    const uint16_t* qbase = p.q + b * p.q_sb + h * p.q_sh;
    const uint16_t* kbase = p.k + b * p.k_sb + h * p.k_sh;
    const uint16_t* vbase = p.v + b * p.v_sb + h * p.v_sh;

    // ---- Q fragments: qf[ks] = Q[q = q0 + wave*32 + lo][d = ks*16 + hi*8 ..]
    // (8-element chunks beyond Dh are zero — Dh % 8 == 0 host-checked) ----
    short8 qf[KS];
    const int64_t qrow = q0 + wave * QW + lo;
    const bool qvalid = qrow < p.Lq;
    {
        const uint16_t* qp = qbase + (qvalid ? qrow : (p.Lq - 1)) * p.q_sl;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
            const int d0 = ks * 16 + hi * 8;
            if (p.Dh == DPAD || d0 + 8 <= p.Dh)
                qf[ks] = *reinterpret_cast<const short8*>(qp + d0);
            else
                qf[ks] = short8{0, 0, 0, 0, 0, 0, 0, 0};
        }
    }

    // Online softmax bookkeeping (exp2 domain, guide §B):
    //   m_raw   running max of RAW scores (max commutes with the positive
    //           scale, so per-element work is p = exp2(fma(s, scale2, -msc))
    //           — ONE v_fma + ONE v_exp per score element)
    //   msc     m_raw * scale2 (refreshed only when m_raw moves)
    //   defer-max (guide T13): skip the O/l rescale while the tile max stays
    //   within DEFER_THR (log2 units) of the running max
    float m_raw = -1e30f;
    float msc = -1e30f;
    float l_run = 0.f;
    float16v ot[DT] = {};  // O^T tiles: [dt] -> rows d = dt*32 + crow(r,hi), col q=lo
    const float scale2 = p.scale * 1.44269504088896340736f;
    const float defer_raw = 11.0f / scale2;

    const int n_tiles = (int)((Lkv + KVB - 1) / KVB);
    for (int tile = 0; tile < n_tiles; ++tile) {
        const int64_t t0 = (int64_t)tile * KVB;
        // ---- cooperative staging ----
        // K [t][d]: thread (t-row, 16B column) -> b128 LDS write.
        // V^T [d][t]: lane = d COLUMN, 8 consecutive tokens per pass — the
        // global reads stay perfectly coalesced (64 lanes x 2B = one 128B row
        // per token) and the LDS write becomes ONE b128 per lane instead of
        // 8 scalar ds_write_b16 (the transpose-write conflicts were 15% of
        // kernel time — see profiles/attention_ladder_r01.md ablation).
        {
            constexpr int KCOLS = DPAD / 8;  // 16 B chunks per K row
            constexpr int NCHUNK = KVB * KCOLS;
#pragma unroll
            for (int c = tid; c < NCHUNK; c += NW * WAVE_SIZE) {
                const int t_local = c / KCOLS;
                const int d8 = c % KCOLS;
                const int64_t t_glob = t0 + t_local;
                uint4 kraw = {0, 0, 0, 0};
                const bool d_ok = (p.Dh == DPAD) || (d8 * 8 + 8 <= p.Dh);
                if (d_ok && (!MASK || t_glob < Lkv)) {
                    const int64_t chunk = t_glob / p.LC;
                    const int64_t tin = t_glob % p.LC;
                    kraw = *reinterpret_cast<const uint4*>(
                        kbase + chunk * p.k_sc + tin * p.k_sl + d8 * 8);
                }
                char* kdst = &k_lds[t_local * K_ROW + d8 * 16];
                *reinterpret_cast<uint2*>(kdst) = uint2{kraw.x, kraw.y};
                *reinterpret_cast<uint2*>(kdst + 8) = uint2{kraw.z, kraw.w};
            }
            // V^T: wave w stages token rows [w*8, w*8+8) of each 8*NW-row pass
            constexpr int VROWS_PER_PASS = NW * 8;
#pragma unroll
            for (int rep = 0; rep < KVB / VROWS_PER_PASS; ++rep) {
                const int tb_local = rep * VROWS_PER_PASS + wave * 8;
                const int64_t tb_glob = t0 + tb_local;
                // tb_glob is a multiple of 8 and LC % 8 == 0 (host-checked),
                // so the 8-token window lies in one chunk
                const int64_t chunk = tb_glob / p.LC;
                const int64_t tin = tb_glob % p.LC;
                for (int d = lane; d < DPAD; d += WAVE_SIZE) {
                    uint16_t ve[8];
                    const uint16_t* vp = vbase + chunk * p.v_sc + tin * p.v_sl + d;
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const bool ok = d < p.Dh && (!MASK || tb_glob + j < Lkv);
                        ve[j] = ok ? vp[j * p.v_sl] : (uint16_t)0;
                    }
                    *reinterpret_cast<uint4*>(
                        &vt_lds[d * VT_ROW + vt_swz<VT_ROW>(d, tb_local * 2)]) =
                        *reinterpret_cast<const uint4*>(ve);
                }
            }
        }
        __syncthreads();

        // T15 att[2] double-pipeline (guide): while subtile st's softmax/PV
        // runs on the VALU, subtile st+1's QK^T fills the OTHER score tile on
        // the MFMA pipe (separate pipes -> free overlap). Static two-state
        // ping-pong via full unroll (rule #20: no dynamic indexing).
        auto qk_tile = [&](int st) {
            float16v s = {};
            __builtin_amdgcn_s_setprio(1);  // favor the MFMA wave (guide T5)
#pragma unroll
            for (int ks = 0; ks < KS; ++ks) {
                const int t = st * 32 + lo;
                const char* ksrc = &k_lds[t * K_ROW + (ks * 16 + hi * 8) * 2];
                const uint2 k0 = *reinterpret_cast<const uint2*>(ksrc);
                const uint2 k1 = *reinterpret_cast<const uint2*>(ksrc + 8);
                const uint4 kk{k0.x, k0.y, k1.x, k1.y};
                const short8 kfrag = __builtin_bit_cast(short8, kk);
                s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag, qf[ks], s, 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
            return s;
        };
        float16v s_cur = qk_tile(0);
#pragma unroll
        for (int st = 0; st < KVB / 32; ++st) {  // 32-token sub-tiles
            float16v s_nxt;
            if (st + 1 < KVB / 32) s_nxt = qk_tile(st + 1);
            const float16v s = s_cur;
            // lane holds S^T rows crow(r) = (r&3)+8*(r>>2)+4*hi for q col lo
            float tm = -1e30f;
            float pv[16];
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                float v = s[r];
                if (MASK) {
                    const int crow = (r & 3) + 8 * (r >> 2) + 4 * hi;
                    if (t0 + st * 32 + crow >= Lkv) v = -1e30f;
                }
                pv[r] = v;
                tm = fmaxf(tm, v);
            }
            tm = fmaxf(tm, __shfl_xor(tm, 32, 64));  // partner holds the other 16 rows

            if (!DEFER || __any(m_raw == -1e30f || tm > m_raw + defer_raw)) {
                const float m_new = fmaxf(m_raw, tm);
                const float msc_new = m_new * scale2;
                const float corr = __builtin_amdgcn_exp2f(msc - msc_new);
                m_raw = m_new;
                msc = msc_new;
                l_run *= corr;
#pragma unroll
                for (int dt = 0; dt < DT; ++dt)
#pragma unroll
                    for (int r = 0; r < 16; ++r) ot[dt][r] *= corr;
            }
            float tsum = 0.f;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                pv[r] = __builtin_amdgcn_exp2f(__builtin_fmaf(pv[r], scale2, -msc));
                tsum += pv[r];
            }
            tsum += __shfl_xor(tsum, 32, 64);
            l_run += tsum;

            // ---- pack P to bf16 B-fragments (guide T12: cvt_pk + partner
            // exchange); B frag j=0..7 -> P^T rows kt*16 + hi*8 + j ----
            uint32_t w[8], wp[8];
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                w[i] = cvt_pk_bf16(pv[2 * i], pv[2 * i + 1]);
                wp[i] = __shfl_xor(w[i], 32, 64);
            }
            short8 pb[2];
#pragma unroll
            for (int kt = 0; kt < 2; ++kt) {
                // own word i holds regs {2i,2i+1} = rows {(2i&3)+8*(2i>>2)+4*hi, +1};
                // hi=0 needs [own 4kt,4kt+1 | partner 4kt,4kt+1], hi=1 the mirror.
                uint32_t* pbw = reinterpret_cast<uint32_t*>(&pb[kt]);
                if (hi == 0) {
                    pbw[0] = w[4 * kt + 0];
                    pbw[1] = w[4 * kt + 1];
                    pbw[2] = wp[4 * kt + 0];
                    pbw[3] = wp[4 * kt + 1];
                } else {
                    pbw[0] = wp[4 * kt + 2];
                    pbw[1] = wp[4 * kt + 3];
                    pbw[2] = w[4 * kt + 2];
                    pbw[3] = w[4 * kt + 3];
                }
            }

            // ---- O^T += V^T x P^T ----
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
                for (int kt = 0; kt < 2; ++kt) {
                    const int d = dt * 32 + lo;
                    short8 vf = *reinterpret_cast<const short8*>(
                        &vt_lds[d * VT_ROW + vt_swz<VT_ROW>(d, (st * 32 + kt * 16 + hi * 8) * 2)]);
                    ot[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pb[kt], ot[dt], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);
            s_cur = s_nxt;
        }
        __syncthreads();
    }

    // ---- epilogue: O[q][d] = O^T / l (only the real head_dim columns) ----
    if (qvalid) {
        const float inv = l_run > 0.f ? 1.f / l_run : 0.f;
        uint16_t* op = p.o + ((int64_t)b * p.Lq + qrow) * (p.H * p.Dh) + (int64_t)h * p.Dh;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                if (p.Dh == DPAD || d < p.Dh)
                    op[d] = __builtin_bit_cast(uint16_t, __float2bfloat16(ot[dt][r] * inv));
            }
    }
}

}  // namespace

template <int NW, int DPAD, int KVB, bool MASK>
static void launch_var(const FlashAttnParams& p, hipStream_t stream) {
    const int qblk = QW * NW;
    dim3 grid((unsigned)((p.Lq + qblk - 1) / qblk), (unsigned)(p.B * p.H));
    dim3 block(NW * WAVE_SIZE);
    flash_attn_kernel<NW, DPAD, KVB, MASK, true><<<grid, block, 0, stream>>>(p);
}

template <int DPAD>
static void launch_dpad(const FlashAttnParams& p, hipStream_t stream) {
    const int64_t Lkv = p.NC * p.LC;
    if (Lkv % 128 == 0)
        launch_var<8, DPAD, 128, false>(p, stream);
    else if (Lkv % 64 == 0)
        launch_var<8, DPAD, 64, false>(p, stream);
    else
        launch_var<8, DPAD, 128, true>(p, stream);
}

void launch_flash_attention_d64(const FlashAttnParams& p, hipStream_t stream) {
    // 8-wave (256-row) blocks measured best across the SD-family shapes;
    // KVB prefers 128-token tiles when the sequence tiles evenly, else 64
    // (still mask-free for Lkv % 64 == 0, e.g. SDXL's 14400-token stage),
    // else the masked variant. DPAD covers the SD-family head dims.
    if (p.Dh <= 64)
        launch_dpad<64>(p, stream);
    else if (p.Dh <= 96)
        launch_dpad<96>(p, stream);
    else if (p.Dh <= 128)
        launch_dpad<128>(p, stream);
    else
        launch_dpad<160>(p, stream);
}

// ---- fragment-layout probes (tests/test_ops_gpu.py) ------------------------
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
    float4v_ acc = {};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) d[lane * 4 + r] = acc[r];
}

__global__ void mfma_probe32_kernel(const float* __restrict__ a, const float* __restrict__ b,
                                    float* __restrict__ d) {
    const int lane = threadIdx.x;
    short8 af, bf;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        af[j] = __builtin_bit_cast(short, __float2bfloat16(a[lane * 8 + j]));
        bf[j] = __builtin_bit_cast(short, __float2bfloat16(b[lane * 8 + j]));
    }
    float16v acc = {};
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 16; ++r) d[lane * 16 + r] = acc[r];
}
}  // namespace

void launch_mfma_probe(const float* a, const float* b, float* d, hipStream_t stream) {
    mfma_probe_kernel<<<1, WAVE_SIZE, 0, stream>>>(a, b, d);
}

void launch_mfma_probe32(const float* a, const float* b, float* d, hipStream_t stream) {
    mfma_probe32_kernel<<<1, WAVE_SIZE, 0, stream>>>(a, b, d);
}
