// Implicit-GEMM 3x3 convolution, gfx950, bf16, NCHW (K4/K5/K10 of SURVEY
// §2.4a): per-tap GEMM accumulation — the conv is 9 shifted GEMMs
// O^T[cout][pix] += W^T_tap[cout][cin] x I[cin][pix + tap-shift] — so no
// im2col materialization and no torch.cat/F.pad halo assembly (the
// reference's concat+pad, /root/reference/distrifuser/modules/pp/conv2d.py:
// 72-88, is pure overhead): halo rows are read IN PLACE from the comm
// buffer via the top/bot pointers.
//
// Structure (mirrors the attention kernel's measured idioms,
// profiles/attention_ladder_r01.md):
// * SWAPPED operands: D[m=cout][n=pixel] = A[cout][cin] x B[cin][pixel]
//   via mfma_f32_32x32x16_bf16, so output stores are x-contiguous
//   (32 lanes = 64 B rows) and the WEIGHT operand is the register-friendly
//   one: weights are prepacked host-side into exact per-lane fragment order
//   ([tap][ks][ct][lane][8] — each fragment load is one coalesced 1 KB
//   wave read, L2-resident across the grid).
// * The input tile is staged once per cin-slice into LDS TRANSPOSED
//   ([pixel-row][cin], b128-packed writes — the attention V^T staging
//   pattern: global reads stay 128 B-coalesced along x, and each lane
//   writes one b128 instead of 8 scalar ds_write_b16).
// * B-fragment reads are row reads of the transposed tile; the XOR window
//   swizzle spreads consecutive-row reads across the 16 B windows of each
//   row (the 4-way floor of 128 B rows, same as the attention K tile).
// * stride 2 splits the staged tile into x-parity PLANES so the
//   stride-2 fragment reads become stride-1 row reads (no extra conflicts).
// * Bias is deferred to the epilogue (fused add on the accumulators), which
//   is also where per-channel SiLU could fuse later.
//
// Reference semantics matched: pp/conv2d.py:20-41 (sliced conv_in — same
// kernel, interior pointer offset into the full latent), :59-112 (halo
// exchange, zero-pad at true image borders).

#include "common.h"
#include "kernels.h"

namespace {

typedef float float16v __attribute__((ext_vector_type(16)));
typedef int int32x4_ck __attribute__((ext_vector_type(4)));

// raw-buffer load with hardware bounds check (OOB -> 0); gfx9 descriptor
// word3 = 0x00020000 (same constant CK uses, /opt/rocm/include/ck/ck.hpp:82)
__device__ int32x4_ck llvm_amdgcn_raw_buffer_load_x4(
    int32x4_ck srsrc, int voffset, int soffset,
    int glc_slc) __asm("llvm.amdgcn.raw.buffer.load.v4i32");

__device__ __forceinline__ int32x4_ck make_srsrc(const void* base, uint32_t bytes) {
    const uint64_t a = (uint64_t)base;
    int32x4_ck r;
    r.x = (int)(uint32_t)a;
    r.y = (int)(uint32_t)(a >> 32);
    r.z = (int)bytes;
    r.w = 0x00020000;
    return r;
}


// S: conv stride; (YB rows) x (XW*32*PW x) output tile per block; CIN_T cins
// per staged LDS tile; each wave owns PW 32-x subtiles x NCT 32-cout tiles
// (A fragments reused across the PW subtiles, B across the NCT tiles).
//
// Measured evolution (kernel_bench, 1280x1280@120^2, see profiles/):
//   v1 scalar-gather staging ............ 120 TF (vmem instr bound)
//   v3 LDS-staged weights + PW=2 ........ 385 TF (A-frag L2 latency fixed)
//   v4 DPP transpose + unrolled staging .. 450 TF
//   v5 (this): staging-volume cuts — NCT=4 halves the input re-read
//   across cout-blocks; 8 B-padded LDS rows (stride 72 B, gcd(18,32)=2)
//   make fragment reads 2-way instead of 8-way conflicted as b64 pairs;
//   load bursts are issued in arrays before the transpose+write pass.
template <int S, int YB, int XW, int CIN_T, int NCT, int PW>
__global__ __launch_bounds__(YB * XW * WAVE_SIZE) void conv3x3_kernel(Conv3x3Params p) {
    constexpr int NW = YB * XW;
    constexpr int XB = XW * 32 * PW;
    constexpr int YIN = (YB - 1) * S + 3;          // staged input rows
    constexpr int XIN = (XB - 1) * S + 3;          // staged input x positions
    constexpr int XP = (S == 1) ? XIN : (XB + 1);  // rows per x-parity plane
    constexpr int KS_T = CIN_T / 16;               // k-slices per staged tile
    constexpr int ROW_P = CIN_T * 2 + 8;           // padded LDS row bytes
    constexpr int NFRAG = 9 * KS_T * NCT;          // weight frags per tile
    constexpr int IN_SZ = S * YIN * XP * ROW_P;
    constexpr int W_SZ = NFRAG * WAVE_SIZE * 16;
    // double-buffered: tile t+1 stages into buffer (t+1)&1 while tile t
    // computes from buffer t&1 (one barrier per tile; the staging loads'
    // HBM/L2 latency hides under the MFMA phase)
    __shared__ char in_lds[2 * IN_SZ];
    __shared__ char w_lds[2 * W_SZ];

    const int tid = threadIdx.x;
    const int wave = tid / WAVE_SIZE;
    const int lane = tid % WAVE_SIZE;
    const int lo = lane & 31;
    const int hi = lane >> 5;
    const int wy = wave / XW;
    const int wx = wave % XW;

    // XCD-aware decode of the 1-D grid: hardware assigns dispatch D to XCD
    // D%8, so giving the NCB cout-blocks of one pixel-tile consecutive D/8
    // slots at the same D%8 keeps them CONCURRENT ON ONE XCD — the staged
    // input tile and the cout-block's weights then live in that XCD's L2.
    // (Without this, PMC showed every staged byte fetched from HBM:
    // TCC_EA_RDREQ == staged volume, 1.45 GB/dispatch at 1280^2@120^2.)
    const int nxb = (p.Wo + XB - 1) / XB;
    const int nyb = (p.Ho + YB - 1) / YB;
    const int npix = nxb * nyb * p.B;  // pixel-tiles x batch
    const int ncb = (p.CT + NCT - 1) / NCT;
    const int D = blockIdx.x;
    const int cb = (D >> 3) % ncb;
    const int pp = (D & 7) + 8 * ((D >> 3) / ncb);
    if (pp >= npix) return;
    const int b = pp / (nxb * nyb);
    const int pt = pp % (nxb * nyb);
    const int xb0 = (pt % nxb) * XB;
    const int yb0 = (pt / nxb) * YB;

    const uint16_t* xin = p.x + (int64_t)b * p.x_sb;
    const uint16_t* top = p.top ? p.top + (int64_t)b * p.t_sb : nullptr;
    const uint16_t* bot = p.bot ? p.bot + (int64_t)b * p.b_sb : nullptr;

    float16v acc[NCT][PW] = {};

    // weights are packed to KS 16-cin slices (KS*16 a multiple of 64, zero
    // padded), so every staged tile is fully covered
    const int n_cin_tiles = (p.KS * 16 + CIN_T - 1) / CIN_T;
    constexpr int WTRIP = (NFRAG + NW - 1) / NW;
    constexpr int CG = CIN_T / 8;              // 8-cin groups
    constexpr int NXW = (XIN - 1 + 7) / 8;     // 8-x windows for xi >= 1
    constexpr int NXS = (NXW + 7) / 8;         // window strips per wave pass
    constexpr int NSLAB = YIN * CG * NXS;
    constexpr int STRIP = (NSLAB + NW - 1) / NW;

    // ---- v7 staging prologue: per-slab state hoisted out of the k-loop.
    // Interior slabs load through a buffer descriptor whose num_records
    // (= Cin * sc * 2 B) zeroes the cin tail in hardware, so the hot loop
    // has no per-lane masks and no 64-bit address math (the v6 staging asm
    // was saveexec/mad_i64-bound: ~21 us per 120 KB tile).
    const int gr = (lane & 3) | ((lane & 8) >> 1);
    const int gw = ((lane >> 2) & 1) | (((lane >> 4) & 3) << 1);
    const int32x4_ck desc0 = make_srsrc(xin, (uint32_t)((int64_t)p.Cin * p.x_sc * 2));
    const uint32_t voff_inc = (uint32_t)(CIN_T * p.x_sc * 2);
    const bool w_vec = (p.W % 8) == 0;
    uint32_t voff[STRIP];
    uint32_t interior_mask = 0, tail_mask = 0, valid_mask = 0;
#pragma unroll
    for (int sit = 0; sit < STRIP; ++sit) {
        const int slab = sit * NW + wave;
        voff[sit] = 0;
        if (NSLAB % NW != 0 && slab >= NSLAB) continue;
        valid_mask |= 1u << sit;
        const int xs = slab % NXS;
        const int cg = (slab / NXS) % CG;
        const int ry = slab / (NXS * CG);
        const int y_in = yb0 * S - 1 + ry;
        const int xw = xs * 8 + gw;
        const int x_in0 = xb0 * S + xw * 8;
        if (y_in >= 0 && y_in < p.H) interior_mask |= 1u << sit;
        // does any lane's window cross the right image edge? (x tail)
        if (xb0 * S + (xs * 8 + 7) * 8 + 8 > p.W) tail_mask |= 1u << sit;
        voff[sit] = (uint32_t)(((int64_t)(cg * 8 + gr) * p.x_sc +
                                (int64_t)(y_in >= 0 ? y_in : 0) * p.W + x_in0) * 2);
    }

    // ---- v8 pipeline phases ------------------------------------------------
    // Per tile t (while tile t-1 computes): weights stream straight to LDS
    // via global_load_lds DMA (no registers, wave-uniform LDS base +
    // lane*16 matches the linear fragment layout exactly); input slabs are
    // buffer-loaded into registers and written to LDS AFTER the compute
    // phase (T14 issue-early/write-late — __syncthreads drains vmcnt(0), so
    // in-flight loads cannot cross a barrier and the only overlap window is
    // within the iteration).
    auto stage_weights = [&](int cint, char* wb) {
        if (p.debug == 3) return;
#pragma unroll
        for (int it = 0; it < WTRIP; ++it) {
            const int frag = it * NW + wave;
            if (NFRAG % NW != 0 && frag >= NFRAG) break;
            const int ct2 = frag % NCT;
            const int ks2 = (frag / NCT) % KS_T;
            const int tap = frag / (NCT * KS_T);
            const int ct = cb * NCT + ct2;
            char* ldst = wb + (int64_t)frag * WAVE_SIZE * 16;
            if (ct < p.CT) {
                const uint16_t* g =
                    p.wp +
                    (((int64_t)(tap * p.KS + cint * KS_T + ks2) * p.CT + ct) * WAVE_SIZE + lane) *
                        8;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) uint32_t*)g,
                    (__attribute__((address_space(3))) uint32_t*)ldst, 16, 0, 0);
            } else {
                *reinterpret_cast<uint4*>(ldst + lane * 16) = uint4{0, 0, 0, 0};
            }
        }
    };
    uint4 iregs[STRIP];
    auto load_input = [&]() {
        if (p.debug == 1) return;
#pragma unroll
        for (int sit = 0; sit < STRIP; ++sit) {
            iregs[sit] = uint4{0, 0, 0, 0};
            if (!((valid_mask >> sit) & 1)) continue;
            if ((interior_mask >> sit) & 1 && w_vec) {
                iregs[sit] = __builtin_bit_cast(
                    uint4, llvm_amdgcn_raw_buffer_load_x4(desc0, (int)voff[sit], 0, 0));
                voff[sit] += voff_inc;
            }
        }
    };
    auto write_input = [&](int cint, char* inb) {
        if (p.debug == 1) return;
        const int cin0 = cint * CIN_T;
#pragma unroll
        for (int sit = 0; sit < STRIP; ++sit) {
            if (!((valid_mask >> sit) & 1)) continue;
            const int slab = sit * NW + wave;
            const int xs = slab % NXS;
            const int cg = (slab / NXS) % CG;
            const int ry = slab / (NXS * CG);
            const int xw = xs * 8 + gw;
            const int x_in0 = xb0 * S + xw * 8;
            uint4 raw = iregs[sit];
            if ((interior_mask >> sit) & 1) {
                if (w_vec) {
                    if ((tail_mask >> sit) & 1) {
                        // zero the x >= W elements of this lane's window
                        const int rem = p.W - x_in0;  // may be <= 0
                        uint32_t* dw = reinterpret_cast<uint32_t*>(&raw);
#pragma unroll
                        for (int d = 0; d < 4; ++d) {
                            const uint32_t m = (rem >= 2 * d + 2)
                                                   ? 0xffffffffu
                                                   : ((rem == 2 * d + 1) ? 0x0000ffffu : 0u);
                            dw[d] &= m;
                        }
                    }
                } else {
                    // W % 8 != 0: rows are not 16 B aligned; guarded scalar
                    // loads (correctness path, non-SD shapes)
                    const int cin = cin0 + cg * 8 + gr;
                    const int y_in = yb0 * S - 1 + ry;
                    if (cin < p.Cin) {
                        const uint16_t* rp = xin + (int64_t)cin * p.x_sc + (int64_t)y_in * p.W;
                        uint16_t vals[8];
#pragma unroll
                        for (int jj = 0; jj < 8; ++jj)
                            vals[jj] = (x_in0 + jj < p.W) ? rp[x_in0 + jj] : (uint16_t)0;
                        raw = *reinterpret_cast<const uint4*>(vals);
                    }
                }
            } else {
                // halo / zero rows (block at a patch boundary): rare
                const int y_in = yb0 * S - 1 + ry;
                const uint16_t* src = nullptr;
                int64_t sc = 0;
                if (y_in == -1 && top) {
                    src = top;
                    sc = p.t_sc;
                } else if (y_in == p.H && bot) {
                    src = bot;
                    sc = p.b_sc;
                }
                const int cin = cin0 + cg * 8 + gr;
                if (src && cin < p.Cin) {
                    const uint16_t* rp = src + (int64_t)cin * sc;
                    uint16_t vals[8];
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj)
                        vals[jj] = (x_in0 + jj < p.W) ? rp[x_in0 + jj] : (uint16_t)0;
                    raw = *reinterpret_cast<const uint4*>(vals);
                }
            }
            const uint4 tr = transpose8x8_bf16(raw, lane);
            const int xi = 1 + xw * 8 + gr;  // lane now owns x = xi
            if (xi < XIN) {
                const int plane = (S == 1) ? 0 : (xi & 1);
                const int row = (S == 1) ? xi : (xi >> 1);
                char* dst = &inb[((plane * YIN + ry) * XP + row) * ROW_P + cg * 16];
                *reinterpret_cast<uint2*>(dst) = uint2{tr.x, tr.y};
                *reinterpret_cast<uint2*>(dst + 8) = uint2{tr.z, tr.w};
            }
        }
        // left halo column xi = 0 (x_in = xb0*S - 1): scalar, tiny
        for (int c = tid; c < YIN * CIN_T; c += NW * WAVE_SIZE) {
            const int ci = c % CIN_T;
            const int ry = c / CIN_T;
            const int y_in = yb0 * S - 1 + ry;
            const int x_in = xb0 * S - 1;
            const int cin = cin0 + ci;
            uint16_t val = 0;
            if (x_in >= 0 && cin < p.Cin) {
                if (y_in >= 0 && y_in < p.H)
                    val = xin[(int64_t)cin * p.x_sc + (int64_t)y_in * p.W + x_in];
                else if (y_in == -1 && top)
                    val = top[(int64_t)cin * p.t_sc + x_in];
                else if (y_in == p.H && bot)
                    val = bot[(int64_t)cin * p.b_sc + x_in];
            }
            // xi = 0: plane 0, row 0 for both strides
            *reinterpret_cast<uint16_t*>(&inb[((0 * YIN + ry) * XP + 0) * ROW_P + ci * 2]) = val;
        }
    };
    auto compute_tile = [&](char* inb, char* wb) {
        if (p.debug != 2) {
            // ---- accumulate 9 taps x KS_T k-slices -----------------------
#pragma unroll
            for (int tap = 0; tap < 9; ++tap) {
                const int dy = tap / 3;
                const int dx = tap % 3;
                const int ry = wy * S + dy;
#pragma unroll
                for (int ks2 = 0; ks2 < KS_T; ++ks2) {
                    short8 bfrag[PW];
#pragma unroll
                    for (int px = 0; px < PW; ++px) {
                        const int xi = ((wx * PW + px) * 32 + lo) * S + dx;
                        const int plane = (S == 1) ? 0 : (xi & 1);
                        const int row = (S == 1) ? xi : (xi >> 1);
                        const char* baddr = &inb[((plane * YIN + ry) * XP + row) * ROW_P +
                                                    (ks2 * 16 + hi * 8) * 2];
                        const uint2 b0 = *reinterpret_cast<const uint2*>(baddr);
                        const uint2 b1 = *reinterpret_cast<const uint2*>(baddr + 8);
                        const uint4 bb{b0.x, b0.y, b1.x, b1.y};
                        bfrag[px] = __builtin_bit_cast(short8, bb);
                    }
#pragma unroll
                    for (int ct2 = 0; ct2 < NCT; ++ct2) {
                        const short8 afrag = *reinterpret_cast<const short8*>(
                            &wb[(((tap * KS_T + ks2) * NCT + ct2) * WAVE_SIZE + lane) * 16]);
#pragma unroll
                        for (int px = 0; px < PW; ++px)
                            acc[ct2][px] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                                afrag, bfrag[px], acc[ct2][px], 0, 0, 0);
                    }
                }
            }
        }
    };

    if (n_cin_tiles > 0) {
        stage_weights(0, w_lds);
        load_input();
        write_input(0, in_lds);
    }
    for (int cint = 0; cint < n_cin_tiles; ++cint) {
        char* inb = in_lds + (cint & 1) * IN_SZ;
        char* wb = w_lds + (cint & 1) * W_SZ;
        __syncthreads();  // staged tile cint visible; buffer cint+1 free
        const bool more = cint + 1 < n_cin_tiles;
        if (more) {
            stage_weights(cint + 1, w_lds + ((cint + 1) & 1) * W_SZ);
            load_input();
        }
        compute_tile(inb, wb);
        if (more) write_input(cint + 1, in_lds + ((cint + 1) & 1) * IN_SZ);
    }


    // ---- epilogue: O[b][cout][y][x] = acc + bias (+ residual) --------------
    const int y = yb0 + wy;
    if (y >= p.Ho) return;
#pragma unroll
    for (int px = 0; px < PW; ++px) {
        const int x = xb0 + (wx * PW + px) * 32 + lo;
        if (x >= p.Wo) continue;
        const int64_t pix_off = (int64_t)b * p.Cout * p.Ho * p.Wo + (int64_t)y * p.Wo + x;
        uint16_t* obase = p.o + pix_off;
        const uint16_t* rbase = p.residual ? p.residual + pix_off : nullptr;
#pragma unroll
        for (int ct2 = 0; ct2 < NCT; ++ct2) {
            const int ct = cb * NCT + ct2;
            if (ct >= p.CT) break;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int cout = ct * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
                if (cout < p.Cout) {
                    float v = acc[ct2][px][r];
                    if (p.bias) v += to_f32(reinterpret_cast<const bf16_t*>(p.bias)[cout]);
                    if (p.bias2)
                        v += to_f32(reinterpret_cast<const bf16_t*>(
                            p.bias2)[b * p.Cout + cout]);
                    if (rbase)
                        v += to_f32(reinterpret_cast<const bf16_t*>(
                            rbase)[(int64_t)cout * p.Ho * p.Wo]);
                    obase[(int64_t)cout * p.Ho * p.Wo] =
                        __builtin_bit_cast(uint16_t, __float2bfloat16(v));
                }
            }
        }
    }
}

}  // namespace

void launch_conv3x3(const Conv3x3Params& p, int stride, hipStream_t stream) {
    if (stride == 1) {
        // 8 waves: 8 output rows x 1 x-wave of 64 px (PW=2 A-frag reuse),
        // 128 couts per block (NCT=4 halves the input re-read);
        // LDS (double-buffered): 2 x (input 10x66x40 B + 36 KB weight frags).
        constexpr int YB = 8, XW = 1, NCT = 4, PW = 2;
        constexpr int XB = XW * 32 * PW;
        const int nxb = (p.Wo + XB - 1) / XB;
        const int nyb = (p.Ho + YB - 1) / YB;
        const int npix = nxb * nyb * p.B;
        const int ncb = (p.CT + NCT - 1) / NCT;
        dim3 grid((unsigned)(8 * ncb * ((npix + 7) / 8)));
        if (p.Wo <= 128) {
            // small-W config (SDXL's 120^2 mid/up shapes): 32-px tiles at
            // 127 VGPRs / 64 KB LDS -> 4 waves/SIMD, 2 blocks/CU; measured
            // +5% over the wide config at 1280@120^2 (the wide config's
            // 64-px tiles also waste 6% on the 120->128 x-pad there)
            constexpr int YB2 = 8, XW2 = 1, NCT2 = 2, PW2 = 1;
            constexpr int XB2 = XW2 * 32 * PW2;
            const int nxb2 = (p.Wo + XB2 - 1) / XB2;
            const int nyb2 = (p.Ho + YB2 - 1) / YB2;
            const int npix2 = nxb2 * nyb2 * p.B;
            const int ncb2 = (p.CT + NCT2 - 1) / NCT2;
            dim3 g2((unsigned)(8 * ncb2 * ((npix2 + 7) / 8)));
            conv3x3_kernel<1, YB2, XW2, 16, NCT2, PW2>
                <<<g2, dim3(YB2 * XW2 * WAVE_SIZE), 0, stream>>>(p);
            return;
        }
        conv3x3_kernel<1, YB, XW, 16, NCT, PW><<<grid, dim3(YB * XW * WAVE_SIZE), 0, stream>>>(p);
    } else {
        // stride 2: 8 waves: 4 output rows x 2 x-waves of 32 px;
        // LDS (double-buffered): 2 x (parity input 46.8 KB + 18.4 KB weights).
        constexpr int YB = 4, XW = 2, NCT = 2, PW = 1;
        constexpr int XB = XW * 32 * PW;
        const int nxb = (p.Wo + XB - 1) / XB;
        const int nyb = (p.Ho + YB - 1) / YB;
        const int npix = nxb * nyb * p.B;
        const int ncb = (p.CT + NCT - 1) / NCT;
        dim3 grid((unsigned)(8 * ncb * ((npix + 7) / 8)));
        conv3x3_kernel<2, YB, XW, 16, NCT, PW><<<grid, dim3(YB * XW * WAVE_SIZE), 0, stream>>>(p);
    }
}
