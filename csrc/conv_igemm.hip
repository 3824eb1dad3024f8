// sodkit — MFMA implicit-GEMM convolution family for CDNA4 (gfx950).
//
// Covers SURVEY §2.3 K1 (3x3), K2 (1x1), K6 (7x7 stem), K10 (dilated),
// K13 (Res2Net splits are just more K1 calls): forward, dgrad and split-K
// wgrad, all dtypes bf16/fp16, fp32 accumulation.
//
// Layouts (chosen for MI355X, not inherited):
//   activations  NHWC            (torch channels_last physical layout)
//   weights      [Cout][R][S][Cin]  == GEMM A-operand [Cout][K] row-major
//   dgrad weight [Cin][R][S][Cout]  (small permuted copy made by the caller)
//
// GEMM mapping (fwd):  D[cout][m] = sum_k W[cout][k] * im2col(x)[k][m]
//   A-operand (W) stages coalesced from global, no transpose.
//   B-operand (X) is the im2col gather: for NHWC, 8 consecutive k are 8
//   consecutive cin of one (r,s) tap -> 16-byte vector gathers when Cin%8==0.
//   D-fragment: each lane holds 4 *consecutive couts* of one output pixel ->
//   packs to one 8-byte NHWC store.  (Operand roles were chosen exactly so
//   the epilogue store is the coalesced one.)
//
// Structure per K-step (BK=64): reg-staged global->LDS with XOR-swizzled
// tiles (bank-conflict-free ds_read_b128 fragment reads, CDNA4 guide T2),
// 2x2 waves x (FCx FM) 16x16x32 MFMA fragments, two barriers per step.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {

// Swizzled byte offset inside a [rows][BK] 16-bit tile, row pitch BK*2 bytes.
// XOR nibble = (row&7) ^ ((row>>3)&7): conflict-FREE for the b128 fragment
// reads (16-lane groups read rows r..r+15: evens/odds each land on 8
// distinct 16B slots) AND <=2-way for the wgrad transpose's b64 writes
// (rows strided by 8 within a lane group, where a plain (row&7) nibble is
// constant across the group -> 16-way conflict, measured 23% of wave
// cycles as SQ_LDS_BANK_CONFLICT).
template <int BK>
SOD_DEVINL int swz(int row, int kbyte) {
    if constexpr (BK == 128) {
        // 256B row pitch: the row contributes nothing to the bank slot, so
        // a 4-bit nibble (row&15)^((row>>4)&15) is needed — conflict-free
        // for both the contiguous-16-row b128 reads and the stride-8-row
        // b64 transpose writes
        return row * 256 + (kbyte ^ ((((row & 15) ^ ((row >> 4) & 15))) << 4));
    } else {
        return row * (BK * 2) + (kbyte ^ ((((row & 7) ^ ((row >> 3) & 7))) << 4));
    }
}

// ---------------------------------------------------------------------------
// Gather address computation.  MODE 0 = fwd (gather x), MODE 1 = dgrad
// (gather dy through the transposed-conv index map).
// Returns element offset into the source, or -1 if out of range / padding.
// ---------------------------------------------------------------------------
template <int MODE>
SOD_DEVINL long gather_off(int n, int p, int q, int r, int s, int c,
                           int Hs, int Ws, int Cs, int ldc,
                           int stride, int pad, int dil) {
    if (MODE == 0) {
        int ih = p * stride + r * dil - pad;
        int iw = q * stride + s * dil - pad;
        if ((unsigned)ih >= (unsigned)Hs || (unsigned)iw >= (unsigned)Ws)
            return -1;
        return (((long)n * Hs + ih) * Ws + iw) * ldc + c;
    } else if (stride == 1) {
        // fast path: most layers are stride 1 and the generic path's
        // runtime %/÷ by stride costs real VALU in the staging loop
        int oh = p + pad - r * dil;
        int ow = q + pad - s * dil;
        if ((unsigned)oh >= (unsigned)Hs || (unsigned)ow >= (unsigned)Ws)
            return -1;
        return (((long)n * Hs + oh) * Ws + ow) * ldc + c;
    } else {
        int hh = p + pad - r * dil;
        int ww = q + pad - s * dil;
        if (hh < 0 || ww < 0 || hh % stride || ww % stride) return -1;
        int oh = hh / stride, ow = ww / stride;
        if (oh >= Hs || ow >= Ws) return -1;
        return (((long)n * Hs + oh) * Ws + ow) * ldc + c;
    }
}

// ---------------------------------------------------------------------------
// fwd / dgrad kernel.
//   A_w  : [Ncols? no — Ncout][K_total] row-major ("weights")
//   B_src: NHWC source for the gather (x for fwd, dy for dgrad)
//   out  : [M_total][Ncout] (NHWC output flattened over pixels)
// ---------------------------------------------------------------------------
// addrspace helpers for global_load_lds (async global->LDS DMA)
typedef __attribute__((address_space(3))) void lds_void_t;
typedef __attribute__((address_space(1))) const void gbl_void_t;

SOD_DEVINL void glds16(const void *gsrc, char *lds_dst) {
    __builtin_amdgcn_global_load_lds((const gbl_void_t *)gsrc,
                                     (lds_void_t *)lds_dst, 16, 0, 0);
}

SOD_DEVINL int swz_nib(int row) { return ((row & 7) ^ ((row >> 3) & 7)) << 4; }

// ES=1 (fwd only): the epilogue also emits per-channel (sum, sumsq) BN
// partials over the ROUNDED output values into stats_out[tiles_m][2*Ncout]
// — the exact input format of bn_reduce_finalize_kernel — so a following
// BatchNorm never re-reads y for its stats pass (bn_stats_vec was ~1.2
// ms/step of pure y re-read traffic).  Cost in the conv: ~8 live VGPRs,
// a 16-lane shuffle tree per fragment and a 2 KB static-LDS combine.
template <typename T, int MODE, int BCOUT, int BM, int ES = 0>
__global__ __launch_bounds__(256, 3)   // force >= 3 waves/SIMD (171-reg
                                       // natural allocation lands at 2)
void igemm_kernel(const T *__restrict__ B_src, const T *__restrict__ A_w,
                  const float *__restrict__ bias, T *__restrict__ out,
                  float *__restrict__ stats_out,
                  const T *__restrict__ zero_page,
                  int Ns, int Hs, int Ws, int Cs,   // gather-source dims
                  int ld_src,                       // pixel pitch of B_src
                  int P, int Q,                     // output spatial
                  int Ncout,                        // output channels
                  int R, int S, int stride, int pad, int dil,
                  int M_total, int K_total, int tiles_m, int dbuf) {
    constexpr int BK = 64;
    constexpr int FC = BCOUT / 32;   // a-fragments per wave (2x2 wave grid)
    constexpr int FM = BM / 32;      // b-fragments per wave
    constexpr int GW = BCOUT * BK * 2 / 4096;  // glds insts/wave, W tile
    constexpr int GX = BM * BK * 2 / 4096;     // glds insts/wave, X tile

    // dynamic LDS: 2 buffers when the glds path double-buffers (K > BK),
    // 1 otherwise — a fixed 2x allocation halves occupancy exactly for the
    // low-arithmetic-intensity single-K-step convs that need it most
    constexpr int TILE_BYTES = (BCOUT + BM) * BK * 2;
    extern __shared__ __attribute__((aligned(16))) char lds[];

    // XCD-aware block remap (T1): the dispatcher places block b on XCD b%8;
    // giving each XCD a contiguous tile range makes neighbor tiles (which
    // share W panels / x rows) hit the same L2.  Bijective variant.
    int bid = blockIdx.x;
    {
        int nwg = gridDim.x;
        int q = nwg >> 3, r = nwg & 7;
        int xcd = bid & 7, idx = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    const int tile_m = (bid % tiles_m) * BM;
    const int tile_c = (bid / tiles_m) * BCOUT;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wv = t >> 6;                  // 4 waves: 2x2
    const int wv_c = wv >> 1, wv_m = wv & 1;

    const bool vec_ok = (Cs % 8 == 0) && (ld_src % 8 == 0);
    const bool wvec_ok = (K_total % 8 == 0);

    // --- per-thread staging geometry (fixed across K-steps) ---
    // W chunks: BCOUT*BK/8 8-elem chunks, 256 threads
    constexpr int WCH = BCOUT * BK / 8 / 256;       // chunks per thread
    constexpr int XCH = BM * BK / 8 / 256;
    // --- glds lane geometry: the LDS image is written lane-linear, so the
    // tile swizzle moves onto the SOURCE address (inverse-swizzle per lane;
    // CDNA4 guide rule 21).  Usable when 16B global chunks are whole
    // (Cs % 8 == 0 and K_total % 8 == 0); OOB/padding lanes read a zero page.
    const bool glds_ok = vec_ok && wvec_ok;
    // per-inst state kept minimal for occupancy: rows/k-offsets are cheap
    // ALU recomputes; only the incremental pixel/tap state lives in regs
    int gx_n[GX], gx_p[GX], gx_q[GX];  // X tile: pixel geometry per row
    int gx_c[GX], gx_r[GX], gx_s[GX];  // incremental (r,s,c) per inst
#pragma unroll
    for (int i = 0; i < GX; ++i) {
        int off = (wv * GX + i) * 1024 + lane * 16;
        int row = off >> 7;
        int kel = ((off & 127) ^ swz_nib(row)) >> 1;
        int rs = kel / Cs;
        gx_c[i] = kel % Cs;
        gx_r[i] = rs / S;
        gx_s[i] = rs % S;
        int m = tile_m + row;
        int mm = m < M_total ? m : 0;
        gx_n[i] = mm / (P * Q);
        int pq = mm % (P * Q);
        gx_p[i] = pq / Q;
        gx_q[i] = pq % Q;
    }

    f32x4 acc[FC][FM];
#pragma unroll
    for (int a = 0; a < FC; ++a)
#pragma unroll
        for (int b = 0; b < FM; ++b) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    auto stage_glds = [&](int buf, int k0) {
        char *w_lds = lds + buf * TILE_BYTES;
        char *x_lds = w_lds + BCOUT * BK * 2;
#pragma unroll
        for (int i = 0; i < GW; ++i) {
            int off = (wv * GW + i) * 1024 + lane * 16;
            int row = off >> 7;
            int cout = tile_c + row;
            int k = k0 + (((off & 127) ^ swz_nib(row)) >> 1);
            const T *src = (cout < Ncout && k < K_total)
                               ? A_w + (long)cout * K_total + k
                               : zero_page;
            glds16(src, w_lds + (wv * GW + i) * 1024);
        }
#pragma unroll
        for (int i = 0; i < GX; ++i) {
            int row = ((wv * GX + i) * 1024 + lane * 16) >> 7;
            const T *src = zero_page;
            // k < K_total  <=>  r < R (k = (r*S+s)*Cs+c, K = R*S*Cs exact)
            if (tile_m + row < M_total && gx_r[i] < R) {
                long off = gather_off<MODE>(gx_n[i], gx_p[i], gx_q[i],
                                            gx_r[i], gx_s[i], gx_c[i],
                                            Hs, Ws, Cs, ld_src, stride, pad, dil);
                if (off >= 0) src = B_src + off;
            }
            glds16(src, x_lds + (wv * GX + i) * 1024);
            // advance (r,s,c) by BK elements for the next K-step
            // (stage_glds is called once per k0 in increasing order)
            int c = gx_c[i] + BK;
            int sS = gx_s[i], rR = gx_r[i];
            while (c >= Cs) { c -= Cs; if (++sS == S) { sS = 0; ++rR; } }
            gx_c[i] = c; gx_s[i] = sS; gx_r[i] = rR;
        }
    };

    auto stage_reg = [&](int k0) {
        char *w_lds = lds;
        char *x_lds = lds + BCOUT * BK * 2;
        // ---- stage W tile: [BCOUT][BK], coalesced rows from A_w ----
#pragma unroll
        for (int i = 0; i < WCH; ++i) {
            int ch = t + i * 256;
            int row = ch / (BK / 8);
            int k = k0 + (ch % (BK / 8)) * 8;
            int cout = tile_c + row;
            s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (cout < Ncout) {
                const T *src = A_w + (long)cout * K_total + k;
                if (wvec_ok && k + 7 < K_total) {
                    v = *reinterpret_cast<const s16x8 *>(src);
                } else {
#pragma unroll
                    for (int e = 0; e < 8; ++e)
                        if (k + e < K_total)
                            v[e] = reinterpret_cast<const short *>(src)[e];
                }
            }
            *reinterpret_cast<s16x8 *>(w_lds + swz<BK>(row, (ch % (BK / 8)) * 16)) = v;
        }
        // ---- stage X tile: [BM][BK], im2col gather ----
        // (fallback path: geometry recomputed per call so its registers
        // never stay live alongside the glds path's state)
#pragma unroll
        for (int i = 0; i < XCH; ++i) {
            int ch = t + i * 256;
            int row = ch / (BK / 8);
            int k = k0 + (ch % (BK / 8)) * 8;
            int m = tile_m + row;
            int mm = m < M_total ? m : 0;
            int xn = mm / (P * Q);
            int pq = mm % (P * Q);
            int xp = pq / Q, xq = pq % Q;
            s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (m < M_total && k < K_total) {
                if (vec_ok && (k % 8 == 0) && (k / Cs == (k + 7) / Cs)) {
                    int rs = k / Cs, c = k % Cs;
                    int r = rs / S, s = rs % S;
                    long off = gather_off<MODE>(xn, xp, xq, r, s, c,
                                                Hs, Ws, Cs, ld_src, stride, pad, dil);
                    if (off >= 0)
                        v = *reinterpret_cast<const s16x8 *>(B_src + off);
                } else {
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        int ke = k + e;
                        if (ke < K_total) {
                            int rs = ke / Cs, c = ke % Cs;
                            int r = rs / S, s = rs % S;
                            long off = gather_off<MODE>(xn, xp, xq, r, s, c,
                                                        Hs, Ws, Cs, ld_src, stride, pad, dil);
                            if (off >= 0)
                                v[e] = reinterpret_cast<const short *>(B_src)[off];
                        }
                    }
                }
            }
            *reinterpret_cast<s16x8 *>(x_lds + swz<BK>(row, (ch % (BK / 8)) * 16)) = v;
        }
    };

    auto mfma_tile = [&](int buf) {
        const char *w_lds = lds + buf * TILE_BYTES;
        const char *x_lds = w_lds + BCOUT * BK * 2;
#pragma unroll
        for (int kk = 0; kk < BK / 32; ++kk) {
            int kbyte = kk * 64 + (lane >> 4) * 16;
            s16x8 a[FC], b[FM];
#pragma unroll
            for (int fc = 0; fc < FC; ++fc) {
                int row = wv_c * (BCOUT / 2) + fc * 16 + (lane & 15);
                a[fc] = *reinterpret_cast<const s16x8 *>(w_lds + swz<BK>(row, kbyte));
            }
#pragma unroll
            for (int fm = 0; fm < FM; ++fm) {
                int row = wv_m * (BM / 2) + fm * 16 + (lane & 15);
                b[fm] = *reinterpret_cast<const s16x8 *>(x_lds + swz<BK>(row, kbyte));
            }
#pragma unroll
            for (int fc = 0; fc < FC; ++fc)
#pragma unroll
                for (int fm = 0; fm < FM; ++fm)
                    acc[fc][fm] = MfmaOp<T>::mma(a[fc], b[fm], acc[fc][fm]);
        }
    };

    const int nsteps = (K_total + BK - 1) / BK;
    if (glds_ok && dbuf) {
        // 2-phase double buffer: tile s+1's async staging issues BEFORE
        // tile s's MFMAs; one barrier per tile (it also drains the DMA)
        stage_glds(0, 0);
        __syncthreads();
        int cur = 0;
        for (int s = 0; s < nsteps; ++s) {
            if (s + 1 < nsteps) stage_glds(cur ^ 1, (s + 1) * BK);
            mfma_tile(cur);
            __syncthreads();
            cur ^= 1;
        }
    } else if (glds_ok) {
        // single-buffer glds (half the LDS -> more blocks/CU; cross-block
        // overlap instead of intra-block pipelining)
        for (int s = 0; s < nsteps; ++s) {
            stage_glds(0, s * BK);
            __syncthreads();
            mfma_tile(0);
            __syncthreads();
        }
    } else {
        for (int s = 0; s < nsteps; ++s) {
            stage_reg(s * BK);
            __syncthreads();
            mfma_tile(0);
            __syncthreads();
        }
    }

    // ---- epilogue: D[cout][m] -> NHWC out[m][cout], 8B packed stores ----
    // ES: static-LDS combine plane for the two wv_m wave-halves that share
    // each channel quad (discarded entirely for ES=0 instantiations)
    struct ESRed { float v[2][2][BCOUT]; };  // [wv_m][sum|ssq][c_local]
    __shared__ __attribute__((aligned(16))) char es_red_raw[ES ? sizeof(ESRed) : 1];
    ESRed *es_red = reinterpret_cast<ESRed *>(es_red_raw);
    if constexpr (ES) __syncthreads();  // LDS tiles done before reuse
#pragma unroll
    for (int fc = 0; fc < FC; ++fc) {
        int cout = tile_c + wv_c * (BCOUT / 2) + fc * 16 + (lane >> 4) * 4;
        float b4[4] = {0.f, 0.f, 0.f, 0.f};
        if (bias != nullptr) {
#pragma unroll
            for (int e = 0; e < 4; ++e)
                if (cout + e < Ncout) b4[e] = bias[cout + e];
        }
        float ssum[4] = {0.f, 0.f, 0.f, 0.f};
        float sssq[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
            int m = tile_m + wv_m * (BM / 2) + fm * 16 + (lane & 15);
            if (m >= M_total) continue;
            f32x4 v = acc[fc][fm];
            if (bias != nullptr) {
#pragma unroll
                for (int e = 0; e < 4; ++e) v[e] += b4[e];
            }
            if constexpr (ES) {
                // stats over the ROUNDED stored values: identical numbers
                // to what bn_stats_vec would re-read from y
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    T te = ElemTraits<T>::from_f32(v[e]);
                    float f = s16_as_f32<T>(
                        *reinterpret_cast<const short *>(&te));
                    ssum[e] += f;
                    sssq[e] += f * f;
                }
            }
            T *dst = out + (long)m * Ncout + cout;
            // vector store needs 8B alignment: holds iff Ncout % 4 == 0
            if ((Ncout % 4 == 0) && cout + 3 < Ncout) {
                *reinterpret_cast<s16x4 *>(dst) = pack4<T>(v);
            } else {
#pragma unroll
                for (int e = 0; e < 4; ++e)
                    if (cout + e < Ncout)
                        dst[e] = ElemTraits<T>::from_f32(v[e]);
            }
        }
        if constexpr (ES) {
            // fold the 16 m-lanes (lane bits 0..3; channel bits untouched)
#pragma unroll
            for (int d = 1; d < 16; d <<= 1)
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    ssum[e] += __shfl_xor(ssum[e], d, 64);
                    sssq[e] += __shfl_xor(sssq[e], d, 64);
                }
            if ((lane & 15) == 0) {
                int cl = wv_c * (BCOUT / 2) + fc * 16 + (lane >> 4) * 4;
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    es_red->v[wv_m][0][cl + e] = ssum[e];
                    es_red->v[wv_m][1][cl + e] = sssq[e];
                }
            }
        }
    }
    if constexpr (ES) {
        __syncthreads();
        // one slab row per block: stats_out[m_tile][{sum,ssq} x Ncout]
        int mtile = tile_m / BM;
        long base = (long)mtile * 2 * Ncout;
        for (int i = t; i < 2 * BCOUT; i += 256) {
            int row = i / BCOUT, c = i % BCOUT;
            if (tile_c + c < Ncout)
                stats_out[base + (long)row * Ncout + tile_c + c] =
                    es_red->v[0][row][c] + es_red->v[1][row][c];
        }
    }
}

// ---------------------------------------------------------------------------
// wgrad kernel: dw[cout][(r,s,cin)] = sum_m dy[m][cout] * im2col(x)[m][(r,s,cin)]
// Split-K over m; fp32 partials [splitk][Cout][Kw].
//   dy^T tile is built by an in-LDS transpose (vector global read, scalar
//   ds writes); the x tile gathers 8 consecutive output-pixels per lane
//   (m-contiguous in LDS -> single b128 write).
// ---------------------------------------------------------------------------
// NT = threads/block.  256 => 4 waves as 2(cout-half) x 2(j-half),
// per-wave acc 64x64.  512 => 8 waves as 4(cout-quarter) x 2(j-half) over a
// 256x256 tile (per-wave 64x128, acc 128 VGPR): doubles the MFMA work per
// staged LDS byte (64->131 flop/B), which is what bounds this kernel —
// measured r2: the big 1x1/deep-3x3 wgrads sat at 111-256 TF on the
// 128-square tile.
template <typename T, int BCOUT, int BJ, int BKM, int NT = 256>
__global__ __launch_bounds__(NT)
void wgrad_kernel(const T *__restrict__ x, const T *__restrict__ dy,
                  float *__restrict__ partial,
                  int N, int H, int W, int C,     // x dims
                  int ldx, int lddy,              // pixel pitches (slices)
                  int P, int Q, int Cout,         // dy dims
                  int R, int S, int stride, int pad, int dil,
                  int Kw,                          // = R*S*C (dw inner size)
                  int M_total,                     // = N*P*Q
                  int m_per_split, int tiles_c) {
    constexpr int CW = (NT / 64) / 2;             // cout wave groups
    constexpr int FC = BCOUT / (16 * CW);
    constexpr int FJ = BJ / 32;

    __shared__ __align__(16) char lds[(BCOUT + BJ) * BKM * 2];
    char *dyT_lds = lds;                      // [BCOUT][BKM]
    char *xT_lds = lds + BCOUT * BKM * 2;     // [BJ][BKM]

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wv = t >> 6;
    const int wv_c = wv >> 1, wv_j = wv & 1;

    const int tile_c = (blockIdx.x % tiles_c) * BCOUT;
    const int tile_j = (blockIdx.x / tiles_c) * BJ;
    const int z = blockIdx.y;
    const int m_begin = z * m_per_split;
    const int m_end = min(m_begin + m_per_split, M_total);

    const bool dyvec_ok = (Cout % 8 == 0) && (lddy % 8 == 0);
    const bool xvec_ok = (C % 8 == 0) && (ldx % 8 == 0);

    // per-thread x-chunk geometry: chunk ch -> (jv8 = ch % (BJ/8),
    // m_quad = ch / (BJ/8)); 8 consecutive j = 8 consecutive cin of one
    // (r,s) tap when C%8==0 -> four 16B gathers (m..m+3), transposed in
    // registers into 8 ds_write_b64 (the scattered-u16 transpose was
    // write-instruction-bound).
    constexpr int XCH_TOT = BJ * (BKM / 4) / 8;   // total x chunks
    constexpr int XCH = (XCH_TOT + NT - 1) / NT;  // per-thread (may idle)
    int xr[XCH], xs[XCH], xc[XCH], xj[XCH], xml[XCH];
    int xgn[XCH], xgp[XCH], xgq[XCH];   // incremental (n,p,q) of the quad base
    bool xact[XCH];                     // chunk exists (XCH_TOT may be < NT)
#pragma unroll
    for (int i = 0; i < XCH; ++i) {
        int ch = t + i * NT;
        xact[i] = ch < XCH_TOT;
        if (!xact[i]) ch = 0;
        int jv8 = ch % (BJ / 8);
        xml[i] = (ch / (BJ / 8)) * 4;          // first m of the quad
        int j = tile_j + jv8 * 8;
        xj[i] = j;
        int jj = j < Kw ? j : 0;
        int rs = jj / C;
        xc[i] = jj % C;
        xr[i] = rs / S;
        xs[i] = rs % S;
        long m = m_begin + xml[i];
        xgn[i] = (int)(m / ((long)P * Q));
        int pq = (int)(m % ((long)P * Q));
        xgp[i] = pq / Q;
        xgq[i] = pq % Q;
    }

    f32x4 acc[FC][FJ];
#pragma unroll
    for (int a = 0; a < FC; ++a)
#pragma unroll
        for (int b = 0; b < FJ; ++b) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    constexpr int DNCH = BCOUT * (BKM / 4) / 8;   // dy chunks (4 m x 8 cout)
    // ---- cross-K-step software pipeline (PIPE variants): the global
    // loads for step t+1 are ISSUED before step t's MFMA, so their ~600ns
    // latency overlaps the math.  PMC evidence (r2): the 128x128 variant
    // ran at 8% MfmaUtil with only 6% LDS-conflict cycles — latency-bound.
    // The staged registers cost ~50 VGPRs across the MFMA region, which
    // only the 128x128 variant affords without spills/occupancy loss; the
    // other variants keep the original interleaved two-barrier body.
    constexpr bool PIPE = (BCOUT == 128 && BJ == 128 && NT == 256);
    // (BCOUT==64 was A/B'd: wash — its 167->206 VGPR cost one wave of
    //  occupancy, exactly cancelling the latency hiding)

    if constexpr (PIPE) {
        static_assert(BCOUT * (BKM / 4) / 8 == NT || !PIPE, "one dy chunk/thread");
        const int dy_ml = (t / (BCOUT / 8)) * 4;
        const int dy_cv = (t % (BCOUT / 8)) * 8;
        s16x8 vdy[4];
        s16x8 vx[XCH][4];

        auto load_dy = [&](long m0) {
            int cout = tile_c + dy_cv;
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                vdy[u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
                long m = m0 + dy_ml + u;
                if (m < m_end && cout < Cout) {
                    const T *src = dy + m * lddy + cout;
                    if (dyvec_ok && cout + 7 < Cout) {
                        vdy[u] = *reinterpret_cast<const s16x8 *>(src);
                    } else {
#pragma unroll
                        for (int e = 0; e < 8; ++e)
                            if (cout + e < Cout)
                                vdy[u][e] = reinterpret_cast<const short *>(src)[e];
                    }
                }
            }
        };
        auto load_x = [&](long m0) {
#pragma unroll
            for (int i = 0; i < XCH; ++i) {
                if (!xact[i]) continue;
#pragma unroll
                for (int u = 0; u < 4; ++u) {
                    vx[i][u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
                    long m = m0 + xml[i] + u;
                    if (xj[i] < Kw && m < m_end) {
                        int n = xgn[i], p = xgp[i], q = xgq[i] + u;
                        if (q >= Q) { q -= Q; if (++p >= P) { p = 0; ++n; } }
                        if (xvec_ok) {
                            long off = gather_off<0>(n, p, q, xr[i], xs[i], xc[i],
                                                     H, W, C, ldx, stride, pad, dil);
                            if (off >= 0)
                                vx[i][u] = *reinterpret_cast<const s16x8 *>(x + off);
                        } else {
#pragma unroll
                            for (int e = 0; e < 8; ++e) {
                                int j = xj[i] + e;
                                if (j < Kw) {
                                    int rs = j / C, c = j % C;
                                    long off = gather_off<0>(n, p, q, rs / S,
                                                             rs % S, c,
                                                             H, W, C, ldx, stride, pad, dil);
                                    if (off >= 0)
                                        vx[i][u][e] = reinterpret_cast<const short *>(x)[off];
                                }
                            }
                        }
                    }
                }
                int q = xgq[i] + BKM, p = xgp[i], n = xgn[i];
                while (q >= Q) { q -= Q; ++p; }
                while (p >= P) { p -= P; ++n; }
                xgq[i] = q; xgp[i] = p; xgn[i] = n;
            }
        };

        load_dy(m_begin); load_x(m_begin);
        for (long m0 = m_begin; m0 < m_end; m0 += BKM) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                s16x4 w4 = {vdy[0][e], vdy[1][e], vdy[2][e], vdy[3][e]};
                *reinterpret_cast<s16x4 *>(
                    dyT_lds + swz<BKM>(dy_cv + e, dy_ml * 2)) = w4;
            }
#pragma unroll
            for (int i = 0; i < XCH; ++i) {
                if (!xact[i]) continue;
                int j_local0 = (xj[i] - tile_j);
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    s16x4 w4 = {vx[i][0][e], vx[i][1][e], vx[i][2][e], vx[i][3][e]};
                    *reinterpret_cast<s16x4 *>(
                        xT_lds + swz<BKM>(j_local0 + e, xml[i] * 2)) = w4;
                }
            }
            __syncthreads();
            if (m0 + BKM < m_end) { load_dy(m0 + BKM); load_x(m0 + BKM); }

#pragma unroll
            for (int kk = 0; kk < BKM / 32; ++kk) {
                int kbyte = kk * 64 + (lane >> 4) * 16;
                s16x8 a[FC], b[FJ];
#pragma unroll
                for (int fc = 0; fc < FC; ++fc) {
                    int row = wv_c * (BCOUT / CW) + fc * 16 + (lane & 15);
                    a[fc] = *reinterpret_cast<const s16x8 *>(dyT_lds + swz<BKM>(row, kbyte));
                }
#pragma unroll
                for (int fj = 0; fj < FJ; ++fj) {
                    int row = wv_j * (BJ / 2) + fj * 16 + (lane & 15);
                    b[fj] = *reinterpret_cast<const s16x8 *>(xT_lds + swz<BKM>(row, kbyte));
                }
#pragma unroll
                for (int fc = 0; fc < FC; ++fc)
#pragma unroll
                    for (int fj = 0; fj < FJ; ++fj)
                        acc[fc][fj] = MfmaOp<T>::mma(a[fc], b[fj], acc[fc][fj]);
            }
            __syncthreads();
        }
    } else {
    for (int m0 = m_begin; m0 < m_end; m0 += BKM) {
        // ---- stage dy^T tile: 4 coalesced 16B reads dy[m..m+3][cout..+7],
        //      register transpose, 8 ds_write_b64 ----
        for (int ch = t; ch < DNCH; ch += NT) {
            int m_local = (ch / (BCOUT / 8)) * 4;
            int cv = (ch % (BCOUT / 8)) * 8;
            int cout = tile_c + cv;
            s16x8 v[4];
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                v[u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
                int m = m0 + m_local + u;
                if (m < m_end && cout < Cout) {
                    const T *src = dy + (long)m * lddy + cout;
                    if (dyvec_ok && cout + 7 < Cout) {
                        v[u] = *reinterpret_cast<const s16x8 *>(src);
                    } else {
#pragma unroll
                        for (int e = 0; e < 8; ++e)
                            if (cout + e < Cout)
                                v[u][e] = reinterpret_cast<const short *>(src)[e];
                    }
                }
            }
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                s16x4 w4 = {v[0][e], v[1][e], v[2][e], v[3][e]};
                *reinterpret_cast<s16x4 *>(
                    dyT_lds + swz<BKM>(cv + e, m_local * 2)) = w4;
            }
        }
        // ---- stage x^T tile: four 16B cin-gathers (m..m+3) per chunk,
        //      register transpose, 8 ds_write_b64 ----
#pragma unroll
        for (int i = 0; i < XCH; ++i) {
            if (!xact[i]) continue;
            int j_local0 = (xj[i] - tile_j);
            s16x8 v[4];
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                v[u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
                int m = m0 + xml[i] + u;
                if (xj[i] < Kw && m < m_end) {
                    // quad-base geometry is maintained incrementally; +u
                    // wraps at most once (Q >= 4 for every supported shape)
                    int n = xgn[i], p = xgp[i], q = xgq[i] + u;
                    if (q >= Q) { q -= Q; if (++p >= P) { p = 0; ++n; } }
                    if (xvec_ok) {
                        long off = gather_off<0>(n, p, q, xr[i], xs[i], xc[i],
                                                 H, W, C, ldx, stride, pad, dil);
                        if (off >= 0)
                            v[u] = *reinterpret_cast<const s16x8 *>(x + off);
                    } else {
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            int j = xj[i] + e;
                            if (j < Kw) {
                                int rs = j / C, c = j % C;
                                long off = gather_off<0>(n, p, q, rs / S,
                                                         rs % S, c,
                                                         H, W, C, ldx, stride, pad, dil);
                                if (off >= 0)
                                    v[u][e] = reinterpret_cast<const short *>(x)[off];
                            }
                        }
                    }
                }
            }
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                s16x4 w4 = {v[0][e], v[1][e], v[2][e], v[3][e]};
                *reinterpret_cast<s16x4 *>(
                    xT_lds + swz<BKM>(j_local0 + e, xml[i] * 2)) = w4;
            }
            // advance the quad base by BKM pixels for the next K-step
            int q = xgq[i] + BKM, p = xgp[i], n = xgn[i];
            while (q >= Q) { q -= Q; ++p; }
            while (p >= P) { p -= P; ++n; }
            xgq[i] = q; xgp[i] = p; xgn[i] = n;
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < BKM / 32; ++kk) {
            int kbyte = kk * 64 + (lane >> 4) * 16;
            s16x8 a[FC], b[FJ];
#pragma unroll
            for (int fc = 0; fc < FC; ++fc) {
                int row = wv_c * (BCOUT / CW) + fc * 16 + (lane & 15);
                a[fc] = *reinterpret_cast<const s16x8 *>(dyT_lds + swz<BKM>(row, kbyte));
            }
#pragma unroll
            for (int fj = 0; fj < FJ; ++fj) {
                int row = wv_j * (BJ / 2) + fj * 16 + (lane & 15);
                b[fj] = *reinterpret_cast<const s16x8 *>(xT_lds + swz<BKM>(row, kbyte));
            }
#pragma unroll
            for (int fc = 0; fc < FC; ++fc)
#pragma unroll
                for (int fj = 0; fj < FJ; ++fj)
                    acc[fc][fj] = MfmaOp<T>::mma(a[fc], b[fj], acc[fc][fj]);
        }
        __syncthreads();
    }

    }

    // ---- store fp32 partial [z][Cout][Kw] ----
    float *dst_base = partial + (long)z * Cout * Kw;
#pragma unroll
    for (int fc = 0; fc < FC; ++fc) {
        int cout = tile_c + wv_c * (BCOUT / CW) + fc * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int fj = 0; fj < FJ; ++fj) {
            int j = tile_j + wv_j * (BJ / 2) + fj * 16 + (lane & 15);
            if (j >= Kw) continue;
#pragma unroll
            for (int e = 0; e < 4; ++e)
                if (cout + e < Cout)
                    dst_base[(long)(cout + e) * Kw + j] = acc[fc][fj][e];
        }
    }
}

template <typename T>
__global__ void splitk_reduce_kernel(const float *__restrict__ partial,
                                     T *__restrict__ dw, long n_elems,
                                     int splitk) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n_elems; i += stride) {
        // 8 independent accumulators: keep loads in flight over large splitk
        float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
        int z = 0;
        for (; z + 8 <= splitk; z += 8)
#pragma unroll
            for (int u = 0; u < 8; ++u)
                a[u] += partial[(long)(z + u) * n_elems + i];
        for (; z < splitk; ++z) a[0] += partial[(long)z * n_elems + i];
        dw[i] = ElemTraits<T>::from_f32(((a[0] + a[1]) + (a[2] + a[3]))
                                        + ((a[4] + a[5]) + (a[6] + a[7])));
    }
}

// channel sum of an NHWC tensor -> fp32 [C]  (conv bias grad; SURVEY K-misc)
template <typename T>
__global__ void channel_sum_kernel(const T *__restrict__ x, float *__restrict__ out,
                                   long M, int C, int ld) {
    // one block per channel-chunk; grid-stride over rows
    int c = blockIdx.y * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float s = 0.f;
    for (long m = blockIdx.x; m < M; m += gridDim.x)
        s += ElemTraits<T>::to_f32(x[m * ld + c]);
    atomicAdd(&out[c], s);
}

// small-C variant (C <= 16): the channel-per-thread kernel leaves most
// threads idle at tiny C (the 1-channel head's bias grad measured 220 us;
// HDFNet's 9-channel DDPM generators 43 us); here every thread strides
// rows, accumulates all C channels in registers, and the block reduces
// through LDS once.
template <typename T, bool FLAT>
__global__ void channel_sum_small_kernel(const T *__restrict__ x,
                                         float *__restrict__ out,
                                         long M, int C, int ld) {
    __shared__ float red[256 * 16];
    float acc[16] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f,
                     0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    if (FLAT) {   // C==1, ld==1: contiguous — vector reads
        const long tot8 = M >> 3;
        for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < tot8;
             i += (long)gridDim.x * blockDim.x) {
            s16x8 v = *reinterpret_cast<const s16x8 *>(x + i * 8);
#pragma unroll
            for (int e = 0; e < 8; ++e) acc[0] += s16_as_f32<T>(v[e]);
        }
        if (blockIdx.x == 0 && threadIdx.x == 0)   // tail
            for (long m = tot8 * 8; m < M; ++m)
                acc[0] += ElemTraits<T>::to_f32(x[m]);
    } else {
        for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
             m += (long)gridDim.x * blockDim.x)
            for (int c = 0; c < C; ++c)
                acc[c] += ElemTraits<T>::to_f32(x[m * ld + c]);
    }
#pragma unroll
    for (int c = 0; c < 16; ++c) red[threadIdx.x * 16 + c] = acc[c];
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
        if (threadIdx.x < w)
#pragma unroll
            for (int c = 0; c < 16; ++c)
                red[threadIdx.x * 16 + c] += red[(threadIdx.x + w) * 16 + c];
        __syncthreads();
    }
    if (threadIdx.x < C) atomicAdd(&out[threadIdx.x], red[threadIdx.x]);
}

}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")

static inline int out_dim(int in, int k, int stride, int pad, int dil) {
    return (in + 2 * pad - dil * (k - 1) - 1) / stride + 1;
}

namespace {

// 128-byte device zero page for glds OOB/padding lanes
static void *zero_page_ptr(const torch::Tensor &like) {
    static torch::Tensor z;
    if (!z.defined() || z.device() != like.device())
        z = torch::zeros({32}, like.options().dtype(torch::kFloat32));
    return z.data_ptr();
}

template <typename scalar_t, int MODE>
void launch_igemm(const torch::Tensor &src, const torch::Tensor &w,
                  const torch::Tensor &bias, torch::Tensor &out,
                  int Ns, int Hs, int Ws, int Cs, int ld_src,
                  int P, int Q, int Ncout,
                  int R, int S, int stride, int pad, int dil,
                  long M_total, int K_total,
                  torch::Tensor *stats = nullptr) {
    using T = scalar_t;
    auto stream = at::hip::getCurrentHIPStream();
    const float *bias_ptr = bias.defined() && bias.numel()
                                ? bias.data_ptr<float>() : nullptr;
    const T *zpage = reinterpret_cast<const T *>(zero_page_ptr(src));
    auto launch = [&](auto bc_tag, auto bm_tag) {
        constexpr int BC = decltype(bc_tag)::value;
        constexpr int BMv = decltype(bm_tag)::value;
        int tiles_m = (M_total + BMv - 1) / BMv;
        int tiles_c = (Ncout + BC - 1) / BC;
        dim3 grid(tiles_m * tiles_c);
        bool glds = (Cs % 8 == 0) && (K_total % 8 == 0);
        int nsteps = (K_total + 63) / 64;
        // 64KB double-buffer caps the largest variant at 2 blocks/CU; the
        // single-buffer form trades intra-block pipelining for occupancy
        // (policy from A/B measurement; SODKIT_IGEMM_SB=1 forces single)
        bool dbuf = glds && nsteps > 1;
        static const bool force_sb = []() {
            const char *e = getenv("SODKIT_IGEMM_SB");
            return e && e[0] == '1';
        }();
        if (force_sb) dbuf = false;
        size_t shmem = (size_t)(dbuf ? 2 : 1) * (BC + BMv) * 64 * 2;
        float *stats_ptr = nullptr;
        if (stats != nullptr) {
            *stats = torch::empty({tiles_m, 2L * Ncout},
                                  src.options().dtype(torch::kFloat32));
            stats_ptr = stats->data_ptr<float>();
        }
        auto kern = [&](auto es_tag) {
            constexpr int ESv = decltype(es_tag)::value;
            hipLaunchKernelGGL((sodkit::igemm_kernel<T, MODE, BC, BMv, ESv>),
                               grid, dim3(256), shmem, stream.stream(),
                               reinterpret_cast<const T *>(src.data_ptr()),
                               reinterpret_cast<const T *>(w.data_ptr()),
                               bias_ptr,
                               reinterpret_cast<T *>(out.data_ptr()),
                               stats_ptr, zpage,
                               Ns, Hs, Ws, Cs, ld_src, P, Q, Ncout, R, S,
                               stride, pad, dil,
                               (int)M_total, K_total, tiles_m, (int)dbuf);
        };
        if constexpr (MODE == 0) {
            if (stats_ptr != nullptr) {
                kern(std::integral_constant<int, 1>{});
                return;
            }
        }
        kern(std::integral_constant<int, 0>{});
    };
    using c64 = std::integral_constant<int, 64>;
    using c128 = std::integral_constant<int, 128>;
    bool small_c = Ncout <= 64;
    long tiles_m128 = (M_total + 127) / 128;
    bool small_m = tiles_m128 * ((Ncout + 127) / 128) < 192;
    if (small_c) {
        if (small_m) launch(c64{}, c64{});
        else launch(c64{}, c128{});
    } else {
        if (small_m) launch(c128{}, c64{});
        else launch(c128{}, c128{});
    }
}

}  // namespace

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad, int64_t dil,
                         int64_t ld) {
    CHECK_GPU(x); CHECK_GPU(w);
    TORCH_CHECK(x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    TORCH_CHECK(ld >= C, "bad channel pitch");
    int Cout = w.size(0), R = w.size(1), S = w.size(2);
    TORCH_CHECK(w.size(3) == C, "weight Cin mismatch");
    TORCH_CHECK(w.is_contiguous());
    int P = out_dim(H, R, stride, pad, dil);
    int Q = out_dim(W_, S, stride, pad, dil);
    long M = (long)N * P * Q;
    int K = R * S * C;
    auto out = torch::empty({N, Cout, P, Q},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    if (bias.defined() && bias.numel()) bias32 = bias.to(torch::kFloat32);
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_igemm<__hip_bfloat16, 0>(x, w, bias32, out, N, H, W_, C,
                                            (int)ld, P, Q, Cout, R, S,
                                            stride, pad, dil, M, K);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_igemm<__half, 0>(x, w, bias32, out, N, H, W_, C, (int)ld,
                                    P, Q, Cout, R, S, stride, pad, dil, M, K);
        }));
    return out;
}

// fwd + fused BN stats partials: returns {y, slabs[tiles_m, 2*Cout]} —
// slabs feed bn_finalize_slabs / bn_slab_sum (csrc/bnrelu.hip) so the BN
// stats pass never re-reads y.
std::vector<torch::Tensor> conv2d_fwd_stats(torch::Tensor x, torch::Tensor w,
                                            torch::Tensor bias,
                                            int64_t stride, int64_t pad,
                                            int64_t dil, int64_t ld) {
    CHECK_GPU(x); CHECK_GPU(w);
    TORCH_CHECK(x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    TORCH_CHECK(ld >= C, "bad channel pitch");
    int Cout = w.size(0), R = w.size(1), S = w.size(2);
    TORCH_CHECK(w.size(3) == C, "weight Cin mismatch");
    TORCH_CHECK(w.is_contiguous());
    int P = out_dim(H, R, stride, pad, dil);
    int Q = out_dim(W_, S, stride, pad, dil);
    long M = (long)N * P * Q;
    int K = R * S * C;
    auto out = torch::empty({N, Cout, P, Q},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    if (bias.defined() && bias.numel()) bias32 = bias.to(torch::kFloat32);
    torch::Tensor slabs;
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd_stats",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_igemm<__hip_bfloat16, 0>(x, w, bias32, out, N, H, W_, C,
                                            (int)ld, P, Q, Cout, R, S,
                                            stride, pad, dil, M, K, &slabs);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_igemm<__half, 0>(x, w, bias32, out, N, H, W_, C, (int)ld,
                                    P, Q, Cout, R, S, stride, pad, dil, M, K,
                                    &slabs);
        }));
    return {out, slabs};
}

torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w_t, int64_t H,
                           int64_t W_, int64_t stride, int64_t pad,
                           int64_t dil, int64_t lddy) {
    CHECK_GPU(dy); CHECK_GPU(w_t);
    int N = dy.size(0), Cout = dy.size(1), P = dy.size(2), Q = dy.size(3);
    TORCH_CHECK(lddy >= Cout, "bad channel pitch");
    // w_t: [Cin][R][S][Cout]
    int Cin = w_t.size(0), R = w_t.size(1), S = w_t.size(2);
    TORCH_CHECK(w_t.size(3) == Cout && w_t.is_contiguous());
    long M = (long)N * H * W_;
    int K = R * S * Cout;
    auto dx = torch::empty({N, Cin, H, W_},
                           dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor nobias;
    AT_DISPATCH_SWITCH(dy.scalar_type(), "conv2d_dgrad",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_igemm<__hip_bfloat16, 1>(dy, w_t, nobias, dx, N, P, Q, Cout,
                                            (int)lddy, H, W_, Cin, R, S,
                                            stride, pad, dil, M, K);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_igemm<__half, 1>(dy, w_t, nobias, dx, N, P, Q, Cout,
                                    (int)lddy, H, W_, Cin, R, S,
                                    stride, pad, dil, M, K);
        }));
    return dx;
}

torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor dy, int64_t R,
                           int64_t S, int64_t stride, int64_t pad,
                           int64_t dil, int64_t ldx, int64_t lddy) {
    CHECK_GPU(x); CHECK_GPU(dy);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    int Cout = dy.size(1), P = dy.size(2), Q = dy.size(3);
    TORCH_CHECK(dy.size(0) == N);
    long M = (long)N * P * Q;
    int Kw = R * S * C;
    auto stream = at::hip::getCurrentHIPStream();

    auto run = [&](auto t_tag, auto bc_tag, auto bj_tag, auto nt_tag)
        -> torch::Tensor {
        using T = typename decltype(t_tag)::type;
        constexpr int BC = decltype(bc_tag)::value;
        constexpr int BJ = decltype(bj_tag)::value;
        constexpr int NT = decltype(nt_tag)::value;
        // the hot BCOUT=64 (decoder) variant runs BKM=128: half the
        // barriers per MFMA at 48KB LDS (3 blocks/CU); BCOUT=128 keeps
        // BKM=64 (64KB would drop it to 2 blocks/CU); the 8-wave 256-tile
        // variants stay at BKM=64 (64KB LDS, 2 blocks/CU)
        constexpr int BKM = (BC == 64) ? 128 : 64;
        int tiles_c = (Cout + BC - 1) / BC;
        int tiles_j = (Kw + BJ - 1) / BJ;
        int tiles = tiles_c * tiles_j;
        // fill the chip: ~2 blocks per CU (256 CUs) regardless of tile count
        int splitk = std::max(1, std::min(512, 512 / std::max(tiles, 1)));
        splitk = (int)std::min<long>(splitk, (M + BKM - 1) / BKM);
        if (NT == 512)   // 8-wave tiles: keep >=8 K-steps per split so the
            //              prologue + 256KB fp32 partial store amortize
            splitk = std::max(1, std::min<int>(splitk, (int)(M / BKM / 8)));
        // (a >=4-K-steps cap on the 4-wave tiles was TRIED and REVERTED:
        //  at tiny-M shapes the lost block parallelism outweighed the
        //  prologue amortization — trans-conv wgrad 43 -> 33 TF, bench
        //  1217 -> 1200.  Do not repeat.)
        long m_per_split = ((M + splitk - 1) / splitk + BKM - 1) / BKM * BKM;
        splitk = (int)((M + m_per_split - 1) / m_per_split);
        auto partial = torch::empty({splitk, (long)Cout * Kw},
                                    x.options().dtype(torch::kFloat32));
        dim3 grid(tiles, splitk);
        hipLaunchKernelGGL((sodkit::wgrad_kernel<T, BC, BJ, BKM, NT>), grid,
                           dim3(NT), 0, stream.stream(),
                           reinterpret_cast<const T *>(x.data_ptr()),
                           reinterpret_cast<const T *>(dy.data_ptr()),
                           partial.data_ptr<float>(),
                           N, H, W_, C, (int)ldx, (int)lddy,
                           P, Q, Cout, (int)R, (int)S,
                           (int)stride, (int)pad, (int)dil, Kw, (int)M,
                           (int)m_per_split, tiles_c);
        auto dw = torch::empty({Cout, R, S, C}, x.options());
        long n_elems = (long)Cout * Kw;
        int blocks = std::min<long>((n_elems + 255) / 256, 2048);
        hipLaunchKernelGGL((sodkit::splitk_reduce_kernel<T>), dim3(blocks),
                           dim3(256), 0, stream.stream(),
                           partial.data_ptr<float>(),
                           reinterpret_cast<T *>(dw.data_ptr()), n_elems, splitk);
        return dw;
    };

    struct BF { using type = __hip_bfloat16; };
    struct FP { using type = __half; };
    using c64 = std::integral_constant<int, 64>;
    using c128 = std::integral_constant<int, 128>;
    using c256 = std::integral_constant<int, 256>;
    using c512 = std::integral_constant<int, 512>;
    torch::Tensor dw;
    // tile policy: intensity (MFMA flops per staged LDS byte) bounds the
    // kernel, BUT a bigger tile means fewer tiles, a larger split-K to
    // fill the chip, and splitk x Cout x Kw x 4B of fp32 partial traffic —
    // measured r2: at this model's small-M big-Cout shapes (M 3.8-62k)
    // the 256-tile variants LOST 30-40% to partial-store blowup (l4_1x1e
    // 118 vs 180 TF).  The 8-wave tiles need M large enough that split-K
    // stays shallow; gate on M >= 200k pixels (512-res class shapes).
    bool small_c = Cout <= 64;
    bool mbig = M >= 200000;
    bool big = Cout >= 256 && Kw >= 256 && mbig;     // 256x256, 8 waves
    bool tall = Cout >= 256 && Kw >= 128 && !big && mbig;  // 256x128
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_wgrad",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            if (big) dw = run(BF{}, c256{}, c256{}, c512{});
            else if (tall) dw = run(BF{}, c256{}, c128{}, c512{});
            else if (small_c) dw = run(BF{}, c64{}, c128{}, c256{});
            else dw = run(BF{}, c128{}, c128{}, c256{});
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            if (big) dw = run(FP{}, c256{}, c256{}, c512{});
            else if (tall) dw = run(FP{}, c256{}, c128{}, c512{});
            else if (small_c) dw = run(FP{}, c64{}, c128{}, c256{});
            else dw = run(FP{}, c128{}, c128{}, c256{});
        }));
    return dw;
}

torch::Tensor channel_sum(torch::Tensor x, int64_t ld) {
    CHECK_GPU(x);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    long M = (long)N * H * W_;
    auto out = torch::zeros({C}, x.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    bool small = C <= 16;
    bool flat = (C == 1 && ld == 1);
    dim3 grid_small(std::min<long>((M + 2047) / 2048, 512));
    dim3 grid(std::min<long>((M + 255) / 256, 1024), (C + 255) / 256);
    AT_DISPATCH_SWITCH(x.scalar_type(), "channel_sum",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            using T = __hip_bfloat16;
            if (small && flat)
                hipLaunchKernelGGL((sodkit::channel_sum_small_kernel<T, true>),
                                   grid_small, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
            else if (small)
                hipLaunchKernelGGL((sodkit::channel_sum_small_kernel<T, false>),
                                   grid_small, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
            else
                hipLaunchKernelGGL((sodkit::channel_sum_kernel<T>),
                                   grid, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            using T = __half;
            if (small && flat)
                hipLaunchKernelGGL((sodkit::channel_sum_small_kernel<T, true>),
                                   grid_small, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
            else if (small)
                hipLaunchKernelGGL((sodkit::channel_sum_small_kernel<T, false>),
                                   grid_small, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
            else
                hipLaunchKernelGGL((sodkit::channel_sum_kernel<T>),
                                   grid, dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   out.data_ptr<float>(), M, C, (int)ld);
        })
        AT_DISPATCH_CASE(at::kFloat, [&] {
            hipLaunchKernelGGL((sodkit::channel_sum_kernel<float>),
                               grid, dim3(256), 0, stream.stream(),
                               reinterpret_cast<const float *>(x.data_ptr()),
                               out.data_ptr<float>(), M, C, (int)ld);
        }));
    return out;
}
