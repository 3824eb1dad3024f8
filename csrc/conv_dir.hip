// sodkit — direct (non-im2col) fwd/dgrad kernel for 3x3 stride-1
// dilation-1 pad-1 convolutions with C == 64, Cout <= 64 on CDNA4: the
// MINet/HDFNet decoder + AIM/SIM class, which is ~18% of the training
// step (kernel_summary r2: the 176/88-square 64-channel convs).
//
// The im2col igemm kernel stages every x element ~9x (once per tap its
// im2col row touches): B-operand traffic = M*K*2 bytes (1.1 GB per
// 176-square conv), which L2 bandwidth bounds at ~400 TF.  Here a block
// stages the three input rows of an output row ONCE in their natural
// [q][c] layout and the nine taps read ROW-shifted windows — for the
// forward direction the MFMA K dimension is j=(r,s,c), which is
// c-contiguous in both operands, so every fragment read is a plain
// aligned b128 (no gather, no transpose, no alignment tricks):
//
//   A = w[cout][(r,s,c)]      KRSC rows ARE the GEMM-A rows (LDS-resident,
//                             staged once per block, ~72 KB)
//   B[q][(r,s,c)] = x[p+r-1][q+s-1][c]   from 3 row images [q][c]
//
// dgrad is the same kernel with A = the permuted [Cin][r][s][Cout] copy
// read at the flipped tap (8-tap) and B = dy rows (same shift form).
//
// Wave layout: 8 waves = 2 cout-halves x 4 q-groups; q-group frags are
// strided (g, g+4, g+8) so ragged W wastes at most the tail fragment.
// Row pitches padded so (pitch/4) % 64 == 4|36 (16-row b128 fragment
// reads hit 16 distinct bank quads).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace cdir {

typedef short s16x4_ __attribute__((ext_vector_type(4)));
typedef __attribute__((address_space(3))) void lds_void_t;
typedef __attribute__((address_space(1))) const void gbl_void_t;

SOD_DEVINL void glds16(const void *gsrc, char *lds_dst) {
    __builtin_amdgcn_global_load_lds((const gbl_void_t *)gsrc,
                                     (lds_void_t *)lds_dst, 16, 0, 0);
}

// x images keep their natural row size (CC*2 bytes) so `global_load_lds`
// can stage them; bank conflicts are avoided by a 16B-block XOR swizzle
// applied to the glds SOURCE address and inverted on the read side:
//   block' = block ^ s(row),  s(row) = (row & (B-1)) ^ ((row >> log2B) & (B-1))
// which makes the 16 consecutive rows of a b128 fragment read hit 16
// distinct bank quads.  The weight image keeps a padded pitch (register-
// staged once; rows are 1152/576 B, not glds-friendly anyway).
template <int CC> struct Pitch;
template <> struct Pitch<64> {
    static constexpr int X = 128;
    static constexpr int W = 1168;  // 1152 data + 16 (stride 292)
    SOD_DEVINL static int swz(int row) { return (row & 7) ^ ((row >> 3) & 7); }
};
template <> struct Pitch<32> {
    static constexpr int X = 64;
    static constexpr int W = 592;   // 576 data + 16 (stride 148)
    SOD_DEVINL static int swz(int row) { return (row & 3) ^ ((row >> 2) & 3); }
};

// MODE 0: fwd (src = x, wts = w KRSC). MODE 1: dgrad (src = dy,
// wts = w_t [Cin][R][S][Cout], taps flipped on the A side).
// ROWS = output rows per sweep: 2 for W <= 96 (H even) amortizes the
// staging and barriers over twice the MFMA — the overlapping 4-row input
// window serves both output rows, and the A fragments are shared.
// ES=1 (fwd): also accumulate per-channel BN stat partials over the
// ROUNDED outputs across the block's whole row sweep; one slab row per
// block in stats_out[blocks][2*Cout] (bn_reduce_finalize layout).
template <typename T, int MODE, int ROWS, int CC, int ES = 0>
__global__ __launch_bounds__(512, 1)
void conv_dir_kernel(const T *__restrict__ src, const T *__restrict__ wts,
                     const float *__restrict__ bias, T *__restrict__ out,
                     float *__restrict__ stats_out,
                     const T *__restrict__ zero_page,
                     int N, int H, int W, int ldsrc, int ldout, int Cout,
                     int rows_per_blk, int total_rows, int nq_row) {
    // q-fragment slots: 4 q-wave-groups x 3 slots = 12.  ROWS==1 assigns
    // ceil(nq_row/4) slots per group (round-robin).  ROWS==2 POOLS the
    // pair's 2*nq_row fragments across all 12 slots (row = fr / nq_row):
    // at W=88 the per-row split left 6 frags on 8 slots (75% of the MFMA
    // lanes doing work); the pool uses 12/12.
    const int nqf = (ROWS == 2) ? 3 : (nq_row + 3) / 4;
    const int pool = ROWS * nq_row;
    constexpr int XPITCH = Pitch<CC>::X;
    constexpr int WPITCH = Pitch<CC>::W;
    constexpr int WROW = 9 * CC * 2;     // weight row bytes
    constexpr int NKT = 9 * CC / 32;     // k-tiles of 32
    extern __shared__ char lds[];
    char *wbuf = lds;                               // [Cout<=64][WPITCH]
    // +16 extra zero rows: the ragged last q-fragment reads rows past
    // W+1; they must stay in-bounds and zero
    const int xrows = W + 18;
    char *xbuf[ROWS + 2];
#pragma unroll
    for (int d = 0; d < ROWS + 2; ++d)
        xbuf[d] = lds + 64 * WPITCH + d * xrows * XPITCH;
    const int lds_bytes = 64 * WPITCH + (ROWS + 2) * xrows * XPITCH;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int wm = wave >> 2;          // cout half
    const int wq = wave & 3;           // q group

    const int r0 = blockIdx.x * rows_per_blk;
    const int r1 = min(r0 + rows_per_blk, total_rows);

    // ---- one-time zero fill (halo rows / pads stay zero) ----
    for (int i = t * 8; i < lds_bytes; i += 512 * 8)
        *reinterpret_cast<long *>(lds + i) = 0;
    __syncthreads();

    // ---- one-time weight staging: [row][WROW bytes] -> padded pitch ----
    {
        int row = t >> 3, b16 = t & 7;
        if (row < Cout) {
            const char *g = reinterpret_cast<const char *>(wts) + (size_t)row * WROW;
            char *d = wbuf + row * WPITCH;
            // WROW/16 chunks per row (NOT necessarily a multiple of 8:
            // CC=32 has 36 — the multiple-of-128 loop dropped the tail)
            for (int c16 = b16; c16 < WROW / 16; c16 += 8) {
                int off = c16 * 16;
                *reinterpret_cast<s16x8 *>(d + off) =
                    *reinterpret_cast<const s16x8 *>(g + off);
            }
        }
    }

    constexpr int PIX_PER_PASS = 512 / (CC / 8);
    const int passes = (W + PIX_PER_PASS - 1) / PIX_PER_PASS;

    // ES: per-lane stat partials, live across the whole row sweep (+16
    // VGPRs; the kernel is LDS-bound at 1 block/CU, not register-bound)
    float dsum[2][4], dssq[2][4];
    if constexpr (ES) {
#pragma unroll
        for (int a = 0; a < 2; ++a)
#pragma unroll
            for (int e = 0; e < 4; ++e) { dsum[a][e] = 0.f; dssq[a][e] = 0.f; }
    }

    for (int g = r0; g < r1; g += ROWS) {
        const int n = g / H;
        const int p = g % H;   // ROWS==2: H even and g even => p, p+1 same image

        __syncthreads();   // previous row's MFMA reads done (also covers
        //                    the one-time staging on the first iteration)

        // ---- stage the ROWS+2 input rows via glds (async DMA): one
        //      instruction fills 1024 B = 1024/XPITCH LDS rows; the
        //      per-lane SOURCE is pre-swizzled so reads are conflict-free
        {
            constexpr int BLKS = CC / 8;            // 16B blocks per row
            constexpr int RPI = 64 / BLKS;          // LDS rows per glds
            const int my_row = lane / BLKS;         // row within the chunk
            const int my_blk = lane % BLKS;
            const int nchunks = (W + RPI - 1) / RPI;
#pragma unroll
            for (int d = 0; d < ROWS + 2; ++d) {
                int ih = p + d - 1;
                bool rv = ih >= 0 && ih < H;
                for (int ch = wave; ch < nchunks; ch += 8) {
                    int q = ch * RPI + my_row;      // pixel this lane feeds
                    int ldsrow = q + 1;             // +halo
                    int sblk = my_blk ^ Pitch<CC>::swz(ldsrow);
                    const void *g = (rv && q < W)
                        ? (const void *)(src + ((size_t)(n * H + ih) * W + q)
                                             * ldsrc + sblk * 8)
                        : (const void *)zero_page;
                    glds16(g, xbuf[d] + (size_t)(ch * RPI + 1) * XPITCH
                                  + lane * 16);
                }
            }
        }
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();

        // ---- MFMA: D[cout][q] per output row over the K = 9*CC dim ----
        f32x4 acc[2][3];
#pragma unroll
        for (int a = 0; a < 2; ++a)
#pragma unroll
            for (int b = 0; b < 3; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

        const int oct16 = ((lane >> 4) << 4);       // k-octet byte
        const int arow = (wm * 32 + (lane & 15)) * WPITCH;
#pragma unroll
        for (int kt = 0; kt < NKT; ++kt) {
            const int tap = (CC == 64) ? (kt >> 1) : kt;
            const int chalf = (CC == 64) ? (kt & 1) : 0;
            const int atap = MODE ? (8 - tap) : tap;
            const int abyte = atap * (CC * 2) + chalf * 64 + oct16;
            s16x8 a0 = *reinterpret_cast<const s16x8 *>(wbuf + arow + abyte);
            s16x8 a1 = *reinterpret_cast<const s16x8 *>(wbuf + arow + 16 * WPITCH
                                                        + abyte);
            const int rr = tap / 3, ss = tap % 3;
            const int bblk = (chalf * 64 + oct16) >> 4;  // 16B block index
#pragma unroll
            for (int i = 0; i < 3; ++i) {
                if (i >= nqf) break;
                const int fr = wq + i * 4;
                if (ROWS == 2 && fr >= pool) continue;
                const int rw = (ROWS == 2 && fr >= nq_row) ? 1 : 0;
                int q0 = (fr - rw * nq_row) * 16;
                // row q + s - 1, +1 halo base; rows beyond W+1 are zero pad
                int row = q0 + (lane & 15) + ss;
                const int bswz = (bblk ^ Pitch<CC>::swz(row)) << 4;
                s16x8 b = *reinterpret_cast<const s16x8 *>(
                    xbuf[rr + rw] + row * XPITCH + bswz);
                acc[0][i] = MfmaOp<T>::mma(a0, b, acc[0][i]);
                acc[1][i] = MfmaOp<T>::mma(a1, b, acc[1][i]);
            }
        }

        // ---- store the output rows (NHWC, 4 couts per lane) ----
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            int cout = wm * 32 + mf * 16 + ((lane >> 4) << 2);
            if (cout >= Cout) continue;
            float b4[4] = {0.f, 0.f, 0.f, 0.f};
            if (bias != nullptr) {
#pragma unroll
                for (int e = 0; e < 4; ++e)
                    if (cout + e < Cout) b4[e] = bias[cout + e];
            }
#pragma unroll
            for (int i = 0; i < 3; ++i) {
                if (i >= nqf) break;
                const int fr = wq + i * 4;
                if (ROWS == 2 && fr >= pool) continue;
                const int rw = (ROWS == 2 && fr >= nq_row) ? 1 : 0;
                int q = (fr - rw * nq_row) * 16 + (lane & 15);
                if (q >= W) continue;
                const size_t obase = (size_t)(n * H + p + rw) * W;
                s16x4_ v;
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    T tv = ElemTraits<T>::from_f32(acc[mf][i][e] + b4[e]);
                    v[e] = *reinterpret_cast<const short *>(&tv);
                    if constexpr (ES) {
                        float fr = s16_as_f32<T>(v[e]);
                        dsum[mf][e] += fr;
                        dssq[mf][e] += fr * fr;
                    }
                }
                *reinterpret_cast<s16x4_ *>(
                    out + (obase + q) * ldout + cout) = v;
            }
        }
    }

    if constexpr (ES) {
        struct ESR { float v[4][2][64]; };          // [wq][sum|ssq][c]
        __shared__ __attribute__((aligned(16))) char esr_raw[sizeof(ESR)];
        ESR *esr = reinterpret_cast<ESR *>(esr_raw);
        __syncthreads();
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            float s4[4], q4[4];
#pragma unroll
            for (int e = 0; e < 4; ++e) { s4[e] = dsum[mf][e]; q4[e] = dssq[mf][e]; }
#pragma unroll
            for (int d = 1; d < 16; d <<= 1)
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    s4[e] += __shfl_xor(s4[e], d, 64);
                    q4[e] += __shfl_xor(q4[e], d, 64);
                }
            int cl = wm * 32 + mf * 16 + ((lane >> 4) << 2);
            if ((lane & 15) == 0 && cl < Cout) {
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    esr->v[wq][0][cl + e] = s4[e];
                    esr->v[wq][1][cl + e] = q4[e];
                }
            }
        }
        __syncthreads();
        long base = (long)blockIdx.x * 2 * Cout;
        for (int i = t; i < 2 * Cout; i += 512)
            stats_out[base + i] =
                ((esr->v[0][i / Cout][i % Cout] + esr->v[1][i / Cout][i % Cout]) +
                 (esr->v[2][i / Cout][i % Cout] + esr->v[3][i / Cout][i % Cout]));
    }
}

}  // namespace cdir
}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

namespace {
void *cdir_zero_page(const torch::Tensor &like) {
    static torch::Tensor z;
    if (!z.defined() || z.device() != like.device())
        z = torch::zeros({32}, like.options().dtype(torch::kFloat32));
    return z.data_ptr();
}
}  // namespace

bool conv_dir_eligible(int64_t R, int64_t S, int64_t stride, int64_t pad,
                       int64_t dil, int64_t Cin, int64_t Cout, int64_t W) {
    if (R != 3 || S != 3 || stride != 1 || pad != 1 || dil != 1) return false;
    if (Cin != 64 && Cin != 32) return false;  // j-chunking needs C in {32,64}
    if (Cout > 64 || Cout % 4 != 0) return false;
    if (W > 184 || W < 16) return false;
    int bufs = (W <= 96) ? 4 : 3;   // ROWS==2 window below 96
    int xp = Cin == 64 ? sodkit::cdir::Pitch<64>::X : sodkit::cdir::Pitch<32>::X;
    int wp = Cin == 64 ? sodkit::cdir::Pitch<64>::W : sodkit::cdir::Pitch<32>::W;
    // margin: the ES=1 fwd variant adds a 2 KB static LDS combine plane
    // and its dynamic opt-in is 160K-4K (see launch_conv_dir_r)
    return 64 * wp + bufs * (W + 18) * xp <= 156 * 1024;
}

namespace {

template <typename T, int MODE, int ROWS, int CC, int ES = 0>
void launch_conv_dir_r(const torch::Tensor &src, const torch::Tensor &wts,
                       const float *bias_ptr, torch::Tensor &out,
                       int N, int H, int W, int ldsrc, int ldout, int Cout,
                       torch::Tensor *stats = nullptr) {
    auto stream = at::hip::getCurrentHIPStream();
    const int total_rows = N * H;
    // ~2 row-sweeping blocks per CU; ROWS==2 keeps block ranges pair-aligned
    int blocks = std::min(512, total_rows / ROWS);
    int rows_per_blk = (total_rows / ROWS + blocks - 1) / blocks * ROWS;
    blocks = (total_rows + rows_per_blk - 1) / rows_per_blk;
    const int nq_row = (W + 15) / 16;        // q fragments per output row
    const size_t shmem = 64 * (size_t)sodkit::cdir::Pitch<CC>::W
                         + (ROWS + 2) * (size_t)(W + 18) * sodkit::cdir::Pitch<CC>::X;
    float *stats_ptr = nullptr;
    if constexpr (ES) {
        *stats = torch::empty({blocks, 2L * Cout},
                              src.options().dtype(torch::kFloat32));
        stats_ptr = stats->data_ptr<float>();
    }
    static bool attr = false;
    if (!attr) {
        // the ES=1 kernel carries a 2 KB static combine plane: the dynamic
        // opt-in must leave room for it or hipFuncSetAttribute FAILS — and
        // a swallowed failure here leaves hipErrorInvalidValue in the
        // thread's error slot, which torch's next launch check reports as
        // a phantom launch failure (first call of each instantiation only)
        (void)hipFuncSetAttribute(
            (const void *)sodkit::cdir::conv_dir_kernel<T, MODE, ROWS, CC, ES>,
            hipFuncAttributeMaxDynamicSharedMemorySize,
            160 * 1024 - (ES ? 4096 : 0));
        (void)hipGetLastError();   // never let an opt-in failure poison it
        attr = true;
    }
    hipLaunchKernelGGL((sodkit::cdir::conv_dir_kernel<T, MODE, ROWS, CC, ES>),
                       dim3(blocks), dim3(512), shmem, stream.stream(),
                       reinterpret_cast<const T *>(src.data_ptr()),
                       reinterpret_cast<const T *>(wts.data_ptr()), bias_ptr,
                       reinterpret_cast<T *>(out.data_ptr()), stats_ptr,
                       reinterpret_cast<const T *>(cdir_zero_page(src)),
                       N, H, W, ldsrc, ldout, Cout, rows_per_blk, total_rows,
                       nq_row);
}

template <typename T, int MODE, int ES = 0>
void launch_conv_dir(const torch::Tensor &src, const torch::Tensor &wts,
                     const float *bias_ptr, torch::Tensor &out,
                     int N, int H, int W, int ldsrc, int ldout, int Cout,
                     int CC, torch::Tensor *stats = nullptr) {
    // pair rows when the window fits and pairs cannot straddle images
    bool pair = W <= 96 && H % 2 == 0;
    if (CC == 64) {
        if (pair) launch_conv_dir_r<T, MODE, 2, 64, ES>(src, wts, bias_ptr, out,
                                                    N, H, W, ldsrc, ldout, Cout, stats);
        else launch_conv_dir_r<T, MODE, 1, 64, ES>(src, wts, bias_ptr, out,
                                               N, H, W, ldsrc, ldout, Cout, stats);
    } else {
        if (pair) launch_conv_dir_r<T, MODE, 2, 32, ES>(src, wts, bias_ptr, out,
                                                    N, H, W, ldsrc, ldout, Cout, stats);
        else launch_conv_dir_r<T, MODE, 1, 32, ES>(src, wts, bias_ptr, out,
                                               N, H, W, ldsrc, ldout, Cout, stats);
    }
}

}  // namespace

torch::Tensor conv2d_fwd_dir(torch::Tensor x, torch::Tensor w,
                             torch::Tensor bias, int64_t ld) {
    TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    int Cout = w.size(0);
    TORCH_CHECK(w.size(1) == 3 && w.size(2) == 3 && w.size(3) == C
                && w.is_contiguous());
    TORCH_CHECK(conv_dir_eligible(3, 3, 1, 1, 1, C, Cout, W_), "not eligible");
    TORCH_CHECK(ld >= C, "bad channel pitch");
    auto out = torch::empty({N, Cout, H, W_},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    const float *bp = nullptr;
    if (bias.defined() && bias.numel()) {
        bias32 = bias.to(torch::kFloat32).contiguous();
        bp = bias32.data_ptr<float>();
    }
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd_dir",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv_dir<__hip_bfloat16, 0>(x, w, bp, out, N, H, W_,
                                               (int)ld, Cout, Cout, C);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv_dir<__half, 0>(x, w, bp, out, N, H, W_, (int)ld,
                                       Cout, Cout, C);
        }));
    return out;
}

// fwd + fused BN stat partials (same contract as conv2d_fwd_stats)
std::vector<torch::Tensor> conv2d_fwd_dir_stats(torch::Tensor x,
                                                torch::Tensor w,
                                                torch::Tensor bias,
                                                int64_t ld) {
    TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    int Cout = w.size(0);
    TORCH_CHECK(w.size(1) == 3 && w.size(2) == 3 && w.size(3) == C
                && w.is_contiguous());
    TORCH_CHECK(conv_dir_eligible(3, 3, 1, 1, 1, C, Cout, W_), "not eligible");
    TORCH_CHECK(ld >= C, "bad channel pitch");
    auto out = torch::empty({N, Cout, H, W_},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    const float *bp = nullptr;
    if (bias.defined() && bias.numel()) {
        bias32 = bias.to(torch::kFloat32).contiguous();
        bp = bias32.data_ptr<float>();
    }
    torch::Tensor slabs;
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd_dir_stats",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv_dir<__hip_bfloat16, 0, 1>(x, w, bp, out, N, H, W_,
                                                  (int)ld, Cout, Cout, C,
                                                  &slabs);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv_dir<__half, 0, 1>(x, w, bp, out, N, H, W_, (int)ld,
                                          Cout, Cout, C, &slabs);
        }));
    return {out, slabs};
}

torch::Tensor conv2d_dgrad_dir(torch::Tensor dy, torch::Tensor w_t,
                               int64_t lddy) {
    TORCH_CHECK(dy.is_cuda() && w_t.is_cuda());
    int N = dy.size(0), Cout = dy.size(1), H = dy.size(2), W_ = dy.size(3);
    int Cin = w_t.size(0);
    TORCH_CHECK(w_t.size(1) == 3 && w_t.size(2) == 3 && w_t.size(3) == Cout
                && w_t.is_contiguous());
    // roles swap: the gemm "C" is Cout(=64), output channels are Cin
    TORCH_CHECK(conv_dir_eligible(3, 3, 1, 1, 1, Cout, Cin, W_), "not eligible");
    TORCH_CHECK(lddy >= Cout, "bad channel pitch");
    auto dx = torch::empty({N, Cin, H, W_},
                           dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    AT_DISPATCH_SWITCH(dy.scalar_type(), "conv2d_dgrad_dir",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv_dir<__hip_bfloat16, 1>(dy, w_t, nullptr, dx, N, H, W_,
                                               (int)lddy, Cin, Cin, Cout);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv_dir<__half, 1>(dy, w_t, nullptr, dx, N, H, W_,
                                       (int)lddy, Cin, Cin, Cout);
        }));
    return dx;
}
