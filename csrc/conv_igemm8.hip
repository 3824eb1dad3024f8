// sodkit — 8-phase 256x256 implicit-GEMM conv (fwd + dgrad), the
// next-generation structure over csrc/conv_igemm.hip's 128-square
// 2-barrier kernel.  OPT-IN this round: dispatched only under
// SODKIT_IGEMM_8P=1 (sodkit/ops/conv.py) for eligible shapes
// (Cout % 256 == 0, Cin % 8 == 0); the 128-square kernel remains the
// default and the fallback.
//
// Structure reconstructed and verified from the CDNA4 guide's 8-phase
// template — derivation and measured status in docs/IGEMM8_PLAN.md:
// the standalone twin (tools/gemm8.hip) hits 1008 TF bf16 @4k^3 GEMM
// with a 25/25 race screen, and the conv adaptation (tools/conv8.hip)
// refcheck-passes fwd (1x1/3x3/stride/dilation/K- and M-tails) and
// dgrad (stride 1/2, dilation 2) on MI355X at 689 TF on the canonical
// 3x3 256ch shape (shipped kernel: 350-500 TF band).
//
// Geometry: C tile 256(cout) x 256(pixels), BK=64, 512 threads as
// 2(cout-half) x 4(pixel-quarter) waves, per-wave 128x64 = 8x4 MFMA
// fragments.  LDS 128 KB = 2 K-tile buffers x {A=weights, B=im2col} x
// {k-half} images of [256 rows][32 k] with the st_16x32 swizzle.  The
// K-HALF image split makes the (p+2)-offset half-image staging and the
// two s_waitcnt vmcnt(6) checkpoints per 4-phase group race-free (the
// full stagger proof is in tools/gemm8.hip's header).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace ig8 {

typedef short s16x4_ __attribute__((ext_vector_type(4)));
typedef __attribute__((address_space(3))) void lds_void8_t;
typedef __attribute__((address_space(1))) const void gbl_void8_t;

SOD_DEVINL void glds16_(const void *gsrc, char *lds_dst) {
    __builtin_amdgcn_global_load_lds((const gbl_void8_t *)gsrc,
                                     (lds_void8_t *)lds_dst, 16, 0, 0);
}

SOD_DEVINL int swz(int x) { return x ^ (((x >> 9) & 1) << 5); }

constexpr int IMG_BYTES = 16384;   // [256 rows][32 k] 16-bit
SOD_DEVINL int img_off(int buf, int op, int kh) {
    return ((buf * 2 + op) * 2 + kh) * IMG_BYTES;
}

// MODE=0 fwd; MODE=1 dgrad (x=dy over the OH x OW grid, w permuted to
// [Cin][R][S][Cout], gather inverts the conv geometry per tap).
// ES=1 (fwd): also emit per-channel BN stat partials over the ROUNDED
// outputs into stats_out[tiles_m][2*Cout] (bn_reduce_finalize layout) —
// same scheme as the 128-square igemm kernel; the l1/l2/l3 1x1 expands
// this backend owns are the LARGEST BatchNorm inputs in the model.
template <typename T, int MODE, int ES = 0>
__global__ __launch_bounds__(512, 1)
void conv8_kernel(const T *__restrict__ x, const T *__restrict__ w,
                  const float *__restrict__ bias, T *__restrict__ out,
                  float *__restrict__ stats_out,
                  const T *__restrict__ zero_page,
                  int N_, int H, int W, int Cin, int ld, int Cout,
                  int R, int S, int stride, int pad, int dil,
                  int OH, int OW) {
    extern __shared__ char lds[];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int wm = wave >> 2, wn = wave & 3;
    const int K = R * S * Cin;
    const long M = (long)N_ * OH * OW;

    int nwg = gridDim.x;
    int q = nwg / 8, r8 = nwg % 8;
    int xcd = blockIdx.x % 8, seq = blockIdx.x / 8;
    int wgid = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q) + seq;
    const int ntile = (int)((M + 255) / 256);
    const int tc = wgid / ntile;
    const long tp = (long)(wgid % ntile);
    const size_t Kb = (size_t)K * 2;
    const T *Abase = w + (size_t)tc * 256 * K;

    const int NKT = (K + 63) / 64;

    int srow[2], scol[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int s = swz((j * 8192 + tid * 16) & (IMG_BYTES - 1));
        srow[j] = s >> 6;
        scol[j] = s & 63;
    }
    int pn[2], poh[2], pow_[2];
    bool pvalid[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        long m = tp * 256 + srow[j];
        pvalid[j] = m < M;
        long mm = pvalid[j] ? m : 0;
        pn[j] = (int)(mm / ((long)OH * OW));
        int rem = (int)(mm % ((long)OH * OW));
        poh[j] = rem / OW;
        pow_[j] = rem % OW;
    }

    auto stage = [&](int kt, int op, int kh) {
        int ktc = kt < NKT ? kt : 0;
        char *dst = lds + img_off(kt & 1, op, kh);
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int k = ktc * 64 + kh * 32 + (scol[j] >> 1);
            const char *g;
            if (op == 0) {
                g = (const char *)(k < K
                        ? (const void *)((const char *)Abase +
                                         (size_t)srow[j] * Kb + (size_t)k * 2)
                        : (const void *)zero_page);
            } else {
                int tap = k / Cin, c = k - tap * Cin;
                int rr = tap / S, ss = tap - rr * S;
                int ih, iw;
                bool ok = pvalid[j] && k < K;
                if constexpr (MODE == 0) {
                    ih = poh[j] * stride - pad + rr * dil;
                    iw = pow_[j] * stride - pad + ss * dil;
                    ok = ok && ih >= 0 && ih < H && iw >= 0 && iw < W;
                } else {
                    int nh = poh[j] + pad - rr * dil;
                    int nw = pow_[j] + pad - ss * dil;
                    ih = nh / stride; iw = nw / stride;
                    ok = ok && nh >= 0 && nw >= 0 &&
                         nh == ih * stride && nw == iw * stride &&
                         ih < H && iw < W;
                }
                g = ok ? (const char *)(x + (((size_t)pn[j] * H + ih) * W + iw)
                                                * ld + c)
                       : (const char *)zero_page;
            }
            glds16_(g, dst + j * 8192 + tid * 16);
        }
    };
    auto stage_for = [&](int a) {      // absolute phase a
        int hp = (a + 2) & 3;
        stage((a + 6) >> 2, hp & 1, hp >> 1);
    };

    int a_off[8], b_off[2][2];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
        a_off[mf] = swz((wm * 128 + mf * 16 + (lane & 15)) * 64 +
                        ((lane >> 4) << 4));
#pragma unroll
    for (int nq = 0; nq < 2; ++nq)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
            b_off[nq][nf] = swz((wn * 64 + nq * 32 + nf * 16 + (lane & 15)) * 64 +
                                ((lane >> 4) << 4));

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    stage(0, 0, 0); stage(0, 1, 0); stage(0, 0, 1);
    stage(0, 1, 1); stage(1, 0, 0); stage(1, 1, 0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __syncthreads();

    s16x8 afrag[8], bfrag[2];
    for (int kt = 0; kt < NKT; ++kt) {
        const int buf = kt & 1;
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int a = kt * 4 + p;
            const int kh = p >> 1, nq = p & 1;
            const char *ai = lds + img_off(buf, 0, kh);
            const char *bi = lds + img_off(buf, 1, kh);
            if (p == 0) {   // checkpoint: glds first, the wait covers the reads
                stage_for(a);
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
#pragma unroll
                for (int mf = 0; mf < 8; ++mf)
                    afrag[mf] = *(const s16x8 *)(ai + a_off[mf]);
#pragma unroll
                for (int nf = 0; nf < 2; ++nf)
                    bfrag[nf] = *(const s16x8 *)(bi + b_off[nq][nf]);
            } else {
                if (nq == 0)
#pragma unroll
                    for (int mf = 0; mf < 8; ++mf)
                        afrag[mf] = *(const s16x8 *)(ai + a_off[mf]);
#pragma unroll
                for (int nf = 0; nf < 2; ++nf)
                    bfrag[nf] = *(const s16x8 *)(bi + b_off[nq][nf]);
                stage_for(a);
            }
            __builtin_amdgcn_s_barrier();
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int mf = 0; mf < 8; ++mf)
#pragma unroll
                for (int nf = 0; nf < 2; ++nf)
                    acc[mf][nq * 2 + nf] = MfmaOp<T>::mma(
                        afrag[mf], bfrag[nf], acc[mf][nq * 2 + nf]);
            __builtin_amdgcn_s_setprio(0);
            __builtin_amdgcn_s_barrier();
        }
    }

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    // ES: [wn][sum|ssq][256 c_local] combine plane (the 4 wn waves share
    // each channel); static LDS on top of the 128 KB dynamic tile = 136 KB
    struct ESR { float v[4][2][256]; };
    __shared__ __attribute__((aligned(16))) char esr_raw[ES ? sizeof(ESR) : 1];
    ESR *esr = reinterpret_cast<ESR *>(esr_raw);
    const int c0 = tc * 256 + wm * 128;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
        float ssum[4] = {0.f, 0.f, 0.f, 0.f};
        float sssq[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
            long pix = tp * 256 + wn * 64 + nf * 16 + (lane & 15);
            if (pix >= M) continue;
            int cc = c0 + mf * 16 + ((lane >> 4) << 2);
            s16x4_ v;
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                float f = acc[mf][nf][e];
                if (bias != nullptr) f += bias[cc + e];
                T t = ElemTraits<T>::from_f32(f);
                v[e] = *reinterpret_cast<const short *>(&t);
                if constexpr (ES) {
                    float fr = s16_as_f32<T>(v[e]);  // the ROUNDED value
                    ssum[e] += fr;
                    sssq[e] += fr * fr;
                }
            }
            *reinterpret_cast<s16x4_ *>(out + pix * Cout + cc) = v;
        }
        if constexpr (ES) {
#pragma unroll
            for (int d = 1; d < 16; d <<= 1)
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    ssum[e] += __shfl_xor(ssum[e], d, 64);
                    sssq[e] += __shfl_xor(sssq[e], d, 64);
                }
            if ((lane & 15) == 0) {
                int cl = wm * 128 + mf * 16 + ((lane >> 4) << 2);
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    esr->v[wn][0][cl + e] = ssum[e];
                    esr->v[wn][1][cl + e] = sssq[e];
                }
            }
        }
    }
    if constexpr (ES) {
        __syncthreads();
        long base = tp * 2 * Cout;   // slab row = this block's m tile
        for (int i = tid; i < 2 * 256; i += 512) {
            int row = i >> 8, c = i & 255;
            stats_out[base + (long)row * Cout + tc * 256 + c] =
                ((esr->v[0][row][c] + esr->v[1][row][c]) +
                 (esr->v[2][row][c] + esr->v[3][row][c]));
        }
    }
}

}  // namespace ig8
}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

namespace {

void *zero_page_ptr8(const torch::Tensor &like) {
    static torch::Tensor z;
    if (!z.defined() || z.device() != like.device())
        z = torch::zeros({32}, like.options().dtype(torch::kFloat32));
    return z.data_ptr();
}

template <typename T, int MODE>
void launch_conv8(const torch::Tensor &src, const torch::Tensor &w,
                  const float *bias_ptr, torch::Tensor &out,
                  int N, int H, int W, int Cin, int ld, int Cout,
                  int R, int S, int stride, int pad, int dil,
                  int OH, int OW, long M, torch::Tensor *stats = nullptr) {
    auto stream = at::hip::getCurrentHIPStream();
    constexpr size_t SHMEM = 128 * 1024;
    int tiles_m = (int)((M + 255) / 256);
    dim3 grid((unsigned)tiles_m * (Cout / 256));
    float *stats_ptr = nullptr;
    if constexpr (MODE == 0) {
        if (stats != nullptr) {
            *stats = torch::empty({tiles_m, 2L * Cout},
                                  src.options().dtype(torch::kFloat32));
            stats_ptr = stats->data_ptr<float>();
        }
    }
    auto kern = [&](auto es_tag) {
        constexpr int ESv = decltype(es_tag)::value;
        static bool attr_set = false;   // per (T, MODE, ES) instantiation
        if (!attr_set) {
            // 128K dynamic + 8K ES static = 136K <= 160K, so this opt-in is
            // valid either way; clear the error slot regardless so a failure
            // can never surface as a phantom launch error downstream
            (void)hipFuncSetAttribute(
                (const void *)sodkit::ig8::conv8_kernel<T, MODE, ESv>,
                hipFuncAttributeMaxDynamicSharedMemorySize, SHMEM);
            (void)hipGetLastError();
            attr_set = true;
        }
        hipLaunchKernelGGL((sodkit::ig8::conv8_kernel<T, MODE, ESv>), grid,
                           dim3(512), SHMEM, stream.stream(),
                           reinterpret_cast<const T *>(src.data_ptr()),
                           reinterpret_cast<const T *>(w.data_ptr()), bias_ptr,
                           reinterpret_cast<T *>(out.data_ptr()), stats_ptr,
                           reinterpret_cast<const T *>(zero_page_ptr8(src)),
                           N, H, W, Cin, ld, Cout, R, S, stride, pad, dil,
                           OH, OW);
    };
    if (stats_ptr != nullptr)
        kern(std::integral_constant<int, 1>{});
    else
        kern(std::integral_constant<int, 0>{});
}

inline int out_dim8(int in, int k, int stride, int pad, int dil) {
    return (in + 2 * pad - dil * (k - 1) - 1) / stride + 1;
}

}  // namespace

bool conv8_eligible(int64_t cout, int64_t cin) {
    return cout % 256 == 0 && cin % 8 == 0;
}

torch::Tensor conv2d_fwd_8p(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t dil, int64_t ld) {
    TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    int Cout = w.size(0), R = w.size(1), S = w.size(2);
    TORCH_CHECK(w.size(3) == C && w.is_contiguous());
    TORCH_CHECK(conv8_eligible(Cout, C), "8p: needs Cout%256==0, Cin%8==0");
    TORCH_CHECK(ld >= C, "bad channel pitch");
    int P = out_dim8(H, R, stride, pad, dil);
    int Q = out_dim8(W_, S, stride, pad, dil);
    long M = (long)N * P * Q;
    auto out = torch::empty({N, Cout, P, Q},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    const float *bp = nullptr;
    if (bias.defined() && bias.numel()) {
        bias32 = bias.to(torch::kFloat32).contiguous();
        bp = bias32.data_ptr<float>();
    }
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd_8p",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv8<__hip_bfloat16, 0>(x, w, bp, out, N, H, W_, C,
                                            (int)ld, Cout, R, S, stride, pad,
                                            dil, P, Q, M);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv8<__half, 0>(x, w, bp, out, N, H, W_, C, (int)ld,
                                    Cout, R, S, stride, pad, dil, P, Q, M);
        }));
    return out;
}

// fwd + fused BN stat partials (same contract as conv2d_fwd_stats)
std::vector<torch::Tensor> conv2d_fwd_8p_stats(torch::Tensor x,
                                               torch::Tensor w,
                                               torch::Tensor bias,
                                               int64_t stride, int64_t pad,
                                               int64_t dil, int64_t ld) {
    TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4 && w.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    int Cout = w.size(0), R = w.size(1), S = w.size(2);
    TORCH_CHECK(w.size(3) == C && w.is_contiguous());
    TORCH_CHECK(conv8_eligible(Cout, C), "8p: needs Cout%256==0, Cin%8==0");
    TORCH_CHECK(ld >= C, "bad channel pitch");
    int P = out_dim8(H, R, stride, pad, dil);
    int Q = out_dim8(W_, S, stride, pad, dil);
    long M = (long)N * P * Q;
    auto out = torch::empty({N, Cout, P, Q},
                            x.options().memory_format(at::MemoryFormat::ChannelsLast));
    torch::Tensor bias32;
    const float *bp = nullptr;
    if (bias.defined() && bias.numel()) {
        bias32 = bias.to(torch::kFloat32).contiguous();
        bp = bias32.data_ptr<float>();
    }
    torch::Tensor slabs;
    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_fwd_8p_stats",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv8<__hip_bfloat16, 0>(x, w, bp, out, N, H, W_, C,
                                            (int)ld, Cout, R, S, stride, pad,
                                            dil, P, Q, M, &slabs);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv8<__half, 0>(x, w, bp, out, N, H, W_, C, (int)ld,
                                    Cout, R, S, stride, pad, dil, P, Q, M,
                                    &slabs);
        }));
    return {out, slabs};
}

torch::Tensor conv2d_dgrad_8p(torch::Tensor dy, torch::Tensor w_t, int64_t H,
                              int64_t W_, int64_t stride, int64_t pad,
                              int64_t dil, int64_t lddy) {
    TORCH_CHECK(dy.is_cuda() && w_t.is_cuda());
    int N = dy.size(0), Cout = dy.size(1), P = dy.size(2), Q = dy.size(3);
    int Cin = w_t.size(0), R = w_t.size(1), S = w_t.size(2);
    TORCH_CHECK(w_t.size(3) == Cout && w_t.is_contiguous());
    TORCH_CHECK(conv8_eligible(Cin, Cout), "8p dgrad: Cin%256==0, Cout%8==0");
    TORCH_CHECK(lddy >= Cout, "bad channel pitch");
    long M = (long)N * H * W_;
    auto dx = torch::empty({N, Cin, (long)H, (long)W_},
                           dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    AT_DISPATCH_SWITCH(dy.scalar_type(), "conv2d_dgrad_8p",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            launch_conv8<__hip_bfloat16, 1>(dy, w_t, nullptr, dx, N, P, Q,
                                            Cout, (int)lddy, Cin, R, S,
                                            stride, pad, dil, (int)H, (int)W_,
                                            M);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            launch_conv8<__half, 1>(dy, w_t, nullptr, dx, N, P, Q, Cout,
                                    (int)lddy, Cin, R, S, stride, pad, dil,
                                    (int)H, (int)W_, M);
        }));
    return dx;
}
