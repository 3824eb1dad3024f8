// 8-phase 256x256 implicit-GEMM conv PROTOTYPE on the verified gemm8
// base (tools/gemm8.hip: 1008 TF, 25/25 race screen — docs/IGEMM8_PLAN.md).
// STANDALONE, NOT wired into sodkit dispatch; round-2 must rerun the full
// verification discipline before any integration.
//   hipcc -O3 --offload-arch=gfx950 tools/conv8.hip -o conv8 && ./conv8
//
// Mapping (same phase schedule / staging / swizzle as gemm8):
//   "A" operand = weights KRSC [Cout][K], K = R*S*Cin  (C rows = cout,
//       so each lane's 4 accum regs are 4 consecutive couts -> packed
//       8 B NHWC stores, identical to the shipped 128-square kernel)
//   "B" operand = im2col gather over NHWC x: pixel row -> (n,oh,ow)
//       precomputed per thread; k -> (tap r,s, cin) per stage call
//       (divisions per stage; shipped kernel's incremental state is the
//       round-2 optimization).  Out-of-image taps and K/M tails read a
//       zero page — contributions multiply to 0.
//   Tails: M%256 via zero-page gather + guarded stores; K%128 via the
//       absolute-phase loop (odd K-tile count runs a 4-phase half
//       iteration); Cout%256==0 required (eligible layers only).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __attribute__((address_space(3))) void lds_void_t;
typedef __attribute__((address_space(1))) const void gbl_void_t;

#define DEVINL __device__ __forceinline__

DEVINL void glds16(const void *gsrc, char *lds_dst) {
    __builtin_amdgcn_global_load_lds((const gbl_void_t *)gsrc,
                                     (lds_void_t *)lds_dst, 16, 0, 0);
}

DEVINL int swz(int x) { return x ^ (((x >> 9) & 1) << 5); }

constexpr int IMG_BYTES = 16384;   // [256 rows][32 k] bf16
DEVINL int img_off(int buf, int op, int kh) {
    return ((buf * 2 + op) * 2 + kh) * IMG_BYTES;
}

struct ConvShape {
    int N, H, W, Cin, Cout, R, S, stride, pad, dil, OH, OW;
    long M() const { return (long)N * OH * OW; }
    int K() const { return R * S * Cin; }
};

// PIPE=0: the verified phase body (reads in the barrier shadow).
// PIPE=1: the docs/IGEMM8_PLAN.md interleave lever — phase p+1's fragment
// reads issue inside phase p's setprio(1) MFMA region (legal for p<3; the
// p3->p0 hop crosses the vmcnt checkpoint and must not prefetch).  Only
// a second B fragment pair is added (+8 VGPRs); the A set is re-read via
// compiler-ordered WAR after its last MFMA use.  UNMEASURED — round-2 A/B.
// MODE=0: forward conv.  MODE=1: dgrad — the caller passes x=dy (NHWC over
// the conv's OUTPUT grid, dims H/W here), w permuted to [Cin][R][S][Cout]
// (so "Cout" here = the conv's Cin = dx channels, "Cin" here = the conv's
// Cout = dy channels), OH/OW = the dx grid; the gather inverts the conv
// geometry per tap: oh = (ih + pad - r*dil)/stride, valid iff exact.
template <int PIPE, int MODE>
__global__ __launch_bounds__(512, 1)
void conv8_kernel(const __hip_bfloat16 *__restrict__ x,     // NHWC
                  const __hip_bfloat16 *__restrict__ w,     // KRSC [Cout][K]
                  __hip_bfloat16 *__restrict__ out,         // [M][Cout]
                  const __hip_bfloat16 *__restrict__ zero_page,
                  int N_, int H, int W, int Cin, int Cout,
                  int R, int S, int stride, int pad, int dil,
                  int OH, int OW) {
    extern __shared__ char lds[];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int wm = wave >> 2, wn = wave & 3;   // wm: cout half, wn: pixel quarter
    const int K = R * S * Cin;
    const long M = (long)N_ * OH * OW;

    // bijective XCD remap
    int nwg = gridDim.x;
    int q = nwg / 8, r8 = nwg % 8;
    int xcd = blockIdx.x % 8, seq = blockIdx.x / 8;
    int wgid = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q) + seq;
    const int ntile = (int)((M + 255) / 256);
    const int tc = wgid / ntile;               // cout tile
    const long tp = (long)(wgid % ntile);      // pixel tile
    const size_t Kb = (size_t)K * 2;
    const __hip_bfloat16 *Abase = w + (size_t)tc * 256 * K;

    const int NKT = (K + 63) / 64;

    // --- staging state ---
    int srow[2], scol[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int s = swz((j * 8192 + tid * 16) & (IMG_BYTES - 1));
        srow[j] = s >> 6;
        scol[j] = s & 63;
    }
    // B gather: pixel row -> (n, oh, ow), fixed per thread
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
            if (op == 0) {                       // weights: linear k-major
                g = (const char *)(k < K ? (const void *)((const char *)Abase +
                        (size_t)srow[j] * Kb + (size_t)k * 2)
                        : (const void *)zero_page);
            } else {                             // im2col gather
                int tap = k / Cin, c = k - tap * Cin;
                int rr = tap / S, ss = tap - rr * S;
                int ih, iw;
                bool ok = pvalid[j] && k < K;
                if constexpr (MODE == 0) {
                    ih = poh[j] * stride - pad + rr * dil;
                    iw = pow_[j] * stride - pad + ss * dil;
                    ok = ok && ih >= 0 && ih < H && iw >= 0 && iw < W;
                } else {                         // dgrad: invert the geometry
                    int nh = poh[j] + pad - rr * dil;
                    int nw = pow_[j] + pad - ss * dil;
                    ih = nh / stride; iw = nw / stride;
                    ok = ok && nh >= 0 && nw >= 0 &&
                         nh == ih * stride && nw == iw * stride &&
                         ih < H && iw < W;
                }
                g = ok ? (const char *)(x + (((size_t)pn[j] * H + ih) * W + iw)
                                                * Cin + c)
                       : (const char *)zero_page;
            }
            glds16(g, dst + j * 8192 + tid * 16);
        }
    };
    // absolute phase a: stage half hp=(a+2)&3 of K-tile (a+6)>>2
    auto stage_for = [&](int a) {
        int hp = (a + 2) & 3;
        stage((a + 6) >> 2, hp & 1, hp >> 1);
    };

    // --- fragment offsets ---
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

    // --- prologue (6 halves, steady-state order) ---
    stage(0, 0, 0); stage(0, 1, 0); stage(0, 0, 1);
    stage(0, 1, 1); stage(1, 0, 0); stage(1, 1, 0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __syncthreads();

    s16x8 afrag[8], bfrag[2], bfrag2[2];
    // --- main loop over absolute phases (handles odd NKT) ---
    for (int kt = 0; kt < NKT; ++kt) {
        const int buf = kt & 1;
        const char *ai1 = lds + img_off(buf, 0, 1);   // kh=1 images (PIPE
        const char *bi1 = lds + img_off(buf, 1, 1);   // prefetch targets)
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int a = kt * 4 + p;
            const int kh = p >> 1, nq = p & 1;
            const char *ai = lds + img_off(buf, 0, kh);
            const char *bi = lds + img_off(buf, 1, kh);
            if (p == 0) {                 // checkpoint phase
                stage_for(a);
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
#pragma unroll
                for (int mf = 0; mf < 8; ++mf)
                    afrag[mf] = *(const s16x8 *)(ai + a_off[mf]);
#pragma unroll
                for (int nf = 0; nf < 2; ++nf)
                    bfrag[nf] = *(const s16x8 *)(bi + b_off[nq][nf]);
            } else {
                if (PIPE == 0) {
                    if (nq == 0)
#pragma unroll
                        for (int mf = 0; mf < 8; ++mf)
                            afrag[mf] = *(const s16x8 *)(ai + a_off[mf]);
#pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        bfrag[nf] = *(const s16x8 *)(bi + b_off[nq][nf]);
                }
                stage_for(a);
            }
            __builtin_amdgcn_s_barrier();
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_setprio(1);
            {
                // PIPE=1: odd phases consume the second B set
                const s16x8 *bsel = (PIPE == 1 && nq == 1) ? bfrag2 : bfrag;
#pragma unroll
                for (int mf = 0; mf < 8; ++mf)
#pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        acc[mf][nq * 2 + nf] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                (bf16x8)afrag[mf], (bf16x8)bsel[nf],
                                acc[mf][nq * 2 + nf], 0, 0, 0);
            }
            if (PIPE == 1) {
                // prefetch phase p+1's fragments inside the MFMA region
                // (compiler orders the afrag re-reads after their last use)
                if (p == 0) {
#pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        bfrag2[nf] = *(const s16x8 *)(bi + b_off[1][nf]);
                } else if (p == 1) {
#pragma unroll
                    for (int mf = 0; mf < 8; ++mf)
                        afrag[mf] = *(const s16x8 *)(ai1 + a_off[mf]);
#pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        bfrag[nf] = *(const s16x8 *)(bi1 + b_off[0][nf]);
                } else if (p == 2) {
#pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        bfrag2[nf] = *(const s16x8 *)(bi1 + b_off[1][nf]);
                }                             // p==3: crosses the checkpoint
            }
            __builtin_amdgcn_s_setprio(0);
            __builtin_amdgcn_s_barrier();
        }
    }

    // --- epilogue: packed 8 B stores of 4 consecutive couts ---
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int c0 = tc * 256 + wm * 128;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
            long pix = tp * 256 + wn * 64 + nf * 16 + (lane & 15);
            if (pix >= M) continue;
            int cc = c0 + mf * 16 + ((lane >> 4) << 2);
            s16x4 v;
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                __hip_bfloat16 t = __float2bfloat16(acc[mf][nf][e]);
                v[e] = *reinterpret_cast<const short *>(&t);
            }
            *reinterpret_cast<s16x4 *>(
                (__hip_bfloat16 *)out + pix * Cout + cc) = v;
        }
}

// ---------------------------------------------------------------------------
#define HIP_CHECK(c) do { hipError_t e = (c); if (e != hipSuccess) { \
    fprintf(stderr, "HIP %s @%d\n", hipGetErrorString(e), __LINE__); \
    exit(1); } } while (0)

static float lcg_float(unsigned &s) {
    s = s * 1664525u + 1013904223u;
    return (float)(s >> 8) / (float)(1u << 24) * 2.f - 1.f;
}

static void run_case(ConvShape cs, int iters, bool check, int pipe = 0) {
    const long M = cs.M();
    const int K = cs.K();
    std::vector<__hip_bfloat16> hx((size_t)cs.N * cs.H * cs.W * cs.Cin);
    std::vector<__hip_bfloat16> hw((size_t)cs.Cout * K);
    unsigned seed = 999;
    for (auto &v : hx) v = __float2bfloat16(lcg_float(seed));
    for (auto &v : hw) v = __float2bfloat16(lcg_float(seed));

    __hip_bfloat16 *dx, *dw, *dy, *dz;
    HIP_CHECK(hipMalloc(&dx, hx.size() * 2));
    HIP_CHECK(hipMalloc(&dw, hw.size() * 2));
    HIP_CHECK(hipMalloc(&dy, (size_t)M * cs.Cout * 2));
    HIP_CHECK(hipMalloc(&dz, 16384));
    HIP_CHECK(hipMemset(dz, 0, 16384));
    HIP_CHECK(hipMemcpy(dx, hx.data(), hx.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dw, hw.data(), hw.size() * 2, hipMemcpyHostToDevice));

    dim3 grid((unsigned)((M + 255) / 256) * (cs.Cout / 256)), block(512);
    size_t shmem = 128 * 1024;
    auto kern = pipe ? conv8_kernel<1, 0> : conv8_kernel<0, 0>;
    HIP_CHECK(hipFuncSetAttribute((const void *)kern,
              hipFuncAttributeMaxDynamicSharedMemorySize, shmem));
    auto launch = [&] {
        hipLaunchKernelGGL(kern, grid, block, shmem, 0, dx, dw, dy, dz,
                           cs.N, cs.H, cs.W, cs.Cin, cs.Cout, cs.R, cs.S,
                           cs.stride, cs.pad, cs.dil, cs.OH, cs.OW);
    };
    launch();
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipGetLastError());

    if (check) {
        std::vector<__hip_bfloat16> hy((size_t)M * cs.Cout);
        HIP_CHECK(hipMemcpy(hy.data(), dy, hy.size() * 2, hipMemcpyDeviceToHost));
        // sampled fp64 reference
        unsigned s2 = 31337;
        double max_rel = 0; int bad = 0; const int samples = 2048;
        for (int t = 0; t < samples; ++t) {
            s2 = s2 * 1664525u + 1013904223u; long pix = (s2 >> 8) % M;
            s2 = s2 * 1664525u + 1013904223u; int co = (s2 >> 8) % cs.Cout;
            int n = (int)(pix / ((long)cs.OH * cs.OW));
            int rem = (int)(pix % ((long)cs.OH * cs.OW));
            int oh = rem / cs.OW, ow = rem % cs.OW;
            double ref = 0;
            for (int rr = 0; rr < cs.R; ++rr)
                for (int ss = 0; ss < cs.S; ++ss) {
                    int ih = oh * cs.stride - cs.pad + rr * cs.dil;
                    int iw = ow * cs.stride - cs.pad + ss * cs.dil;
                    if (ih < 0 || ih >= cs.H || iw < 0 || iw >= cs.W) continue;
                    for (int c = 0; c < cs.Cin; ++c)
                        ref += (double)__bfloat162float(
                                   hx[(((size_t)n * cs.H + ih) * cs.W + iw) *
                                      cs.Cin + c]) *
                               (double)__bfloat162float(
                                   hw[(size_t)co * K + (rr * cs.S + ss) *
                                      cs.Cin + c]);
                }
            double got = __bfloat162float(hy[(size_t)pix * cs.Cout + co]);
            double rel = fabs(got - ref) / (fabs(ref) + 1e-3);
            if (rel > max_rel) max_rel = rel;
            if (rel > 0.05) ++bad;
        }
        printf("conv8 refcheck N%d %dx%d Ci%d Co%d %dx%d s%d d%d: "
               "max_rel %.4g bad %d/%d -> %s\n",
               cs.N, cs.H, cs.W, cs.Cin, cs.Cout, cs.R, cs.S, cs.stride,
               cs.dil, max_rel, bad, samples, bad == 0 ? "PASS" : "FAIL");
    }
    if (iters > 0) {
        hipEvent_t t0, t1;
        HIP_CHECK(hipEventCreate(&t0)); HIP_CHECK(hipEventCreate(&t1));
        for (int i = 0; i < 3; ++i) launch();
        HIP_CHECK(hipEventRecord(t0));
        for (int i = 0; i < iters; ++i) launch();
        HIP_CHECK(hipEventRecord(t1));
        HIP_CHECK(hipEventSynchronize(t1));
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
        double tf = 2.0 * M * cs.Cout * K * iters / (ms * 1e-3) / 1e12;
        printf("conv8 time  N%d %dx%d Ci%d Co%d %dx%d: %.3f ms, %.1f TF\n",
               cs.N, cs.H, cs.W, cs.Cin, cs.Cout, cs.R, cs.S, ms / iters, tf);
    }
    hipFree(dx); hipFree(dw); hipFree(dy); hipFree(dz);
}

// dgrad refcheck: dx = conv8_kernel<.,1> over dy with permuted weights,
// checked against a sampled fp64 transposed-conv reference.  Requires the
// conv's Cin % 256 == 0 (dx channels are the kernel's cout tile).
static void run_dgrad_case(ConvShape cs, int iters, bool check) {
    const long Mo = cs.M();                       // dy pixels
    const long Mi = (long)cs.N * cs.H * cs.W;     // dx pixels
    const int K = cs.R * cs.S * cs.Cout;          // dgrad GEMM K
    std::vector<__hip_bfloat16> hdy((size_t)Mo * cs.Cout);
    std::vector<__hip_bfloat16> hw((size_t)cs.Cout * cs.R * cs.S * cs.Cin);
    unsigned seed = 4242;
    for (auto &v : hdy) v = __float2bfloat16(lcg_float(seed));
    for (auto &v : hw) v = __float2bfloat16(lcg_float(seed));
    // permute KRSC [Co][R][S][Ci] -> [Ci][R][S][Co]
    std::vector<__hip_bfloat16> hwt((size_t)cs.Cin * cs.R * cs.S * cs.Cout);
    for (int co = 0; co < cs.Cout; ++co)
        for (int t = 0; t < cs.R * cs.S; ++t)
            for (int ci = 0; ci < cs.Cin; ++ci)
                hwt[((size_t)ci * cs.R * cs.S + t) * cs.Cout + co] =
                    hw[((size_t)co * cs.R * cs.S + t) * cs.Cin + ci];

    __hip_bfloat16 *ddy, *dwt, *ddx, *dz;
    HIP_CHECK(hipMalloc(&ddy, hdy.size() * 2));
    HIP_CHECK(hipMalloc(&dwt, hwt.size() * 2));
    HIP_CHECK(hipMalloc(&ddx, (size_t)Mi * cs.Cin * 2));
    HIP_CHECK(hipMalloc(&dz, 16384));
    HIP_CHECK(hipMemset(dz, 0, 16384));
    HIP_CHECK(hipMemcpy(ddy, hdy.data(), hdy.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dwt, hwt.data(), hwt.size() * 2, hipMemcpyHostToDevice));

    dim3 grid((unsigned)((Mi + 255) / 256) * (cs.Cin / 256)), block(512);
    size_t shmem = 128 * 1024;
    auto kern = conv8_kernel<0, 1>;
    HIP_CHECK(hipFuncSetAttribute((const void *)kern,
              hipFuncAttributeMaxDynamicSharedMemorySize, shmem));
    auto launch = [&] {
        // swapped geometry: x=dy (its grid is OHxOW), out grid = HxW
        hipLaunchKernelGGL(kern, grid, block, shmem, 0, ddy, dwt, ddx, dz,
                           cs.N, cs.OH, cs.OW, cs.Cout, cs.Cin, cs.R, cs.S,
                           cs.stride, cs.pad, cs.dil, cs.H, cs.W);
    };
    launch();
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipGetLastError());

    if (check) {
        std::vector<__hip_bfloat16> hdx((size_t)Mi * cs.Cin);
        HIP_CHECK(hipMemcpy(hdx.data(), ddx, hdx.size() * 2,
                            hipMemcpyDeviceToHost));
        unsigned s2 = 515151;
        double max_rel = 0; int bad = 0; const int samples = 2048;
        for (int t = 0; t < samples; ++t) {
            s2 = s2 * 1664525u + 1013904223u; long pix = (s2 >> 8) % Mi;
            s2 = s2 * 1664525u + 1013904223u; int ci = (s2 >> 8) % cs.Cin;
            int n = (int)(pix / ((long)cs.H * cs.W));
            int rem = (int)(pix % ((long)cs.H * cs.W));
            int ih = rem / cs.W, iw = rem % cs.W;
            double ref = 0;
            for (int rr = 0; rr < cs.R; ++rr)
                for (int ss = 0; ss < cs.S; ++ss) {
                    int nh = ih + cs.pad - rr * cs.dil;
                    int nw = iw + cs.pad - ss * cs.dil;
                    if (nh < 0 || nw < 0 || nh % cs.stride || nw % cs.stride)
                        continue;
                    int oh = nh / cs.stride, ow = nw / cs.stride;
                    if (oh >= cs.OH || ow >= cs.OW) continue;
                    for (int co = 0; co < cs.Cout; ++co)
                        ref += (double)__bfloat162float(
                                   hdy[(((size_t)n * cs.OH + oh) * cs.OW + ow) *
                                       cs.Cout + co]) *
                               (double)__bfloat162float(
                                   hw[((size_t)co * cs.R * cs.S +
                                       rr * cs.S + ss) * cs.Cin + ci]);
                }
            double got = __bfloat162float(hdx[(size_t)pix * cs.Cin + ci]);
            double rel = fabs(got - ref) / (fabs(ref) + 1e-3);
            if (rel > max_rel) max_rel = rel;
            if (rel > 0.05) ++bad;
        }
        printf("conv8 DGRAD refcheck N%d %dx%d Ci%d Co%d %dx%d s%d d%d: "
               "max_rel %.4g bad %d/%d -> %s\n",
               cs.N, cs.H, cs.W, cs.Cin, cs.Cout, cs.R, cs.S, cs.stride,
               cs.dil, max_rel, bad, samples, bad == 0 ? "PASS" : "FAIL");
    }
    if (iters > 0) {
        hipEvent_t t0, t1;
        HIP_CHECK(hipEventCreate(&t0)); HIP_CHECK(hipEventCreate(&t1));
        for (int i = 0; i < 3; ++i) launch();
        HIP_CHECK(hipEventRecord(t0));
        for (int i = 0; i < iters; ++i) launch();
        HIP_CHECK(hipEventRecord(t1));
        HIP_CHECK(hipEventSynchronize(t1));
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
        double tf = 2.0 * Mi * cs.Cin * K * iters / (ms * 1e-3) / 1e12;
        printf("conv8 DGRAD time N%d %dx%d Ci%d Co%d %dx%d: %.3f ms, %.1f TF\n",
               cs.N, cs.H, cs.W, cs.Cin, cs.Cout, cs.R, cs.S, ms / iters, tf);
    }
    hipFree(ddy); hipFree(dwt); hipFree(ddx); hipFree(dz);
}

static ConvShape mk(int N, int HW, int Ci, int Co, int k, int st, int dl) {
    ConvShape c{N, HW, HW, Ci, Co, k, k, st, (k / 2) * dl, dl, 0, 0};
    c.OH = (c.H + 2 * c.pad - dl * (k - 1) - 1) / st + 1;
    c.OW = c.OH;
    return c;
}

int main(int argc, char **argv) {
    int iters = argc > 1 ? atoi(argv[1]) : 20;
    int pipe = argc > 2 ? atoi(argv[2]) : 0;   // 1 = interleave A/B variant
    // refcheck: small shapes covering 1x1 / 3x3 / odd-NKT K / M-tail /
    // stride / dilation
    run_case(mk(2, 22, 64, 256, 3, 1, 1), 0, true, pipe);  // K=576 odd NKT
    run_case(mk(2, 16, 256, 256, 1, 1, 1), 0, true, pipe); // pure GEMM 1x1
    run_case(mk(2, 23, 72, 256, 3, 2, 1), 0, true, pipe);  // stride 2
    run_case(mk(2, 20, 64, 512, 3, 1, 2), 0, true, pipe);  // dilation 2
    // dgrad (MODE=1): cs.Cin is the kernel's cout tile -> Cin % 256
    run_dgrad_case(mk(2, 22, 256, 64, 3, 1, 1), 0, true);   // K=576 odd NKT
    run_dgrad_case(mk(2, 18, 256, 128, 3, 2, 1), 0, true);  // stride 2
    run_dgrad_case(mk(2, 20, 512, 64, 3, 1, 2), 0, true);   // dilation 2
    // timing: MINet-R50 hot shapes (batch 32)
    run_case(mk(32, 44, 256, 256, 3, 1, 1), iters, false, pipe);
    run_case(mk(32, 44, 1024, 256, 1, 1, 1), iters, false, pipe);
    run_case(mk(32, 22, 512, 512, 3, 1, 1), iters, false, pipe);
    run_dgrad_case(mk(32, 44, 256, 256, 3, 1, 1), iters, false);
    return 0;
}
