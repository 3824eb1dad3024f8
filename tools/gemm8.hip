// 8-phase 256x256 bf16 GEMM microbenchmark — the round-2 conv-structure
// base (docs/IGEMM8_PLAN.md).  STANDALONE: hipcc -O3 --offload-arch=gfx950
// tools/gemm8.hip -o gemm8 && ./gemm8 [M N K iters]
//
// Reconstructed from the guide's textual spec (the example file is not in
// this image), so this is a NEW template per the guide's two-lane
// discipline: refcheck at 256/512 + race screens REQUIRED before any use
// in the framework.  It is NOT wired into sodkit dispatch.
//
// C[M][N] = A[M][K] . B[N][K]^T, bf16 inputs, fp32 accumulation.
//
// Structure (derivation in docs/IGEMM8_PLAN.md):
//   tile 256x256, BK=64, 512 threads = 8 waves as 2(M) x 4(N),
//   per-wave C = 128x64 = 8m x 4n fragments of 16x16 (acc = 128 VGPRs).
//   LDS 128 KB = 2 K-tile buffers x {A,B} x {k-half 0,1} images of
//   [256 rows][32 k] bf16 (16 KB each).  The K-HALF image split is what
//   makes the staging/consumption stagger race-free with only two
//   s_waitcnt vmcnt checkpoints per 8 phases:
//     phase p of an iteration (2 K-tiles kt=2i, 2i+1):
//       p:      0     1     2     3     4     5     6     7
//       uses:   kt.k0 kt.k0 kt.k1 kt.k1 k1.k0 k1.k0 k1.k1 k1.k1   (k1=kt+1)
//       stages: k1A1  k1B1  k2A0  k2B0  k2A1  k2B1  k3A0  k3B0
//     Every staged image is written >= 1 phase after its region's last
//     reader passed a barrier, and every consumed image is >= 4 phases
//     (>= 3 staged pairs) older than the newest in-flight load, so
//     `s_waitcnt vmcnt(6)` placed right after the glds of phases 0 and 4
//     retires exactly the halves the next 4 phases read.
//   Per phase: [nq==0: 8 A ds_read_b128] + 2 B ds_read_b128, one
//   half-image glds prefetch (2 x global_load_lds 16B/lane), raw
//   s_barrier, s_waitcnt lgkmcnt(0), setprio(1), 16 MFMAs, setprio(0),
//   s_barrier.
//
// LDS swizzle: st_16x32 per the guide, byte ^= ((byte>>9)&1)<<5 within
// each 1024 B (16 row x 64 B) subtile; SWZ=2 additionally XORs bit8->bit4
// (16 B granule, keeps b128 contiguity) which the bank model says removes
// the residual 2-way conflict — MEASURED: swz1 1008 TF beats swz2 940 TF
// at 4k^3, keep the guide's swizzle (profiles/gemm8.log).
//
// The pipelined-fragment-read interleave experiment (issue phase p+1's
// ds_reads inside phase p's MFMA region) lives in tools/conv8.hip as the
// PIPE=1 template arm — A/B it there (a 1x1 conv IS this GEMM), keeping
// this file frozen as the verified reference implementation.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __attribute__((address_space(3))) void lds_void_t;
typedef __attribute__((address_space(1))) const void gbl_void_t;

#define DEVINL __device__ __forceinline__

DEVINL void glds16(const void *gsrc, char *lds_dst) {
    __builtin_amdgcn_global_load_lds((const gbl_void_t *)gsrc,
                                     (lds_void_t *)lds_dst, 16, 0, 0);
}

template <int SWZ>
DEVINL int swz(int x) {
    x ^= ((x >> 9) & 1) << 5;
    if constexpr (SWZ == 2) x ^= ((x >> 8) & 1) << 4;
    return x;
}

// image geometry: [256 rows][32 k] bf16 = 64 B rows, 16 KB
constexpr int IMG_BYTES = 16384;
// lds offset of image (buf, op, khalf)
DEVINL int img_off(int buf, int op, int kh) {
    return ((buf * 2 + op) * 2 + kh) * IMG_BYTES;
}

template <int SWZ>
__global__ __launch_bounds__(512, 1)
void gemm8_kernel(const __hip_bfloat16 *__restrict__ A,
                  const __hip_bfloat16 *__restrict__ B,
                  __hip_bfloat16 *__restrict__ C,
                  int M, int N, int K) {
    extern __shared__ char lds[];
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int wm = wave >> 2, wn = wave & 3;

    // bijective XCD remap (guide formula)
    int nwg = gridDim.x;
    int q = nwg / 8, r = nwg % 8;
    int xcd = blockIdx.x % 8, seq = blockIdx.x / 8;
    int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + seq;
    const int ntile = N / 256;
    const int tm = wgid / ntile, tn = wgid % ntile;
    const size_t Kb = (size_t)K * 2;          // row pitch in bytes
    const __hip_bfloat16 *Abase = A + (size_t)tm * 256 * K;
    const __hip_bfloat16 *Bbase = B + (size_t)tn * 256 * K;

    const int NKT = K / 64;                   // K-tiles (K % 128 == 0)

    // --- glds staging state: two 16 B pieces per thread per half-image ---
    // dest piece j: byte d = j*8192 + tid*16 -> swizzled source (row, col)
    int srow[2], scol[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int s = swz<SWZ>((j * 8192 + tid * 16) & (IMG_BYTES - 1));
        srow[j] = s >> 6;
        scol[j] = s & 63;
    }
    // stage half-image (op, kh) of K-tile kt into buffer (kt & 1)
    auto stage = [&](int kt, int op, int kh) {
        int ktc = kt < NKT ? kt : 0;          // clamp: dummy (dead region)
        const __hip_bfloat16 *src = op == 0 ? Abase : Bbase;
        char *dst = lds + img_off(kt & 1, op, kh);
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            const char *g = (const char *)src + (size_t)srow[j] * Kb +
                            (size_t)ktc * 128 + kh * 64 + scol[j];
            glds16(g, dst + j * 8192 + tid * 16);
        }
    };
    // phase p staging target: hp=(p+2)&3 -> op=hp&1, kh=hp>>1,
    // kt = base + 1 + ((p+2)>>2)
    auto stage_for_phase = [&](int ktbase, int p) {
        int hp = (p + 2) & 3;
        stage(ktbase + 1 + ((p + 2) >> 2), hp & 1, hp >> 1);
    };

    // --- fragment read addresses (byte offsets inside an image) ---
    // A frag mf (m = wm*128 + mf*16 + (lane&15)), k piece (lane>>4)*16
    int a_off[8], b_off[2][2];                // b_off[nq][nf]
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
        int row = wm * 128 + mf * 16 + (lane & 15);
        a_off[mf] = swz<SWZ>(row * 64 + ((lane >> 4) << 4));
    }
#pragma unroll
    for (int nq = 0; nq < 2; ++nq)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
            int row = wn * 64 + nq * 32 + nf * 16 + (lane & 15);
            b_off[nq][nf] = swz<SWZ>(row * 64 + ((lane >> 4) << 4));
        }

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // --- prologue: 6 halves in steady-state order, drain K-tile 0 ---
    stage(0, 0, 0); stage(0, 1, 0); stage(0, 0, 1);
    stage(0, 1, 1); stage(1, 0, 0); stage(1, 1, 0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __syncthreads();

    s16x8 afrag[8], bfrag[2];
    // --- main loop: one iteration = 2 K-tiles = 8 phases ---
    for (int it = 0; it < NKT / 2; ++it) {
        const int ktbase = 2 * it;
#pragma unroll
        for (int p = 0; p < 8; ++p) {
            const int ktc = ktbase + (p >> 2);     // consumed K-tile
            const int kh = (p >> 1) & 1;           // its k-half
            const int nq = p & 1;                  // n-quadrant
            const int buf = ktc & 1;
            const char *ai = lds + img_off(buf, 0, kh);
            const char *bi = lds + img_off(buf, 1, kh);
            if (p == 0 || p == 4) {
                // checkpoint phases: glds FIRST, then the counted wait —
                // the wait must cover this phase's own ds_reads
                stage_for_phase(ktbase, p);
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
                if (nq == 0)
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
                stage_for_phase(ktbase, p);
            }
            __builtin_amdgcn_s_barrier();
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int mf = 0; mf < 8; ++mf)
#pragma unroll
                for (int nf = 0; nf < 2; ++nf)
                    acc[mf][nq * 2 + nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        (bf16x8)afrag[mf], (bf16x8)bfrag[nf],
                        acc[mf][nq * 2 + nf], 0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
            __builtin_amdgcn_s_barrier();
        }
    }

    // --- epilogue: drain, then write C (row-major bf16) ---
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int m0 = tm * 256 + wm * 128, n0 = tn * 256 + wn * 64;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
            int mm = m0 + mf * 16 + ((lane >> 4) << 2);
            int nn = n0 + nf * 16 + (lane & 15);
#pragma unroll
            for (int e = 0; e < 4; ++e)
                C[(size_t)(mm + e) * N + nn] =
                    __float2bfloat16(acc[mf][nf][e]);
        }
}

// ---------------------------------------------------------------------------
#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), \
            __FILE__, __LINE__); exit(1); } } while (0)

static float lcg_float(unsigned &s) {          // uniform [-1, 1)
    s = s * 1664525u + 1013904223u;
    return (float)(s >> 8) / (float)(1u << 24) * 2.f - 1.f;
}

int main(int argc, char **argv) {
    int M = argc > 1 ? atoi(argv[1]) : 4096;
    int N = argc > 2 ? atoi(argv[2]) : M;
    int K = argc > 3 ? atoi(argv[3]) : M;
    int iters = argc > 4 ? atoi(argv[4]) : 20;
    int swzmode = argc > 5 ? atoi(argv[5]) : 1;
    if (M % 256 || N % 256 || K % 128) {
        fprintf(stderr, "need M,N %% 256 == 0 and K %% 128 == 0\n");
        return 1;
    }
    std::vector<__hip_bfloat16> hA((size_t)M * K), hB((size_t)N * K);
    unsigned seed = 12345;
    for (auto &v : hA) v = __float2bfloat16(lcg_float(seed));
    for (auto &v : hB) v = __float2bfloat16(lcg_float(seed));

    __hip_bfloat16 *dA, *dB, *dC;
    HIP_CHECK(hipMalloc(&dA, hA.size() * 2));
    HIP_CHECK(hipMalloc(&dB, hB.size() * 2));
    HIP_CHECK(hipMalloc(&dC, (size_t)M * N * 2));
    HIP_CHECK(hipMemcpy(dA, hA.data(), hA.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dB, hB.data(), hB.size() * 2, hipMemcpyHostToDevice));

    dim3 grid((M / 256) * (N / 256)), block(512);
    size_t shmem = 128 * 1024;
    auto kern = swzmode == 2 ? gemm8_kernel<2> : gemm8_kernel<1>;
    HIP_CHECK(hipFuncSetAttribute(
        (const void *)kern, hipFuncAttributeMaxDynamicSharedMemorySize, shmem));

    hipLaunchKernelGGL(kern, grid, block, shmem, 0, dA, dB, dC, M, N, K);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipGetLastError());

    // refcheck: sampled fp64 dot products (exhaustive at M<=512)
    std::vector<__hip_bfloat16> hC((size_t)M * N);
    HIP_CHECK(hipMemcpy(hC.data(), dC, hC.size() * 2, hipMemcpyDeviceToHost));
    int samples = (M <= 512 && N <= 512) ? M * N : 4096;
    unsigned s2 = 777;
    double max_rel = 0;
    int bad = 0;
    for (int t = 0; t < samples; ++t) {
        int i, j;
        if (samples == M * N) { i = t / N; j = t % N; }
        else {
            s2 = s2 * 1664525u + 1013904223u; i = (s2 >> 8) % M;
            s2 = s2 * 1664525u + 1013904223u; j = (s2 >> 8) % N;
        }
        double ref = 0;
        for (int k = 0; k < K; ++k)
            ref += (double)__bfloat162float(hA[(size_t)i * K + k]) *
                   (double)__bfloat162float(hB[(size_t)j * K + k]);
        double got = __bfloat162float(hC[(size_t)i * N + j]);
        double rel = fabs(got - ref) / (fabs(ref) + 1e-3);
        if (rel > max_rel) max_rel = rel;
        if (rel > 0.05) ++bad;
    }
    printf("refcheck %dx%dx%d swz%d: max_rel %.4g, bad %d/%d -> %s\n",
           M, N, K, swzmode, max_rel, bad, samples,
           bad == 0 ? "PASS" : "FAIL");
    if (bad) return 2;

    // timing
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0)); HIP_CHECK(hipEventCreate(&t1));
    for (int w = 0; w < 3; ++w)
        hipLaunchKernelGGL(kern, grid, block, shmem, 0, dA, dB, dC, M, N, K);
    HIP_CHECK(hipEventRecord(t0));
    for (int w = 0; w < iters; ++w)
        hipLaunchKernelGGL(kern, grid, block, shmem, 0, dA, dB, dC, M, N, K);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    double tf = 2.0 * M * N * K * iters / (ms * 1e-3) / 1e12;
    printf("gemm8 %dx%dx%d swz%d: %.2f ms/iter, %.1f TF bf16 (random data)\n",
           M, N, K, swzmode, ms / iters, tf);
    return 0;
}
