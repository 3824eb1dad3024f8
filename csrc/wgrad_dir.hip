// sodkit — direct (non-im2col) weight-gradient kernel for 3x3 stride-1
// dilation-1 pad-1 convolutions on CDNA4 (SURVEY §2.3 K1 wgrad).
//
// Why it exists (measured, r2 ubench): the im2col split-K wgrad tops out
// at 150-260 TF because every x element is re-GATHERED and re-staged once
// per tap it participates in (9x for 3x3), so LDS-write/gather traffic —
// not MFMA — bounds the kernel.  For the model's dominant wgrad shapes
// (64-channel decoder/AIM convs at 88..176 square, layer1 3x3) this
// kernel instead stages each input ROW once as an LDS image x^T[c][q]
// and computes all nine taps from SHIFTED reads of the same image:
//
//   dw[cout][r][s][c] = sum_{n,p,q} dy[n,p,q,cout] * x[n,p+r-1,q+s-1,c]
//
// One workgroup (512 threads, 8 waves) owns a [64 cout][9*64 c] output
// tile and marches over a split-K range of output rows (n,p).  Per row
// it stages x rows p-1,p,p+1 and dy row p (register transpose -> b64
// LDS writes; ~90 KB of LDS writes per 1728 MFMA cycles, so staging is
// ~10% even un-overlapped).  The GEMM-K dimension is q: MFMA k-tiles of
// 32 columns; the tap shift s-1 moves the x read window by one ELEMENT,
// which breaks ds_read_b128 alignment — shifted taps read the two
// neighbouring aligned 16B blocks and recombine with v_alignbit
// (2 reads + 4 VALU vs 1 read; VALU runs beside MFMA).
//
// LDS pitches are chosen so (pitch/4) % 64 == 4: the 16 lane rows of a
// b128 fragment read land on 16 distinct bank quads (no swizzle needed,
// and no bank-XOR to break the shifted-window addressing).
//
// Split-K partials are fp32 [z][Cout][9C]; the same unrolled reduce as
// the im2col path finishes dw.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace wdir {

typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
typedef unsigned int u32x8 __attribute__((ext_vector_type(8)));

// Read a 16-byte fragment at element offset (q0 + s - 1) from an x^T row:
// s==1 is 16B-aligned; s==0/2 recombine two aligned reads (+-2 bytes).
template <int S>
SOD_DEVINL s16x8 shifted_read(const char *row, int base_byte) {
    if constexpr (S == 1) {
        return *reinterpret_cast<const s16x8 *>(row + base_byte);
    } else if constexpr (S == 2) {
        u32x4 a = *reinterpret_cast<const u32x4 *>(row + base_byte);
        u32x4 b = *reinterpret_cast<const u32x4 *>(row + base_byte + 16);
        u32x4 o;
        o[0] = (a[0] >> 16) | (a[1] << 16);
        o[1] = (a[1] >> 16) | (a[2] << 16);
        o[2] = (a[2] >> 16) | (a[3] << 16);
        o[3] = (a[3] >> 16) | (b[0] << 16);
        union { u32x4 u; s16x8 s; } r; r.u = o; return r.s;
    } else {  // S == 0: window starts 2 bytes BEFORE the aligned base
        u32x4 a = *reinterpret_cast<const u32x4 *>(row + base_byte - 16);
        u32x4 b = *reinterpret_cast<const u32x4 *>(row + base_byte);
        u32x4 o;
        o[0] = (a[3] >> 16) | (b[0] << 16);
        o[1] = (b[0] >> 16) | (b[1] << 16);
        o[2] = (b[1] >> 16) | (b[2] << 16);
        o[3] = (b[2] >> 16) | (b[3] << 16);
        union { u32x4 u; s16x8 s; } r; r.u = o; return r.s;
    }
}

// dy is consumed un-shifted; plain aligned b128.
SOD_DEVINL s16x8 read16(const char *row, int base_byte) {
    return *reinterpret_cast<const s16x8 *>(row + base_byte);
}

template <typename T>
__global__ __launch_bounds__(512, 1)
void wgrad_dir_kernel(const T *__restrict__ x, const T *__restrict__ dy,
                      float *__restrict__ partial,
                      int N, int H, int W, int C,   // x dims (P=H, Q=W)
                      int ldx, int lddy, int Cout,
                      int ktiles, int xpitch, int dypitch,
                      int rows_per_z, int total_rows, int cc_chunks,
                      int Kw) {
    extern __shared__ char lds[];
    char *xbuf[3] = {lds, lds + 64 * xpitch, lds + 2 * 64 * xpitch};
    char *dybuf = lds + 3 * 64 * xpitch;
    const int lds_bytes = 3 * 64 * xpitch + 64 * dypitch;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int wm = wave >> 2;        // cout half (2 x 32)
    const int wj = wave & 3;         // c quarter (4 x 16)

    const int cq = blockIdx.x / cc_chunks;   // cout chunk (64)
    const int cc = blockIdx.x % cc_chunks;   // cin chunk (64)
    const int z = blockIdx.y;
    const int r0 = z * rows_per_z;
    const int r1 = min(r0 + rows_per_z, total_rows);

    const int Wc4 = (W + 3) & ~3;            // staged q extent (b64 granule)

    // one-time zero fill: halo columns / k-tail / bank pads stay zero
    for (int i = t * 8; i < lds_bytes; i += 512 * 8)
        *reinterpret_cast<long *>(lds + i) = 0;

    // per-thread stage chunk: 4 consecutive pixels x 8 consecutive channels
    // (chunk id bits [c8:3][pq:5] keep write conflicts <= 2-way, see header)
    const int n_pq = Wc4 >> 2;
    const int CH_X = 8 * n_pq;               // 64 c rows / 8 per chunk

    f32x4 acc[2][9];
#pragma unroll
    for (int a = 0; a < 2; ++a)
#pragma unroll
        for (int b = 0; b < 9; ++b) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    // fragment read geometry (lane&15 = row-in-16, lane>>4 = k octet)
    const int arow = wm * 32 + (lane & 15);          // dy rows for mf=0
    const int brow = wj * 16 + (lane & 15);          // x^T c row

    __syncthreads();

    // thread -> (c8 group, pixel quad); this split keeps the b64
    // transpose-write bank conflicts <= 2-way (header)
    const int s_c8 = (t >> 5) & 7;
    const int s_pq = (t & 31) + ((t >> 8) << 5);
    const bool s_on = s_pq < n_pq;

    // staged registers for the cross-row software pipeline: row g+1's
    // global loads are issued before row g's MFMA so their latency
    // overlaps the math (same lever as the im2col wgrad PIPE variant;
    // +64 VGPRs stays within the 2-waves/SIMD budget)
    s16x8 xv[3][4], dyv[4];

    auto load_rows = [&](int g) {
        const int n = g / H;
        const int p = g % H;
#pragma unroll
        for (int d = 0; d < 3; ++d) {
            int ih = p + d - 1;
            bool rv = s_on && ih >= 0 && ih < H;
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                int q = s_pq * 4 + u;
                xv[d][u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
                if (rv && q < W)
                    xv[d][u] = *reinterpret_cast<const s16x8 *>(
                        x + ((size_t)(n * H + ih) * W + q) * ldx +
                        cc * 64 + s_c8 * 8);
            }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
            int q = s_pq * 4 + u;
            dyv[u] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
            if (s_on && q < W && cq * 64 + s_c8 * 8 < Cout)
                dyv[u] = *reinterpret_cast<const s16x8 *>(
                    dy + ((size_t)(n * H + p) * W + q) * lddy +
                    cq * 64 + s_c8 * 8);
        }
        // cout tail (Cout % 8 != 0 within the last vector): zero lanes
        if (s_on && cq * 64 + s_c8 * 8 < Cout && cq * 64 + s_c8 * 8 + 8 > Cout) {
#pragma unroll
            for (int u = 0; u < 4; ++u)
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (cq * 64 + s_c8 * 8 + e >= Cout) dyv[u][e] = 0;
        }
    };

    if (r0 < r1) load_rows(r0);
    for (int g = r0; g < r1; ++g) {
        __syncthreads();   // previous row's MFMA reads are done

        // ---- register transpose -> b64 LDS writes ([row][q] images) ----
        if (s_on) {
            const int qb = (8 + s_pq * 4) * 2;   // q=0 lives at element 8
#pragma unroll
            for (int d = 0; d < 3; ++d) {
                char *base = xbuf[d] + (s_c8 * 8) * xpitch + qb;
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    s16x4 w4 = {xv[d][0][e], xv[d][1][e], xv[d][2][e],
                                xv[d][3][e]};
                    *reinterpret_cast<s16x4 *>(base + e * xpitch) = w4;
                }
            }
            char *base = dybuf + (s_c8 * 8) * dypitch + s_pq * 4 * 2;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                s16x4 w4 = {dyv[0][e], dyv[1][e], dyv[2][e], dyv[3][e]};
                *reinterpret_cast<s16x4 *>(base + e * dypitch) = w4;
            }
        }

        __syncthreads();
        if (g + 1 < r1) load_rows(g + 1);

        // ---- MFMA sweep over the q k-tiles ----
        for (int kt = 0; kt < ktiles; ++kt) {
            const int kb = kt * 64 + ((lane >> 4) << 4);   // dy k byte
            s16x8 a0 = read16(dybuf + arow * dypitch, kb);
            s16x8 a1 = read16(dybuf + (arow + 16) * dypitch, kb);
            const int xb = kb + 16;                        // + q0 element pad
#pragma unroll
            for (int r = 0; r < 3; ++r) {
                const char *br = xbuf[r] + brow * xpitch;
                s16x8 b0 = shifted_read<0>(br, xb);
                s16x8 b1 = shifted_read<1>(br, xb);
                s16x8 b2 = shifted_read<2>(br, xb);
                acc[0][r * 3 + 0] = MfmaOp<T>::mma(a0, b0, acc[0][r * 3 + 0]);
                acc[1][r * 3 + 0] = MfmaOp<T>::mma(a1, b0, acc[1][r * 3 + 0]);
                acc[0][r * 3 + 1] = MfmaOp<T>::mma(a0, b1, acc[0][r * 3 + 1]);
                acc[1][r * 3 + 1] = MfmaOp<T>::mma(a1, b1, acc[1][r * 3 + 1]);
                acc[0][r * 3 + 2] = MfmaOp<T>::mma(a0, b2, acc[0][r * 3 + 2]);
                acc[1][r * 3 + 2] = MfmaOp<T>::mma(a1, b2, acc[1][r * 3 + 2]);
            }
        }
    }

    // ---- store fp32 partials [z][Cout][Kw], Kw = 9*C ----
    float *dst = partial + (size_t)z * Cout * Kw;
    const int c_lane = cc * 64 + wj * 16 + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
        int cout = cq * 64 + wm * 32 + mf * 16 + ((lane >> 4) << 2);
#pragma unroll
        for (int tap = 0; tap < 9; ++tap) {
            int j = tap * C + c_lane;   // dw [Cout][r][s][C] flattened
#pragma unroll
            for (int e = 0; e < 4; ++e)
                if (cout + e < Cout)
                    dst[(size_t)(cout + e) * Kw + j] = acc[mf][tap][e];
        }
    }
}

template <typename T>
__global__ void wdir_reduce_kernel(const float *__restrict__ partial,
                                   T *__restrict__ dw, long n_elems,
                                   int splitk) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < n_elems; i += stride) {
        float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
        int zi = 0;
        for (; zi + 8 <= splitk; zi += 8)
#pragma unroll
            for (int u = 0; u < 8; ++u)
                a[u] += partial[(long)(zi + u) * n_elems + i];
        for (; zi < splitk; ++zi) a[0] += partial[(long)zi * n_elems + i];
        dw[i] = ElemTraits<T>::from_f32(((a[0] + a[1]) + (a[2] + a[3]))
                                        + ((a[4] + a[5]) + (a[6] + a[7])));
    }
}

}  // namespace wdir
}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

namespace {

// smallest pitch >= need with (pitch/4) % 64 == 4  (conflict-free b128
// fragment reads across 16 consecutive rows; see header)
inline int bankfree_pitch(int need) {
    int p = ((need + 255) / 256) * 256 + 16;
    if (p - 256 >= need) p -= 256;
    return p;
}

}  // namespace

bool wgrad_dir_eligible(int64_t R, int64_t S, int64_t stride, int64_t pad,
                        int64_t dil, int64_t C, int64_t W) {
    if (R != 3 || S != 3 || stride != 1 || pad != 1 || dil != 1) return false;
    if (C % 64 != 0) return false;   // stage reads 8c vectors across a full
    //                                  64-row chunk; partial chunks would
    //                                  read past the channel range
    if (W > 184 || W < 8) return false;
    int ktiles = (int)((W + 31) / 32);
    int xpitch = bankfree_pitch((ktiles * 32 + 24) * 2);
    int dypitch = bankfree_pitch(ktiles * 32 * 2 + 32);
    return 3 * 64 * xpitch + 64 * dypitch <= 160 * 1024;
}

torch::Tensor conv2d_wgrad_dir(torch::Tensor x, torch::Tensor dy,
                               int64_t ldx, int64_t lddy) {
    TORCH_CHECK(x.is_cuda() && dy.is_cuda());
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int Cout = dy.size(1);
    TORCH_CHECK(dy.size(0) == N && dy.size(2) == H && dy.size(3) == W,
                "wgrad_dir: 3x3 s1 d1 p1 only (P=H, Q=W)");
    TORCH_CHECK(wgrad_dir_eligible(3, 3, 1, 1, 1, C, W), "shape not eligible");
    const int Kw = 9 * C;
    const int total_rows = N * H;
    const int cq_chunks = (Cout + 63) / 64;
    const int chunks = cq_chunks * (C / 64);
    int splitk = std::max(1, std::min(512, 512 / std::max(chunks, 1)));
    splitk = std::min(splitk, total_rows);
    int rows_per_z = (total_rows + splitk - 1) / splitk;
    splitk = (total_rows + rows_per_z - 1) / rows_per_z;

    const int ktiles = (W + 31) / 32;
    const int xpitch = bankfree_pitch((ktiles * 32 + 24) * 2);
    const int dypitch = bankfree_pitch(ktiles * 32 * 2 + 32);
    const size_t shmem = (size_t)3 * 64 * xpitch + 64 * dypitch;

    auto stream = at::hip::getCurrentHIPStream();
    auto partial = torch::empty({splitk, (long)Cout * Kw},
                                x.options().dtype(torch::kFloat32));
    auto dw = torch::empty({Cout, 3, 3, C}, x.options());
    long n_elems = (long)Cout * Kw;
    int rblocks = std::min<long>((n_elems + 255) / 256, 2048);
    dim3 grid(cq_chunks * (C / 64), splitk);

    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_wgrad_dir",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            using T = __hip_bfloat16;
            static bool attr = false;
            if (!attr) {
                (void)hipFuncSetAttribute(
                    (const void *)sodkit::wdir::wgrad_dir_kernel<T>,
                    hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
                attr = true;
            }
            hipLaunchKernelGGL((sodkit::wdir::wgrad_dir_kernel<T>), grid,
                               dim3(512), shmem, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               reinterpret_cast<const T *>(dy.data_ptr()),
                               partial.data_ptr<float>(), N, H, W, C,
                               (int)ldx, (int)lddy, Cout, ktiles, xpitch,
                               dypitch, rows_per_z, total_rows,
                               C / 64, Kw);
            hipLaunchKernelGGL((sodkit::wdir::wdir_reduce_kernel<T>),
                               dim3(rblocks), dim3(256), 0, stream.stream(),
                               partial.data_ptr<float>(),
                               reinterpret_cast<T *>(dw.data_ptr()),
                               n_elems, splitk);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            using T = __half;
            static bool attr = false;
            if (!attr) {
                (void)hipFuncSetAttribute(
                    (const void *)sodkit::wdir::wgrad_dir_kernel<T>,
                    hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
                attr = true;
            }
            hipLaunchKernelGGL((sodkit::wdir::wgrad_dir_kernel<T>), grid,
                               dim3(512), shmem, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               reinterpret_cast<const T *>(dy.data_ptr()),
                               partial.data_ptr<float>(), N, H, W, C,
                               (int)ldx, (int)lddy, Cout, ktiles, xpitch,
                               dypitch, rows_per_z, total_rows,
                               C / 64, Kw);
            hipLaunchKernelGGL((sodkit::wdir::wdir_reduce_kernel<T>),
                               dim3(rblocks), dim3(256), 0, stream.stream(),
                               partial.data_ptr<float>(),
                               reinterpret_cast<T *>(dw.data_ptr()),
                               n_elems, splitk);
        }));
    return dw;
}
