// sodkit — VALU weight-gradient kernel for 1x1 convolutions (SURVEY §2.3
// K2 wgrad).  dw[cout][c] = sum_m dy[m][cout] * x[m][c] — a tall-skinny
// reduction (K = M pixels is 10^4..10^6, the output is tiny).
//
// Why not MFMA: both operands need k(=pixel)-contiguous fragments, which
// forces the register->LDS transpose staging that bounds the im2col
// split-K wgrad at 108-187 TF on these shapes (measured r2 ubench).  A
// register-accumulating VALU kernel has NO transpose: each thread owns a
// [4 cout][16 c] patch of the output, the block stages [MB][cout-tile] +
// [MB][c-tile] rows in LDS (coalesced, layout-preserving), and the inner
// loop is pure broadcast-read + FMA.  The arithmetic runs at VALU rate
// (~320 TF f32-FMA peak) but the kernel is staging/issue-bound far below
// the transpose-staged MFMA alternative's measured rates.
//
// Stride-2 1x1 (ResNet downsample) is the same kernel with the x pixel
// index mapped through the stride.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace wk1 {

constexpr int MB = 32;          // pixels per LDS stage
constexpr int TC = 64;          // cout tile
constexpr int TJ = 256;         // c tile

template <typename T>
__global__ __launch_bounds__(256, 2)
void wgrad_k1_kernel(const T *__restrict__ x, const T *__restrict__ dy,
                     float *__restrict__ partial,
                     int N, int H, int W,        // x spatial dims
                     int ldx, int lddy, int C, int Cout,
                     int P, int Q, int stride,
                     long M_total, long m_per_split, int cc_chunks) {
    __shared__ __align__(16) short dy_lds[MB * TC];
    __shared__ __align__(16) short x_lds[MB * TJ];

    const int t = threadIdx.x;
    const int cq = blockIdx.x / cc_chunks;    // cout chunk
    const int cc = blockIdx.x % cc_chunks;    // c chunk
    const int z = blockIdx.y;
    const long m0 = z * m_per_split;
    const long m1 = min(m0 + m_per_split, M_total);

    const int tc16 = t & 15;                  // c group  (16 x 16)
    const int tm4 = t >> 4;                   // cout group (16 x 4)
    const int my_c = cc * TJ + tc16 * 16;
    const int my_co = cq * TC + tm4 * 4;

    float acc[4][16];
#pragma unroll
    for (int a = 0; a < 4; ++a)
#pragma unroll
        for (int b = 0; b < 16; ++b) acc[a][b] = 0.f;

    // stage chunks: dy 2048 elems -> 1 vector/thread; x 8192 -> 4
    const int sdm = t >> 3, sdg = t & 7;          // dy: [m][8-group]
    const int sxm = t >> 5, sxg = t & 31;         // x:  [m][32-group], 4 m-passes

    for (long mc = m0; mc < m1; mc += MB) {
        __syncthreads();
        {
            long m = mc + sdm;
            s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (m < m1 && cq * TC + sdg * 8 < Cout) {
                v = *reinterpret_cast<const s16x8 *>(
                    dy + m * lddy + cq * TC + sdg * 8);
                if (cq * TC + sdg * 8 + 8 > Cout) {   // cout tail
#pragma unroll
                    for (int e = 0; e < 8; ++e)
                        if (cq * TC + sdg * 8 + e >= Cout) v[e] = 0;
                }
            }
            *reinterpret_cast<s16x8 *>(dy_lds + sdm * TC + sdg * 8) = v;
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                long mm = mc + sxm + u * 8;
                s16x8 w = {0, 0, 0, 0, 0, 0, 0, 0};
                if (mm < m1 && cc * TJ + sxg * 8 < C) {
                    long px = mm;
                    if (stride != 1) {
                        // output pixel -> strided input pixel
                        long n = mm / ((long)P * Q);
                        int pq = (int)(mm % ((long)P * Q));
                        px = (n * H + (pq / Q) * stride) * W
                             + (pq % Q) * stride;
                    }
                    w = *reinterpret_cast<const s16x8 *>(
                        x + px * ldx + cc * TJ + sxg * 8);
                    if (cc * TJ + sxg * 8 + 8 > C) {  // c tail
#pragma unroll
                        for (int e = 0; e < 8; ++e)
                            if (cc * TJ + sxg * 8 + e >= C) w[e] = 0;
                    }
                }
                *reinterpret_cast<s16x8 *>(
                    x_lds + (sxm + u * 8) * TJ + sxg * 8) = w;
            }
        }
        __syncthreads();

        const int mcnt = (int)min((long)MB, m1 - mc);
        for (int m = 0; m < mcnt; ++m) {
            s16x4 vdy = *reinterpret_cast<const s16x4 *>(
                dy_lds + m * TC + tm4 * 4);
            s16x8 vx0 = *reinterpret_cast<const s16x8 *>(
                x_lds + m * TJ + tc16 * 16);
            s16x8 vx1 = *reinterpret_cast<const s16x8 *>(
                x_lds + m * TJ + tc16 * 16 + 8);
            float g[4], xv[16];
#pragma unroll
            for (int a = 0; a < 4; ++a) g[a] = s16_as_f32<T>(vdy[a]);
#pragma unroll
            for (int b = 0; b < 8; ++b) {
                xv[b] = s16_as_f32<T>(vx0[b]);
                xv[8 + b] = s16_as_f32<T>(vx1[b]);
            }
#pragma unroll
            for (int a = 0; a < 4; ++a)
#pragma unroll
                for (int b = 0; b < 16; ++b) acc[a][b] += g[a] * xv[b];
        }
    }

    // partial [z][Cout][C] fp32
    float *dst = partial + (size_t)z * Cout * C;
#pragma unroll
    for (int a = 0; a < 4; ++a) {
        if (my_co + a >= Cout) break;
        float *row = dst + (size_t)(my_co + a) * C + my_c;
#pragma unroll
        for (int b = 0; b < 16; ++b)
            if (my_c + b < C) row[b] = acc[a][b];
    }
}

template <typename T>
__global__ void wk1_reduce_kernel(const float *__restrict__ partial,
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

}  // namespace wk1
}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

bool wgrad_k1_eligible(int64_t R, int64_t S, int64_t dil, int64_t pad,
                       int64_t C, int64_t Cout) {
    return R == 1 && S == 1 && dil == 1 && pad == 0
           && C % 16 == 0 && Cout % 4 == 0 && Cout >= 4;
}

torch::Tensor conv2d_wgrad_k1(torch::Tensor x, torch::Tensor dy,
                              int64_t stride, int64_t ldx, int64_t lddy) {
    TORCH_CHECK(x.is_cuda() && dy.is_cuda());
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    int Cout = dy.size(1), P = dy.size(2), Q = dy.size(3);
    TORCH_CHECK(dy.size(0) == N);
    TORCH_CHECK(wgrad_k1_eligible(1, 1, 1, 0, C, Cout), "not eligible");
    long M = (long)N * P * Q;
    const int cq_chunks = (Cout + sodkit::wk1::TC - 1) / sodkit::wk1::TC;
    const int cc_chunks = (C + sodkit::wk1::TJ - 1) / sodkit::wk1::TJ;
    const int chunks = cq_chunks * cc_chunks;
    // 256-thread blocks, 2/CU residency -> ~1024 blocks fill the chip
    int splitk = std::max(1, std::min(1024, 1024 / std::max(chunks, 1)));
    splitk = (int)std::min<long>(splitk, (M + sodkit::wk1::MB - 1)
                                             / sodkit::wk1::MB);
    long m_per_split = ((M + splitk - 1) / splitk + sodkit::wk1::MB - 1)
                       / sodkit::wk1::MB * sodkit::wk1::MB;
    splitk = (int)((M + m_per_split - 1) / m_per_split);

    auto stream = at::hip::getCurrentHIPStream();
    auto partial = torch::empty({splitk, (long)Cout * C},
                                x.options().dtype(torch::kFloat32));
    auto dw = torch::empty({Cout, 1, 1, C}, x.options());
    long n_elems = (long)Cout * C;
    int rblocks = (int)std::min<long>((n_elems + 255) / 256, 2048);
    dim3 grid(chunks, splitk);

    AT_DISPATCH_SWITCH(x.scalar_type(), "conv2d_wgrad_k1",
        AT_DISPATCH_CASE(at::kBFloat16, [&] {
            using T = __hip_bfloat16;
            hipLaunchKernelGGL((sodkit::wk1::wgrad_k1_kernel<T>), grid,
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               reinterpret_cast<const T *>(dy.data_ptr()),
                               partial.data_ptr<float>(), N, H, W,
                               (int)ldx, (int)lddy, C, Cout, P, Q,
                               (int)stride, M, m_per_split, cc_chunks);
            hipLaunchKernelGGL((sodkit::wk1::wk1_reduce_kernel<T>),
                               dim3(rblocks), dim3(256), 0, stream.stream(),
                               partial.data_ptr<float>(),
                               reinterpret_cast<T *>(dw.data_ptr()),
                               n_elems, splitk);
        })
        AT_DISPATCH_CASE(at::kHalf, [&] {
            using T = __half;
            hipLaunchKernelGGL((sodkit::wk1::wgrad_k1_kernel<T>), grid,
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               reinterpret_cast<const T *>(dy.data_ptr()),
                               partial.data_ptr<float>(), N, H, W,
                               (int)ldx, (int)lddy, C, Cout, P, Q,
                               (int)stride, M, m_per_split, cc_chunks);
            hipLaunchKernelGGL((sodkit::wk1::wk1_reduce_kernel<T>),
                               dim3(rblocks), dim3(256), 0, stream.stream(),
                               partial.data_ptr<float>(),
                               reinterpret_cast<T *>(dw.data_ptr()),
                               n_elems, splitk);
        }));
    return dw;
}
