// sodkit — NHWC 3x3 stride-2 pad-1 max pool fwd/bwd.
//
// ATen's max_pool*_nhwc pair measured 86 us fwd / 190 us bwd on the stem
// tensor (~0.3 TB/s in backward: it re-derives windows and scatters).
// Here the forward records a packed 4-bit argmax per channel (8 channels
// -> one u32) and the backward is a deterministic per-INPUT-pixel gather
// over the <= 4 output windows that cover it — no atomics, no re-scan.
// Tap iteration order matches ATen (row-major, strict >) so tie-broken
// gradients agree with the torch reference bit-for-bit.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {
namespace pool {

template <typename T>
__global__ void maxpool_fwd_kernel(const T *__restrict__ x,
                                   T *__restrict__ y,
                                   unsigned *__restrict__ idx,
                                   int N, int C, int H, int W, int P, int Q) {
    int C8 = C >> 3;
    long total = (long)N * P * Q * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long step = (long)gridDim.x * blockDim.x;
    for (; i < total; i += step) {
        int c8 = (int)(i % C8);
        long m = i / C8;
        int q = (int)(m % Q);
        long m2 = m / Q;
        int p = (int)(m2 % P);
        int n = (int)(m2 / P);
        float best[8];
        int bi[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) { best[e] = -1e30f; bi[e] = 0; }
        for (int r = 0; r < 3; ++r) {
            int ih = 2 * p - 1 + r;
            if (ih < 0 || ih >= H) continue;
            for (int s = 0; s < 3; ++s) {
                int iw = 2 * q - 1 + s;
                if (iw < 0 || iw >= W) continue;
                s16x8 v = *reinterpret_cast<const s16x8 *>(
                    x + (((long)n * H + ih) * W + iw) * C + c8 * 8);
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    float f = s16_as_f32<T>(v[e]);
                    if (f > best[e]) { best[e] = f; bi[e] = r * 3 + s; }
                }
            }
        }
        s16x8 o;
        unsigned pack = 0;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            T t = ElemTraits<T>::from_f32(best[e]);
            o[e] = *reinterpret_cast<const short *>(&t);
            pack |= (unsigned)bi[e] << (4 * e);
        }
        *reinterpret_cast<s16x8 *>(y + i * 8) = o;
        idx[i] = pack;
    }
}

template <typename T>
__global__ void maxpool_bwd_kernel(const T *__restrict__ dy,
                                   const unsigned *__restrict__ idx,
                                   T *__restrict__ dx,
                                   int N, int C, int H, int W, int P, int Q) {
    int C8 = C >> 3;
    long total = (long)N * H * W * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long step = (long)gridDim.x * blockDim.x;
    for (; i < total; i += step) {
        int c8 = (int)(i % C8);
        long m = i / C8;
        int w = (int)(m % W);
        long m2 = m / W;
        int h = (int)(m2 % H);
        int n = (int)(m2 / H);
        float acc[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] = 0.f;
        // output rows covering input row h: ih = 2p-1+r, r in 0..2
#pragma unroll
        for (int r = 0; r < 3; ++r) {
            int ph2 = h + 1 - r;              // = 2p
            if (ph2 < 0 || (ph2 & 1)) continue;
            int p = ph2 >> 1;
            if (p >= P) continue;
#pragma unroll
            for (int s = 0; s < 3; ++s) {
                int qw2 = w + 1 - s;
                if (qw2 < 0 || (qw2 & 1)) continue;
                int q = qw2 >> 1;
                if (q >= Q) continue;
                long o = (((long)n * P + p) * Q + q) * C8 + c8;
                unsigned pack = idx[o];
                int tap = r * 3 + s;
                s16x8 g = *reinterpret_cast<const s16x8 *>(dy + o * 8);
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    if (((pack >> (4 * e)) & 15u) == (unsigned)tap)
                        acc[e] += s16_as_f32<T>(g[e]);
            }
        }
        s16x8 o8;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            T t = ElemTraits<T>::from_f32(acc[e]);
            o8[e] = *reinterpret_cast<const short *>(&t);
        }
        *reinterpret_cast<s16x8 *>(dx + i * 8) = o8;
    }
}

}  // namespace pool
}  // namespace sodkit

#define DISPATCH_T2(TENSOR, NAME, ...)                                         \
    AT_DISPATCH_SWITCH((TENSOR).scalar_type(), NAME,                           \
        AT_DISPATCH_CASE(at::kBFloat16, [&] { using T = __hip_bfloat16; __VA_ARGS__(); }) \
        AT_DISPATCH_CASE(at::kHalf,     [&] { using T = __half; __VA_ARGS__(); }))

std::vector<torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    TORCH_CHECK(C % 8 == 0 && x.element_size() == 2);
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
    int P = (H + 2 - 3) / 2 + 1, Q = (W + 2 - 3) / 2 + 1;
    auto y = torch::empty({N, C, P, Q},
                          x.options().memory_format(at::MemoryFormat::ChannelsLast));
    auto idx = torch::empty({(long)N * P * Q * (C / 8)},
                            x.options().dtype(torch::kInt32));
    long total = (long)N * P * Q * (C / 8);
    int blocks = (int)std::min<long>((total + 255) / 256, 8192);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_T2(x, "maxpool3x3s2_fwd", [&] {
        hipLaunchKernelGGL((sodkit::pool::maxpool_fwd_kernel<T>),
                           dim3(blocks), dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(x.data_ptr()),
                           reinterpret_cast<T *>(y.data_ptr()),
                           reinterpret_cast<unsigned *>(idx.data_ptr()),
                           N, C, H, W, P, Q);
    });
    return {y, idx};
}

torch::Tensor maxpool3x3s2_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t H, int64_t W) {
    TORCH_CHECK(dy.is_cuda() && dy.dim() == 4);
    int N = dy.size(0), C = dy.size(1), P = dy.size(2), Q = dy.size(3);
    dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
    auto dx = torch::empty({N, C, (long)H, (long)W},
                           dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    long total = (long)N * H * W * (C / 8);
    int blocks = (int)std::min<long>((total + 255) / 256, 8192);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_T2(dy, "maxpool3x3s2_bwd", [&] {
        hipLaunchKernelGGL((sodkit::pool::maxpool_bwd_kernel<T>),
                           dim3(blocks), dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(dy.data_ptr()),
                           reinterpret_cast<const unsigned *>(idx.data_ptr()),
                           reinterpret_cast<T *>(dx.data_ptr()),
                           N, C, (int)H, (int)W, P, Q);
    });
    return dx;
}
