// sodkit — dynamic (spatially-variant) 3x3 filtering for the HDFNet DDPM
// (SURVEY §2.3 K10): y[n,h,w,c] = sum_e k[n,h,w,e] * x[n,h+dh_e,w+dw_e,c]
// at a given dilation, zero padding.  NHWC; per-pixel kernels k have E=9
// taps.  Replaces the unfold+mul composition (9x materialized columns).
//
// fwd: channel-vectorized gather (8 ch / thread, 9 taps).
// dx : mirrored gather (k read at the shifted source pixel).
// dk : per-pixel C-reduction; block = rows_pb pixels x C/8 lanes, subgroup
//      shuffle reduce (C/8 must be a power of two <= 64 for the fast path;
//      the python wrapper falls back to the composed form otherwise).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {

constexpr int E = 9;

template <typename T>
__global__ void dynfilter_fwd_kernel(const T *__restrict__ x,
                                     const T *__restrict__ k,
                                     T *__restrict__ y, int N, int C8, int H,
                                     int W, int dil) {
    int C = C8 * 8;
    long total8 = (long)N * H * W * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total8; i += stride) {
        int c8 = (int)(i % C8);
        long pix = i / C8;
        int w = (int)(pix % W);
        long t2 = pix / W;
        int h = (int)(t2 % H);
        int n = (int)(t2 / H);
        float kv[E];
#pragma unroll
        for (int e = 0; e < E; ++e)
            kv[e] = ElemTraits<T>::to_f32(k[pix * E + e]);
        float acc[8];
#pragma unroll
        for (int c = 0; c < 8; ++c) acc[c] = 0.f;
        const T *xb = x + ((long)n * H) * W * C + c8 * 8;
#pragma unroll
        for (int e = 0; e < E; ++e) {
            int hh = h + (e / 3 - 1) * dil;
            int ww = w + (e % 3 - 1) * dil;
            if ((unsigned)hh >= (unsigned)H || (unsigned)ww >= (unsigned)W)
                continue;
            s16x8 v = *reinterpret_cast<const s16x8 *>(xb + ((long)hh * W + ww) * C);
#pragma unroll
            for (int c = 0; c < 8; ++c) acc[c] += kv[e] * s16_as_f32<T>(v[c]);
        }
        s16x8 vo;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
            T tv = ElemTraits<T>::from_f32(acc[c]);
            vo[c] = *reinterpret_cast<const short *>(&tv);
        }
        *reinterpret_cast<s16x8 *>(y + i * 8) = vo;
    }
}

template <typename T>
__global__ void dynfilter_dx_kernel(const T *__restrict__ dy,
                                    const T *__restrict__ k,
                                    T *__restrict__ dx, int N, int C8, int H,
                                    int W, int dil) {
    int C = C8 * 8;
    long total8 = (long)N * H * W * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total8; i += stride) {
        int c8 = (int)(i % C8);
        long pix = i / C8;
        int w = (int)(pix % W);
        long t2 = pix / W;
        int h = (int)(t2 % H);
        int n = (int)(t2 / H);
        float acc[8];
#pragma unroll
        for (int c = 0; c < 8; ++c) acc[c] = 0.f;
        const T *dyb = dy + ((long)n * H) * W * C + c8 * 8;
        const T *kb = k + ((long)n * H) * W * E;
#pragma unroll
        for (int e = 0; e < E; ++e) {
            int hh = h - (e / 3 - 1) * dil;   // mirrored: source pixel
            int ww = w - (e % 3 - 1) * dil;
            if ((unsigned)hh >= (unsigned)H || (unsigned)ww >= (unsigned)W)
                continue;
            long spix = (long)hh * W + ww;
            float kv = ElemTraits<T>::to_f32(kb[spix * E + e]);
            s16x8 v = *reinterpret_cast<const s16x8 *>(dyb + spix * C);
#pragma unroll
            for (int c = 0; c < 8; ++c) acc[c] += kv * s16_as_f32<T>(v[c]);
        }
        s16x8 vo;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
            T tv = ElemTraits<T>::from_f32(acc[c]);
            vo[c] = *reinterpret_cast<const short *>(&tv);
        }
        *reinterpret_cast<s16x8 *>(dx + i * 8) = vo;
    }
}

// dk[n,h,w,e] = sum_c x[n,h+dh,w+dw,c] * dy[n,h,w,c]
template <typename T>
__global__ void dynfilter_dk_kernel(const T *__restrict__ x,
                                    const T *__restrict__ dy,
                                    T *__restrict__ dk, long npix, int C8,
                                    int H, int W, int dil) {
    int C = C8 * 8;
    int nch = min(C8, 64);             // power of two by wrapper contract
    int rows_pb = blockDim.x / nch;
    int lane_c = threadIdx.x % nch;
    int row_off = threadIdx.x / nch;
    for (long pix = (long)blockIdx.x * rows_pb + row_off; pix < npix;
         pix += (long)gridDim.x * rows_pb) {
        int w = (int)(pix % W);
        long t2 = pix / W;
        int h = (int)(t2 % H);
        long nimg = t2 / H;
        float part[E];
#pragma unroll
        for (int e = 0; e < E; ++e) part[e] = 0.f;
        const T *xb = x + nimg * H * W * C;
        for (int cc = lane_c; cc < C8; cc += nch) {
            s16x8 dv = *reinterpret_cast<const s16x8 *>(
                dy + pix * C + cc * 8);
            float dyv[8];
#pragma unroll
            for (int c = 0; c < 8; ++c) dyv[c] = s16_as_f32<T>(dv[c]);
#pragma unroll
            for (int e = 0; e < E; ++e) {
                int hh = h + (e / 3 - 1) * dil;
                int ww = w + (e % 3 - 1) * dil;
                if ((unsigned)hh >= (unsigned)H || (unsigned)ww >= (unsigned)W)
                    continue;
                s16x8 v = *reinterpret_cast<const s16x8 *>(
                    xb + ((long)hh * W + ww) * C + cc * 8);
                float s = 0.f;
#pragma unroll
                for (int c = 0; c < 8; ++c) s += s16_as_f32<T>(v[c]) * dyv[c];
                part[e] += s;
            }
        }
        // subgroup reduce over the pixel's nch lanes (contiguous in tid)
#pragma unroll
        for (int e = 0; e < E; ++e)
            for (int off = nch >> 1; off > 0; off >>= 1)
                part[e] += __shfl_down(part[e], off, 64);
        if (lane_c == 0) {
#pragma unroll
            for (int e = 0; e < E; ++e)
                dk[pix * E + e] = ElemTraits<T>::from_f32(part[e]);
        }
    }
}

}  // namespace sodkit

#define DISPATCH_DF(TENSOR, NAME, ...)                                         \
    AT_DISPATCH_SWITCH((TENSOR).scalar_type(), NAME,                           \
        AT_DISPATCH_CASE(at::kBFloat16, [&] { using T = __hip_bfloat16; __VA_ARGS__(); }) \
        AT_DISPATCH_CASE(at::kHalf,     [&] { using T = __half; __VA_ARGS__(); }))

torch::Tensor dynfilter_fwd(torch::Tensor x, torch::Tensor k, int64_t dil) {
    TORCH_CHECK(x.is_cuda() && x.size(1) % 8 == 0 && k.size(1) == 9);
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    auto y = torch::empty_like(x);
    long total8 = x.numel() / 8;
    int blocks = (int)std::min<long>((total8 + 255) / 256, 4096);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_DF(x, "dynfilter_fwd", [&] {
        hipLaunchKernelGGL((sodkit::dynfilter_fwd_kernel<T>), dim3(blocks),
                           dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(x.data_ptr()),
                           reinterpret_cast<const T *>(k.data_ptr()),
                           reinterpret_cast<T *>(y.data_ptr()),
                           N, C / 8, H, W, (int)dil);
    });
    return y;
}

std::vector<torch::Tensor> dynfilter_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor k, int64_t dil) {
    int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    auto dx = torch::empty_like(x);
    auto dk = torch::empty_like(k);
    long total8 = x.numel() / 8;
    long npix = (long)N * H * W;
    int C8 = C / 8;
    TORCH_CHECK((C8 & (C8 - 1)) == 0 && C8 <= 64, "C/8 must be pow2 <= 64");
    auto stream = at::hip::getCurrentHIPStream();
    int b1 = (int)std::min<long>((total8 + 255) / 256, 4096);
    int rows_pb = 256 / std::min(C8, 64);
    int b2 = (int)std::min<long>((npix + rows_pb - 1) / rows_pb, 4096);
    DISPATCH_DF(x, "dynfilter_bwd", [&] {
        hipLaunchKernelGGL((sodkit::dynfilter_dx_kernel<T>), dim3(b1),
                           dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(dy.data_ptr()),
                           reinterpret_cast<const T *>(k.data_ptr()),
                           reinterpret_cast<T *>(dx.data_ptr()),
                           N, C8, H, W, (int)dil);
        hipLaunchKernelGGL((sodkit::dynfilter_dk_kernel<T>), dim3(b2),
                           dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(x.data_ptr()),
                           reinterpret_cast<const T *>(dy.data_ptr()),
                           reinterpret_cast<T *>(dk.data_ptr()),
                           npix, C8, H, W, (int)dil);
    });
    return {dx, dk};
}
