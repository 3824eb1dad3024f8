// sodkit — bilinear resize (+fused add) fwd and atomic-free gather bwd for
// CDNA4 (the decoder top-down path and `upsample_add`).
// NHWC; index math matches torch's align_corners=False upsample exactly
// (area_pixel source index with max(0, .) clamp).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {

SOD_DEVINL void src_index(int o, float scale, int in_size, int &i0, int &i1,
                          float &l) {
    float src = fmaxf(0.f, ((float)o + 0.5f) * scale - 0.5f);
    i0 = (int)src;
    if (i0 > in_size - 1) i0 = in_size - 1;
    i1 = i0 + (i0 < in_size - 1 ? 1 : 0);
    l = src - (float)i0;
    if (l < 0.f) l = 0.f;
    if (l > 1.f) l = 1.f;
}

// vectorized: 8 channels per thread (C % 8 == 0), same 4-tap weights
template <typename T, bool ADD>
__global__ void bilinear_fwd_vec_kernel(const T *__restrict__ x,
                                        const T *__restrict__ add,
                                        T *__restrict__ y, int N, int C8,
                                        int Hi, int Wi, int Ho, int Wo,
                                        float sh, float sw) {
    int C = C8 * 8;
    long total8 = (long)N * Ho * Wo * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total8; i += stride) {
        int c8 = (int)(i % C8);
        long pix = i / C8;
        int ow = (int)(pix % Wo);
        long t2 = pix / Wo;
        int oh = (int)(t2 % Ho);
        int n = (int)(t2 / Ho);
        int h0, h1, w0, w1;
        float lh, lw;
        src_index(oh, sh, Hi, h0, h1, lh);
        src_index(ow, sw, Wi, w0, w1, lw);
        const T *base = x + ((long)n * Hi) * Wi * C + c8 * 8;
        s16x8 v00 = *reinterpret_cast<const s16x8 *>(base + ((long)h0 * Wi + w0) * C);
        s16x8 v01 = *reinterpret_cast<const s16x8 *>(base + ((long)h0 * Wi + w1) * C);
        s16x8 v10 = *reinterpret_cast<const s16x8 *>(base + ((long)h1 * Wi + w0) * C);
        s16x8 v11 = *reinterpret_cast<const s16x8 *>(base + ((long)h1 * Wi + w1) * C);
        s16x8 va;
        if (ADD) va = *reinterpret_cast<const s16x8 *>(add + i * 8);
        s16x8 vo;
        float w00 = (1 - lh) * (1 - lw), w01 = (1 - lh) * lw;
        float w10 = lh * (1 - lw), w11 = lh * lw;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            float v = w00 * s16_as_f32<T>(v00[e]) + w01 * s16_as_f32<T>(v01[e])
                    + w10 * s16_as_f32<T>(v10[e]) + w11 * s16_as_f32<T>(v11[e]);
            if (ADD) v += s16_as_f32<T>(va[e]);
            T tv = ElemTraits<T>::from_f32(v);
            vo[e] = *reinterpret_cast<const short *>(&tv);
        }
        *reinterpret_cast<s16x8 *>(y + i * 8) = vo;
    }
}

template <typename T, bool ADD>
__global__ void bilinear_fwd_kernel(const T *__restrict__ x,
                                    const T *__restrict__ add,
                                    T *__restrict__ y, int N, int C, int Hi,
                                    int Wi, int Ho, int Wo, float sh, float sw) {
    long total = (long)N * Ho * Wo * C;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = (int)(i % C);
        long pix = i / C;
        int ow = (int)(pix % Wo);
        long t2 = pix / Wo;
        int oh = (int)(t2 % Ho);
        int n = (int)(t2 / Ho);
        int h0, h1, w0, w1;
        float lh, lw;
        src_index(oh, sh, Hi, h0, h1, lh);
        src_index(ow, sw, Wi, w0, w1, lw);
        const T *base = x + ((long)n * Hi) * Wi * C + c;
        float v00 = ElemTraits<T>::to_f32(base[((long)h0 * Wi + w0) * C]);
        float v01 = ElemTraits<T>::to_f32(base[((long)h0 * Wi + w1) * C]);
        float v10 = ElemTraits<T>::to_f32(base[((long)h1 * Wi + w0) * C]);
        float v11 = ElemTraits<T>::to_f32(base[((long)h1 * Wi + w1) * C]);
        float v = (1 - lh) * ((1 - lw) * v00 + lw * v01)
                + lh * ((1 - lw) * v10 + lw * v11);
        if (ADD) v += ElemTraits<T>::to_f32(add[i]);
        y[i] = ElemTraits<T>::from_f32(v);
    }
}

// Deterministic gather backward: each input pixel sums the output pixels
// whose interpolation stencil touches it (bounded range; ~2x upsampling in
// the decoder means <= 4 taps per dim).
template <typename T>
__global__ void bilinear_bwd_vec_kernel(const T *__restrict__ dy,
                                        T *__restrict__ dx, int N, int C8,
                                        int Hi, int Wi, int Ho, int Wo,
                                        float sh, float sw) {
    int C = C8 * 8;
    long total8 = (long)N * Hi * Wi * C8;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total8; i += stride) {
        int c8 = (int)(i % C8);
        long pix = i / C8;
        int iw = (int)(pix % Wi);
        long t2 = pix / Wi;
        int ih = (int)(t2 % Hi);
        int n = (int)(t2 / Hi);
        // exact contribution range (the per-tap weight==0 guard below keeps
        // one element of float-rounding slack); the clamped source index
        // means the LAST input row/col receives from every larger oh/ow
        int oh_lo = max(0, (int)floorf(((float)ih - 0.5f) / sh - 0.5f));
        int oh_hi = (ih == Hi - 1) ? (Ho - 1)
                  : min(Ho - 1, (int)ceilf(((float)ih + 1.5f) / sh - 0.5f));
        int ow_lo = max(0, (int)floorf(((float)iw - 0.5f) / sw - 0.5f));
        int ow_hi = (iw == Wi - 1) ? (Wo - 1)
                  : min(Wo - 1, (int)ceilf(((float)iw + 1.5f) / sw - 0.5f));
        float acc[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] = 0.f;
        const T *dybase = dy + (long)n * Ho * Wo * C + c8 * 8;
        for (int oh = oh_lo; oh <= oh_hi; ++oh) {
            int h0, h1;
            float lh;
            src_index(oh, sh, Hi, h0, h1, lh);
            float wh = 0.f;
            if (h0 == ih) wh += 1.f - lh;
            if (h1 == ih) wh += lh;
            if (wh == 0.f) continue;
            for (int ow = ow_lo; ow <= ow_hi; ++ow) {
                int w0, w1;
                float lw;
                src_index(ow, sw, Wi, w0, w1, lw);
                float ww = 0.f;
                if (w0 == iw) ww += 1.f - lw;
                if (w1 == iw) ww += lw;
                if (ww == 0.f) continue;
                float wgt = wh * ww;
                s16x8 v = *reinterpret_cast<const s16x8 *>(
                    dybase + ((long)oh * Wo + ow) * C);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[e] += wgt * s16_as_f32<T>(v[e]);
            }
        }
        s16x8 vo;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            T tv = ElemTraits<T>::from_f32(acc[e]);
            vo[e] = *reinterpret_cast<const short *>(&tv);
        }
        *reinterpret_cast<s16x8 *>(dx + i * 8) = vo;
    }
}

template <typename T>
__global__ void bilinear_bwd_kernel(const T *__restrict__ dy,
                                    T *__restrict__ dx, int N, int C, int Hi,
                                    int Wi, int Ho, int Wo, float sh, float sw) {
    long total = (long)N * Hi * Wi * C;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = (int)(i % C);
        long pix = i / C;
        int iw = (int)(pix % Wi);
        long t2 = pix / Wi;
        int ih = (int)(t2 % Hi);
        int n = (int)(t2 / Hi);
        int oh_lo = max(0, (int)floorf(((float)ih - 0.5f) / sh - 0.5f));
        int oh_hi = (ih == Hi - 1) ? (Ho - 1)
                  : min(Ho - 1, (int)ceilf(((float)ih + 1.5f) / sh - 0.5f));
        int ow_lo = max(0, (int)floorf(((float)iw - 0.5f) / sw - 0.5f));
        int ow_hi = (iw == Wi - 1) ? (Wo - 1)
                  : min(Wo - 1, (int)ceilf(((float)iw + 1.5f) / sw - 0.5f));
        float acc = 0.f;
        const T *dybase = dy + (long)n * Ho * Wo * C + c;
        for (int oh = oh_lo; oh <= oh_hi; ++oh) {
            int h0, h1;
            float lh;
            src_index(oh, sh, Hi, h0, h1, lh);
            float wh = 0.f;
            if (h0 == ih) wh += 1.f - lh;
            if (h1 == ih) wh += lh;
            if (wh == 0.f) continue;
            for (int ow = ow_lo; ow <= ow_hi; ++ow) {
                int w0, w1;
                float lw;
                src_index(ow, sw, Wi, w0, w1, lw);
                float ww = 0.f;
                if (w0 == iw) ww += 1.f - lw;
                if (w1 == iw) ww += lw;
                if (ww == 0.f) continue;
                acc += wh * ww
                     * ElemTraits<T>::to_f32(dybase[((long)oh * Wo + ow) * C]);
            }
        }
        dx[i] = ElemTraits<T>::from_f32(acc);
    }
}

}  // namespace sodkit

#define DISPATCH_T2(TENSOR, NAME, ...)                                         \
    AT_DISPATCH_SWITCH((TENSOR).scalar_type(), NAME,                           \
        AT_DISPATCH_CASE(at::kBFloat16, [&] { using T = __hip_bfloat16; __VA_ARGS__(); }) \
        AT_DISPATCH_CASE(at::kHalf,     [&] { using T = __half; __VA_ARGS__(); })          \
        AT_DISPATCH_CASE(at::kFloat,    [&] { using T = float; __VA_ARGS__(); }))

torch::Tensor bilinear_fwd(torch::Tensor x, int64_t Ho, int64_t Wo,
                           torch::Tensor add) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int N = x.size(0), C = x.size(1), Hi = x.size(2), Wi = x.size(3);
    bool has_add = add.defined() && add.numel() > 0;
    auto y = torch::empty({N, C, Ho, Wo},
                          x.options().memory_format(at::MemoryFormat::ChannelsLast));
    float sh = (float)Hi / (float)Ho, sw = (float)Wi / (float)Wo;
    long total = y.numel();
    int blocks = (int)std::min<long>((total + 255) / 256, 4096);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_T2(x, "bilinear_fwd", [&] {
        if (C % 8 == 0 && sizeof(T) == 2) {
            long total8 = total / 8;
            int b8 = (int)std::min<long>((total8 + 255) / 256, 4096);
            if (has_add)
                hipLaunchKernelGGL((sodkit::bilinear_fwd_vec_kernel<T, true>),
                                   dim3(b8), dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()),
                                   reinterpret_cast<const T *>(add.data_ptr()),
                                   reinterpret_cast<T *>(y.data_ptr()),
                                   N, C / 8, Hi, Wi, (int)Ho, (int)Wo, sh, sw);
            else
                hipLaunchKernelGGL((sodkit::bilinear_fwd_vec_kernel<T, false>),
                                   dim3(b8), dim3(256), 0, stream.stream(),
                                   reinterpret_cast<const T *>(x.data_ptr()), nullptr,
                                   reinterpret_cast<T *>(y.data_ptr()),
                                   N, C / 8, Hi, Wi, (int)Ho, (int)Wo, sh, sw);
        } else if (has_add)
            hipLaunchKernelGGL((sodkit::bilinear_fwd_kernel<T, true>), dim3(blocks),
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               reinterpret_cast<const T *>(add.data_ptr()),
                               reinterpret_cast<T *>(y.data_ptr()),
                               N, C, Hi, Wi, (int)Ho, (int)Wo, sh, sw);
        else
            hipLaunchKernelGGL((sodkit::bilinear_fwd_kernel<T, false>), dim3(blocks),
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()), nullptr,
                               reinterpret_cast<T *>(y.data_ptr()),
                               N, C, Hi, Wi, (int)Ho, (int)Wo, sh, sw);
    });
    return y;
}

torch::Tensor bilinear_bwd(torch::Tensor dy, int64_t Hi, int64_t Wi) {
    TORCH_CHECK(dy.is_cuda() && dy.dim() == 4);
    int N = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
    auto dx = torch::empty({N, C, Hi, Wi},
                           dy.options().memory_format(at::MemoryFormat::ChannelsLast));
    float sh = (float)Hi / (float)Ho, sw = (float)Wi / (float)Wo;
    long total = dx.numel();
    int blocks = (int)std::min<long>((total + 255) / 256, 4096);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_T2(dy, "bilinear_bwd", [&] {
        if (C % 8 == 0 && sizeof(T) == 2) {
            long total8 = total / 8;
            int b8 = (int)std::min<long>((total8 + 255) / 256, 4096);
            hipLaunchKernelGGL((sodkit::bilinear_bwd_vec_kernel<T>), dim3(b8),
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(dy.data_ptr()),
                               reinterpret_cast<T *>(dx.data_ptr()),
                               N, C / 8, (int)Hi, (int)Wi, Ho, Wo, sh, sw);
            return;
        }
        hipLaunchKernelGGL((sodkit::bilinear_bwd_kernel<T>), dim3(blocks),
                           dim3(256), 0, stream.stream(),
                           reinterpret_cast<const T *>(dy.data_ptr()),
                           reinterpret_cast<T *>(dx.data_ptr()),
                           N, C, (int)Hi, (int)Wi, Ho, Wo, sh, sw);
    });
    return dx;
}
