// sodkit — fused BatchNorm(+residual)(+ReLU) kernels for CDNA4 (SURVEY §2.3
// K3 + K8 epilogue).  NHWC layout; fp32 stats/params, bf16/fp16/fp32 data.
// Split into stats-reduce / apply (and bwd-reduce / bwd-apply) so SyncBN is
// one coalesced all-reduce between the halves with an identical kernel path.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {

// ---- per-channel sum + sumsq -> per-block slabs out[nb][2C] (fp32) ----
// Vector path (C % 8 == 0): each thread owns an 8-channel chunk; the 256
// threads cover min(C/8, 256) chunks x rows_pb rows per iteration, so a
// whole iteration's loads are one contiguous coalesced span.  Cross-row
// partials combine in LDS; each block writes a plain [2C] slab (the
// scalar fallback below is the only atomicAdd path).
template <typename T>
__global__ void bn_stats_vec_kernel(const T *__restrict__ x,
                                    float *__restrict__ out, long M, int C) {
    __shared__ float red[256 * 16];
    int C8 = C >> 3;
    int cc0 = blockIdx.y * 256;
    int nch = min(C8 - cc0, 256);
    int rows_pb = 256 / nch;
    int lane_c = threadIdx.x % nch;
    int row_off = threadIdx.x / nch;
    float acc[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[e] = 0.f;
    if (row_off < rows_pb) {
        const T *base = x + (long)(cc0 + lane_c) * 8;
        long mstep = (long)gridDim.x * rows_pb;
        long m = (long)blockIdx.x * rows_pb + row_off;
        // 4-deep unroll: 4 independent loads in flight per wave (the
        // dependent single-load chain leaves HBM ~8x under-subscribed)
        for (; m + 3 * mstep < M; m += 4 * mstep) {
            s16x8 v0 = *reinterpret_cast<const s16x8 *>(base + m * C);
            s16x8 v1 = *reinterpret_cast<const s16x8 *>(base + (m + mstep) * C);
            s16x8 v2 = *reinterpret_cast<const s16x8 *>(base + (m + 2 * mstep) * C);
            s16x8 v3 = *reinterpret_cast<const s16x8 *>(base + (m + 3 * mstep) * C);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float f0 = s16_as_f32<T>(v0[e]), f1 = s16_as_f32<T>(v1[e]);
                float f2 = s16_as_f32<T>(v2[e]), f3 = s16_as_f32<T>(v3[e]);
                acc[e] += (f0 + f1) + (f2 + f3);
                acc[8 + e] += (f0 * f0 + f1 * f1) + (f2 * f2 + f3 * f3);
            }
        }
        for (; m < M; m += mstep) {
            s16x8 v = *reinterpret_cast<const s16x8 *>(base + m * C);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float f = s16_as_f32<T>(v[e]);
                acc[e] += f;
                acc[8 + e] += f * f;
            }
        }
    }
#pragma unroll
    for (int e = 0; e < 16; ++e) red[threadIdx.x * 16 + e] = acc[e];
    __syncthreads();
    if (row_off == 0) {
        for (int r = 1; r < rows_pb; ++r)
#pragma unroll
            for (int e = 0; e < 16; ++e)
                acc[e] += red[(r * nch + lane_c) * 16 + e];
        int c0 = (cc0 + lane_c) * 8;
        // per-block partial [2C] slab: plain stores; a second kernel reduces
        // (same-address atomicAdd serializes ~linearly in block count)
        float *slab = out + (long)blockIdx.x * 2 * C;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            slab[c0 + e] = acc[e];
            slab[C + c0 + e] = acc[8 + e];
        }
    }
}

// partial[nb][R] -> out[R] (deterministic fixed-order sum).  The slab
// dimension is split across threadIdx.y and combined with a fixed-order
// LDS tree: a flat R-threads grid is only R/256 blocks (1-2 for BN-sized
// R), which leaves one CU chewing through nb*R floats at latency — the
// y-split is ~TY x more parallelism at identical determinism.  8
// independent accumulators keep 8 loads in flight per thread.
template <int TX, int TY>
__global__ void slab_reduce_kernel(const float *__restrict__ partial,
                                   float *__restrict__ out, int nb, long R) {
    __shared__ float red[TY][TX];
    long r = (long)blockIdx.x * TX + threadIdx.x;
    float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    if (r < R) {
        int b = threadIdx.y;
        for (; b + 7 * TY < nb; b += 8 * TY)
#pragma unroll
            for (int u = 0; u < 8; ++u)
                a[u] += partial[(long)(b + u * TY) * R + r];
        for (; b < nb; b += TY) a[0] += partial[(long)b * R + r];
    }
    red[threadIdx.y][threadIdx.x] =
        ((a[0] + a[1]) + (a[2] + a[3])) + ((a[4] + a[5]) + (a[6] + a[7]));
    __syncthreads();
#pragma unroll
    for (int s = TY / 2; s > 0; s >>= 1) {
        if (threadIdx.y < s)
            red[threadIdx.y][threadIdx.x] += red[threadIdx.y + s][threadIdx.x];
        __syncthreads();
    }
    if (threadIdx.y == 0 && r < R) out[r] = red[0][threadIdx.x];
}

// fused: reduce the per-block stat slabs AND produce mean/invstd/running
// stats in one kernel (halves the per-BN-layer launch count on the
// non-SyncBN forward path).  Same y-split as slab_reduce, but each thread
// carries the (sum, sumsq) PAIR for its channel so the ty==0 row can
// finalize without a cross-column exchange.
template <int TX, int TY>
__global__ void bn_reduce_finalize_kernel(const float *__restrict__ partial,
                                          int nb, int C, float n,
                                          float momentum, float eps,
                                          float *__restrict__ running_mean,
                                          float *__restrict__ running_var,
                                          float *__restrict__ mean_out,
                                          float *__restrict__ invstd_out) {
    __shared__ float red[TY][TX][2];
    int c = blockIdx.x * TX + threadIdx.x;
    float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    float q[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    long R = 2L * C;
    if (c < C) {
        int b = threadIdx.y;
        for (; b + 7 * TY < nb; b += 8 * TY)
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                a[u] += partial[(long)(b + u * TY) * R + c];
                q[u] += partial[(long)(b + u * TY) * R + C + c];
            }
        for (; b < nb; b += TY) {
            a[0] += partial[(long)b * R + c];
            q[0] += partial[(long)b * R + C + c];
        }
    }
    red[threadIdx.y][threadIdx.x][0] =
        ((a[0] + a[1]) + (a[2] + a[3])) + ((a[4] + a[5]) + (a[6] + a[7]));
    red[threadIdx.y][threadIdx.x][1] =
        ((q[0] + q[1]) + (q[2] + q[3])) + ((q[4] + q[5]) + (q[6] + q[7]));
    __syncthreads();
#pragma unroll
    for (int s = TY / 2; s > 0; s >>= 1) {
        if (threadIdx.y < s) {
            red[threadIdx.y][threadIdx.x][0] += red[threadIdx.y + s][threadIdx.x][0];
            red[threadIdx.y][threadIdx.x][1] += red[threadIdx.y + s][threadIdx.x][1];
        }
        __syncthreads();
    }
    if (threadIdx.y != 0 || c >= C) return;
    float sum = red[0][threadIdx.x][0];
    float ssq = red[0][threadIdx.x][1];
    float mean = sum / n;
    // two-pass variance can go (slightly) negative by cancellation at
    // large-mean fp16 scales; rsqrt(negative) would poison the tensor
    float var = fmaxf(ssq / n - mean * mean, 0.f);
    float unbiased = var * (n / fmaxf(n - 1.f, 1.f));
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    mean_out[c] = mean;
    invstd_out[c] = rsqrtf(var + eps);
}

template <typename T>
__global__ void bn_stats_kernel(const T *__restrict__ x, float *__restrict__ out,
                                long M, int C) {
    int c = blockIdx.y * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float s = 0.f, ss = 0.f;
    for (long m = blockIdx.x; m < M; m += gridDim.x) {
        float v = ElemTraits<T>::to_f32(x[m * C + c]);
        s += v;
        ss += v * v;
    }
    atomicAdd(&out[c], s);
    atomicAdd(&out[C + c], ss);
}

// ---- stats -> (mean, invstd) + running-stat update, one tiny kernel
// (replaces ~8 ATen [C]-sized ops per BN layer per step) ----
__global__ void bn_finalize_kernel(const float *__restrict__ stats, float n,
                                   float momentum, float eps,
                                   float *__restrict__ running_mean,
                                   float *__restrict__ running_var,
                                   float *__restrict__ mean_out,
                                   float *__restrict__ invstd_out, int C,
                                   bool training) {
    int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float mean, var;
    if (training) {
        mean = stats[c] / n;
        var = fmaxf(stats[C + c] / n - mean * mean, 0.f);
        float unbiased = var * (n / fmaxf(n - 1.f, 1.f));
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    } else {
        mean = running_mean[c];
        var = fmaxf(running_var[c], 0.f);
    }
    mean_out[c] = mean;
    invstd_out[c] = rsqrtf(var + eps);
}

// ---- vectorized apply: 8 channels per thread (C % 8 == 0) ----
template <typename T, bool RELU, bool RES>
__global__ void bn_apply_vec_kernel(const T *__restrict__ x,
                                    const T *__restrict__ res,
                                    const float *__restrict__ mean,
                                    const float *__restrict__ invstd,
                                    const float *__restrict__ gamma,
                                    const float *__restrict__ beta,
                                    T *__restrict__ y,
                                    unsigned char *__restrict__ mask,
                                    long M, int C8) {
    // mask != nullptr (training, RELU): emit a 1-bit-per-element ReLU-mask
    // plane (one byte per 8-channel vector) so the backward kernels read
    // 1/16 of y's bytes instead of y itself.
    // COLUMN thread mapping (bn_bwd_reduce's pattern): each thread owns a
    // fixed 8-channel group and strides rows — the 4 per-channel
    // coefficient vectors load ONCE instead of per element-group (the
    // flat mapping ran at ~1.9 TB/s, coefficient-load bound).
    int cc0 = blockIdx.y * 256;
    int nch = min(C8 - cc0, 256);
    int rows_pb = 256 / nch;
    int lane_c = threadIdx.x % nch;
    int row_off = threadIdx.x / nch;
    if (row_off >= rows_pb) return;
    int c0 = (cc0 + lane_c) * 8;
    const long gcol = cc0 + lane_c;
    float sc[8], sh[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        sc[e] = invstd[c0 + e] * gamma[c0 + e];
        sh[e] = beta[c0 + e] - mean[c0 + e] * sc[e];
    }
    long mstep = (long)gridDim.x * rows_pb;
    for (long m = (long)blockIdx.x * rows_pb + row_off; m < M; m += mstep) {
        long i = m * C8 + gcol;
        s16x8 vx = *reinterpret_cast<const s16x8 *>(x + i * 8);
        s16x8 vr;
        if (RES) vr = *reinterpret_cast<const s16x8 *>(res + i * 8);
        s16x8 vo;
        unsigned int mb = 0;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            float v = s16_as_f32<T>(vx[e]) * sc[e] + sh[e];
            if (RES) v += s16_as_f32<T>(vr[e]);
            if (RELU) v = fmaxf(v, 0.f);
            T tv = ElemTraits<T>::from_f32(v);
            vo[e] = *reinterpret_cast<const short *>(&tv);
            if (RELU && mask != nullptr)
                // bit from the ROUNDED value: bit-exact with the y>0 path
                mb |= (ElemTraits<T>::to_f32(tv) > 0.f ? 1u : 0u) << e;
        }
        *reinterpret_cast<s16x8 *>(y + i * 8) = vo;
        if (RELU && mask != nullptr) mask[i] = (unsigned char)mb;
    }
}

template <typename T, bool RELU, bool RES>
__global__ void bn_apply_kernel(const T *__restrict__ x,
                                const T *__restrict__ res,
                                const float *__restrict__ mean,
                                const float *__restrict__ invstd,
                                const float *__restrict__ gamma,
                                const float *__restrict__ beta,
                                T *__restrict__ y, long total, int C) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = (int)(i % C);
        float sc = invstd[c] * gamma[c];
        float sh = beta[c] - mean[c] * sc;
        float v = ElemTraits<T>::to_f32(x[i]) * sc + sh;
        if (RES) v += ElemTraits<T>::to_f32(res[i]);
        if (RELU) v = fmaxf(v, 0.f);
        y[i] = ElemTraits<T>::from_f32(v);
    }
}

// ---- backward reduce: g = dy * relumask(y); out[0][c]=sum g,
//      out[1][c]=sum g*xhat ----
template <typename T, bool RELU, bool MASKED>
__global__ void bn_bwd_reduce_vec_kernel(const T *__restrict__ dy,
                                         const T *__restrict__ y,
                                         const T *__restrict__ x,
                                         const unsigned char *__restrict__ mask,
                                         const float *__restrict__ mean,
                                         const float *__restrict__ invstd,
                                         float *__restrict__ out, long M, int C) {
    // MASKED: the ReLU mask comes from bn_apply's bit plane (1/16 of y's
    // bytes) instead of re-reading y — drops a third of the kernel's
    // memory traffic (BASELINE "bn_bwd_reduce" lever).
    __shared__ float red[256 * 16];
    int C8 = C >> 3;
    int cc0 = blockIdx.y * 256;
    int nch = min(C8 - cc0, 256);
    int rows_pb = 256 / nch;
    int lane_c = threadIdx.x % nch;
    int row_off = threadIdx.x / nch;
    float acc[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[e] = 0.f;
    int c0 = (cc0 + lane_c) * 8;
    const long moff = cc0 + lane_c;          // mask plane column
    float mu[8], is[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) { mu[e] = mean[c0 + e]; is[e] = invstd[c0 + e]; }
    if (row_off < rows_pb) {
        long off0 = (long)c0;
        long mstep = (long)gridDim.x * rows_pb;
        long m = (long)blockIdx.x * rows_pb + row_off;
        for (; m + mstep < M; m += 2 * mstep) {
            long i0 = m * C + off0, i1 = (m + mstep) * C + off0;
            s16x8 vdy0 = *reinterpret_cast<const s16x8 *>(dy + i0);
            s16x8 vdy1 = *reinterpret_cast<const s16x8 *>(dy + i1);
            s16x8 vx0 = *reinterpret_cast<const s16x8 *>(x + i0);
            s16x8 vx1 = *reinterpret_cast<const s16x8 *>(x + i1);
            s16x8 vy0, vy1;
            unsigned int mb0 = 0xff, mb1 = 0xff;
            if (RELU) {
                if (MASKED) {
                    mb0 = mask[m * C8 + moff];
                    mb1 = mask[(m + mstep) * C8 + moff];
                } else {
                    vy0 = *reinterpret_cast<const s16x8 *>(y + i0);
                    vy1 = *reinterpret_cast<const s16x8 *>(y + i1);
                }
            }
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float g0 = s16_as_f32<T>(vdy0[e]), g1 = s16_as_f32<T>(vdy1[e]);
                if (RELU) {
                    if (MASKED) {
                        if (!((mb0 >> e) & 1)) g0 = 0.f;
                        if (!((mb1 >> e) & 1)) g1 = 0.f;
                    } else {
                        if (s16_as_f32<T>(vy0[e]) <= 0.f) g0 = 0.f;
                        if (s16_as_f32<T>(vy1[e]) <= 0.f) g1 = 0.f;
                    }
                }
                float xh0 = (s16_as_f32<T>(vx0[e]) - mu[e]) * is[e];
                float xh1 = (s16_as_f32<T>(vx1[e]) - mu[e]) * is[e];
                acc[e] += g0 + g1;
                acc[8 + e] += g0 * xh0 + g1 * xh1;
            }
        }
        for (; m < M; m += mstep) {
            long i = m * C + off0;
            s16x8 vdy = *reinterpret_cast<const s16x8 *>(dy + i);
            s16x8 vy, vx = *reinterpret_cast<const s16x8 *>(x + i);
            unsigned int mb = 0xff;
            if (RELU) {
                if (MASKED) mb = mask[m * C8 + moff];
                else vy = *reinterpret_cast<const s16x8 *>(y + i);
            }
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float g = s16_as_f32<T>(vdy[e]);
                if (RELU) {
                    if (MASKED) { if (!((mb >> e) & 1)) g = 0.f; }
                    else if (s16_as_f32<T>(vy[e]) <= 0.f) g = 0.f;
                }
                float xh = (s16_as_f32<T>(vx[e]) - mu[e]) * is[e];
                acc[e] += g;
                acc[8 + e] += g * xh;
            }
        }
    }
#pragma unroll
    for (int e = 0; e < 16; ++e) red[threadIdx.x * 16 + e] = acc[e];
    __syncthreads();
    if (row_off == 0) {
        for (int r = 1; r < rows_pb; ++r)
#pragma unroll
            for (int e = 0; e < 16; ++e)
                acc[e] += red[(r * nch + lane_c) * 16 + e];
        float *slab = out + (long)blockIdx.x * 2 * C;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            slab[c0 + e] = acc[e];
            slab[C + c0 + e] = acc[8 + e];
        }
    }
}

template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T *__restrict__ dy,
                                     const T *__restrict__ y,
                                     const T *__restrict__ x,
                                     const float *__restrict__ mean,
                                     const float *__restrict__ invstd,
                                     float *__restrict__ out, long M, int C) {
    int c = blockIdx.y * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float mu = mean[c], is = invstd[c];
    float sg = 0.f, sgx = 0.f;
    for (long m = blockIdx.x; m < M; m += gridDim.x) {
        long i = m * C + c;
        float g = ElemTraits<T>::to_f32(dy[i]);
        if (RELU && ElemTraits<T>::to_f32(y[i]) <= 0.f) g = 0.f;
        float xh = (ElemTraits<T>::to_f32(x[i]) - mu) * is;
        sg += g;
        sgx += g * xh;
    }
    atomicAdd(&out[c], sg);
    atomicAdd(&out[C + c], sgx);
}

// ---- vectorized backward apply (C % 8 == 0) ----
template <typename T, bool RELU, bool NEED_DRES, bool MASKED>
__global__ void bn_bwd_apply_vec_kernel(const T *__restrict__ dy,
                                        const T *__restrict__ y,
                                        const T *__restrict__ x,
                                        const unsigned char *__restrict__ mask,
                                        const float *__restrict__ mean,
                                        const float *__restrict__ invstd,
                                        const float *__restrict__ gamma,
                                        const float *__restrict__ sums,
                                        float inv_n, T *__restrict__ dx,
                                        T *__restrict__ dres, long M,
                                        int C8) {
    // column thread mapping; see bn_apply_vec_kernel
    int C = C8 * 8;
    int cc0 = blockIdx.y * 256;
    int nch = min(C8 - cc0, 256);
    int rows_pb = 256 / nch;
    int lane_c = threadIdx.x % nch;
    int row_off = threadIdx.x / nch;
    if (row_off >= rows_pb) return;
    int c0 = (cc0 + lane_c) * 8;
    const long gcol = cc0 + lane_c;
    float is8[8], mu8[8], gis8[8], s0[8], s1[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        is8[e] = invstd[c0 + e];
        mu8[e] = mean[c0 + e];
        gis8[e] = gamma[c0 + e] * is8[e];
        s0[e] = sums[c0 + e] * inv_n;
        s1[e] = sums[C + c0 + e] * inv_n;
    }
    long mstep = (long)gridDim.x * rows_pb;
    for (long m = (long)blockIdx.x * rows_pb + row_off; m < M; m += mstep) {
        long i = m * C8 + gcol;
        s16x8 vdy = *reinterpret_cast<const s16x8 *>(dy + i * 8);
        s16x8 vx = *reinterpret_cast<const s16x8 *>(x + i * 8);
        s16x8 vy;
        unsigned int mb = 0xff;
        if (RELU) {
            if (MASKED) mb = mask[i];
            else vy = *reinterpret_cast<const s16x8 *>(y + i * 8);
        }
        s16x8 vo, vdr;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            float g = s16_as_f32<T>(vdy[e]);
            if (RELU) {
                if (MASKED) { if (!((mb >> e) & 1)) g = 0.f; }
                else if (s16_as_f32<T>(vy[e]) <= 0.f) g = 0.f;
            }
            if (NEED_DRES) {
                T tg = ElemTraits<T>::from_f32(g);
                vdr[e] = *reinterpret_cast<const short *>(&tg);
            }
            float xh = (s16_as_f32<T>(vx[e]) - mu8[e]) * is8[e];
            float v = g - (s0[e] + xh * s1[e]);
            T tv = ElemTraits<T>::from_f32(gis8[e] * v);
            vo[e] = *reinterpret_cast<const short *>(&tv);
        }
        *reinterpret_cast<s16x8 *>(dx + i * 8) = vo;
        if (NEED_DRES) *reinterpret_cast<s16x8 *>(dres + i * 8) = vdr;
    }
}

// ---- backward apply:
//   train: dx = gamma*invstd * (g - (sum_g + xhat*sum_gxhat)/n)
//   eval (inv_n == 0): dx = gamma*invstd * g
//   optional residual grad dres = g ----
template <typename T, bool RELU, bool NEED_DRES>
__global__ void bn_bwd_apply_kernel(const T *__restrict__ dy,
                                    const T *__restrict__ y,
                                    const T *__restrict__ x,
                                    const float *__restrict__ mean,
                                    const float *__restrict__ invstd,
                                    const float *__restrict__ gamma,
                                    const float *__restrict__ sums,
                                    float inv_n, T *__restrict__ dx,
                                    T *__restrict__ dres, long total, int C) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int c = (int)(i % C);
        float g = ElemTraits<T>::to_f32(dy[i]);
        if (RELU && ElemTraits<T>::to_f32(y[i]) <= 0.f) g = 0.f;
        if (NEED_DRES) dres[i] = ElemTraits<T>::from_f32(g);
        float is = invstd[c];
        float xh = (ElemTraits<T>::to_f32(x[i]) - mean[c]) * is;
        float v = g - (sums[c] + xh * sums[C + c]) * inv_n;
        dx[i] = ElemTraits<T>::from_f32(gamma[c] * is * v);
    }
}

}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

namespace {

inline dim3 stats_grid(long M, int C) {
    return dim3(std::min<long>(M, 512), (C + 255) / 256);
}

inline int ew_blocks(long total) {
    return (int)std::min<long>((total + 255) / 256, 4096);
}

#define DISPATCH_T(TENSOR, NAME, ...)                                          \
    AT_DISPATCH_SWITCH((TENSOR).scalar_type(), NAME,                           \
        AT_DISPATCH_CASE(at::kBFloat16, [&] { using T = __hip_bfloat16; __VA_ARGS__(); }) \
        AT_DISPATCH_CASE(at::kHalf,     [&] { using T = __half; __VA_ARGS__(); })          \
        AT_DISPATCH_CASE(at::kFloat,    [&] { using T = float; __VA_ARGS__(); }))

}  // namespace

// column-mapped elementwise grid: ~4 m-iterations per thread (hoisted
// coefficients amortize; loads pipeline across the loop)
static inline dim3 vec_ew_grid(long M, int C) {
    int C8 = C / 8;
    int nch = std::min(C8, 256);
    int rows_pb = 256 / nch;
    long xblocks = std::max<long>(1,
        std::min<long>((M + rows_pb * 4 - 1) / (rows_pb * 4), 2048));
    return dim3((unsigned)xblocks, (C8 + 255) / 256);
}

static inline dim3 vec_stats_grid(long M, int C) {
    int C8 = C / 8;
    int nch = std::min(C8, 256);
    int rows_pb = 256 / nch;
    // target >=8 loop iterations per thread: small tensors get fewer blocks
    // (=> fewer partial slabs for the reduce pass), big ones fill the chip
    long xblocks = std::min<long>((M + rows_pb * 8 - 1) / (rows_pb * 8), 512);
    if (xblocks < 64) xblocks = std::min<long>((M + rows_pb - 1) / rows_pb, 64);
    return dim3((unsigned)xblocks, (C8 + 255) / 256);
}

torch::Tensor bn_stats(torch::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int C = x.size(1);
    long M = x.numel() / C;
    bool vec = (C % 8 == 0) && x.element_size() == 2;
    // the vec path's slab_reduce overwrites out fully -> empty, not zeros
    auto out = vec ? torch::empty({2, C}, x.options().dtype(torch::kFloat32))
                   : torch::zeros({2, C}, x.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_T(x, "bn_stats", [&] {
        if (C % 8 == 0 && sizeof(T) == 2) {
            dim3 g = vec_stats_grid(M, C);
            // y-chunks (C8 > 256) share x-indexed slabs: disjoint channel
            // ranges, no collision
            int nb = g.x;
            auto partial = torch::empty({nb, 2L * C},
                                        x.options().dtype(torch::kFloat32));
            hipLaunchKernelGGL((sodkit::bn_stats_vec_kernel<T>), g, dim3(256),
                               0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               partial.data_ptr<float>(), M, C);
            hipLaunchKernelGGL((sodkit::slab_reduce_kernel<16, 16>),
                               dim3((2 * C + 15) / 16), dim3(16, 16), 0,
                               stream.stream(), partial.data_ptr<float>(),
                               out.data_ptr<float>(), nb, 2L * C);
        } else
            hipLaunchKernelGGL((sodkit::bn_stats_kernel<T>), stats_grid(M, C),
                               dim3(256), 0, stream.stream(),
                               reinterpret_cast<const T *>(x.data_ptr()),
                               out.data_ptr<float>(), M, C);
    });
    return out;
}

std::vector<torch::Tensor> bn_finalize(torch::Tensor stats, double n,
                                       double momentum, double eps,
                                       torch::Tensor running_mean,
                                       torch::Tensor running_var,
                                       bool training) {
    int C = running_mean.size(0);
    auto mean = torch::empty({C}, running_mean.options());
    auto invstd = torch::empty({C}, running_mean.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(sodkit::bn_finalize_kernel, dim3((C + 255) / 256),
                       dim3(256), 0, stream.stream(),
                       training ? stats.data_ptr<float>() : nullptr, (float)n,
                       (float)momentum, (float)eps,
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), C, training);
    return {mean, invstd};
}

std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor x, double n,
                                              double momentum, double eps,
                                              torch::Tensor running_mean,
                                              torch::Tensor running_var) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int C = x.size(1);
    long M = x.numel() / C;
    auto stream = at::hip::getCurrentHIPStream();
    auto mean = torch::empty({C}, running_mean.options());
    auto invstd = torch::empty({C}, running_mean.options());
    bool vec = (C % 8 == 0) && x.element_size() == 2;
    if (!vec) {
        // fallback: plain stats (atomics) + finalize
        auto stats = bn_stats(x);
        return bn_finalize(stats, n, momentum, eps, running_mean, running_var,
                           true);
    }
    dim3 g = vec_stats_grid(M, C);
    int nb = g.x;
    auto partial = torch::empty({nb, 2L * C},
                                x.options().dtype(torch::kFloat32));
    DISPATCH_T(x, "bn_stats_finalize", [&] {
        hipLaunchKernelGGL((sodkit::bn_stats_vec_kernel<T>), g, dim3(256), 0,
                           stream.stream(),
                           reinterpret_cast<const T *>(x.data_ptr()),
                           partial.data_ptr<float>(), M, C);
    });
    hipLaunchKernelGGL((sodkit::bn_reduce_finalize_kernel<16, 16>),
                       dim3((C + 15) / 16), dim3(16, 16), 0, stream.stream(),
                       partial.data_ptr<float>(), nb, C, (float)n,
                       (float)momentum, (float)eps,
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
    return {mean, invstd};
}

// finalize conv-emitted stat slabs (conv2d_fwd_stats): partial[nb][2C]
// has the same layout as bn_stats_vec's output, so this is exactly the
// second half of bn_stats_finalize — the stats pass never touches y.
std::vector<torch::Tensor> bn_finalize_slabs(torch::Tensor partial, double n,
                                             double momentum, double eps,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var) {
    TORCH_CHECK(partial.is_cuda() && partial.dim() == 2
                && partial.scalar_type() == at::kFloat);
    int nb = partial.size(0);
    int C = (int)(partial.size(1) / 2);
    TORCH_CHECK(C == running_mean.size(0), "slab/param channel mismatch");
    auto mean = torch::empty({C}, running_mean.options());
    auto invstd = torch::empty({C}, running_mean.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL((sodkit::bn_reduce_finalize_kernel<16, 16>),
                       dim3((C + 15) / 16), dim3(16, 16), 0, stream.stream(),
                       partial.data_ptr<float>(), nb, C, (float)n,
                       (float)momentum, (float)eps,
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
    return {mean, invstd};
}

// reduce conv-emitted stat slabs to the coalesced [2, C] stats tensor
// (SyncBN: this is what gets all-reduced before bn_finalize)
torch::Tensor bn_slab_sum(torch::Tensor partial) {
    TORCH_CHECK(partial.is_cuda() && partial.dim() == 2
                && partial.scalar_type() == at::kFloat);
    int nb = partial.size(0);
    long R = partial.size(1);
    auto out = torch::empty({2, R / 2},
                            partial.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL((sodkit::slab_reduce_kernel<16, 16>),
                       dim3((R + 15) / 16), dim3(16, 16), 0, stream.stream(),
                       partial.data_ptr<float>(), out.data_ptr<float>(), nb,
                       R);
    return out;
}

std::vector<torch::Tensor> bn_apply(torch::Tensor x, torch::Tensor mean,
                                    torch::Tensor invstd, torch::Tensor gamma,
                                    torch::Tensor beta, bool relu,
                                    torch::Tensor residual, bool want_mask) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int C = x.size(1);
    long total = x.numel();
    auto y = torch::empty_like(x);
    torch::Tensor mask;   // ReLU bit plane for the backward kernels
    bool emit_mask = want_mask && relu && C % 8 == 0 && x.element_size() == 2;
    if (emit_mask)
        mask = torch::empty({total / 8}, x.options().dtype(torch::kByte));
    auto stream = at::hip::getCurrentHIPStream();
    bool has_res = residual.defined() && residual.numel() > 0;
    auto meanf = mean.contiguous(), invstdf = invstd.contiguous();
    auto gammaf = gamma.to(torch::kFloat32).contiguous();
    auto betaf = beta.to(torch::kFloat32).contiguous();
    DISPATCH_T(x, "bn_apply", [&] {
        const T *resp = has_res ? reinterpret_cast<const T *>(residual.data_ptr())
                                : nullptr;
        auto launch = [&](auto rl, auto rs) {
            if (C % 8 == 0 && sizeof(T) == 2) {
                long M = total / C;
                hipLaunchKernelGGL(
                    (sodkit::bn_apply_vec_kernel<T, decltype(rl)::value, decltype(rs)::value>),
                    vec_ew_grid(M, C), dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(x.data_ptr()), resp,
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    gammaf.data_ptr<float>(), betaf.data_ptr<float>(),
                    reinterpret_cast<T *>(y.data_ptr()),
                    emit_mask ? mask.data_ptr<unsigned char>() : nullptr,
                    M, C / 8);
            } else
                hipLaunchKernelGGL(
                    (sodkit::bn_apply_kernel<T, decltype(rl)::value, decltype(rs)::value>),
                    dim3(ew_blocks(total)), dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(x.data_ptr()), resp,
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    gammaf.data_ptr<float>(), betaf.data_ptr<float>(),
                    reinterpret_cast<T *>(y.data_ptr()), total, C);
        };
        using tt = std::true_type; using ff = std::false_type;
        if (relu) { if (has_res) launch(tt{}, tt{}); else launch(tt{}, ff{}); }
        else      { if (has_res) launch(ff{}, tt{}); else launch(ff{}, ff{}); }
    });
    if (!mask.defined())
        mask = torch::empty({0}, x.options().dtype(torch::kByte));
    return {y, mask};
}

torch::Tensor bn_bwd_reduce(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                            torch::Tensor mean, torch::Tensor invstd, bool relu,
                            torch::Tensor mask) {
    int C = x.size(1);
    long M = x.numel() / C;
    bool vec = (C % 8 == 0) && x.element_size() == 2;
    bool masked = relu && vec && mask.defined() && mask.numel() == x.numel() / 8;
    auto out = vec ? torch::empty({2, C}, x.options().dtype(torch::kFloat32))
                   : torch::zeros({2, C}, x.options().dtype(torch::kFloat32));
    auto stream = at::hip::getCurrentHIPStream();
    auto meanf = mean.contiguous(), invstdf = invstd.contiguous();
    DISPATCH_T(x, "bn_bwd_reduce", [&] {
        auto launch = [&](auto rl, auto mk) {
            if (C % 8 == 0 && sizeof(T) == 2) {
                dim3 g = vec_stats_grid(M, C);
                int nb = g.x;
                auto partial = torch::empty({nb, 2L * C},
                                            x.options().dtype(torch::kFloat32));
                hipLaunchKernelGGL(
                    (sodkit::bn_bwd_reduce_vec_kernel<T, decltype(rl)::value,
                                                      decltype(mk)::value>),
                    g, dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(dy.data_ptr()),
                    reinterpret_cast<const T *>(y.data_ptr()),
                    reinterpret_cast<const T *>(x.data_ptr()),
                    decltype(mk)::value ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    partial.data_ptr<float>(), M, C);
                hipLaunchKernelGGL((sodkit::slab_reduce_kernel<16, 16>),
                                   dim3((2 * C + 15) / 16), dim3(16, 16), 0,
                                   stream.stream(), partial.data_ptr<float>(),
                                   out.data_ptr<float>(), nb, 2L * C);
            } else
                hipLaunchKernelGGL(
                    (sodkit::bn_bwd_reduce_kernel<T, decltype(rl)::value>),
                    stats_grid(M, C), dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(dy.data_ptr()),
                    reinterpret_cast<const T *>(y.data_ptr()),
                    reinterpret_cast<const T *>(x.data_ptr()),
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    out.data_ptr<float>(), M, C);
        };
        using tt = std::true_type; using ff = std::false_type;
        if (relu) { if (masked) launch(tt{}, tt{}); else launch(tt{}, ff{}); }
        else launch(ff{}, ff{});
    });
    return out;
}

std::vector<torch::Tensor> bn_bwd_apply(torch::Tensor dy, torch::Tensor y,
                                        torch::Tensor x, torch::Tensor mean,
                                        torch::Tensor invstd, torch::Tensor gamma,
                                        torch::Tensor sums, double n_total,
                                        bool relu, bool need_dres,
                                        torch::Tensor mask) {
    int C = x.size(1);
    long total = x.numel();
    auto dx = torch::empty_like(x);
    torch::Tensor dres;
    if (need_dres) dres = torch::empty_like(x);
    float inv_n = n_total > 0 ? (float)(1.0 / n_total) : 0.f;
    bool vec8 = (C % 8 == 0) && x.element_size() == 2;
    bool masked = relu && vec8 && mask.defined() && mask.numel() == total / 8;
    auto stream = at::hip::getCurrentHIPStream();
    auto meanf = mean.contiguous(), invstdf = invstd.contiguous();
    auto gammaf = gamma.to(torch::kFloat32).contiguous();
    auto sumsf = sums.contiguous();
    DISPATCH_T(x, "bn_bwd_apply", [&] {
        auto launch = [&](auto rl, auto dr, auto mk) {
            if (C % 8 == 0 && sizeof(T) == 2) {
                long M = total / C;
                hipLaunchKernelGGL(
                    (sodkit::bn_bwd_apply_vec_kernel<T, decltype(rl)::value,
                                                     decltype(dr)::value,
                                                     decltype(mk)::value>),
                    vec_ew_grid(M, C), dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(dy.data_ptr()),
                    reinterpret_cast<const T *>(y.data_ptr()),
                    reinterpret_cast<const T *>(x.data_ptr()),
                    decltype(mk)::value ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    gammaf.data_ptr<float>(), sumsf.data_ptr<float>(), inv_n,
                    reinterpret_cast<T *>(dx.data_ptr()),
                    need_dres ? reinterpret_cast<T *>(dres.data_ptr()) : nullptr,
                    M, C / 8);
            } else
                hipLaunchKernelGGL(
                    (sodkit::bn_bwd_apply_kernel<T, decltype(rl)::value, decltype(dr)::value>),
                    dim3(ew_blocks(total)), dim3(256), 0, stream.stream(),
                    reinterpret_cast<const T *>(dy.data_ptr()),
                    reinterpret_cast<const T *>(y.data_ptr()),
                    reinterpret_cast<const T *>(x.data_ptr()),
                    meanf.data_ptr<float>(), invstdf.data_ptr<float>(),
                    gammaf.data_ptr<float>(), sumsf.data_ptr<float>(), inv_n,
                    reinterpret_cast<T *>(dx.data_ptr()),
                    need_dres ? reinterpret_cast<T *>(dres.data_ptr()) : nullptr,
                    total, C);
        };
        using tt = std::true_type; using ff = std::false_type;
        if (relu && masked) {
            if (need_dres) launch(tt{}, tt{}, tt{}); else launch(tt{}, ff{}, tt{});
        } else if (relu) {
            if (need_dres) launch(tt{}, tt{}, ff{}); else launch(tt{}, ff{}, ff{});
        } else {
            if (need_dres) launch(ff{}, tt{}, ff{}); else launch(ff{}, ff{}, ff{});
        }
    });
    if (!need_dres) dres = torch::Tensor();
    return {dx, dres};
}
