// sodkit — single-launch fused BatchNorm forward (stats + finalize +
// apply) via a cooperative grid.
//
// The split BN forward costs three launches per layer (stats partials,
// reduce+finalize, apply); across ~106 BN calls per training step the two
// small kernels sit at their ~5 us latency floors (~1 ms/step of pure
// floor).  A cooperative launch runs all three phases in ONE kernel with
// grid.sync() between them — the math, the slab layout and the
// deterministic fixed-order reduction are IDENTICAL to the split path
// (which remains the SyncBN / non-vec / fallback route: SyncBN needs the
// RCCL all-reduce between stats and apply).
//
// Eligibility: training, C % 8 == 0, C <= 2048 (single channel-chunk),
// 16-bit data, device supports cooperative launch, and the whole grid is
// co-resident (grid size comes from the occupancy query).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_cooperative_groups.h>

#include "common.h"

namespace cg = cooperative_groups;

#define DISPATCH_T2(TENSOR, NAME, ...)                                        \
    AT_DISPATCH_SWITCH((TENSOR).scalar_type(), NAME,                          \
        AT_DISPATCH_CASE(at::kBFloat16,                                       \
                         [&] { using T = __hip_bfloat16; __VA_ARGS__(); })    \
        AT_DISPATCH_CASE(at::kHalf, [&] { using T = __half; __VA_ARGS__(); }))

namespace sodkit {
namespace bnf {

template <typename T, bool RELU, bool RES>
__global__ __launch_bounds__(256)
void bn_fwd_fused_kernel(const T *__restrict__ x, const T *__restrict__ res,
                         const float *__restrict__ gamma,
                         const float *__restrict__ beta,
                         float n, float momentum, float eps,
                         float *__restrict__ running_mean,
                         float *__restrict__ running_var,
                         float *__restrict__ mean_out,
                         float *__restrict__ invstd_out,
                         T *__restrict__ y, unsigned char *__restrict__ mask,
                         float *__restrict__ partial,
                         long M, int C) {
    cg::grid_group grid = cg::this_grid();
    const int C8 = C >> 3;
    const int nch = min(C8, 256);
    const int rows_pb = 256 / nch;
    const int lane_c = threadIdx.x % nch;
    const int row_off = threadIdx.x / nch;
    const int c0 = lane_c * 8;

    // ---- phase 1: per-block stat partials (bn_stats_vec body) ----
    {
        __shared__ float red[256 * 16];
        float acc[16];
#pragma unroll
        for (int e = 0; e < 16; ++e) acc[e] = 0.f;
        if (row_off < rows_pb) {
            const T *base = x + (long)c0;
            long mstep = (long)gridDim.x * rows_pb;
            long m = (long)blockIdx.x * rows_pb + row_off;
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
            float *slab = partial + (long)blockIdx.x * 2 * C;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                slab[c0 + e] = acc[e];
                slab[C + c0 + e] = acc[8 + e];
            }
        }
    }
    grid.sync();

    // ---- phase 2: fixed-order slab reduce + finalize (grid-stride c) ----
    const int nb = gridDim.x;
    for (int c = blockIdx.x * 256 + threadIdx.x; c < C;
         c += gridDim.x * 256) {
        float a[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
        float q[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
        long R = 2L * C;
        int b = 0;
        for (; b + 8 <= nb; b += 8)
#pragma unroll
            for (int u = 0; u < 8; ++u) {
                a[u] += partial[(long)(b + u) * R + c];
                q[u] += partial[(long)(b + u) * R + C + c];
            }
        for (; b < nb; ++b) {
            a[0] += partial[(long)b * R + c];
            q[0] += partial[(long)b * R + C + c];
        }
        float sum = ((a[0] + a[1]) + (a[2] + a[3])) + ((a[4] + a[5]) + (a[6] + a[7]));
        float ssq = ((q[0] + q[1]) + (q[2] + q[3])) + ((q[4] + q[5]) + (q[6] + q[7]));
        float mean = sum / n;
        float var = fmaxf(ssq / n - mean * mean, 0.f);
        float unbiased = var * (n / fmaxf(n - 1.f, 1.f));
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
        mean_out[c] = mean;
        invstd_out[c] = rsqrtf(var + eps);
    }
    grid.sync();

    // ---- phase 3: apply (+optional ReLU-mask plane), column mapping ----
    {
        if (row_off >= rows_pb) return;
        float sc[8], sh[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            float is = invstd_out[c0 + e];
            sc[e] = is * gamma[c0 + e];
            sh[e] = beta[c0 + e] - mean_out[c0 + e] * sc[e];
        }
        long mstep = (long)gridDim.x * rows_pb;
        for (long m = (long)blockIdx.x * rows_pb + row_off; m < M; m += mstep) {
            long i = m * C8 + lane_c;
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
                    mb |= (ElemTraits<T>::to_f32(tv) > 0.f ? 1u : 0u) << e;
            }
            *reinterpret_cast<s16x8 *>(y + i * 8) = vo;
            if (RELU && mask != nullptr) mask[i] = (unsigned char)mb;
        }
    }
}

}  // namespace bnf
}  // namespace sodkit

// ===========================================================================
// Host glue
// ===========================================================================

namespace {

bool coop_supported() {
    static int sup = -1;
    if (sup < 0) {
        int dev = 0;
        (void)hipGetDevice(&dev);
        (void)hipDeviceGetAttribute(&sup, hipDeviceAttributeCooperativeLaunch,
                                    dev);
    }
    return sup > 0;
}

int coop_blocks(const void *kernel) {
    int per_cu = 0;
    (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(&per_cu, kernel, 256, 0);
    int dev = 0, cus = 0;
    (void)hipGetDevice(&dev);
    (void)hipDeviceGetAttribute(&cus, hipDeviceAttributeMultiprocessorCount,
                                dev);
    return std::max(1, per_cu * cus);
}

}  // namespace

bool bn_fwd_fused_eligible(int64_t C, int64_t elem_size) {
    return coop_supported() && C % 8 == 0 && C <= 2048 && elem_size == 2;
}

// returns {y, mask (empty if not emitted), mean, invstd}
std::vector<torch::Tensor> bn_fwd_fused(torch::Tensor x, torch::Tensor gamma,
                                        torch::Tensor beta, double n,
                                        double momentum, double eps,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var, bool relu,
                                        torch::Tensor residual,
                                        bool want_mask) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 4);
    int C = x.size(1);
    long M = x.numel() / C;
    TORCH_CHECK(bn_fwd_fused_eligible(C, x.element_size()), "not eligible");
    auto y = torch::empty_like(x);
    auto mean = torch::empty({C}, running_mean.options());
    auto invstd = torch::empty({C}, running_mean.options());
    torch::Tensor mask;
    bool emit_mask = want_mask && relu;
    if (emit_mask)
        mask = torch::empty({M * (C / 8)}, x.options().dtype(torch::kByte));
    bool has_res = residual.defined() && residual.numel() > 0;
    auto gammaf = gamma.to(torch::kFloat32).contiguous();
    auto betaf = beta.to(torch::kFloat32).contiguous();
    auto stream = at::hip::getCurrentHIPStream();

    DISPATCH_T2(x, "bn_fwd_fused", [&] {
        auto launch = [&](auto rl, auto rs) {
            constexpr bool RL = decltype(rl)::value;
            constexpr bool RS = decltype(rs)::value;
            const void *kern =
                (const void *)sodkit::bnf::bn_fwd_fused_kernel<T, RL, RS>;
            int nb = std::min(coop_blocks(kern), 512);
            // enough rows of work per block for phase 1
            int C8 = C / 8;
            int rows_pb = 256 / std::min(C8, 256);
            nb = (int)std::min<long>(nb, std::max<long>(1,
                     (M + rows_pb - 1) / rows_pb));
            auto partial = torch::empty({nb, 2L * C},
                                        x.options().dtype(torch::kFloat32));
            const T *xp = reinterpret_cast<const T *>(x.data_ptr());
            const T *rp = RS ? reinterpret_cast<const T *>(residual.data_ptr())
                             : nullptr;
            const float *gp = gammaf.data_ptr<float>();
            const float *bp = betaf.data_ptr<float>();
            float nf = (float)n, mo = (float)momentum, ep = (float)eps;
            float *rmp = running_mean.data_ptr<float>();
            float *rvp = running_var.data_ptr<float>();
            float *mp = mean.data_ptr<float>();
            float *ip = invstd.data_ptr<float>();
            T *yp = reinterpret_cast<T *>(y.data_ptr());
            unsigned char *kp = emit_mask ? mask.data_ptr<unsigned char>()
                                          : nullptr;
            float *pp = partial.data_ptr<float>();
            void *args[] = {&xp, &rp, &gp, &bp, &nf, &mo, &ep, &rmp, &rvp,
                            &mp, &ip, &yp, &kp, &pp, &M, &C};
            hipError_t err = hipLaunchCooperativeKernel(
                kern, dim3(nb), dim3(256), args, 0, stream.stream());
            TORCH_CHECK(err == hipSuccess, "bn_fwd_fused coop launch: ",
                        hipGetErrorString(err));
        };
        using tt = std::true_type; using ff = std::false_type;
        if (relu) { if (has_res) launch(tt{}, tt{}); else launch(tt{}, ff{}); }
        else      { if (has_res) launch(ff{}, tt{}); else launch(ff{}, ff{}); }
    });
    if (!mask.defined())
        mask = torch::empty({0}, x.options().dtype(torch::kByte));
    return {y, mask, mean, invstd};
}
