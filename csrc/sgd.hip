// sodkit — fused multi-tensor SGD for CDNA4.
// One launch sweeps every parameter chunk: bf16/fp16 grad upcast, grad scale
// (1/world * 1/loss_scale folded), weight decay, momentum, fp32 master
// update, and the fp32->bf16/fp16 param write-back — all in a single pass.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace sodkit {

constexpr int SGD_CHUNK = 16384;

// ptrs  [T][4]: param, grad, master, momentum  (device addresses)
// chunks[K][2]: tensor index, element offset
template <typename T, bool NESTEROV>
__global__ void sgd_fused_kernel(const long *__restrict__ ptrs,
                                 const long *__restrict__ numels,
                                 const float *__restrict__ wds,
                                 const long *__restrict__ chunks,
                                 float lr, float mu, float gscale,
                                 const float *__restrict__ scale,
                                 const float *__restrict__ lrp) {
    if (lrp != nullptr) lr = lrp[0];  // device-resident LR (hipGraph replay:
    //                                   the host updates a tensor, not a
    //                                   frozen kernel argument)
    if (scale != nullptr) {
        // device-side dynamic-loss-scale multiplier (fp16 path, sync-free):
        // 0.0 == overflow detected this step -> skip the whole update
        // (weight decay / momentum must not advance on a skipped step)
        float s = scale[0];
        if (s == 0.0f) return;
        gscale *= s;
    }
    long ti = chunks[blockIdx.x * 2 + 0];
    long off = chunks[blockIdx.x * 2 + 1];
    T *param = reinterpret_cast<T *>(ptrs[ti * 4 + 0]);
    const T *grad = reinterpret_cast<const T *>(ptrs[ti * 4 + 1]);
    float *master = reinterpret_cast<float *>(ptrs[ti * 4 + 2]);
    float *mom = reinterpret_cast<float *>(ptrs[ti * 4 + 3]);
    float wd = wds[ti];
    long end = min(off + (long)SGD_CHUNK, numels[ti]);
    for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
        float g = ElemTraits<T>::to_f32(grad[i]) * gscale;
        float w = master[i];
        g += wd * w;
        float m = mu * mom[i] + g;
        mom[i] = m;
        float upd = NESTEROV ? (g + mu * m) : m;
        w -= lr * upd;
        master[i] = w;
        param[i] = ElemTraits<T>::from_f32(w);
    }
}

}  // namespace sodkit

void sgd_fused(torch::Tensor ptrs, torch::Tensor numels, torch::Tensor wds,
               torch::Tensor chunks, int64_t dtype_code, double lr, double mu,
               double gscale, bool nesterov, torch::Tensor scale,
               torch::Tensor lr_t) {
    int K = chunks.size(0);
    const float *scale_ptr = nullptr;
    if (scale.defined() && scale.numel() > 0) {
        TORCH_CHECK(scale.is_cuda() && scale.scalar_type() == torch::kFloat32,
                    "sgd scale multiplier must be a device fp32 scalar");
        scale_ptr = scale.data_ptr<float>();
    }
    const float *lr_ptr = nullptr;
    if (lr_t.defined() && lr_t.numel() > 0) {
        TORCH_CHECK(lr_t.is_cuda() && lr_t.scalar_type() == torch::kFloat32,
                    "device lr must be a device fp32 scalar");
        lr_ptr = lr_t.data_ptr<float>();
    }
    auto stream = at::hip::getCurrentHIPStream();
    auto launch = [&](auto t_tag, auto nv) {
        using T = typename decltype(t_tag)::type;
        hipLaunchKernelGGL((sodkit::sgd_fused_kernel<T, decltype(nv)::value>),
                           dim3(K), dim3(256), 0, stream.stream(),
                           ptrs.data_ptr<long>(), numels.data_ptr<long>(),
                           wds.data_ptr<float>(), chunks.data_ptr<long>(),
                           (float)lr, (float)mu, (float)gscale, scale_ptr,
                           lr_ptr);
    };
    struct F32 { using type = float; };
    struct BF { using type = __hip_bfloat16; };
    struct F16 { using type = __half; };
    using tt = std::true_type; using ff = std::false_type;
    if (dtype_code == 0) { nesterov ? launch(F32{}, tt{}) : launch(F32{}, ff{}); }
    else if (dtype_code == 1) { nesterov ? launch(BF{}, tt{}) : launch(BF{}, ff{}); }
    else { nesterov ? launch(F16{}, tt{}) : launch(F16{}, ff{}); }
}
