// sodkit._C — python bindings for the CDNA4 kernel suite.
#include <torch/extension.h>

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad, int64_t dil, int64_t ld);
torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w_t, int64_t H,
                           int64_t W, int64_t stride, int64_t pad, int64_t dil,
                           int64_t lddy);
torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor dy, int64_t R,
                           int64_t S, int64_t stride, int64_t pad, int64_t dil,
                           int64_t ldx, int64_t lddy);
torch::Tensor channel_sum(torch::Tensor x, int64_t ld);
// fwd + fused BN stat partials (csrc/conv_igemm.hip ES epilogue)
std::vector<torch::Tensor> conv2d_fwd_stats(torch::Tensor x, torch::Tensor w,
                                            torch::Tensor bias,
                                            int64_t stride, int64_t pad,
                                            int64_t dil, int64_t ld);
std::vector<torch::Tensor> bn_finalize_slabs(torch::Tensor partial, double n,
                                             double momentum, double eps,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var);
torch::Tensor bn_slab_sum(torch::Tensor partial);
// 8-phase opt-in conv backend (SODKIT_IGEMM_8P=1; docs/IGEMM8_PLAN.md)
bool conv8_eligible(int64_t cout, int64_t cin);
torch::Tensor conv2d_fwd_8p(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t dil, int64_t ld);
std::vector<torch::Tensor> conv2d_fwd_8p_stats(torch::Tensor x,
                                               torch::Tensor w,
                                               torch::Tensor bias,
                                               int64_t stride, int64_t pad,
                                               int64_t dil, int64_t ld);
torch::Tensor conv2d_dgrad_8p(torch::Tensor dy, torch::Tensor w_t, int64_t H,
                              int64_t W, int64_t stride, int64_t pad,
                              int64_t dil, int64_t lddy);
torch::Tensor im2col_nhwc(torch::Tensor x, int64_t R, int64_t S,
                          int64_t stride, int64_t pad, int64_t dil,
                          int64_t ldx);
// direct 3x3 s1 d1 p1 64-channel fwd/dgrad (csrc/conv_dir.hip)
bool conv_dir_eligible(int64_t R, int64_t S, int64_t stride, int64_t pad,
                       int64_t dil, int64_t Cin, int64_t Cout, int64_t W);
torch::Tensor conv2d_fwd_dir(torch::Tensor x, torch::Tensor w,
                             torch::Tensor bias, int64_t ld);
std::vector<torch::Tensor> conv2d_fwd_dir_stats(torch::Tensor x,
                                                torch::Tensor w,
                                                torch::Tensor bias,
                                                int64_t ld);
torch::Tensor conv2d_dgrad_dir(torch::Tensor dy, torch::Tensor w_t,
                               int64_t lddy);
// VALU 1x1 wgrad (csrc/wgrad_k1.hip)
bool wgrad_k1_eligible(int64_t R, int64_t S, int64_t dil, int64_t pad,
                       int64_t C, int64_t Cout);
torch::Tensor conv2d_wgrad_k1(torch::Tensor x, torch::Tensor dy,
                              int64_t stride, int64_t ldx, int64_t lddy);
// direct 3x3 s1 d1 p1 wgrad (csrc/wgrad_dir.hip)
bool wgrad_dir_eligible(int64_t R, int64_t S, int64_t stride, int64_t pad,
                        int64_t dil, int64_t C, int64_t W);
torch::Tensor conv2d_wgrad_dir(torch::Tensor x, torch::Tensor dy,
                               int64_t ldx, int64_t lddy);
// batched grad->bucket gather for deferred-grad mode (csrc/multicopy.hip)
int64_t multicopy_chunk_bytes();
void multicopy(torch::Tensor src, torch::Tensor dst, torch::Tensor nbytes,
               torch::Tensor segmap, torch::Tensor chunkmap);
void multipermute(torch::Tensor src, torch::Tensor dst, torch::Tensor K,
                  torch::Tensor C, torch::Tensor RS, torch::Tensor seg,
                  torch::Tensor rsb, torch::Tensor ktb, torch::Tensor ctb);
// NHWC 3x3 s2 p1 max pool with packed argmax (csrc/maxpool.hip)
std::vector<torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x);
torch::Tensor maxpool3x3s2_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t H, int64_t W);

torch::Tensor bn_stats(torch::Tensor x);
bool bn_fwd_fused_eligible(int64_t C, int64_t elem_size);
std::vector<torch::Tensor> bn_fwd_fused(torch::Tensor x, torch::Tensor gamma,
                                        torch::Tensor beta, double n,
                                        double momentum, double eps,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var, bool relu,
                                        torch::Tensor residual,
                                        bool want_mask);
std::vector<torch::Tensor> bn_finalize(torch::Tensor stats, double n,
                                       double momentum, double eps,
                                       torch::Tensor running_mean,
                                       torch::Tensor running_var, bool training);
std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor x, double n,
                                             double momentum, double eps,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var);
std::vector<torch::Tensor> bn_apply(torch::Tensor x, torch::Tensor mean,
                                    torch::Tensor invstd, torch::Tensor gamma,
                                    torch::Tensor beta, bool relu,
                                    torch::Tensor residual, bool want_mask);
torch::Tensor bn_bwd_reduce(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                            torch::Tensor mean, torch::Tensor invstd, bool relu,
                            torch::Tensor mask);
std::vector<torch::Tensor> bn_bwd_apply(torch::Tensor dy, torch::Tensor y,
                                        torch::Tensor x, torch::Tensor mean,
                                        torch::Tensor invstd, torch::Tensor gamma,
                                        torch::Tensor sums, double n_total,
                                        bool relu, bool need_dres,
                                        torch::Tensor mask);

torch::Tensor bilinear_fwd(torch::Tensor x, int64_t Ho, int64_t Wo,
                           torch::Tensor add);
torch::Tensor bilinear_bwd(torch::Tensor dy, int64_t Hi, int64_t Wi);

std::vector<torch::Tensor> loss_fwd(torch::Tensor z, torch::Tensor t,
                                    bool need_ssim);
torch::Tensor loss_bwd(torch::Tensor p, torch::Tensor t, torch::Tensor img_sums,
                       torch::Tensor stats, double wb, double wi, double ws,
                       torch::Tensor go);

torch::Tensor dynfilter_fwd(torch::Tensor x, torch::Tensor k, int64_t dil);
std::vector<torch::Tensor> dynfilter_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor k, int64_t dil);

void sgd_fused(torch::Tensor ptrs, torch::Tensor numels, torch::Tensor wds,
               torch::Tensor chunks, int64_t dtype_code, double lr, double mu,
               double gscale, bool nesterov, torch::Tensor scale,
               torch::Tensor lr_t);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "sodkit CDNA4 (gfx950/MI355X) kernels";
    m.def("conv2d_fwd", &conv2d_fwd);
    m.def("conv2d_dgrad", &conv2d_dgrad);
    m.def("conv2d_wgrad", &conv2d_wgrad);
    m.def("channel_sum", &channel_sum);
    m.def("conv8_eligible", &conv8_eligible);
    m.def("conv2d_fwd_8p", &conv2d_fwd_8p);
    m.def("conv2d_dgrad_8p", &conv2d_dgrad_8p);
    m.def("im2col_nhwc", &im2col_nhwc);
    m.def("conv_dir_eligible", &conv_dir_eligible);
    m.def("conv2d_fwd_dir", &conv2d_fwd_dir);
    m.def("conv2d_dgrad_dir", &conv2d_dgrad_dir);
    m.def("wgrad_k1_eligible", &wgrad_k1_eligible);
    m.def("conv2d_wgrad_k1", &conv2d_wgrad_k1);
    m.def("wgrad_dir_eligible", &wgrad_dir_eligible);
    m.def("conv2d_wgrad_dir", &conv2d_wgrad_dir);
    m.def("bn_stats", &bn_stats);
    m.def("conv2d_fwd_stats", &conv2d_fwd_stats);
    m.def("conv2d_fwd_8p_stats", &conv2d_fwd_8p_stats);
    m.def("conv2d_fwd_dir_stats", &conv2d_fwd_dir_stats);
    m.def("bn_finalize_slabs", &bn_finalize_slabs);
    m.def("bn_slab_sum", &bn_slab_sum);
    m.def("bn_fwd_fused_eligible", &bn_fwd_fused_eligible);
    m.def("bn_fwd_fused", &bn_fwd_fused);
    m.def("bn_finalize", &bn_finalize);
    m.def("bn_stats_finalize", &bn_stats_finalize);
    m.def("bn_apply", &bn_apply);
    m.def("bn_bwd_reduce", &bn_bwd_reduce);
    m.def("bn_bwd_apply", &bn_bwd_apply);
    m.def("bilinear_fwd", &bilinear_fwd);
    m.def("bilinear_bwd", &bilinear_bwd);
    m.def("loss_fwd", &loss_fwd);
    m.def("loss_bwd", &loss_bwd);
    m.def("dynfilter_fwd", &dynfilter_fwd);
    m.def("dynfilter_bwd", &dynfilter_bwd);
    m.def("sgd_fused", &sgd_fused);
    m.def("multicopy_chunk_bytes", &multicopy_chunk_bytes);
    m.def("multicopy", &multicopy);
    m.def("multipermute", &multipermute);
    m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd);
    m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd);
    // build metadata (diagnostics: confirms the in-tree gfx950 build is
    // the one loaded, e.g. in the driver's native-code check)
    m.attr("BUILD_ARCH") = "gfx950";
    m.attr("BUILD_DATE") = __DATE__ " " __TIME__;
}
