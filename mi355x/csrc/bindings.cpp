// pybind bindings for the mi355x gfx950 kernel library.
#include <torch/extension.h>

#include <vector>

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y);
at::Tensor mfma_probe32(at::Tensor A, at::Tensor B);
void sgd_step(at::Tensor p, at::Tensor g, at::Tensor m, double lr, double mu,
              double wd, double gscale);
at::Tensor cast_to_16(at::Tensor src, at::Tensor like);
at::Tensor cast_permute_krsc(at::Tensor w, at::Tensor like);
at::Tensor cast_permute_krsc_pad(at::Tensor w, at::Tensor like);
at::Tensor cast_permute_rsck(at::Tensor w, at::Tensor like);

at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                      long stride, long pad, long act, long kR, long kS);
at::Tensor conv2d_dgrad(at::Tensor dy, at::Tensor wflip, long stride,
                        long pad, long H, long W, at::Tensor addin);
at::Tensor conv2d_wgrad(at::Tensor x, at::Tensor dy, long R, long S,
                        long stride, long pad);
std::vector<at::Tensor> conv2d_fwd_stats(at::Tensor x, at::Tensor w,
                                         long stride, long pad, long kR,
                                         long kS);
std::vector<at::Tensor> conv2d_fwd_scaled(at::Tensor x, at::Tensor w,
                                          at::Tensor asc, at::Tensor ash,
                                          long stride, long pad,
                                          bool want_stats);
at::Tensor conv2d_wgrad_scaled(at::Tensor x, at::Tensor asc, at::Tensor ash,
                               at::Tensor dy, long R, long S, long stride,
                               long pad);

at::Tensor bn_stats(at::Tensor x);
at::Tensor bn_finalize(at::Tensor stats, at::Tensor running_mean,
                       at::Tensor running_var, double m_total,
                       double momentum, double eps);
std::vector<at::Tensor> bn_apply(at::Tensor x, at::Tensor mean, at::Tensor invstd,
                    at::Tensor gamma, at::Tensor beta, at::Tensor res,
                    long act, bool want_mask);
at::Tensor bn_bwd_reduce(at::Tensor x, at::Tensor dy, at::Tensor y,
                         at::Tensor mean, at::Tensor invstd,
                         at::Tensor mask, at::Tensor asc, at::Tensor ash);
std::vector<at::Tensor> bn_bwd_dx(at::Tensor x, at::Tensor dy, at::Tensor y,
                                  at::Tensor mean, at::Tensor invstd,
                                  at::Tensor gamma, at::Tensor dgamma,
                                  at::Tensor dbeta, double m_total,
                                  bool want_dres, at::Tensor mask,
                                  at::Tensor asc, at::Tensor ash);

std::vector<at::Tensor> maxpool_fwd(at::Tensor x, long kernel, long stride,
                                    long pad);
at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                       long kernel, long stride, long pad);
at::Tensor global_avg_pool(at::Tensor x);

at::Tensor linear_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, long act);
at::Tensor linear_dgrad(at::Tensor dy, at::Tensor w, at::Tensor wt);
at::Tensor linear_wgrad(at::Tensor x, at::Tensor dy);
at::Tensor gemm_nt_mfma(at::Tensor A, at::Tensor B, at::Tensor bias, long act,
                        bool out32);

std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits,
                                          at::Tensor target);
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor target,
                             at::Tensor lse, at::Tensor dloss);

void register_rccl(py::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_rccl(m);
  m.def("relu_bwd", &relu_bwd);
  m.def("mfma_probe32", &mfma_probe32);
  m.def("sgd_step", &sgd_step);
  m.def("cast_to_16", &cast_to_16);
  m.def("cast_permute_krsc", &cast_permute_krsc);
  m.def("cast_permute_krsc_pad", &cast_permute_krsc_pad);
  m.def("cast_permute_rsck", &cast_permute_rsck);
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_wgrad", &conv2d_wgrad);
  m.def("conv2d_fwd_stats", &conv2d_fwd_stats);
  m.def("conv2d_fwd_scaled", &conv2d_fwd_scaled);
  m.def("conv2d_wgrad_scaled", &conv2d_wgrad_scaled);
  m.def("bn_stats", &bn_stats);
  m.def("bn_finalize", &bn_finalize);
  m.def("bn_apply", &bn_apply);
  m.def("bn_bwd_reduce", &bn_bwd_reduce);
  m.def("bn_bwd_dx", &bn_bwd_dx);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("global_avg_pool", &global_avg_pool);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_dgrad", &linear_dgrad);
  m.def("linear_wgrad", &linear_wgrad);
  m.def("gemm_nt_mfma", &gemm_nt_mfma);
  m.def("cross_entropy_fwd", &cross_entropy_fwd);
  m.def("cross_entropy_bwd", &cross_entropy_bwd);
}
