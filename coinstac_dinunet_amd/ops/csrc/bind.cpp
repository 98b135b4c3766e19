// pybind bindings for the gfx950 kernel library (_hip_ops).
#include <torch/extension.h>

#include <vector>

// adam.hip
void fused_adam_flat(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                     torch::Tensor v, double lr, double beta1, double beta2,
                     double eps, double weight_decay, int64_t step);
void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                int64_t step);
void fused_sgd(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> bufs, double lr, double momentum,
               double weight_decay);
// pack.hip
void pack_tensors(std::vector<torch::Tensor> tensors, torch::Tensor flat);
void unpack_tensors(torch::Tensor flat, std::vector<torch::Tensor> tensors);
// lsnll.hip
std::vector<torch::Tensor> logsoftmax_nll_fwd(torch::Tensor logits,
                                              torch::Tensor target);
torch::Tensor logsoftmax_nll_bwd(torch::Tensor logprobs, torch::Tensor target,
                                 torch::Tensor grad_out,
                                 torch::ScalarType out_dtype);
torch::Tensor argmax_rows(torch::Tensor x);
// metrics.hip
torch::Tensor prf1a_counts(torch::Tensor pred, torch::Tensor true_);
torch::Tensor confusion_matrix(torch::Tensor pred, torch::Tensor true_,
                               int64_t num_classes);
// linear.hip
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor weight,
                         torch::Tensor bias, bool relu);
torch::Tensor linear_dgrad(torch::Tensor go, torch::Tensor weight);
torch::Tensor linear_wgrad(torch::Tensor go, torch::Tensor x);
torch::Tensor colsum(torch::Tensor g);
void gram_schmidt(torch::Tensor m, double eps);
// conv3d.hip
torch::Tensor mfma_probe_gemm(torch::Tensor A, torch::Tensor B);
torch::Tensor conv3d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         c10::optional<torch::Tensor> bn_ab);
torch::Tensor conv3d_dgrad(torch::Tensor go, torch::Tensor w,
                           std::vector<int64_t> in_shape, int64_t stride);
torch::Tensor conv3d_wgrad(torch::Tensor x, torch::Tensor go, int64_t stride,
                           int64_t variant, c10::optional<torch::Tensor> bn_ab);
torch::Tensor channel_sum(torch::Tensor go);
// conv3d_spatial.hip
torch::Tensor conv3d_fwd_spatial(torch::Tensor x, torch::Tensor w,
                                 int64_t stride, int64_t ctile_opt,
                                 c10::optional<torch::Tensor> bn_ab);
std::vector<torch::Tensor> conv3d_fwd_spatial_stats(
    torch::Tensor x, torch::Tensor w, int64_t stride,
    c10::optional<torch::Tensor> bn_ab);
torch::Tensor conv3d_dgrad_spatial(torch::Tensor go, torch::Tensor w,
                                   std::vector<int64_t> in_shape);
std::vector<torch::Tensor> conv3d_dgrad_spatial_bnbwd(
    torch::Tensor go, torch::Tensor w, std::vector<int64_t> in_shape,
    torch::Tensor xraw, torch::Tensor bn_prm);
torch::Tensor conv3d_dgrad_s2_spatial(torch::Tensor go, torch::Tensor w,
                                      std::vector<int64_t> in_shape);
// conv2d.hip
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         c10::optional<torch::Tensor> bn_ab);
torch::Tensor conv2d_dgrad(torch::Tensor go, torch::Tensor w,
                           std::vector<int64_t> in_shape, int64_t stride);
torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor go,
                           int64_t stride, c10::optional<torch::Tensor> bn_ab,
                           int64_t ks);
// rankdad.hip
std::vector<torch::Tensor> power_iter_bc(torch::Tensor B, torch::Tensor C,
                                         int64_t rank, int64_t iters,
                                         double tol, torch::Tensor starts);
torch::Tensor rowsum(torch::Tensor m);
// pointwise.hip
torch::Tensor conv3d_pw_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias);
torch::Tensor conv3d_pw_dgrad(torch::Tensor go, torch::Tensor w);
torch::Tensor conv3d_pw_wgrad(torch::Tensor x, torch::Tensor go);
// bnorm.hip
std::vector<torch::Tensor> bn3d_fwd(torch::Tensor x, torch::Tensor gamma,
                                    torch::Tensor beta, double eps, bool relu);
std::vector<torch::Tensor> bn3d_stats(torch::Tensor x, double eps);
std::vector<torch::Tensor> bn3d_bwd_pre(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor mean_rstd,
                                        torch::Tensor gamma,
                                        torch::Tensor beta, bool relu,
                                        torch::Tensor sums2);
torch::Tensor bn3d_normalize(torch::Tensor x, torch::Tensor mean_rstd,
                             torch::Tensor gamma, torch::Tensor beta,
                             bool relu);
std::vector<torch::Tensor> bn3d_fwd_res(torch::Tensor x, torch::Tensor res,
                                        torch::Tensor gamma,
                                        torch::Tensor beta, double eps);
std::vector<torch::Tensor> bn3d_bwd_res(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor res,
                                        torch::Tensor mean_rstd,
                                        torch::Tensor gamma,
                                        torch::Tensor beta);
torch::Tensor bn3d_infer(torch::Tensor x, torch::Tensor gamma,
                         torch::Tensor beta, torch::Tensor running_mean,
                         torch::Tensor running_var, double eps, bool relu);
std::vector<torch::Tensor> bn3d_bwd(torch::Tensor dy, torch::Tensor x,
                                    torch::Tensor mean_rstd,
                                    torch::Tensor gamma, torch::Tensor beta,
                                    bool relu);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_adam_flat", &fused_adam_flat, "single-kernel Adam on flat arenas");
  m.def("fused_adam", &fused_adam, "per-tensor fused Adam");
  m.def("fused_sgd", &fused_sgd, "per-tensor fused SGD");
  m.def("pack_tensors", &pack_tensors, "pack tensor list into flat buffer");
  m.def("unpack_tensors", &unpack_tensors, "scatter flat buffer into tensors");
  m.def("logsoftmax_nll_fwd", &logsoftmax_nll_fwd);
  m.def("logsoftmax_nll_bwd", &logsoftmax_nll_bwd);
  m.def("argmax_rows", &argmax_rows);
  m.def("prf1a_counts", &prf1a_counts);
  m.def("confusion_matrix", &confusion_matrix);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_dgrad", &linear_dgrad);
  m.def("linear_wgrad", &linear_wgrad);
  m.def("colsum", &colsum);
  m.def("gram_schmidt", &gram_schmidt);
  m.def("mfma_probe_gemm", &mfma_probe_gemm);
  m.def("conv3d_fwd", &conv3d_fwd, py::arg("x"), py::arg("w"),
        py::arg("stride"), py::arg("bn_ab") = py::none());
  m.def("conv3d_dgrad", &conv3d_dgrad);
  m.def("conv3d_wgrad", &conv3d_wgrad, py::arg("x"), py::arg("go"),
        py::arg("stride"), py::arg("variant") = 0,
        py::arg("bn_ab") = py::none());
  m.def("channel_sum", &channel_sum);
  m.def("conv3d_fwd_spatial", &conv3d_fwd_spatial,
        py::arg("x"), py::arg("w"), py::arg("stride"),
        py::arg("ctile_opt") = 0, py::arg("bn_ab") = py::none());
  m.def("conv3d_fwd_spatial_stats", &conv3d_fwd_spatial_stats,
        py::arg("x"), py::arg("w"), py::arg("stride"),
        py::arg("bn_ab") = py::none());
  m.def("conv3d_dgrad_spatial", &conv3d_dgrad_spatial);
  m.def("conv3d_dgrad_spatial_bnbwd", &conv3d_dgrad_spatial_bnbwd);
  m.def("conv3d_dgrad_s2_spatial", &conv3d_dgrad_s2_spatial);
  m.def("conv2d_fwd", &conv2d_fwd, py::arg("x"), py::arg("w"),
        py::arg("stride"), py::arg("bn_ab") = py::none());
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_wgrad", &conv2d_wgrad, py::arg("x"), py::arg("go"),
        py::arg("stride"), py::arg("bn_ab") = py::none(),
        py::arg("ks") = 3);
  m.def("power_iter_bc", &power_iter_bc);
  m.def("rowsum", &rowsum);
  m.def("conv3d_pw_fwd", &conv3d_pw_fwd);
  m.def("conv3d_pw_dgrad", &conv3d_pw_dgrad);
  m.def("conv3d_pw_wgrad", &conv3d_pw_wgrad);
  m.def("bn3d_fwd", &bn3d_fwd);
  m.def("bn3d_stats", &bn3d_stats);
  m.def("bn3d_normalize", &bn3d_normalize);
  m.def("bn3d_bwd_pre", &bn3d_bwd_pre);
  m.def("bn3d_fwd_res", &bn3d_fwd_res);
  m.def("bn3d_bwd_res", &bn3d_bwd_res);
  m.def("bn3d_infer", &bn3d_infer);
  m.def("bn3d_bwd", &bn3d_bwd);
}
