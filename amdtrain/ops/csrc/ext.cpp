// Python bindings for the amdtrain gfx950 HIP kernels (amdtrain._C).
#include <torch/extension.h>

#include <optional>
#include <vector>

// sgd.hip
void multi_tensor_sgd(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> momenta, double lr,
                      double momentum, double weight_decay, bool nesterov,
                      bool first, bool zero_grad);

// cross_entropy.hip
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits, at::Tensor target);
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor target,
                             at::Tensor lse, at::Tensor upstream);

// elementwise.hip
at::Tensor normalize_u8(at::Tensor x, std::vector<double> mean,
                        std::vector<double> std_, long dtype_code);
at::Tensor topk_ranks(at::Tensor logits, at::Tensor target);
at::Tensor scatter_rows_x2(at::Tensor src2d, long Nn, long H, long W,
                           long Hs, long Ws);
void multi_tensor_scale_check(std::vector<at::Tensor> tensors, double scale,
                              at::Tensor found_inf);
void multi_tensor_cast(std::vector<at::Tensor> src,
                       std::vector<at::Tensor> dst);

// batchnorm.hip
std::vector<at::Tensor> batch_norm_fwd_train(
    at::Tensor x, at::Tensor weight, at::Tensor bias, at::Tensor running_mean,
    at::Tensor running_var, double momentum, double eps, bool relu,
    std::optional<at::Tensor> addend);
std::vector<at::Tensor> batch_norm_fwd_train_from_parts(
    at::Tensor x, at::Tensor parts, at::Tensor weight, at::Tensor bias,
    at::Tensor running_mean, at::Tensor running_var, double momentum,
    double eps, bool relu, std::optional<at::Tensor> addend);
at::Tensor batch_norm_fwd_eval(at::Tensor x, at::Tensor weight,
                               at::Tensor bias, at::Tensor running_mean,
                               at::Tensor running_var, double eps, bool relu,
                               std::optional<at::Tensor> addend);
std::vector<at::Tensor> batch_norm_bwd(at::Tensor x, at::Tensor grad_out,
                                       at::Tensor y, at::Tensor weight,
                                       at::Tensor mean, at::Tensor invstd,
                                       bool relu, bool need_ghat,
                                       std::optional<at::Tensor> scale,
                                       std::optional<at::Tensor> shift,
                                       std::optional<at::Tensor> grad_out2,
                                       std::optional<at::Tensor> relu_mask);

// gemm.hip
at::Tensor gemm_bt(at::Tensor A, at::Tensor B, bool f32_out,
                   std::optional<at::Tensor> addend,
                   std::optional<at::Tensor> bias);
at::Tensor gemm_tn(at::Tensor dY, at::Tensor X, long msplit);
at::Tensor gemm_bt_strided(at::Tensor A, at::Tensor B, long Nn, long H,
                           long W, long stride);
at::Tensor gemm_tn_strided(at::Tensor dY, at::Tensor X, long Nn, long H,
                           long W, long stride);
at::Tensor transpose_2d(at::Tensor x);

// conv3x3.hip
at::Tensor conv3x3_fwd(at::Tensor x2d, long Nn, long H, long W, long stride,
                       at::Tensor w2d);
std::vector<at::Tensor> conv3x3_fwd_stats(at::Tensor x2d, long Nn, long H,
                                          long W, long stride,
                                          at::Tensor w2d, bool banded);
std::vector<at::Tensor> gemm_bt_stats(at::Tensor A, at::Tensor B);

// gemm8p.hip
at::Tensor gemm_bt_8p(at::Tensor A, at::Tensor B);
at::Tensor gemm_bt_8p3(at::Tensor A, at::Tensor B);
at::Tensor conv3x3_8p(at::Tensor A2d, long Nn, long H, long W, long stride,
                      at::Tensor w2d, bool dgrad);
at::Tensor conv3x3_dgrad(at::Tensor dy2d, long Nn, long H, long W,
                         long stride, at::Tensor w2d, bool banded);
at::Tensor conv3x3_wgrad(at::Tensor dy2d, at::Tensor x2d, long Nn, long H,
                         long W, long stride);

// wgrad.hip
at::Tensor tn2_wgrad(at::Tensor dY, at::Tensor X, long taps, long Nn, long H,
                     long W, long stride, long gmode, long kh, long kw,
                     long pad);
at::Tensor tn2_wgrad_banded(at::Tensor dY, at::Tensor X, long Nn, long H,
                            long W, long stride);
at::Tensor tr16_probe(at::Tensor in);

// conv_stem.hip
at::Tensor conv_generic_fwd(at::Tensor x2d, long Nn, long H, long W, long KH,
                            long KW, long stride, long pad, at::Tensor w2);
at::Tensor conv_generic_wgrad(at::Tensor dy2d, at::Tensor x2d, long Nn,
                              long H, long W, long KH, long KW, long stride,
                              long pad);
at::Tensor conv_generic_dgrad(at::Tensor dy2d, at::Tensor w2p, long Nn,
                              long H, long W, long KH, long KW, long stride,
                              long pad);

// pool.hip
std::vector<at::Tensor> max_pool_3x3_s2_fwd(at::Tensor x);
at::Tensor max_pool_3x3_s2_bwd(at::Tensor grad_y, at::Tensor idx, long H,
                               long W);
at::Tensor global_avg_pool_fwd(at::Tensor x);
at::Tensor global_avg_pool_bwd(at::Tensor grad_y, long H, long W);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "amdtrain hand-written gfx950 (CDNA4) HIP kernels";
  m.def("multi_tensor_sgd", &multi_tensor_sgd, "fused multi-tensor SGD");
  m.def("cross_entropy_fwd", &cross_entropy_fwd);
  m.def("cross_entropy_bwd", &cross_entropy_bwd);
  m.def("normalize_u8", &normalize_u8);
  m.def("topk_ranks", &topk_ranks);
  m.def("scatter_rows_x2", &scatter_rows_x2);
  m.def("multi_tensor_scale_check", &multi_tensor_scale_check);
  m.def("multi_tensor_cast", &multi_tensor_cast);
  m.def("batch_norm_fwd_train", &batch_norm_fwd_train,
        py::arg("x"), py::arg("weight"), py::arg("bias"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("relu"),
        py::arg("addend") = std::nullopt);
  m.def("batch_norm_fwd_train_from_parts", &batch_norm_fwd_train_from_parts,
        py::arg("x"), py::arg("parts"), py::arg("weight"), py::arg("bias"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("relu"), py::arg("addend") = std::nullopt);
  m.def("batch_norm_fwd_eval", &batch_norm_fwd_eval,
        py::arg("x"), py::arg("weight"), py::arg("bias"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("eps"),
        py::arg("relu"), py::arg("addend") = std::nullopt);
  m.def("batch_norm_bwd", &batch_norm_bwd, py::arg("x"), py::arg("grad_out"),
        py::arg("y"), py::arg("weight"), py::arg("mean"), py::arg("invstd"),
        py::arg("relu"), py::arg("need_ghat"),
        py::arg("scale") = std::nullopt, py::arg("shift") = std::nullopt,
        py::arg("grad_out2") = std::nullopt,
        py::arg("relu_mask") = std::nullopt);
  m.def("gemm_bt", &gemm_bt, py::arg("A"), py::arg("B"),
        py::arg("f32_out") = false, py::arg("addend") = std::nullopt,
        py::arg("bias") = std::nullopt);
  m.def("gemm_tn", &gemm_tn, py::arg("dY"), py::arg("X"),
        py::arg("msplit") = 0);
  m.def("gemm_bt_strided", &gemm_bt_strided);
  m.def("gemm_tn_strided", &gemm_tn_strided);
  m.def("transpose_2d", &transpose_2d);
  m.def("conv3x3_fwd", &conv3x3_fwd);
  m.def("conv3x3_fwd_stats", &conv3x3_fwd_stats, py::arg("x2d"),
        py::arg("Nn"), py::arg("H"), py::arg("W"), py::arg("stride"),
        py::arg("w2d"), py::arg("banded") = false);
  m.def("gemm_bt_stats", &gemm_bt_stats);
  m.def("gemm_bt_8p", &gemm_bt_8p);
  m.def("gemm_bt_8p3", &gemm_bt_8p3);
  m.def("conv3x3_8p", &conv3x3_8p);
  m.def("conv3x3_dgrad", &conv3x3_dgrad, py::arg("dy2d"), py::arg("Nn"),
        py::arg("H"), py::arg("W"), py::arg("stride"), py::arg("w2d"),
        py::arg("banded") = false);
  m.def("conv3x3_wgrad", &conv3x3_wgrad);
  m.def("tn2_wgrad", &tn2_wgrad, py::arg("dY"), py::arg("X"),
        py::arg("taps") = 1, py::arg("Nn") = 0, py::arg("H") = 0,
        py::arg("W") = 0, py::arg("stride") = 1, py::arg("gmode") = 0,
        py::arg("kh") = 3, py::arg("kw") = 3, py::arg("pad") = 1);
  m.def("tn2_wgrad_banded", &tn2_wgrad_banded);
  m.def("tr16_probe", &tr16_probe);
  m.def("conv_generic_fwd", &conv_generic_fwd);
  m.def("conv_generic_wgrad", &conv_generic_wgrad);
  m.def("conv_generic_dgrad", &conv_generic_dgrad);
  m.def("max_pool_3x3_s2_fwd", &max_pool_3x3_s2_fwd);
  m.def("max_pool_3x3_s2_bwd", &max_pool_3x3_s2_bwd);
  m.def("global_avg_pool_fwd", &global_avg_pool_fwd);
  m.def("global_avg_pool_bwd", &global_avg_pool_bwd);
}
