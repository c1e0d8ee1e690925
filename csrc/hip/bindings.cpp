// pybind11 bindings for the bflc_amd gfx950 HIP kernels.
#include <torch/extension.h>

#include <tuple>
#include <vector>

namespace bflc {

// elementwise.hip
void axpy_(torch::Tensor y, torch::Tensor x, double alpha);
void sgd_step_(torch::Tensor p, torch::Tensor g, double lr);
void adam_step_(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, long step, double lr, double beta1,
                double beta2, double eps);
torch::Tensor weighted_fedavg(torch::Tensor deltas, torch::Tensor w);
void sgd_master_(torch::Tensor p, torch::Tensor shadow, torch::Tensor g,
                 double lr);
void adam_master_(torch::Tensor p, torch::Tensor shadow, torch::Tensor g,
                  torch::Tensor m, torch::Tensor v, long step, double lr,
                  double beta1, double beta2, double eps);
void adam_master_graph_(torch::Tensor p, torch::Tensor shadow,
                        torch::Tensor g, torch::Tensor m, torch::Tensor v,
                        torch::Tensor step, torch::Tensor bc, double lr,
                        double beta1, double beta2, double eps);
void refresh_shadow_(torch::Tensor p, torch::Tensor shadow);
void score_load_(torch::Tensor shadow, torch::Tensor global_flat,
                 torch::Tensor delta, double lr);
void delta_extract_(torch::Tensor out, torch::Tensor global_flat,
                    torch::Tensor w, double lr);
torch::Tensor relu_fwd(torch::Tensor x);
torch::Tensor relu_bwd(torch::Tensor y, torch::Tensor dy);
std::tuple<torch::Tensor, torch::Tensor> relu_bwd_colsum(
    const torch::Tensor& y, const torch::Tensor& dy);

// softmax_ce.hip
std::tuple<torch::Tensor, torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                                        torch::Tensor target);
torch::Tensor softmax_ce_bwd(torch::Tensor probs, torch::Tensor target,
                             torch::Tensor gloss);

// reduce.hip
double accuracy(torch::Tensor logits, torch::Tensor target);
torch::Tensor accuracy_t(torch::Tensor logits, torch::Tensor target);

// gemm_bf16.hip
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         bool relu);
torch::Tensor gemm_raw(torch::Tensor A, torch::Tensor B, bool ta, bool tb);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> linear_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, bool want_dx);

// conv_im2col.hip
std::tuple<torch::Tensor, torch::Tensor> conv2d_fwd_col(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, long stride, long pad,
    bool relu, bool want_col);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
conv2d_fwd_bn(torch::Tensor x, torch::Tensor w, long stride, long pad);
std::tuple<torch::Tensor, torch::Tensor> bn_stats(torch::Tensor x,
                                                  double eps);
std::tuple<torch::Tensor, torch::Tensor> bn_stats_finalize(
    torch::Tensor psum, torch::Tensor psq, double count, double eps);
torch::Tensor batchnorm_norm(torch::Tensor x, torch::Tensor gamma,
                             torch::Tensor beta, torch::Tensor mean,
                             torch::Tensor invstd, bool relu,
                             c10::optional<torch::Tensor> residual);
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         long stride, long pad, bool relu);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> conv2d_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, long stride,
    long pad, c10::optional<torch::Tensor> col_cache, bool want_db,
    bool want_dx);
std::tuple<torch::Tensor, torch::Tensor> maxpool2d_fwd(torch::Tensor x,
                                                       long kernel,
                                                       long stride,
                                                       bool want_idx);
torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx,
                            std::vector<long> in_shape, long kernel,
                            long stride);

// batchnorm.hip
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta, double eps,
    bool relu, c10::optional<torch::Tensor> residual);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_bwd(
    torch::Tensor x, torch::Tensor dy, torch::Tensor mean,
    torch::Tensor invstd, torch::Tensor gamma,
    c10::optional<torch::Tensor> y_relu);
torch::Tensor global_avgpool_fwd(torch::Tensor x);
torch::Tensor global_avgpool_bwd(torch::Tensor dy, long H, long W);
torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b);
torch::Tensor add_relu_bwd(torch::Tensor y, torch::Tensor dy);

}  // namespace bflc

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "bflc_amd hand-written gfx950 (CDNA4) HIP kernels";
  m.def("axpy_", &bflc::axpy_, "y += alpha*x (fp32, in place)");
  m.def("sgd_step_", &bflc::sgd_step_, "fused flat SGD");
  m.def("adam_step_", &bflc::adam_step_, "fused flat Adam");
  m.def("weighted_fedavg", &bflc::weighted_fedavg,
        "fixed-order weighted FedAvg reduce");
  m.def("sgd_master_", &bflc::sgd_master_,
        "fused fp32-master SGD + bf16 shadow refresh");
  m.def("adam_master_", &bflc::adam_master_);
  m.def("adam_master_graph_", &bflc::adam_master_graph_,
        "hipGraph-capturable Adam: device step counter + on-device bias "
        "corrections (tick kernel), fp32 master + bf16 shadow");
  m.def("refresh_shadow_", &bflc::refresh_shadow_, "shadow = bf16(master)");
  m.def("score_load_", &bflc::score_load_,
        "shadow = compute_dtype(global - lr*delta): one-pass scoring "
        "candidate load (replaces copy+axpy+set_flat)");
  m.def("delta_extract_", &bflc::delta_extract_,
        "out = (global - w)/lr: one-pass pseudo-gradient extraction");
  m.def("relu_fwd", &bflc::relu_fwd);
  m.def("relu_bwd", &bflc::relu_bwd);
  m.def("relu_bwd_colsum", &bflc::relu_bwd_colsum,
        "fused relu backward + bias-grad column sum: (dx, db)");
  m.def("softmax_ce_fwd", &bflc::softmax_ce_fwd,
        "fused softmax cross-entropy fwd (loss, probs)");
  m.def("softmax_ce_bwd", &bflc::softmax_ce_bwd);
  m.def("accuracy", &bflc::accuracy, "fused argmax-compare-reduce");
  m.def("accuracy_t", &bflc::accuracy_t,
        "accuracy as a device tensor (no host sync)");
  m.def("linear_fwd", &bflc::linear_fwd, "MFMA bf16 GEMM + bias (+relu)",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("relu") = false);
  m.def("gemm_raw", &bflc::gemm_raw, "raw GEMM (bench/ablation)");
  m.def("linear_bwd", &bflc::linear_bwd,
        "(dx, dw, db) - dx empty unless want_dx", py::arg("x"),
        py::arg("w"), py::arg("dy"), py::arg("want_dx") = true);
  m.def("conv2d_fwd", &bflc::conv2d_fwd,
        "NHWC implicit-GEMM convolution (MFMA); inference entry, no col",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("stride"),
        py::arg("pad"), py::arg("relu") = false);
  m.def("conv2d_fwd_col", &bflc::conv2d_fwd_col, "(y, col) - col for bwd",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("stride"),
        py::arg("pad"), py::arg("relu") = false,
        py::arg("want_col") = true);
  m.def("conv2d_fwd_bn", &bflc::conv2d_fwd_bn,
        "(y, col, psum, psq) - conv with fused epilogue BN stats");
  m.def("bn_stats", &bflc::bn_stats, "(mean, invstd) of x [.., C]");
  m.def("bn_stats_finalize", &bflc::bn_stats_finalize,
        "(mean, invstd) from [chunks][C] partials");
  m.def("batchnorm_norm", &bflc::batchnorm_norm,
        py::arg("x"), py::arg("gamma"), py::arg("beta"), py::arg("mean"),
        py::arg("invstd"), py::arg("relu"),
        py::arg("residual") = py::none(),
        "normalization pass with known stats");
  m.def("conv2d_bwd", &bflc::conv2d_bwd,
        "(dx, dw, db) - db empty unless want_db, dx empty unless "
        "want_dx (first-layer convs need no input gradient)",
        py::arg("x"), py::arg("w"), py::arg("dy"), py::arg("stride"),
        py::arg("pad"), py::arg("col_cache") = py::none(),
        py::arg("want_db") = true, py::arg("want_dx") = true);
  m.def("maxpool2d_fwd", &bflc::maxpool2d_fwd, py::arg("x"),
        py::arg("kernel"), py::arg("stride"),
        py::arg("want_idx") = true);
  m.def("maxpool2d_bwd", &bflc::maxpool2d_bwd);
  m.def("batchnorm_fwd", &bflc::batchnorm_fwd,
        py::arg("x"), py::arg("gamma"), py::arg("beta"), py::arg("eps"),
        py::arg("relu"), py::arg("residual") = py::none(),
        "(y, mean, invstd) — batch-stats BN, optional fused relu");
  m.def("batchnorm_bwd", &bflc::batchnorm_bwd, "(dx, dgamma, dbeta)",
        py::arg("x"), py::arg("dy"), py::arg("mean"), py::arg("invstd"),
        py::arg("gamma"), py::arg("y_relu") = py::none());
  m.def("global_avgpool_fwd", &bflc::global_avgpool_fwd);
  m.def("global_avgpool_bwd", &bflc::global_avgpool_bwd);
  m.def("add_relu_fwd", &bflc::add_relu_fwd, "fused residual add + relu");
  m.def("add_relu_bwd", &bflc::add_relu_bwd);
}
