// torch library registration for the gfx950 HIP kernels.
//
// Registered under torch.ops.bodywork_hip.* so the ops are first-class
// torch ops (dispatcher + hipGraph capture compatible).
#include <torch/extension.h>

// datagen.hip
std::tuple<at::Tensor, at::Tensor> datagen_hip(int64_t n, int64_t seed,
                                               int64_t stream_offset,
                                               double alpha, double beta,
                                               double sigma);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> random_split_hip(
    const at::Tensor& X, const at::Tensor& y, double test_frac, int64_t seed);
// linreg.hip
at::Tensor linreg_stats_hip(const at::Tensor& x, const at::Tensor& y);
at::Tensor linear_score_hip(const at::Tensor& x, const at::Tensor& ab);
at::Tensor poly_stats_hip(const at::Tensor& x, const at::Tensor& y,
                          int64_t nf, double mu, double s);
at::Tensor poly_score_hip(const at::Tensor& x, const at::Tensor& coef,
                          double mu, double s);
at::Tensor regression_metrics_hip(const at::Tensor& y, const at::Tensor& yhat);
at::Tensor score_label_metrics_hip(const at::Tensor& s, const at::Tensor& l);
// mlp_small.hip
std::tuple<at::Tensor, at::Tensor> expand1d_bf16_hip(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& b, bool relu,
    const c10::optional<at::Tensor>& mask, bool emit_mask);
at::Tensor rowdot_bf16_hip(const at::Tensor& h, const at::Tensor& w,
                           const at::Tensor& bias);
at::Tensor coldot_bf16_hip(const at::Tensor& m, const at::Tensor& v,
                           bool also_colsum);
at::Tensor colsum_bf16_hip(const at::Tensor& m);
// gemm.hip
at::Tensor linear_bf16_hip(const at::Tensor& x, const at::Tensor& w,
                           const c10::optional<at::Tensor>& bias, bool relu,
                           const c10::optional<at::Tensor>& mask,
                           bool out_fp32);
std::tuple<at::Tensor, at::Tensor> linear_relu_mask_bf16_hip(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& bias);
at::Tensor gemm_tn_bf16_hip(const at::Tensor& a, const at::Tensor& b,
                            bool out_fp32);
// gemm8.hip
at::Tensor gemm8_relu_dot_bf16_hip(const at::Tensor& x, const at::Tensor& w,
                                   const at::Tensor& b2,
                                   const at::Tensor& w3);
// gemm_mx8.hip — MX-fp8 (e4m3) K=128 scaled-MFMA path
at::Tensor quantize_e4m3_hip(const at::Tensor& x, int64_t e);
at::Tensor expand1d_e4m3_hip(const at::Tensor& x, const at::Tensor& w,
                             const c10::optional<at::Tensor>& b, int64_t e);
at::Tensor gemm_mx8_nt_hip(const at::Tensor& a8, int64_t ea,
                           const at::Tensor& b8, int64_t eb,
                           const c10::optional<at::Tensor>& bias, bool relu,
                           bool out_fp32);
at::Tensor gemm_mx8_relu_dot_hip(const at::Tensor& a8, int64_t ea,
                                 const at::Tensor& b8, int64_t eb,
                                 const at::Tensor& b2, const at::Tensor& w3);
// optim.hip
void adam_step_hip(at::Tensor p, const at::Tensor& g, at::Tensor m,
                   at::Tensor v, const c10::optional<at::Tensor>& p_bf16,
                   double lr, double beta1, double beta2, double eps,
                   int64_t t, const c10::optional<at::Tensor>& bc);
at::Tensor batch_indices_hip(at::Tensor ctr, const at::Tensor& n_dev,
                             int64_t bs, int64_t seed);
at::Tensor transpose_to_bf16_hip(const at::Tensor& src);
at::Tensor transpose_bf16_hip(const at::Tensor& src);

TORCH_LIBRARY(bodywork_hip, m) {
  m.def("datagen(int n, int seed, int stream_offset, float alpha, float beta, "
        "float sigma) -> (Tensor, Tensor)");
  m.def("random_split(Tensor X, Tensor y, float test_frac, int seed) -> "
        "(Tensor, Tensor, Tensor, Tensor)");
  m.def("linreg_stats(Tensor x, Tensor y) -> Tensor");
  m.def("linear_score(Tensor x, Tensor ab) -> Tensor");
  m.def("poly_stats(Tensor x, Tensor y, int nf, float mu, float s) -> Tensor");
  m.def("poly_score(Tensor x, Tensor coef, float mu, float s) -> Tensor");
  m.def("regression_metrics(Tensor y, Tensor yhat) -> Tensor");
  m.def("score_label_metrics(Tensor s, Tensor l) -> Tensor");
  m.def("expand1d_bf16(Tensor x, Tensor w, Tensor? b, bool relu, "
        "Tensor? mask, bool emit_mask) -> (Tensor, Tensor)");
  m.def("rowdot_bf16(Tensor h, Tensor w, Tensor bias) -> Tensor");
  m.def("coldot_bf16(Tensor m, Tensor v, bool also_colsum) -> Tensor");
  m.def("colsum_bf16(Tensor m) -> Tensor");
  m.def("linear_bf16(Tensor x, Tensor w, Tensor? bias, bool relu, "
        "Tensor? mask, bool out_fp32) -> Tensor");
  m.def("linear_relu_mask_bf16(Tensor x, Tensor w, Tensor? bias) -> "
        "(Tensor, Tensor)");
  m.def("gemm_tn_bf16(Tensor a, Tensor b, bool out_fp32) -> Tensor");
  m.def("quantize_e4m3(Tensor x, int e) -> Tensor");
  m.def("expand1d_e4m3(Tensor x, Tensor w, Tensor? b, int e) -> Tensor");
  m.def("gemm_mx8_nt(Tensor a8, int ea, Tensor b8, int eb, Tensor? bias, "
        "bool relu, bool out_fp32) -> Tensor");
  m.def("gemm_mx8_relu_dot(Tensor a8, int ea, Tensor b8, int eb, "
        "Tensor b2, Tensor w3) -> Tensor");
  m.def("gemm8_relu_dot_bf16(Tensor x, Tensor w, Tensor b2, Tensor w3) "
        "-> Tensor");
  m.def("adam_step(Tensor(a!) p, Tensor g, Tensor(b!) m, Tensor(c!) v, "
        "Tensor(d!)? p_bf16, float lr, float beta1, float beta2, float eps, "
        "int t, Tensor? bc) -> ()");
  m.def("batch_indices(Tensor(a!) ctr, Tensor n_dev, int bs, int seed) "
        "-> Tensor");
  m.def("transpose_to_bf16(Tensor src) -> Tensor");
  m.def("transpose_bf16(Tensor src) -> Tensor");
}

TORCH_LIBRARY_IMPL(bodywork_hip, CUDA, m) {
  m.impl("datagen", datagen_hip);
  m.impl("random_split", random_split_hip);
  m.impl("linreg_stats", linreg_stats_hip);
  m.impl("linear_score", linear_score_hip);
  m.impl("poly_stats", poly_stats_hip);
  m.impl("poly_score", poly_score_hip);
  m.impl("regression_metrics", regression_metrics_hip);
  m.impl("score_label_metrics", score_label_metrics_hip);
  m.impl("expand1d_bf16", expand1d_bf16_hip);
  m.impl("rowdot_bf16", rowdot_bf16_hip);
  m.impl("coldot_bf16", coldot_bf16_hip);
  m.impl("colsum_bf16", colsum_bf16_hip);
  m.impl("linear_bf16", linear_bf16_hip);
  m.impl("linear_relu_mask_bf16", linear_relu_mask_bf16_hip);
  m.impl("gemm_tn_bf16", gemm_tn_bf16_hip);
  m.impl("quantize_e4m3", quantize_e4m3_hip);
  m.impl("expand1d_e4m3", expand1d_e4m3_hip);
  m.impl("gemm_mx8_nt", gemm_mx8_nt_hip);
  m.impl("gemm_mx8_relu_dot", gemm_mx8_relu_dot_hip);
  m.impl("gemm8_relu_dot_bf16", gemm8_relu_dot_bf16_hip);
  m.impl("adam_step", adam_step_hip);
  m.impl("batch_indices", batch_indices_hip);
  m.impl("transpose_to_bf16", transpose_to_bf16_hip);
  m.impl("transpose_bf16", transpose_bf16_hip);
}

// datagen takes no tensor argument, so the dispatcher cannot route it by
// device -- register a CompositeExplicitAutograd fallback that forwards
// to the CUDA implementation (it allocates on the current CUDA device).
TORCH_LIBRARY_IMPL(bodywork_hip, CompositeExplicitAutograd, m) {
  m.impl("datagen", datagen_hip);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "bodywork_mlops_demo_amd gfx950 HIP kernels";
}
