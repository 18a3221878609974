// pybind11 bindings for the hypha_amd CDNA4 HIP extension (_C).

#include <torch/extension.h>

#include <vector>

// elementwise.hip
void adamw_step_(torch::Tensor master, torch::Tensor param, torch::Tensor grad,
                 torch::Tensor m, torch::Tensor v, double lr, double beta1, double beta2,
                 double eps, double wd, long step);
void nesterov_step_(torch::Tensor theta, torch::Tensor delta, torch::Tensor mom, double lr,
                    double mu);
void extract_delta(torch::Tensor master, torch::Tensor theta0, torch::Tensor out);
// adamw8.hip
void adamw8_step_(torch::Tensor master, torch::Tensor param, torch::Tensor grad,
                  torch::Tensor m8, torch::Tensor v8, torch::Tensor m_scale,
                  torch::Tensor v_scale, double lr, double beta1, double beta2,
                  double eps, double wd, long step);
torch::Tensor swiglu_fwd(torch::Tensor gate, torch::Tensor up);
torch::Tensor grad_norm_sq(torch::Tensor x);
// grouped_gemm.hip
torch::Tensor grouped_gemm(torch::Tensor x, torch::Tensor w, torch::Tensor group_offsets);
// lean_opt.hip
void adamw8_lean_(torch::Tensor param, torch::Tensor grad, torch::Tensor m8,
                  torch::Tensor v8, torch::Tensor m_scale, torch::Tensor v_scale,
                  double lr, double beta1, double beta2, double eps, double wd,
                  long step, long seed);
void extract_delta_bf16(torch::Tensor theta_t, torch::Tensor theta0, torch::Tensor out);
void nesterov_bf16_(torch::Tensor theta, torch::Tensor delta, torch::Tensor mom,
                    double lr, double mu, long seed);
std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dout, torch::Tensor gate,
                                      torch::Tensor up);
// rmsnorm.hip
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
torch::Tensor gelu_fwd(torch::Tensor x);
torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x);
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor a, torch::Tensor b,
                                           torch::Tensor w, double eps);
std::vector<torch::Tensor> add_rmsnorm_bwd(torch::Tensor dy, torch::Tensor dh,
                                           torch::Tensor h, torch::Tensor w,
                                           torch::Tensor rstd);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                       torch::Tensor rstd);
// rope.hip
std::vector<torch::Tensor> rope_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
                                    torch::Tensor sin_t, bool inverse);
std::vector<torch::Tensor> rope_fwd_ex(torch::Tensor q, torch::Tensor k,
                                       torch::Tensor cos_t, torch::Tensor sin_t,
                                       bool inverse, const std::string& layout);
// cross_entropy.hip
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets);
torch::Tensor ce_bwd_(torch::Tensor logits, torch::Tensor targets, torch::Tensor lse,
                      double scale);
// attention.hip
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    bool causal);
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor dout, torch::Tensor lse,
                                    bool causal);
std::vector<torch::Tensor> attn_fwd_ex(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       bool causal, const std::string& layout);
std::vector<torch::Tensor> attn_bwd_ex(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       torch::Tensor o, torch::Tensor dout,
                                       torch::Tensor lse, bool causal,
                                       const std::string& layout);
// probe.hip (MFMA fragment-layout verification)
torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b);
torch::Tensor mfma_probe_16x16x32(torch::Tensor a, torch::Tensor b);
// fp8_cast.hip
std::vector<torch::Tensor> fp8_cast_transpose(torch::Tensor x, torch::Tensor scale,
                                              torch::Tensor partials, long skip_t);
void fp8_scale_update_(torch::Tensor partials, torch::Tensor scale, double margin,
                       long n);
long long fp8_cast_grid_size(long long R, long long C);
// fp8_lean.hip (fp8 weight storage, BASELINE config 5)
void adamw8_fp8_lean_(torch::Tensor w8, torch::Tensor wscale, torch::Tensor grad,
                      torch::Tensor m8, torch::Tensor v8, torch::Tensor m_scale,
                      torch::Tensor v_scale, double lr, double beta1, double beta2,
                      double eps, double wd, long step, long seed);
void fp8_extract_delta(torch::Tensor w8, torch::Tensor wscale, torch::Tensor theta0,
                       torch::Tensor out);
void fp8_requant_(torch::Tensor theta, torch::Tensor w8, torch::Tensor wscale,
                  long seed);
std::vector<torch::Tensor> fp8_weight_cast_transpose(torch::Tensor w8s,
                                                     torch::Tensor wscale,
                                                     torch::Tensor scale,
                                                     torch::Tensor partials);
// decode.hip (flash-decoding KV-cache attention)
torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kcache, torch::Tensor vcache,
                          long t);
torch::Tensor attn_decode_graph(torch::Tensor q, torch::Tensor kcache,
                                torch::Tensor vcache, torch::Tensor t_dev);
torch::Tensor attn_decode_fp8(torch::Tensor q, torch::Tensor kcache,
                              torch::Tensor vcache, torch::Tensor kscale,
                              torch::Tensor vscale, long t);
torch::Tensor attn_decode_fp8_graph(torch::Tensor q, torch::Tensor kcache,
                                    torch::Tensor vcache, torch::Tensor kscale,
                                    torch::Tensor vscale, torch::Tensor t_dev);
void kv_append_fp8_(torch::Tensor k, torch::Tensor v, torch::Tensor k8,
                    torch::Tensor v8, torch::Tensor kscale, torch::Tensor vscale,
                    torch::Tensor pos);
// skinny_gemm.hip (decode GEMV, M <= 8)
torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("adamw_step_", &adamw_step_);
  m.def("nesterov_step_", &nesterov_step_);
  m.def("extract_delta", &extract_delta);
  m.def("adamw8_step_", &adamw8_step_);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("grad_norm_sq", &grad_norm_sq);
  m.def("grouped_gemm", &grouped_gemm);
  m.def("adamw8_lean_", &adamw8_lean_);
  m.def("extract_delta_bf16", &extract_delta_bf16);
  m.def("nesterov_bf16_", &nesterov_bf16_);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("add_rmsnorm_bwd", &add_rmsnorm_bwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("rope_fwd_ex", &rope_fwd_ex);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd_", &ce_bwd_);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_fwd_ex", &attn_fwd_ex);
  m.def("attn_bwd_ex", &attn_bwd_ex);
  m.def("mfma_probe_32x32x16", &mfma_probe_32x32x16);
  m.def("mfma_probe_16x16x32", &mfma_probe_16x16x32);
  m.def("fp8_cast_transpose", &fp8_cast_transpose, pybind11::arg("x"),
        pybind11::arg("scale"), pybind11::arg("partials"),
        pybind11::arg("skip_t") = 0);
  m.def("fp8_scale_update_", &fp8_scale_update_, pybind11::arg("partials"),
        pybind11::arg("scale"), pybind11::arg("margin"), pybind11::arg("n") = 0);
  m.def("fp8_cast_grid_size", &fp8_cast_grid_size);
  m.def("adamw8_fp8_lean_", &adamw8_fp8_lean_);
  m.def("fp8_extract_delta", &fp8_extract_delta);
  m.def("fp8_requant_", &fp8_requant_);
  m.def("fp8_weight_cast_transpose", &fp8_weight_cast_transpose);
  m.def("attn_decode", &attn_decode);
  m.def("attn_decode_graph", &attn_decode_graph);
  m.def("attn_decode_fp8", &attn_decode_fp8);
  m.def("attn_decode_fp8_graph", &attn_decode_fp8_graph);
  m.def("kv_append_fp8_", &kv_append_fp8_);
  m.def("skinny_gemm", &skinny_gemm);
}
