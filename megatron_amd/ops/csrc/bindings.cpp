// Python bindings for the MI355X ops extension.

#include <torch/extension.h>

#include <vector>

torch::Tensor gemv_bf16(torch::Tensor weight, torch::Tensor x);
torch::Tensor decode_attn(torch::Tensor q, torch::Tensor k_cache,
                          torch::Tensor v_cache, torch::Tensor pos,
                          double scale, int64_t window);
torch::Tensor decode_rope_append(torch::Tensor q, torch::Tensor k,
                                 torch::Tensor v, torch::Tensor cos,
                                 torch::Tensor sin, torch::Tensor pos,
                                 torch::Tensor k_cache,
                                 torch::Tensor v_cache);
bool wgrad_gemm_hand(torch::Tensor input, torch::Tensor grad_output,
                     torch::Tensor main_grad);

std::vector<torch::Tensor> fp8_cast_transpose(torch::Tensor x,
                                              torch::Tensor scale);

// norms.hip
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor weight,
                                       double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor weight,
                                       torch::Tensor inv,
                                       c10::optional<torch::Tensor> dres);
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor weight,
                                         torch::Tensor bias, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor weight,
                                         torch::Tensor mean,
                                         torch::Tensor inv,
                                         c10::optional<torch::Tensor> dres);

// softmax.hip
torch::Tensor scaled_masked_softmax_fwd(torch::Tensor x,
                                        c10::optional<torch::Tensor> mask,
                                        double scale, bool causal);
torch::Tensor scaled_masked_softmax_bwd(torch::Tensor dy, torch::Tensor y,
                                        double scale);

// elementwise.hip
torch::Tensor glu_fwd(torch::Tensor x, int64_t mode);
torch::Tensor glu_bwd(torch::Tensor dy, torch::Tensor x, int64_t mode);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cosT,
                       torch::Tensor sinT);
torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cosT,
                       torch::Tensor sinT);
torch::Tensor rope_bwd_into(torch::Tensor dy, torch::Tensor cosT,
                            torch::Tensor sinT, torch::Tensor out);
std::vector<torch::Tensor> bias_dropout_add_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> bias,
    torch::Tensor residual, double p, int64_t seed, int64_t offset);
torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p);

// adam.hip
void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double wd,
                int64_t step, int64_t adam_w_mode, double grad_scale);
void fused_adam_with_model_copy(std::vector<torch::Tensor> params,
                                std::vector<torch::Tensor> grads,
                                std::vector<torch::Tensor> exp_avgs,
                                std::vector<torch::Tensor> exp_avg_sqs,
                                std::vector<torch::Tensor> model_params,
                                double lr, double beta1, double beta2,
                                double eps, double wd, int64_t step,
                                int64_t adam_w_mode, double grad_scale);

// wgrad.hip
std::vector<torch::Tensor> fp8_quantize(torch::Tensor x,
                                        torch::Tensor scale);
void wgrad_gemm_accum_fp32(torch::Tensor input, torch::Tensor grad_output,
                           torch::Tensor main_grad);

// data_helpers.cpp (CPU)
torch::Tensor build_sample_idx(torch::Tensor sizes, torch::Tensor doc_idx,
                               int64_t seq_length, int64_t num_samples);
void build_blending_indices(torch::Tensor dataset_index,
                            torch::Tensor dataset_sample_index,
                            torch::Tensor weights, int64_t num_datasets,
                            int64_t size, bool verbose);

// flash_attn.hip
std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double softmax_scale,
                                          int64_t window_size,
                                          double dropout_p, int64_t drop_seed,
                                          int64_t drop_offset);
std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor out, torch::Tensor lse,
                                          bool causal, double softmax_scale,
                                          int64_t window_size,
                                          double dropout_p, int64_t drop_seed,
                                          int64_t drop_offset);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("wgrad_gemm_hand", &wgrad_gemm_hand);
  m.def("fp8_cast_transpose", &fp8_cast_transpose);
  m.def("gemv_bf16", &gemv_bf16);
  m.def("decode_attn", &decode_attn, py::arg("q"),
        py::arg("k_cache"), py::arg("v_cache"), py::arg("pos"),
        py::arg("scale"), py::arg("window") = -1);
  m.def("decode_rope_append", &decode_rope_append);
  m.def("rmsnorm_bwd", &rmsnorm_bwd, py::arg("dy"), py::arg("x"),
        py::arg("weight"), py::arg("inv"),
        py::arg("dres") = py::none());
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd, py::arg("dy"), py::arg("x"),
        py::arg("weight"), py::arg("mean"), py::arg("inv"),
        py::arg("dres") = py::none());
  m.def("scaled_masked_softmax_fwd", &scaled_masked_softmax_fwd);
  m.def("scaled_masked_softmax_bwd", &scaled_masked_softmax_bwd);
  m.def("glu_fwd", &glu_fwd);
  m.def("glu_bwd", &glu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("rope_bwd", &rope_bwd);
  m.def("rope_bwd_into", &rope_bwd_into);
  m.def("bias_dropout_add_fwd", &bias_dropout_add_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("fused_adam", &fused_adam);
  m.def("fused_adam_with_model_copy", &fused_adam_with_model_copy);
  m.def("wgrad_gemm_accum_fp32", &wgrad_gemm_accum_fp32);
  m.def("build_sample_idx", &build_sample_idx);
  m.def("build_blending_indices", &build_blending_indices);
  m.def("fp8_quantize", &fp8_quantize);
  m.def("flash_attn_fwd", &flash_attn_fwd, py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("causal"),
        py::arg("softmax_scale"), py::arg("window_size"),
        py::arg("dropout_p") = 0.0, py::arg("drop_seed") = 0,
        py::arg("drop_offset") = 0);
  m.def("flash_attn_bwd", &flash_attn_bwd, py::arg("dout"),
        py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("out"), py::arg("lse"), py::arg("causal"),
        py::arg("softmax_scale"), py::arg("window_size"),
        py::arg("dropout_p") = 0.0, py::arg("drop_seed") = 0,
        py::arg("drop_offset") = 0);
}
