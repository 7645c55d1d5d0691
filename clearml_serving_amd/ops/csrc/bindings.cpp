// pybind11 bindings for the gfx950 kernel library (_hip_ops.so).
#include <torch/extension.h>

torch::Tensor bias_relu_add(torch::Tensor x, c10::optional<torch::Tensor> bias,
                            c10::optional<torch::Tensor> residual);
torch::Tensor bias_gelu(torch::Tensor x, c10::optional<torch::Tensor> bias);
torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up);
torch::Tensor layernorm(torch::Tensor x, torch::Tensor weight,
                        torch::Tensor bias, double eps,
                        c10::optional<torch::Tensor> residual);
torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor weight, double eps,
                      c10::optional<torch::Tensor> residual);
torch::Tensor softmax_lastdim(torch::Tensor x);
torch::Tensor attention_prefill(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, bool causal, double scale,
                                c10::optional<torch::Tensor> seq_lens,
                                bool bshd);
torch::Tensor sample_top_k_top_p(torch::Tensor logits, double temperature,
                                 long top_k, double top_p, long seed);
torch::Tensor attention_decode(torch::Tensor q, torch::Tensor k_cache,
                               torch::Tensor v_cache,
                               torch::Tensor block_table,
                               torch::Tensor seq_lens, double scale,
                               c10::optional<torch::Tensor> k_scale,
                               c10::optional<torch::Tensor> v_scale);
void kv_cache_write(torch::Tensor knew, torch::Tensor vnew,
                    torch::Tensor k_cache, torch::Tensor v_cache,
                    torch::Tensor slot_mapping,
                    c10::optional<torch::Tensor> k_scale,
                    c10::optional<torch::Tensor> v_scale);
torch::Tensor gemm_bf16(torch::Tensor a, torch::Tensor b);
torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w, int64_t variant);
torch::Tensor attention_prefill_paged(torch::Tensor q, torch::Tensor k_cache,
                                      torch::Tensor v_cache,
                                      torch::Tensor block_table,
                                      torch::Tensor kv_lens,
                                      torch::Tensor q_lens, double scale);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  double theta);
std::vector<torch::Tensor> rmsnorm_fp8(torch::Tensor x, torch::Tensor weight,
                                       double eps,
                                       c10::optional<torch::Tensor> residual);
std::vector<torch::Tensor> silu_mul_fp8(torch::Tensor gate, torch::Tensor up);
std::vector<torch::Tensor> quant_fp8(torch::Tensor x);
torch::Tensor skinny_gemm_fp8(torch::Tensor a8, torch::Tensor a_scale,
                              torch::Tensor w8, torch::Tensor w_scale);
torch::Tensor skinny_gemm_fp8_v2(torch::Tensor a8, torch::Tensor a_scale,
                                 torch::Tensor w8, torch::Tensor w_scale);
torch::Tensor conv3x3_nhwc(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, bool relu,
                           c10::optional<torch::Tensor> residual);
torch::Tensor attention_prefill_v2(torch::Tensor q, torch::Tensor k,
                                   torch::Tensor v, bool causal, double scale,
                                   c10::optional<torch::Tensor> seq_lens,
                                   bool bshd, bool kdirect, bool pipe);
torch::Tensor attention_prefill_paged_v2(
    torch::Tensor q, torch::Tensor k_cache, torch::Tensor v_cache,
    torch::Tensor block_table, torch::Tensor kv_lens, torch::Tensor q_lens,
    double scale, c10::optional<torch::Tensor> k_scale,
    c10::optional<torch::Tensor> v_scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "clearml-serving-amd gfx950 kernel library";
  m.def("bias_relu_add", &bias_relu_add, py::arg("x"),
        py::arg("bias") = py::none(), py::arg("residual") = py::none());
  m.def("bias_gelu", &bias_gelu, py::arg("x"), py::arg("bias") = py::none());
  m.def("silu_mul", &silu_mul);
  m.def("layernorm", &layernorm, py::arg("x"), py::arg("weight"),
        py::arg("bias"), py::arg("eps") = 1e-5,
        py::arg("residual") = py::none());
  m.def("rmsnorm", &rmsnorm, py::arg("x"), py::arg("weight"),
        py::arg("eps") = 1e-6, py::arg("residual") = py::none());
  m.def("softmax_lastdim", &softmax_lastdim);
  m.def("attention_prefill", &attention_prefill, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("causal"), py::arg("scale"),
        py::arg("seq_lens") = py::none(), py::arg("bshd") = false);
  m.def("sample_top_k_top_p", &sample_top_k_top_p);
  m.def("rope_inplace", &rope_inplace);
  m.def("attention_decode", &attention_decode, py::arg("q"),
        py::arg("k_cache"), py::arg("v_cache"), py::arg("block_table"),
        py::arg("seq_lens"), py::arg("scale"),
        py::arg("k_scale") = py::none(), py::arg("v_scale") = py::none());
  m.def("kv_cache_write", &kv_cache_write, py::arg("knew"), py::arg("vnew"),
        py::arg("k_cache"), py::arg("v_cache"), py::arg("slot_mapping"),
        py::arg("k_scale") = py::none(), py::arg("v_scale") = py::none());
  m.def("gemm_bf16", &gemm_bf16);
  m.def("skinny_gemm", &skinny_gemm, py::arg("a"), py::arg("w"),
        py::arg("variant") = 0);
  m.def("attention_prefill_paged", &attention_prefill_paged);
  m.def("rmsnorm_fp8", &rmsnorm_fp8, py::arg("x"), py::arg("weight"),
        py::arg("eps") = 1e-6, py::arg("residual") = py::none());
  m.def("silu_mul_fp8", &silu_mul_fp8);
  m.def("quant_fp8", &quant_fp8);
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8);
  m.def("skinny_gemm_fp8_v2", &skinny_gemm_fp8_v2);
  m.def("conv3x3_nhwc", &conv3x3_nhwc, py::arg("x"), py::arg("w"),
        py::arg("bias") = py::none(), py::arg("relu") = false,
        py::arg("residual") = py::none());
  m.def("attention_prefill_v2", &attention_prefill_v2, py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("causal"), py::arg("scale"),
        py::arg("seq_lens") = py::none(), py::arg("bshd") = false,
        py::arg("kdirect") = false, py::arg("pipe") = false);
  m.def("attention_prefill_paged_v2", &attention_prefill_paged_v2,
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_table"), py::arg("kv_lens"), py::arg("q_lens"),
        py::arg("scale"), py::arg("k_scale") = py::none(),
        py::arg("v_scale") = py::none());
}
