// saturn_amd._C: pybind entry for the hand-written CDNA4 kernels.

#include <torch/extension.h>
#include <vector>

namespace samd {
void fused_sgd(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> moms, std::vector<at::Tensor> masters,
               double lr, double momentum, double weight_decay);
void fused_adam(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                std::vector<at::Tensor> masters, double lr, double beta1,
                double beta2, double eps, double weight_decay, double bc1,
                double bc2);
std::vector<at::Tensor> norm_fwd(at::Tensor x, at::Tensor w,
                                 c10::optional<at::Tensor> b, double eps,
                                 bool rms);
std::vector<at::Tensor> norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                 at::Tensor mean, at::Tensor rstd, bool rms,
                                 bool needs_db);
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets,
                               int64_t Tr, int64_t ignore_index);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                  at::Tensor dloss, int64_t Tr, int64_t ignore_index);
void rope_apply(at::Tensor y, at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                int64_t t_len, int64_t t_off, bool half_style, bool backward);
at::Tensor gelu_fwd(at::Tensor x);
at::Tensor gelu_bwd(at::Tensor dy, at::Tensor x);
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 bool causal, int64_t kv_len);
std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor o, at::Tensor lse,
                                 bool causal, int64_t kv_len);
at::Tensor add3(at::Tensor a, at::Tensor b, at::Tensor c);
at::Tensor swiglu_fwd(at::Tensor gate, at::Tensor up);
std::vector<at::Tensor> swiglu_bwd(at::Tensor dout, at::Tensor gate,
                                   at::Tensor up);
at::Tensor embed_fwd(at::Tensor weight, at::Tensor idx);
at::Tensor embed_bwd(at::Tensor dout, at::Tensor idx, long vocab);
at::Tensor dropout_fwd(at::Tensor x, double p, long seed);
at::Tensor dropout_bwd(at::Tensor dout, double p, long seed);
}  // namespace samd

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "saturn_amd hand-written CDNA4 (gfx950) kernels";
  m.def("fused_sgd", &samd::fused_sgd, "fused multi-tensor SGD");
  m.def("fused_adam", &samd::fused_adam, "fused multi-tensor AdamW");
  m.def("norm_fwd", &samd::norm_fwd, "LayerNorm/RMSNorm forward");
  m.def("norm_bwd", &samd::norm_bwd, "LayerNorm/RMSNorm backward");
  m.def("ce_fwd", &samd::ce_fwd, "fused cross-entropy forward");
  m.def("ce_bwd", &samd::ce_bwd, "fused cross-entropy backward");
  m.def("rope_apply", &samd::rope_apply,
        "fused rotary embedding (out-of-place, offset table rows)");
  m.def("gelu_fwd", &samd::gelu_fwd, "fused tanh-approx GELU forward");
  m.def("gelu_bwd", &samd::gelu_bwd, "fused tanh-approx GELU backward");
  m.def("attn_fwd", &samd::attn_fwd, "fused flash attention forward",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"),
        py::arg("kv_len") = -1);
  m.def("attn_bwd", &samd::attn_bwd, "fused flash attention backward",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("causal"),
        py::arg("kv_len") = -1);
  m.def("add3", &samd::add3, "fused 3-way residual add");
  m.def("swiglu_fwd", &samd::swiglu_fwd, "fused silu(gate)*up");
  m.def("swiglu_bwd", &samd::swiglu_bwd, "fused SwiGLU backward");
  m.def("embed_fwd", &samd::embed_fwd, "token embedding gather");
  m.def("embed_bwd", &samd::embed_bwd, "embedding scatter-add backward");
  m.def("dropout_fwd", &samd::dropout_fwd, "counter-based fused dropout");
  m.def("dropout_bwd", &samd::dropout_bwd, "dropout backward (mask recompute)");
}
