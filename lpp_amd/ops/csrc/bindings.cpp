// Python bindings for the lpp_amd gfx950 kernel extension.
#include <torch/extension.h>

#include <string>
#include <tuple>
#include <vector>

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor weight, double eps);
std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor weight,
                                    at::Tensor invrms);
at::Tensor rope_fwd(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, int64_t pos_offset);
at::Tensor rope_bwd(at::Tensor dy, at::Tensor cos_t, at::Tensor sin_t, int64_t pos_offset);
at::Tensor swiglu_fwd(at::Tensor gate, at::Tensor up);
std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor gate, at::Tensor up);
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits, at::Tensor labels);
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor labels, at::Tensor lse,
                             double scale);
std::vector<at::Tensor> attention_fwd(at::Tensor q, at::Tensor k, at::Tensor v);
std::vector<at::Tensor> attention_bwd(at::Tensor dO, at::Tensor q, at::Tensor k,
                                      at::Tensor v, at::Tensor o, at::Tensor lse2);
at::Tensor mfma_test_16x16x32(at::Tensor A, at::Tensor B);
void wgrad_f32_accum(at::Tensor x, at::Tensor dy, at::Tensor dw);
void wgrad_f32_accum_pre(at::Tensor xT, at::Tensor dyT, at::Tensor dw);
void wgrad_bf16d_pre(at::Tensor xT, at::Tensor dyT, at::Tensor out);
at::Tensor transpose2d(at::Tensor in);
void accum_bf16_f32(at::Tensor dst, at::Tensor src);
std::vector<std::tuple<int64_t, double, std::string>> wgrad_tune(int64_t T, int64_t in,
                                                                 int64_t out,
                                                                 int64_t reps,
                                                                 int64_t kind);
void wgrad_set_algo(int64_t T, int64_t in, int64_t out, int64_t index, int64_t kind);
std::tuple<int64_t, std::string> wgrad_current_algo(int64_t T, int64_t in, int64_t out,
                                                    int64_t kind);
at::Tensor mfma_test_32x32x16(at::Tensor A, at::Tensor B);
void fused_adamw(std::vector<at::Tensor> params, std::vector<at::Tensor> masters,
                 std::vector<at::Tensor> grads, std::vector<at::Tensor> exp_avg,
                 std::vector<at::Tensor> exp_avg_sq, double lr, double beta1, double beta2,
                 double eps, double weight_decay, double bias1, double bias2,
                 double grad_scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lpp_amd gfx950 (MI355X/CDNA4) kernels";
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (y, invrms)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (dx, dw)");
  m.def("rope_fwd", &rope_fwd, "RoPE forward");
  m.def("rope_bwd", &rope_bwd, "RoPE backward");
  m.def("swiglu_fwd", &swiglu_fwd, "SwiGLU forward");
  m.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward (dgate, dup)");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused CE forward (loss_sum, lse, count)");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused CE backward (dlogits)");
  m.def("fused_adamw", &fused_adamw, "fused mixed-precision AdamW");
  m.def("attention_fwd", &attention_fwd, "flash causal attention forward (O, LSE2)");
  m.def("attention_bwd", &attention_bwd, "flash causal attention backward (dQ, dK, dV)");
  m.def("wgrad_f32_accum", &wgrad_f32_accum, "dW_f32 += dY^T @ X (hipBLASLt, beta=1)");
  m.def("wgrad_f32_accum_pre", &wgrad_f32_accum_pre,
        "dW_f32 += dyT @ xT^T, pre-transposed k-contiguous operands (hipBLASLt, beta=1)");
  m.def("transpose2d", &transpose2d, "bf16/fp16 [R,C] -> [C,R] LDS-tiled transpose");
  m.def("wgrad_bf16d_pre", &wgrad_bf16d_pre,
        "bf16-D pre-transposed wgrad GEMM into a reusable scratch (beta 0)");
  m.def("accum_bf16_f32", &accum_bf16_f32, "dst_f32 += src_bf16 (vectorised)");
  m.def("wgrad_tune", &wgrad_tune,
        "exhaustive hipBLASLt solution sweep for a wgrad shape -> [(index, ms, name)]",
        py::arg("T"), py::arg("in_dim"), py::arg("out"), py::arg("reps"), py::arg("kind") = 0);
  m.def("wgrad_set_algo", &wgrad_set_algo, "pin a hipBLASLt solution index for a shape",
        py::arg("T"), py::arg("in_dim"), py::arg("out"), py::arg("index"), py::arg("kind") = 0);
  m.def("wgrad_current_algo", &wgrad_current_algo,
        "current (heuristic-picked) solution for a shape -> (index, name)",
        py::arg("T"), py::arg("in_dim"), py::arg("out"), py::arg("kind") = 0);
  m.def("mfma_test_16x16x32", &mfma_test_16x16x32, "MFMA layout validation");
  m.def("mfma_test_32x32x16", &mfma_test_32x32x16, "MFMA 32x32x16 layout validation");
}
