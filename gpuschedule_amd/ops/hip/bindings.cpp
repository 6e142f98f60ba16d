// Python bindings for the gfx950 kernel library.
#include <torch/extension.h>

std::vector<torch::Tensor> norm_fwd(torch::Tensor x,
                                    c10::optional<torch::Tensor> residual,
                                    torch::Tensor gamma,
                                    c10::optional<torch::Tensor> beta,
                                    double eps, bool rms);
std::vector<torch::Tensor> norm_bwd(torch::Tensor dy, torch::Tensor x,
                                    c10::optional<torch::Tensor> mean,
                                    torch::Tensor rstd, torch::Tensor gamma,
                                    bool rms,
                                    c10::optional<torch::Tensor> dadd);
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu,
                      c10::optional<torch::Tensor> add);
torch::Tensor gemm_nn(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu);
torch::Tensor transpose2d(torch::Tensor x);
torch::Tensor transpose2d_bhsd(torch::Tensor x);
std::vector<torch::Tensor> transpose2d_colsum(torch::Tensor x);
void adamw_step(torch::Tensor p_bf16, torch::Tensor p32, torch::Tensor grad,
                torch::Tensor m, torch::Tensor v, double lr, double beta1,
                double beta2, double eps, double wd, long step);
void sgd_step(torch::Tensor p32, torch::Tensor p_bf16, torch::Tensor grad,
              torch::Tensor buf, double lr, double momentum, double wd);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal);
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout,
                                    torch::Tensor lse, bool causal);
std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  c10::optional<torch::Tensor> use_mean,
                                  c10::optional<torch::Tensor> use_rstd,
                                  c10::optional<torch::Tensor> run_mean,
                                  c10::optional<torch::Tensor> run_var,
                                  c10::optional<torch::Tensor> residual,
                                  double momentum, double eps, bool relu);
void sgd_step_fp32(torch::Tensor p, torch::Tensor grad, torch::Tensor buf,
                   double lr, double momentum, double wd);
void adamw_multi(torch::Tensor ptrs, torch::Tensor sizes,
                 torch::Tensor chunk_t, torch::Tensor chunk_off,
                 torch::Tensor stepbuf, double lr, double beta1,
                 double beta2, double eps, double wd);
void sgd_multi(torch::Tensor ptrs, torch::Tensor sizes,
               torch::Tensor chunk_t, torch::Tensor chunk_off,
               double lr, double momentum, double wd, bool bf16);
torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cosb,
                         torch::Tensor sinb, bool backward);
torch::Tensor silu_mul_fwd(torch::Tensor g, torch::Tensor u);
torch::Tensor gelu_fwd(torch::Tensor x);
torch::Tensor swiglu_fwd(torch::Tensor gu);
torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu);
torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x);
std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor g,
                                        torch::Tensor u);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits,
                                  torch::Tensor labels);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor lse, torch::Tensor gscale);
torch::Tensor embed_fwd(torch::Tensor weight, torch::Tensor idx);
torch::Tensor embed_bwd(torch::Tensor dy, torch::Tensor idx, long V);
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  c10::optional<torch::Tensor> y,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  torch::Tensor gamma, bool relu,
                                  bool want_dres,
                                  c10::optional<torch::Tensor> beta);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("norm_fwd", &norm_fwd, "fused LayerNorm/RMSNorm forward (bf16)");
  m.def("norm_bwd", &norm_bwd,
        "fused LayerNorm/RMSNorm backward (bf16); optional dadd is a "
        "downstream residual-stream grad added into dx in-kernel",
        py::arg("dy"), py::arg("x"), py::arg("mean"), py::arg("rstd"),
        py::arg("gamma"), py::arg("rms"),
        py::arg("dadd") = c10::optional<torch::Tensor>());
  m.def("gemm_nt", &gemm_nt,
        "MFMA bf16 GEMM C = A @ B^T (+bias, +relu, +fused addend)",
        py::arg("a"), py::arg("b"), py::arg("bias"), py::arg("relu"),
        py::arg("add") = c10::optional<torch::Tensor>());
  m.def("gemm_nn", &gemm_nn,
        "MFMA bf16 GEMM C = A @ B, B red-major (tr_b16 fragment reads)");
  m.def("transpose2d", &transpose2d, "bf16 2-D transpose");
  m.def("transpose2d_bhsd", &transpose2d_bhsd,
        "strided [B,H,S,D] -> contiguous [B,H,D,S] bf16 transpose");
  m.def("transpose2d_colsum", &transpose2d_colsum,
        "bf16 transpose + bf16 column sums (bias grad)");
  m.def("adamw_step", &adamw_step, "fused AdamW step");
  m.def("sgd_step", &sgd_step, "fused SGD-momentum step");
  m.def("attn_fwd", &attn_fwd, "flash-attention forward (bf16, online softmax); returns (o, lse)");
  m.def("attn_bwd", &attn_bwd, "flash-attention backward (FA2 two-kernel)");
  m.def("rope_apply", &rope_apply, "fused rotate-half RoPE fwd/bwd");
  m.def("silu_mul_fwd", &silu_mul_fwd, "fused SwiGLU gate fwd");
  m.def("gelu_fwd", &gelu_fwd, "fused tanh-GELU fwd (bf16)");
  m.def("swiglu_fwd", &swiglu_fwd,
        "packed SwiGLU fwd: y = silu(gu[..,:F]) * gu[..,F:]");
  m.def("swiglu_bwd", &swiglu_bwd,
        "packed SwiGLU bwd: one [rows,2F] grad buffer, no cat");
  m.def("gelu_bwd", &gelu_bwd, "fused tanh-GELU bwd (bf16)");
  m.def("silu_mul_bwd", &silu_mul_bwd, "fused SwiGLU gate bwd");
  m.def("ce_fwd", &ce_fwd, "fused cross-entropy forward (bf16 logits)");
  m.def("ce_bwd", &ce_bwd, "fused cross-entropy backward");
  m.def("sgd_multi", &sgd_multi, "multi-tensor fused SGD (one launch/dtype, zeroes grads)");
  m.def("adamw_multi", &adamw_multi, "multi-tensor fused AdamW (one launch, zeroes grads)");
  m.def("sgd_step_fp32", &sgd_step_fp32, "fused fp32 SGD step");
  m.def("embed_fwd", &embed_fwd, "embedding gather (bf16)");
  m.def("embed_bwd", &embed_bwd,
        "embedding dense grad: fp32 atomic scatter + bf16 cast");
  m.def("bn_fwd", &bn_fwd, "fused BatchNorm(+ReLU) forward, channels_last");
  m.def("bn_bwd", &bn_bwd,
        "fused BatchNorm(+ReLU) backward, channels_last; pass beta "
        "instead of y to recompute the relu mask from x",
        py::arg("dy"), py::arg("x"), py::arg("y"), py::arg("mean"),
        py::arg("rstd"), py::arg("gamma"), py::arg("relu"),
        py::arg("want_dres"),
        py::arg("beta") = c10::optional<torch::Tensor>());
}
