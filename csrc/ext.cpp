// pybind11 glue for the scaffold's gfx950 HIP kernels.
#include <torch/extension.h>

#include <vector>

torch::Tensor adamw_build_meta(std::vector<torch::Tensor> masters,
                               std::vector<torch::Tensor> grads,
                               std::vector<torch::Tensor> exp_avgs,
                               std::vector<torch::Tensor> exp_avg_sqs,
                               std::vector<torch::Tensor> outs);
void fused_adamw_run(torch::Tensor meta_dev, long n, long total_vec4,
                     double lr, double beta1, double beta2, double eps,
                     double weight_decay, long step, double grad_scale,
                     bool grad_bf16, bool has_out,
                     c10::optional<torch::Tensor> step_dev);
void fused_adamw(std::vector<torch::Tensor> masters,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> outs, double lr, double beta1,
                 double beta2, double eps, double weight_decay, long step,
                 double grad_scale);
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor targets);
torch::Tensor softmax_xent_bwd(torch::Tensor logits, torch::Tensor targets,
                               torch::Tensor lse, double dloss);
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd);
torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cos_t,
                         torch::Tensor sin_t, bool inverse);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal,
                                    c10::optional<torch::Tensor> dlse);
torch::Tensor probe_mfma(torch::Tensor A, torch::Tensor B, long which);
torch::Tensor probe_tr(torch::Tensor offsets);
torch::Tensor rccl_unique_id();
long rccl_init(long rank, long world, torch::Tensor id_bytes);
void rccl_all_reduce(long handle, torch::Tensor t);
void rccl_join(long handle);
double rccl_comm_time_ms(long handle);
void rccl_destroy(long handle);
torch::Tensor swiglu_fwd(torch::Tensor g, torch::Tensor u);
torch::Tensor swiglu_packed_fwd(torch::Tensor gu);
torch::Tensor swiglu_packed_bwd(torch::Tensor dh, torch::Tensor gu);
torch::Tensor colsum(torch::Tensor x);
std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dh, torch::Tensor g,
                                      torch::Tensor u);
void save_shard_pinned(std::string path, std::vector<std::string> names,
                       std::vector<torch::Tensor> tensors,
                       std::vector<std::string> meta_keys,
                       std::vector<std::string> meta_vals);
std::vector<torch::Tensor> linear_fwd(torch::Tensor x, torch::Tensor w,
                                      c10::optional<torch::Tensor> bias,
                                      long epilogue,
                                      c10::optional<torch::Tensor> residual);
std::vector<torch::Tensor> bn_fwd(torch::Tensor x,
                                  c10::optional<torch::Tensor> res,
                                  torch::Tensor weight, torch::Tensor bias,
                                  c10::optional<torch::Tensor> running_mean,
                                  c10::optional<torch::Tensor> running_var,
                                  double momentum, double eps, bool relu);
torch::Tensor decode_gemv(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  c10::optional<torch::Tensor> res,
                                  torch::Tensor weight, torch::Tensor bias,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  bool relu);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_adamw", &fused_adamw, "fused multi-tensor AdamW (K1)");
  m.def("adamw_build_meta", &adamw_build_meta,
        "one-time device meta table for the cached AdamW path");
  m.def("fused_adamw_run", &fused_adamw_run,
        "AdamW step on a prebuilt meta (graph-capturable; optional device step)");
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (K2)");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (K2)");
  m.def("softmax_xent_fwd", &softmax_xent_fwd, "softmax cross-entropy fwd (K3)");
  m.def("softmax_xent_bwd", &softmax_xent_bwd, "softmax cross-entropy bwd (K3)");
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (K7)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (K7)");
  m.def("rope_apply", &rope_apply, "RoPE apply/inverse (K7)");
  m.def("attn_fwd", &attn_fwd, "flash attention forward (K4)");
  m.def("attn_bwd", &attn_bwd, "flash attention backward (K4)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("causal"),
        py::arg("dlse") = py::none());
  m.def("probe_mfma", &probe_mfma, "MFMA fragment-layout probe (tests)");
  m.def("probe_tr", &probe_tr, "ds_read_b64_tr_b16 semantics probe (tests)");
  m.def("rccl_unique_id", &rccl_unique_id, "RCCL bootstrap id (rank 0)");
  m.def("rccl_init", &rccl_init, "init direct-RCCL bucket communicator");
  m.def("rccl_all_reduce", &rccl_all_reduce,
        "bucket SUM all-reduce on the comm stream (async)");
  m.def("rccl_join", &rccl_join, "compute stream waits comm stream");
  m.def("rccl_comm_time_ms", &rccl_comm_time_ms,
        "drain measured collective time (ms)");
  m.def("rccl_destroy", &rccl_destroy, "destroy communicator");
  m.def("swiglu_fwd", &swiglu_fwd, "fused silu(g)*u forward");
  m.def("swiglu_packed_fwd", &swiglu_packed_fwd,
        "fused SwiGLU on a packed gate|up buffer");
  m.def("swiglu_packed_bwd", &swiglu_packed_bwd,
        "packed SwiGLU backward (dgu)");
  m.def("colsum", &colsum, "bf16 column sum (bias gradients)");
  m.def("swiglu_bwd", &swiglu_bwd, "fused SwiGLU backward (dg, du)");
  m.def("save_shard_pinned", &save_shard_pinned,
        "stream tensors to a safetensors shard via pinned D2H (C6)");
  m.def("linear_fwd", &linear_fwd,
        "MFMA linear fwd, fused bias+GELU/residual epilogue (K5)");
  m.def("bn_fwd", &bn_fwd,
        "fused bf16 NHWC BatchNorm fwd (+ReLU, +residual)",
        py::arg("x"), py::arg("res"), py::arg("weight"), py::arg("bias"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("relu"));
  m.def("decode_gemv", &decode_gemv,
        "decode-regime GEMV y = x W^T (+bias), B<=8",
        py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
  m.def("bn_bwd", &bn_bwd, "fused bf16 NHWC BatchNorm bwd",
        py::arg("dy"), py::arg("x"), py::arg("res"), py::arg("weight"),
        py::arg("bias"), py::arg("mean"), py::arg("rstd"), py::arg("relu"));
}
