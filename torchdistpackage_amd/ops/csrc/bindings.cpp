// Python bindings for the in-tree gfx950 HIP extension (_tdpa_hip).
#include <torch/extension.h>

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd);
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias);
torch::Tensor bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                            torch::Tensor bias);
void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, long step, double lr, double beta1,
                double beta2, double eps, double wd);
void multi_adamw_step(torch::Tensor cpid, torch::Tensor coff,
                      torch::Tensor pptrs, torch::Tensor gptrs,
                      torch::Tensor mptrs, torch::Tensor moffs,
                      torch::Tensor numels,
                      torch::Tensor m, torch::Tensor v,
                      long step, double lr, double beta1, double beta2,
                      double eps, double wd, bool param_bf16, bool grad_bf16);
void ema_update(torch::Tensor ema, torch::Tensor p, double decay);
torch::Tensor l2norm_sq(torch::Tensor x);
void scale_inplace(torch::Tensor x, double s);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    bool causal, double scale);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    torch::Tensor dq, torch::Tensor dk,
                                    torch::Tensor dv,
                                    bool causal, double scale);
torch::Tensor mfma_probe_16x16x32(torch::Tensor a, torch::Tensor b);
torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b);
torch::Tensor tr16_probe(long addr_mode);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets);
torch::Tensor gemm_fprop(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias);
torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cs, torch::Tensor sn,
                         long pos0, bool fwd);
torch::Tensor swiglu_fwd(torch::Tensor a, torch::Tensor b);
torch::Tensor bgelu_b_fwd(torch::Tensor x, torch::Tensor bias);
torch::Tensor bgelu_b_bwd(torch::Tensor dy, torch::Tensor x,
                          torch::Tensor bias);
std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dy, torch::Tensor a,
                                      torch::Tensor b);
torch::Tensor gemm_dgrad(torch::Tensor dy, torch::Tensor w, bool kswz);
torch::Tensor gemm_wgrad(torch::Tensor dy, torch::Tensor x, long splitk,
                         bool kswz);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor lse, torch::Tensor grad_out);
std::vector<torch::Tensor> ce_partial_fwd(torch::Tensor logits,
                                          torch::Tensor targets);
torch::Tensor gemv_bf16(torch::Tensor x, torch::Tensor W,
                        c10::optional<torch::Tensor> bias);
torch::Tensor rope_apply_pos(torch::Tensor x, torch::Tensor cs,
                             torch::Tensor sn, torch::Tensor pos, bool fwd);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("bias_gelu_fwd", &bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bias_gelu_bwd);
  m.def("adamw_step", &adamw_step);
  m.def("multi_adamw_step", &multi_adamw_step);
  m.def("ema_update", &ema_update);
  m.def("l2norm_sq", &l2norm_sq);
  m.def("scale_inplace", &scale_inplace);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("mfma_probe_16x16x32", &mfma_probe_16x16x32);
  m.def("mfma_probe_32x32x16", &mfma_probe_32x32x16);
  m.def("tr16_probe", &tr16_probe);
  m.def("ce_fwd", &ce_fwd);
  m.def("gemm_fprop", &gemm_fprop);
  m.def("rope_apply", &rope_apply);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("bgelu_b_fwd", &bgelu_b_fwd);
  m.def("bgelu_b_bwd", &bgelu_b_bwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("gemm_dgrad", &gemm_dgrad);
  m.def("gemm_wgrad", &gemm_wgrad);
  m.def("ce_bwd", &ce_bwd);
  m.def("ce_partial_fwd", &ce_partial_fwd);
  m.def("gemv_bf16", &gemv_bf16);
  m.def("rope_apply_pos", &rope_apply_pos);
}
