// Python bindings for the relora_amd gfx950 HIP kernels.

#include <torch/extension.h>

// norms.hip
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_fwd_add(torch::Tensor x, torch::Tensor res,
                                           torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd_add(torch::Tensor x, torch::Tensor w,
                                           torch::Tensor invrms, torch::Tensor dy,
                                           torch::Tensor dsum);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor x, torch::Tensor w,
                                       torch::Tensor invrms, torch::Tensor dy);
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor mean, torch::Tensor invstd,
                                         torch::Tensor dy);
// rope.hip
std::vector<torch::Tensor> rope_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor cos_t, torch::Tensor sin_t,
                                    bool inverse);
// swiglu.hip
torch::Tensor swiglu_fwd(torch::Tensor g, torch::Tensor u);
torch::Tensor gelu_fwd(torch::Tensor x);
torch::Tensor gelu_bwd(torch::Tensor x, torch::Tensor dy);
std::vector<torch::Tensor> swiglu_bwd(torch::Tensor g, torch::Tensor u, torch::Tensor dy);
// ce.hip
std::vector<torch::Tensor> ce_row_stats(torch::Tensor logits, torch::Tensor labels,
                                        long ignore_index);
void ce_grad_(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
              double gscale, long ignore_index);
// adamw.hip
void fused_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                 double lr, double beta1, double beta2, double eps, double wd, long step);
torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> grads);
void multi_tensor_scale_(std::vector<torch::Tensor> grads, double scale);
// lora_gemm.hip
std::vector<torch::Tensor> dropout_mask_fwd(torch::Tensor x, double p, int64_t seed);
torch::Tensor dropout_mask_bwd(torch::Tensor dy, torch::Tensor mask, double p);
void lora_add_nt_(torch::Tensor out, torch::Tensor P, torch::Tensor Q);

// fused_gemm.hip
torch::Tensor fused_lora_gemm(torch::Tensor x, torch::Tensor w, torch::Tensor t,
                              torch::Tensor bw, torch::Tensor bias, double lora_scale);
torch::Tensor fused_lora_gemm3(torch::Tensor x, torch::Tensor w, torch::Tensor t,
                               torch::Tensor bw, torch::Tensor bias, double lora_scale);
torch::Tensor fused_lora_gemm4(torch::Tensor x, torch::Tensor w, torch::Tensor t,
                               torch::Tensor bw, torch::Tensor bias, double lora_scale);
torch::Tensor fused_int8_gemm(torch::Tensor x, torch::Tensor qw, torch::Tensor amax,
                              long N, torch::Tensor t, torch::Tensor bw,
                              torch::Tensor bias, double lora_scale);
torch::Tensor fused_nf4_gemm(torch::Tensor x, torch::Tensor qw, torch::Tensor amax,
                             long N, torch::Tensor t, torch::Tensor bw,
                             torch::Tensor bias, double lora_scale);
void lora_add_nn_(torch::Tensor out, torch::Tensor P, torch::Tensor Q,
                  torch::Tensor mask, double inv_keep);
torch::Tensor skinny_grad(torch::Tensor P, torch::Tensor X, torch::Tensor xmask,
                          double inv_keep, double scale,
                          bool transpose_out, torch::ScalarType dtype);
// quantize.hip
std::vector<torch::Tensor> quantize_nf4(torch::Tensor x);
torch::Tensor dequantize_nf4(torch::Tensor q, torch::Tensor absmax, long n, torch::ScalarType dtype);
std::vector<torch::Tensor> quantize_int8(torch::Tensor x);
torch::Tensor dequantize_int8(torch::Tensor q, torch::Tensor absmax, long n, torch::ScalarType dtype);
// attention.hip
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    double scale);
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse, torch::Tensor dout,
                                    double scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (gfx950)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (gfx950)");
  m.def("rmsnorm_fwd_add", &rmsnorm_fwd_add, "fused residual-add + RMSNorm fwd (gfx950)");
  m.def("rmsnorm_bwd_add", &rmsnorm_bwd_add, "RMSNorm bwd with +dsum fused into dx (gfx950)");
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (gfx950)");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (gfx950)");
  m.def("rope_fwd", &rope_fwd, "RoPE apply fwd/inverse (gfx950)");
  m.def("swiglu_fwd", &swiglu_fwd, "SwiGLU forward (gfx950)");
  m.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward (gfx950)");
  m.def("gelu_fwd", &gelu_fwd, "exact-erf GELU forward (gfx950)");
  m.def("gelu_bwd", &gelu_bwd, "exact-erf GELU backward (gfx950)");
  m.def("ce_row_stats", &ce_row_stats, "CE row lse/target (gfx950)");
  m.def("ce_grad_", &ce_grad_, "CE in-place softmax-onehot grad (gfx950)");
  m.def("fused_adamw", &fused_adamw, "multi-tensor AdamW step (gfx950)");
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm, "multi-tensor L2 norm (gfx950)");
  m.def("multi_tensor_scale_", &multi_tensor_scale_, "multi-tensor scale (gfx950)");
  m.def("dropout_mask_fwd", &dropout_mask_fwd, "fused dropout + packed mask (gfx950)");
  m.def("dropout_mask_bwd", &dropout_mask_bwd, "packed-mask dropout backward (gfx950)");
  m.def("lora_add_nt_", &lora_add_nt_, "out += P @ Q^T rank-r MFMA accumulate (gfx950)");
  m.def("fused_lora_gemm", &fused_lora_gemm,
        "y = x@W^T (+bias) + s*t@Bw^T fused MFMA GEMM, 256^2 glds tile (gfx950)");
  m.def("fused_lora_gemm3", &fused_lora_gemm3,
        "3-buffer counted-vmcnt deep-pipelined fused GEMM variant (gfx950)");
  m.def("fused_lora_gemm4", &fused_lora_gemm4,
        "BK=32 4-buffer counted-vmcnt fused GEMM (glds-span tier, gfx950)");
  m.def("fused_int8_gemm", &fused_int8_gemm,
        "y = x@dequant_int8(W)^T + s*t@Bw^T, dequant fused into staging (gfx950)");
  m.def("fused_nf4_gemm", &fused_nf4_gemm,
        "y = x@dequant(W)^T + s*t@Bw^T with NF4 dequant fused into LDS staging (gfx950)");
  m.def("lora_add_nn_", &lora_add_nn_, "out += maskscale*(P @ Q) rank-r MFMA accumulate (gfx950)");
  m.def("skinny_grad", &skinny_grad, "P^T @ X chunked fp32 reduction (gfx950)");
  m.def("quantize_nf4", &quantize_nf4, "NF4 blockwise quantize (gfx950)");
  m.def("dequantize_nf4", &dequantize_nf4, "NF4 blockwise dequantize (gfx950)");
  m.def("quantize_int8", &quantize_int8, "int8 blockwise quantize (gfx950)");
  m.def("dequantize_int8", &dequantize_int8, "int8 blockwise dequantize (gfx950)");
  m.def("attn_fwd", &attn_fwd, "causal flash attention forward (gfx950 MFMA)");
  m.def("attn_bwd", &attn_bwd, "causal flash attention backward (gfx950 MFMA)");
}
