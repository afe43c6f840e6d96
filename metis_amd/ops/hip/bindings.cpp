// Python bindings for the metis_amd gfx950 HIP kernels.
#include <torch/extension.h>

#include <vector>

std::vector<torch::Tensor> layernorm_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta, double eps);
std::vector<torch::Tensor> layernorm_bwd(
    torch::Tensor dy, torch::Tensor x, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd);
void adamw_step(
    torch::Tensor master, torch::Tensor model, torch::Tensor grad,
    torch::Tensor m, torch::Tensor v,
    double lr, double beta1, double beta2, double eps,
    double weight_decay, long step, double grad_scale);
torch::Tensor attn_decode(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, double scale);
torch::Tensor ce_row_max(torch::Tensor logits);
torch::Tensor ce_row_sumexp(torch::Tensor logits, torch::Tensor m);
std::vector<torch::Tensor> qkv_rope_split(
    torch::Tensor qkv, long nq, long nkv, long head_dim,
    torch::Tensor cos_t, torch::Tensor sin_t);
torch::Tensor qkv_rope_split_bwd(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv, long head_dim,
    torch::Tensor cos_t, torch::Tensor sin_t);
torch::Tensor lt_fp8_matmul(
    torch::Tensor x, torch::Tensor w,
    torch::Tensor scale_x, torch::Tensor scale_w);
std::vector<torch::Tensor> lt_fc1_forward(
    torch::Tensor x, torch::Tensor w, torch::Tensor bias);
std::vector<torch::Tensor> lt_matmul_dgelu_bgrad(
    torch::Tensor dy, torch::Tensor w2, torch::Tensor pre_gelu);
std::vector<torch::Tensor> attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, double scale);
std::vector<torch::Tensor> attn_bwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor d_o, torch::Tensor lse, torch::Tensor delta,
    double scale);
torch::Tensor gemm_bf16(
    torch::Tensor a, torch::Tensor b_nk, c10::optional<torch::Tensor> bias,
    long epilogue);
torch::Tensor mfma_tile_probe(torch::Tensor a, torch::Tensor b);
torch::Tensor gemm8_bf16(torch::Tensor x, torch::Tensor w);
std::vector<torch::Tensor> rmsnorm_fwd(
    torch::Tensor x, torch::Tensor gamma, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(
    torch::Tensor dy, torch::Tensor x, torch::Tensor gamma, torch::Tensor rstd);
torch::Tensor rope_apply(
    torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t, bool backward);
torch::Tensor swiglu_fwd(torch::Tensor a, torch::Tensor b);
std::vector<torch::Tensor> swiglu_bwd(
    torch::Tensor dy, torch::Tensor a, torch::Tensor b);
std::vector<torch::Tensor> cross_entropy_fwd(
    torch::Tensor logits, torch::Tensor labels);
torch::Tensor cross_entropy_bwd(
    torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
    torch::Tensor grad_scale);
std::vector<torch::Tensor> qkv_split_transpose(
    torch::Tensor qkv, long nq, long nkv, long head_dim);
torch::Tensor qkv_split_transpose_bwd(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv, long head_dim);
torch::Tensor heads_merge(torch::Tensor x);
torch::Tensor heads_unmerge(torch::Tensor y, long H);
torch::Tensor tr16_probe(long stride_elems);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("layernorm_fwd", &layernorm_fwd,
          "fused LayerNorm forward (bf16, fp32 stats)");
    m.def("layernorm_bwd", &layernorm_bwd,
          "fused LayerNorm backward (dx + dgamma/dbeta)");
    m.def("adamw_step", &adamw_step, "fused AdamW step");
    m.def("attn_fwd", &attn_fwd,
          "flash attention forward (causal, GQA) -> (O, LSE)");
    m.def("attn_bwd", &attn_bwd,
          "flash attention backward -> (dQ, dK_per_head, dV_per_head)");
    m.def("gemm_bf16", &gemm_bf16,
          "MFMA bf16 GEMM A[M,K] @ B[N,K]^T with fused epilogue "
          "(0=none, 1=bias, 2=bias+gelu)");
    m.def("gemm8_bf16", &gemm8_bf16,
          "256^2-tile 2-buffer glds bf16 GEMM x[M,K] @ w[N,K]^T");
    m.def("mfma_tile_probe", &mfma_tile_probe,
          "single-wave 16x16x32 MFMA layout probe");
    m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm forward");
    m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm backward");
    m.def("rope_apply", &rope_apply, "rotary embedding (fwd/bwd by flag)");
    m.def("swiglu_fwd", &swiglu_fwd, "fused silu(a)*b");
    m.def("swiglu_bwd", &swiglu_bwd, "fused SwiGLU backward");
    m.def("cross_entropy_fwd", &cross_entropy_fwd,
          "fused CE over bf16 logits -> (per-row loss, lse)");
    m.def("cross_entropy_bwd", &cross_entropy_bwd,
          "fused CE backward -> bf16 dlogits");
    m.def("qkv_split_transpose", &qkv_split_transpose,
          "[B,S,(nq+2nkv)D] -> q/k/v [B,h,S,D] in one pass");
    m.def("qkv_split_transpose_bwd", &qkv_split_transpose_bwd,
          "dq/dk/dv -> fused dqkv layout");
    m.def("heads_merge", &heads_merge, "[B,H,S,D] -> [B,S,H*D]");
    m.def("heads_unmerge", &heads_unmerge, "[B,S,H*D] -> [B,H,S,D]");
    m.def("attn_decode", &attn_decode,
          "single-query attention over cached K/V (GQA, online softmax)");
    m.def("ce_row_max", &ce_row_max, "per-row max of bf16 logits (fp32)");
    m.def("ce_row_sumexp", &ce_row_sumexp,
          "per-row sum(exp(x - m)) of bf16 logits (fp32)");
    m.def("qkv_rope_split", &qkv_rope_split,
          "fused QKV relayout + RoPE (neox) forward");
    m.def("qkv_rope_split_bwd", &qkv_rope_split_bwd,
          "fused QKV relayout + RoPE backward gather");
    m.def("lt_fp8_matmul", &lt_fp8_matmul,
          "fp8 e4m3 GEMM with per-tensor scales, bf16 out");
    m.def("lt_fc1_forward", &lt_fc1_forward,
          "hipblaslt GEMM with GELU_AUX_BIAS epilogue (fc1 fused)");
    m.def("lt_matmul_dgelu_bgrad", &lt_matmul_dgelu_bgrad,
          "hipblaslt GEMM with DGELU_BGRAD epilogue (fc2 data grad fused)");
    m.def("tr16_probe", &tr16_probe,
          "ds_read_b64_tr_b16 lane-semantics probe (debug)");
}
