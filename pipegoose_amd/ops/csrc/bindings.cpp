// Python bindings for the pipegoose_amd CDNA4 kernel extension.
#include <torch/extension.h>

std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor w,
                                          torch::Tensor b, double eps);
std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w, torch::Tensor mean,
                                          torch::Tensor rstd);
std::vector<torch::Tensor> layer_norm_res_fwd(torch::Tensor x, torch::Tensor r,
                                              torch::Tensor w, torch::Tensor b,
                                              double eps);
torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias);
std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias);
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits, torch::Tensor targets,
                                             int64_t vocab_start, int64_t vocab_end);
torch::Tensor cross_entropy_bwd(torch::Tensor logits, torch::Tensor targets,
                                torch::Tensor row_max, torch::Tensor row_sumexp,
                                torch::Tensor gscale, int64_t vocab_start,
                                int64_t vocab_end);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor slopes,
                                    double scale, int64_t kv_off);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    torch::Tensor slopes, double scale,
                                    int64_t kv_off);
void attn_bwd_into(torch::Tensor dout, torch::Tensor q,
                   torch::Tensor k, torch::Tensor v,
                   torch::Tensor o, torch::Tensor lse,
                   torch::Tensor slopes, double scale,
                   torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                   int64_t kv_off);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor Bt);
torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor Bt);
torch::Tensor tr16_probe(torch::Tensor tile);
torch::Tensor gemm_bt(torch::Tensor A, torch::Tensor B,
                      c10::optional<torch::Tensor> bias, bool gelu);
void adamw_fused_step(std::vector<torch::Tensor> params,
                      std::vector<torch::Tensor> grads,
                      std::vector<torch::Tensor> exp_avgs,
                      std::vector<torch::Tensor> exp_avg_sqs,
                      torch::Tensor meta_dev, int64_t n_slabs,
                      double lr, double beta1, double beta2, double eps,
                      double weight_decay, int64_t step);
std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w,
                                        double eps);
std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, torch::Tensor rstd);
torch::Tensor rope_apply(torch::Tensor x, double theta_base, bool backward,
                         int64_t pos_offset);
std::vector<torch::Tensor> router_topk(torch::Tensor logits, int64_t k);
torch::Tensor silu_mul_fwd(torch::Tensor gate, torch::Tensor up);
std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor gate,
                                        torch::Tensor up);
std::vector<torch::Tensor> fp8_quant(torch::Tensor x, bool e5m2);
std::vector<torch::Tensor> fp8_quant_dual(torch::Tensor x, bool e5m2);
torch::Tensor fp8_transpose(torch::Tensor q);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("layer_norm_fwd", &layer_norm_fwd, "fused LayerNorm forward (gfx950)");
    m.def("layer_norm_bwd", &layer_norm_bwd, "fused LayerNorm backward (gfx950)");
    m.def("layer_norm_res_fwd", &layer_norm_res_fwd,
          "fused residual-add + LayerNorm forward (gfx950)");
    m.def("bias_gelu_fwd", &bias_gelu_fwd, "fused bias+GeLU forward (gfx950)");
    m.def("bias_gelu_bwd", &bias_gelu_bwd, "fused bias+GeLU backward (gfx950)");
    m.def("cross_entropy_fwd", &cross_entropy_fwd,
          "fused online-softmax CE forward over a vocab shard (gfx950)");
    m.def("cross_entropy_bwd", &cross_entropy_bwd,
          "fused CE backward: (softmax - onehot) * g (gfx950)");
    m.def("attn_fwd", &attn_fwd,
          "flash attention fwd, causal + in-kernel ALiBi (gfx950 MFMA)");
    m.def("attn_bwd", &attn_bwd,
          "flash attention bwd (two-pass, no atomics) (gfx950 MFMA)");
    m.def("attn_bwd_into", &attn_bwd_into,
          "flash attention bwd writing grads into caller buffers "
          "(e.g. fused-qkv gradient slices)");
    m.def("mfma_probe", &mfma_probe,
          "16x16x32 bf16 MFMA fragment-layout probe");
    m.def("mfma_probe32", &mfma_probe32,
          "32x32x16 bf16 MFMA fragment-layout probe");
    m.def("tr16_probe", &tr16_probe,
          "ds_read_b64_tr_b16 semantics probe ([4][16] tile -> per-lane)");
    m.def("gemm_bt", &gemm_bt,
          "hand-written MFMA bf16 GEMM C = A @ B^T (+bias, +gelu) (gfx950)",
          py::arg("A"), py::arg("B"), py::arg("bias") = c10::nullopt,
          py::arg("gelu") = false);
    m.def("adamw_fused_step", &adamw_fused_step,
          "whole AdamW update, one chunked kernel per step (gfx950)");
    m.def("rms_norm_fwd", &rms_norm_fwd, "fused RMSNorm forward (gfx950)");
    m.def("rms_norm_bwd", &rms_norm_bwd, "fused RMSNorm backward (gfx950)");
    m.def("rope_apply", &rope_apply,
          "rotary embedding with in-kernel cos/sin (gfx950)");
    m.def("router_topk", &router_topk,
          "fused Switch router: softmax+topk+colsum+lse in one pass (gfx950)");
    m.def("fp8_quant", &fp8_quant,
          "dynamic per-tensor fp8 quantize: one amax pass + one cast pass");
    m.def("fp8_quant_dual", &fp8_quant_dual,
          "fp8 quantize emitting row-major AND transposed images in one "
          "read (kills the separate backward-layout transpose pass)");
    m.def("fp8_transpose", &fp8_transpose,
          "LDS-tiled byte transpose for fp8 GEMM operand layouts");
    m.def("silu_mul_fwd", &silu_mul_fwd, "fused SwiGLU silu(g)*u (gfx950)");
    m.def("silu_mul_bwd", &silu_mul_bwd, "fused SwiGLU backward (gfx950)");
}
