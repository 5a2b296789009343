// Python bindings for the gfx950 kernel library.
#include <torch/extension.h>

void rms_norm(torch::Tensor out, torch::Tensor x, torch::Tensor weight, double eps);
void rms_norm_fused_add(torch::Tensor out, torch::Tensor x, torch::Tensor residual,
                        torch::Tensor weight, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up);
void gelu_tanh_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up);
void rope_apply(torch::Tensor q, torch::Tensor k, torch::Tensor cosb, torch::Tensor sinb);
void kv_cache_update(torch::Tensor k_cache, torch::Tensor v_cache, torch::Tensor k_new,
                     torch::Tensor v_new, torch::Tensor positions, torch::Tensor seq_ids);
void block_kv_cache_update(torch::Tensor k_cache, torch::Tensor v_cache, torch::Tensor k_new,
                           torch::Tensor v_new, torch::Tensor slot_mapping);
void attention_decode(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                      torch::Tensor v_cache, torch::Tensor seq_lens, torch::Tensor seq_ids,
                      double scale, long window, c10::optional<torch::Tensor> sinks,
                      c10::optional<torch::Tensor> lse, long ring);
void attention_decode_paged(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                            torch::Tensor v_cache, torch::Tensor seq_lens,
                            torch::Tensor block_table, double scale, long window);
void topk_topp_sample(torch::Tensor out, torch::Tensor vals, torch::Tensor top_k,
                      torch::Tensor top_p, torch::Tensor temp, torch::Tensor rand,
                      long global_topk);
void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, double scale, long window,
                   c10::optional<torch::Tensor> sinks);
void flash_prefill_cached(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                          torch::Tensor v_cache, torch::Tensor seq_ids, double scale,
                          long window, c10::optional<torch::Tensor> sinks);
void qkv_rope_update(torch::Tensor q_out, torch::Tensor qkv, torch::Tensor cosb,
                     torch::Tensor sinb, torch::Tensor k_cache, torch::Tensor v_cache,
                     torch::Tensor positions, torch::Tensor seq_ids, long Hq, long Hkv);
void silu_mul_fused(torch::Tensor out, torch::Tensor gate_up);
void skinny_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> bias);
void skinny_gemm2(torch::Tensor y, torch::Tensor y32, torch::Tensor x,
                  torch::Tensor w, c10::optional<torch::Tensor> bias, long ksplit);
void skinny_gemm3(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                  c10::optional<torch::Tensor> bias);
void skinny_gemm4(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                  c10::optional<torch::Tensor> bias);
void skinny_gemm5(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                  c10::optional<torch::Tensor> bias,
                  c10::optional<torch::Tensor> rs,
                  c10::optional<torch::Tensor> part, long splitk, long mode, long mt);
void skinny_gemm5_fp8(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                      torch::Tensor wscale, c10::optional<torch::Tensor> bias,
                      c10::optional<torch::Tensor> part, long splitk);
void fp8_quant(torch::Tensor x, torch::Tensor out, torch::Tensor scale,
               torch::Tensor amax_ws);
void fp8_epilogue(torch::Tensor out, torch::Tensor y32, torch::Tensor w_scale,
                  c10::optional<torch::Tensor> bias);
void moe_grouped_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                      torch::Tensor expert_ids,
                      c10::optional<torch::Tensor> bias, long bn,
                      c10::optional<torch::Tensor> x_gather,
                      c10::optional<torch::Tensor> combine_out,
                      c10::optional<torch::Tensor> combine_scale,
                      c10::optional<torch::Tensor> combine_gather, long bk,
                      long mt);
void moe_grouped_tables_gpu(torch::Tensor order, torch::Tensor dest,
                            torch::Tensor expert_ids, torch::Tensor gtab,
                            torch::Tensor idx_flat, long num_experts,
                            long topk, long tile);
void moe_grouped_gemm_fp8(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                          torch::Tensor wscale, torch::Tensor expert_ids,
                          c10::optional<torch::Tensor> bias, long bn,
                          c10::optional<torch::Tensor> x_gather,
                          c10::optional<torch::Tensor> combine_out,
                          c10::optional<torch::Tensor> combine_scale,
                          c10::optional<torch::Tensor> combine_gather);
void moe_grouped_gemm_mx4(torch::Tensor y, torch::Tensor x, torch::Tensor wq,
                          torch::Tensor ws, torch::Tensor expert_ids,
                          c10::optional<torch::Tensor> bias, long bn,
                          c10::optional<torch::Tensor> x_gather,
                          c10::optional<torch::Tensor> combine_out,
                          c10::optional<torch::Tensor> combine_scale,
                          c10::optional<torch::Tensor> combine_gather);
void router_topk(torch::Tensor vals, torch::Tensor idx, torch::Tensor logits,
                 long top_k, long act_sigmoid, long pre_softmax,
                 long normalize);
bool lt_available();
void lt_linear(torch::Tensor y, torch::Tensor x, torch::Tensor w,
               c10::optional<torch::Tensor> bias);
int lt_tune(int64_t M, int64_t N, int64_t K, bool has_bias,
            torch::Tensor sx, torch::Tensor sw, torch::Tensor sy,
            c10::optional<torch::Tensor> bias);
void gelu_tanh_mul_fused(torch::Tensor out, torch::Tensor gate_up);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rms_norm", &rms_norm, "RMSNorm (CDNA4)");
    m.def("rms_norm_fused_add", &rms_norm_fused_add, "RMSNorm + residual add (CDNA4)");
    m.def("silu_mul", &silu_mul, "SiLU(gate)*up (CDNA4)");
    m.def("gelu_tanh_mul", &gelu_tanh_mul, "GELU-tanh(gate)*up (CDNA4)");
    m.def("rope_apply", &rope_apply, "RoPE rotate-half in place (CDNA4)");
    m.def("kv_cache_update", &kv_cache_update, "linear KV cache scatter (CDNA4)");
    m.def("block_kv_cache_update", &block_kv_cache_update, "paged KV cache scatter (CDNA4)");
    m.def("attention_decode", &attention_decode, "split-KV decode attention (CDNA4)",
          py::arg("out"), py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
          py::arg("seq_lens"), py::arg("seq_ids"), py::arg("scale"), py::arg("window"),
          py::arg("sinks") = py::none(), py::arg("lse") = py::none(),
          py::arg("ring") = 0);
    m.def("attention_decode_paged", &attention_decode_paged, "paged split-KV decode attention (CDNA4)");
    m.def("topk_topp_sample", &topk_topp_sample, "fused top-k/top-p sampling (CDNA4)");
    m.def("flash_prefill", &flash_prefill, "MFMA causal flash prefill (CDNA4)",
          py::arg("out"), py::arg("q"), py::arg("k"), py::arg("v"), py::arg("scale"),
          py::arg("window"), py::arg("sinks") = py::none());
    m.def("flash_prefill_cached", &flash_prefill_cached,
          "MFMA causal flash prefill reading K/V from the cache (CDNA4)",
          py::arg("out"), py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
          py::arg("seq_ids"), py::arg("scale"), py::arg("window"),
          py::arg("sinks") = py::none());
    m.def("qkv_rope_update", &qkv_rope_update, "fused qkv->rope->cache-write (CDNA4)");
    m.def("skinny_gemm", &skinny_gemm, "skinny-M weight-streaming GEMM (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
    m.def("skinny_gemm4", &skinny_gemm4,
          "skinny-M GEMM v4: asm loads + hand-counted vmcnt pipeline (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
    m.def("skinny_gemm5", &skinny_gemm5,
          "skinny-M GEMM v5: all-glds staging, raw-barrier 3-buffer ring, "
          "counted vmcnt, optional split-K + row-scale/bias fold (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("bias") = py::none(),
          py::arg("rs") = py::none(), py::arg("part") = py::none(),
          py::arg("splitk") = 1, py::arg("mode") = 0, py::arg("mt") = 0);
    m.def("skinny_gemm5_fp8", &skinny_gemm5_fp8,
          "fp8-weight skinny decode GEMM: double-depth chunks, LDS-read "
          "dequant, per-channel scale (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("wq"), py::arg("wscale"),
          py::arg("bias") = py::none(), py::arg("part") = py::none(),
          py::arg("splitk") = 1);
    m.def("skinny_gemm3", &skinny_gemm3,
          "skinny-M GEMM v3: full-K register accum, depth-3 prefetch (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
    m.def("skinny_gemm2", &skinny_gemm2, "blocked skinny-M GEMM, K-split (CDNA4)",
          py::arg("y"), py::arg("y32"), py::arg("x"), py::arg("w"),
          py::arg("bias") = py::none(), py::arg("ksplit") = 1);
    m.def("fp8_quant", &fp8_quant, "dynamic per-tensor fp8 quant (CDNA4)");
    m.def("fp8_epilogue", &fp8_epilogue, "scaled-mm row-scale epilogue (CDNA4)",
          py::arg("out"), py::arg("y32"), py::arg("w_scale"),
          py::arg("bias") = py::none());
    m.def("moe_grouped_tables", &moe_grouped_tables_gpu,
          "fused MoE segment tables: one launch vs the torch op chain",
          py::arg("order"), py::arg("dest"), py::arg("expert_ids"),
          py::arg("gtab"), py::arg("idx_flat"), py::arg("num_experts"),
          py::arg("topk"), py::arg("tile") = 64);
    m.def("moe_grouped_gemm", &moe_grouped_gemm,
          "grouped expert GEMM over sorted padded token segments (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("expert_ids"),
          py::arg("bias") = py::none(), py::arg("bn") = 0,
          py::arg("x_gather") = py::none(),
          py::arg("combine_out") = py::none(),
          py::arg("combine_scale") = py::none(),
          py::arg("combine_gather") = py::none(), py::arg("bk") = 0,
          py::arg("mt") = 64);
    m.def("moe_grouped_gemm_fp8", &moe_grouped_gemm_fp8,
          "grouped expert GEMM, fp8-e4m3 weights dequantized at the LDS "
          "fragment read, per-channel scale in the epilogue (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("wq"), py::arg("wscale"),
          py::arg("expert_ids"), py::arg("bias") = py::none(),
          py::arg("bn") = 0, py::arg("x_gather") = py::none(),
          py::arg("combine_out") = py::none(),
          py::arg("combine_scale") = py::none(),
          py::arg("combine_gather") = py::none());
    m.def("moe_grouped_gemm_mx4", &moe_grouped_gemm_mx4,
          "grouped expert GEMM, MXFP4 packed weights dequantized in-kernel (CDNA4)",
          py::arg("y"), py::arg("x"), py::arg("wq"), py::arg("ws"),
          py::arg("expert_ids"), py::arg("bias") = py::none(),
          py::arg("bn") = 0, py::arg("x_gather") = py::none(),
          py::arg("combine_out") = py::none(),
          py::arg("combine_scale") = py::none(),
          py::arg("combine_gather") = py::none());
    m.def("router_topk", &router_topk,
          "fused MoE router top-k + activation + normalize (CDNA4)",
          py::arg("vals"), py::arg("idx"), py::arg("logits"),
          py::arg("top_k"), py::arg("act_sigmoid"), py::arg("pre_softmax"),
          py::arg("normalize"));
    m.def("lt_available", &lt_available, "hipblaslt usable on this device");
    m.def("lt_linear", &lt_linear, "autotuned hipBLASLt linear (graph-safe)",
          py::arg("y"), py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
    m.def("lt_tune", &lt_tune, "pre-tune one GEMM shape",
          py::arg("M"), py::arg("N"), py::arg("K"), py::arg("has_bias"),
          py::arg("sx"), py::arg("sw"), py::arg("sy"), py::arg("bias") = py::none());
    m.def("silu_mul_fused", &silu_mul_fused, "SiLU on fused gate|up (CDNA4)");
    m.def("gelu_tanh_mul_fused", &gelu_tanh_mul_fused, "GELU on fused gate|up (CDNA4)");
}
