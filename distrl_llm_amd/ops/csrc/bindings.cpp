// Python bindings for the gfx950 kernel library.
#include <torch/extension.h>
#include <vector>

torch::Tensor rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
torch::Tensor rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                          double eps);
std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor h,
                                           torch::Tensor w, double eps);
torch::Tensor silu_mul_fwd(torch::Tensor gate, torch::Tensor up);
std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor gate,
                                        torch::Tensor up);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor inv_freq);
void rope_scatter_qkv(torch::Tensor qkv, torch::Tensor positions,
                      torch::Tensor slots, torch::Tensor inv_freq,
                      torch::Tensor key_cache, torch::Tensor value_cache,
                      int64_t H, int64_t KV, int64_t D);
torch::Tensor silu_mul_packed(torch::Tensor gu);
torch::Tensor prefill_attention(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, torch::Tensor tile_q0,
                                torch::Tensor tile_rows, torch::Tensor tile_kv0,
                                int64_t max_len, double scale);
torch::Tensor paged_attention_decode_strided(
    torch::Tensor q, int64_t n_heads, int64_t head_dim, int64_t q_row_stride,
    torch::Tensor kcache, torch::Tensor vcache, torch::Tensor block_tables,
    torch::Tensor ctx_lens, double scale);
void kv_cache_scatter(torch::Tensor k, torch::Tensor v,
                      torch::Tensor key_cache, torch::Tensor value_cache,
                      torch::Tensor slot_mapping);
torch::Tensor paged_attention_decode(torch::Tensor q, torch::Tensor kcache,
                                     torch::Tensor vcache,
                                     torch::Tensor block_tables,
                                     torch::Tensor ctx_lens, double scale);
torch::Tensor sample_tokens(torch::Tensor logits, double temperature,
                            double top_p, int64_t top_k, torch::Tensor seeds,
                            torch::Tensor step);
torch::Tensor sample_tokens2(torch::Tensor logits, double temperature,
                             double top_p, int64_t top_k, torch::Tensor seeds,
                             torch::Tensor step);
std::vector<torch::Tensor> logprob_lse_fwd(torch::Tensor logits,
                                           torch::Tensor targets);
torch::Tensor logprob_loss_bwd(torch::Tensor logits, torch::Tensor targets,
                               torch::Tensor w, torch::Tensor lse);
torch::Tensor nf4_gemm(torch::Tensor x, torch::Tensor w4f, torch::Tensor amaxf,
                       c10::optional<torch::Tensor> bias,
                       c10::optional<torch::Tensor> u,
                       c10::optional<torch::Tensor> bfrag,
                       int64_t N, int64_t K, int64_t r);
void lora_u(torch::Tensor x, torch::Tensor afrag, torch::Tensor u,
            int64_t r, int64_t ksplit);
torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b);
std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, double scale);
std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor o, torch::Tensor lse,
                                          double scale);
void adam8bit_step(torch::Tensor p, torch::Tensor g, torch::Tensor m_q,
                   torch::Tensor v_q, torch::Tensor m_absmax,
                   torch::Tensor v_absmax, double lr, double b1, double b2,
                   double eps, double wd, int64_t step);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (gfx950)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward dx (gfx950)");
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd,
        "fused residual add + RMSNorm forward");
  m.def("silu_mul_fwd", &silu_mul_fwd, "fused SiLU*mul forward");
  m.def("silu_mul_bwd", &silu_mul_bwd, "fused SiLU*mul backward");
  m.def("rope_inplace", &rope_inplace, "fused in-place RoPE (q,k)");
  m.def("rope_scatter_qkv", &rope_scatter_qkv,
        "fused packed-qkv RoPE + paged KV scatter");
  m.def("silu_mul_packed", &silu_mul_packed, "SiLU*mul on packed [gate|up]");
  m.def("paged_attention_decode_strided", &paged_attention_decode_strided,
        "paged GQA decode attention (packed-q rows)");
  m.def("prefill_attention", &prefill_attention,
        "varlen causal prefill attention (prompt phase)");
  m.def("kv_cache_scatter", &kv_cache_scatter, "paged KV cache scatter");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged GQA decode attention");
  m.def("sample_tokens", &sample_tokens,
        "fused temperature/top-k/top-p categorical sampling");
  m.def("sample_tokens2", &sample_tokens2,
        "two-stage fused sampler (slice-parallel; same semantics/RNG)");
  m.def("logprob_lse_fwd", &logprob_lse_fwd,
        "per-token log p(target) + LSE, one streaming pass");
  m.def("logprob_loss_bwd", &logprob_loss_bwd,
        "dlogits for the masked PG/GRPO loss");
  m.def("adam8bit_step", &adam8bit_step, "fused blockwise 8-bit Adam step");
  m.def("nf4_gemm", &nf4_gemm,
        "fused nf4-dequant MFMA GEMM with bias + LoRA-B epilogue");
  m.def("lora_u", &lora_u, "split-K LoRA A projection (u = x @ A^T)");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA mapping probe");
  m.def("flash_attn_fwd", &flash_attn_fwd,
        "causal GQA flash attention forward (returns o, lse)");
  m.def("flash_attn_bwd", &flash_attn_bwd,
        "causal GQA flash attention backward (returns dq, dk, dv)");
}
