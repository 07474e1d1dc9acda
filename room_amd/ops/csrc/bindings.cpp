// Python bindings for room_amd._C — the CDNA4 kernel library.
#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight, double eps);
void fused_add_rmsnorm(torch::Tensor out, torch::Tensor residual, torch::Tensor in,
                       torch::Tensor weight, double eps);
void qk_norm_rope(torch::Tensor q, torch::Tensor k, torch::Tensor q_w,
                  torch::Tensor k_w, torch::Tensor cos_t, torch::Tensor sin_t,
                  torch::Tensor positions, int64_t n_qheads, int64_t n_kvheads,
                  int64_t head_dim, double eps);
void qk_rope_write_kv(torch::Tensor qkv, torch::Tensor kcache,
                      torch::Tensor vcache, torch::Tensor q_w, torch::Tensor k_w,
                      torch::Tensor cos_t, torch::Tensor sin_t,
                      torch::Tensor block_table, torch::Tensor seq_ids,
                      torch::Tensor positions, int64_t n_qheads, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gateup);
void paged_attention(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor block_table,
                     torch::Tensor seq_ids, torch::Tensor q_pos, double scale);
void write_kv(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
              torch::Tensor v, torch::Tensor block_table, torch::Tensor seq_ids,
              torch::Tensor q_pos);
void moe_router(torch::Tensor topk_ids, torch::Tensor topk_w, torch::Tensor logits,
                int64_t K);
void router_topk(torch::Tensor topk_ids, torch::Tensor topk_w, torch::Tensor x,
                 torch::Tensor wr, int64_t K);
void router_gemv_topk(torch::Tensor topk_ids, torch::Tensor topk_w,
                      torch::Tensor x, torch::Tensor wr, int64_t K);
void moe_gemv_h(torch::Tensor h, torch::Tensor x, torch::Tensor w13,
                torch::Tensor pair_token, torch::Tensor pair_expert,
                torch::Tensor out_zero);
void moe_gemv_down(torch::Tensor out, torch::Tensor h, torch::Tensor w2,
                   torch::Tensor pair_w, torch::Tensor pair_token,
                   torch::Tensor pair_expert);
void moe_grouped_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                      torch::Tensor pair_token, torch::Tensor tile_desc);
void moe_grouped_gemm128(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                         torch::Tensor pair_token, torch::Tensor tile_desc,
                         int64_t bm);
void moe_gemv_dedup(torch::Tensor out, torch::Tensor x, torch::Tensor w13,
                    torch::Tensor w2, torch::Tensor topk_ids,
                    torch::Tensor topk_w, torch::Tensor counts,
                    torch::Tensor tok_list, torch::Tensor w_list,
                    torch::Tensor h, torch::Tensor active);
void moe_build_desc(torch::Tensor desc, torch::Tensor counts, int64_t bm);
void moe_combine(torch::Tensor out, torch::Tensor z, torch::Tensor pair_w,
                 torch::Tensor pair_token);
void moe_combine_gather(torch::Tensor out, torch::Tensor z, torch::Tensor topk_w,
                        torch::Tensor inv_order);
void sample_tokens(torch::Tensor out_tokens, torch::Tensor logits,
                   torch::Tensor seeds, int64_t top_k, double temperature,
                   double top_p);
void paged_attention_split(torch::Tensor out, torch::Tensor q,
                           torch::Tensor kcache, torch::Tensor vcache,
                           torch::Tensor block_table, torch::Tensor seq_ids,
                           torch::Tensor q_pos, torch::Tensor part,
                           torch::Tensor part_ml, double scale,
                           int64_t splits);
int64_t attn_nsplits();
void paged_attention_splitk(torch::Tensor part, torch::Tensor part_ml,
                            torch::Tensor q, torch::Tensor kcache,
                            torch::Tensor vcache, torch::Tensor block_table,
                            torch::Tensor seq_ids, torch::Tensor q_pos,
                            double scale, torch::Tensor o_zero,
                            int64_t splits);
void attn_merge_o(torch::Tensor o_accum, torch::Tensor part,
                  torch::Tensor part_ml, torch::Tensor wo, int64_t splits);
void router_addnorm(torch::Tensor y, torch::Tensor x, torch::Tensor delta,
                    torch::Tensor x_out, torch::Tensor xn_out,
                    torch::Tensor gamma, torch::Tensor w, double eps);
int64_t flash_prefill_qtile();
void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                   torch::Tensor vcache, torch::Tensor block_table,
                   torch::Tensor seq_ids, torch::Tensor q_pos,
                   torch::Tensor tile_desc, double scale);
void gemv(torch::Tensor y, torch::Tensor x, torch::Tensor w);
void gemv_addnorm(torch::Tensor y, torch::Tensor x, torch::Tensor delta,
                  torch::Tensor x_out, torch::Tensor gamma, torch::Tensor w,
                  double eps);
void sample_tokens_v3(torch::Tensor out_tokens, torch::Tensor logits,
                      torch::Tensor seeds, int64_t top_k, double temperature,
                      double top_p);
void sample_tokens_v4(torch::Tensor out_tokens, torch::Tensor logits,
                      torch::Tensor seeds, int64_t top_k, double temperature,
                      double top_p);
void sample_scan_probe(torch::Tensor out, torch::Tensor logits);
void hist_append(torch::Tensor hist, torch::Tensor ctr, torch::Tensor toks,
                 int64_t kmax);
void vs_topk(torch::Tensor out_v, torch::Tensor out_i, torch::Tensor cand_v,
             torch::Tensor cand_i, torch::Tensor mat, torch::Tensor query,
             int64_t K);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm, "residual += x; rmsnorm(residual)");
  m.def("qk_norm_rope", &qk_norm_rope, "fused per-head QK RMSNorm + RoPE");
  m.def("qk_rope_write_kv", &qk_rope_write_kv,
        "fused QK-norm + RoPE + paged KV write");
  m.def("silu_mul", &silu_mul, "silu(gate) * up");
  m.def("paged_attention", &paged_attention, "paged causal attention (GQA 8:1)");
  m.def("write_kv", &write_kv, "scatter K/V rows into paged cache");
  m.def("moe_router", &moe_router, "softmax top-k router");
  m.def("router_topk", &router_topk, "fused router GEMV + softmax top-k (decode)");
  m.def("router_gemv_topk", &router_gemv_topk,
        "H-split router GEMV + top-k (two kernels, wide grid)");
  m.def("moe_gemv_h", &moe_gemv_h, "MoE gate/up GEMV + silu-mul (decode)");
  m.def("moe_gemv_down", &moe_gemv_down, "MoE down GEMV + weighted scatter-add");
  m.def("moe_grouped_gemm", &moe_grouped_gemm, "grouped MFMA GEMM (prefill)");
  m.def("moe_grouped_gemm128", &moe_grouped_gemm128, "BM=128 grouped MFMA GEMM");
  m.def("moe_gemv_dedup", &moe_gemv_dedup, "expert-deduped decode MoE");
  m.def("moe_build_desc", &moe_build_desc, "device-side tile desc builder");
  m.def("moe_combine", &moe_combine, "weighted scatter-add combine");
  m.def("moe_combine_gather", &moe_combine_gather, "atomics-free MoE combine");
  m.def("sample_tokens", &sample_tokens, "fused temperature/top-k/top-p sampling");
  m.def("attn_nsplits", &attn_nsplits, "NSPLITS constant");
  m.def("paged_attention_splitk", &paged_attention_splitk,
        "split-KV decode attention, partials only (+O-accum zero side-job)");
  m.def("attn_merge_o", &attn_merge_o,
        "fused split-partial merge + O-projection (f32 atomic accum)");
  m.def("router_addnorm", &router_addnorm,
        "fused residual-add + RMSNorm + router logits (+residual/xn out)");
  m.def("paged_attention_split", &paged_attention_split,
        "split-KV flash-decode paged attention");
  m.def("flash_prefill", &flash_prefill, "MFMA flash-attention prefill");
  m.def("flash_prefill_qtile", &flash_prefill_qtile, "q-tile rows constant");
  m.def("gemv", &gemv, "dense skinny-batch GEMV (decode projections)");
  m.def("gemv_addnorm", &gemv_addnorm,
        "fused residual-add + RMSNorm + GEMV (decode)");
  m.def("sample_tokens_v3", &sample_tokens_v3, "register top-8 sampler");
  m.def("sample_tokens_v4", &sample_tokens_v4,
        "two-stage parallel-scan sampler");
  m.def("sample_scan_probe", &sample_scan_probe, "scan-cost probe");
  m.def("hist_append", &hist_append, "token-history append (multi-step decode)");
  m.def("vs_topk", &vs_topk, "vector-store cosine top-k over bf16 matrix");
}
