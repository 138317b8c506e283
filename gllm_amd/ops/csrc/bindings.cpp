// Python bindings for the gfx950 kernel extension.
#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight,
             double eps);
void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps);
void silu_and_mul(torch::Tensor out, torch::Tensor x);
void gelu_and_mul(torch::Tensor out, torch::Tensor x);
void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, long head_dim,
                      torch::Tensor cos_sin_cache, bool is_neox);
void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping);
void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_table, torch::Tensor seq_lens,
                            double scale, long sliding_window);
void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> bias,
                 torch::Tensor workspace, long splitk);
void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_table,
                             torch::Tensor seq_lens,
                             torch::Tensor query_start_loc,
                             long max_query_len, double scale,
                             long sliding_window);
void mla_paged_attention(torch::Tensor out, torch::Tensor q,
                         torch::Tensor k_cache, torch::Tensor block_table,
                         torch::Tensor seq_lens,
                         torch::Tensor query_start_loc, long max_query_len,
                         double scale, long max_seq_len);
void cache_latent(torch::Tensor k, torch::Tensor k_cache,
                  torch::Tensor slot_mapping);
void moe_align(torch::Tensor topk_ids, long E_local, long expert_start,
               long block_m, torch::Tensor sorted_ids,
               torch::Tensor expert_blocks, torch::Tensor n_post_pad);
void moe_gemm(torch::Tensor C, torch::Tensor A, torch::Tensor W,
              torch::Tensor sorted_ids, torch::Tensor expert_blocks,
              torch::Tensor n_post_pad,
              c10::optional<torch::Tensor> topk_weights, long n_pairs,
              long topk, long block_m, bool scatter);
void moe_sum(torch::Tensor out, torch::Tensor pair_out, long topk);
void per_token_group_quant_fp8(torch::Tensor x, torch::Tensor q,
                               torch::Tensor scales,
                               c10::optional<torch::Tensor> scales_t,
                               bool ue8m0);
void fp8_skinny_gemm(torch::Tensor out, torch::Tensor aq, torch::Tensor as,
                     torch::Tensor w, torch::Tensor ws,
                     c10::optional<torch::Tensor> bias,
                     torch::Tensor workspace, long splitk);
void moe_gemm_fp8(torch::Tensor C, torch::Tensor A, torch::Tensor As,
                  torch::Tensor W, torch::Tensor Ws,
                  torch::Tensor sorted_ids, torch::Tensor expert_blocks,
                  torch::Tensor n_post_pad,
                  c10::optional<torch::Tensor> topk_weights, long n_pairs,
                  long topk, long block_m, bool scatter);
void int4_skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor wq,
                      torch::Tensor sb, c10::optional<torch::Tensor> bias,
                      torch::Tensor workspace, long group);
void moe_gemm_int4(torch::Tensor C, torch::Tensor A, torch::Tensor W,
                   torch::Tensor SB, torch::Tensor sorted_ids,
                   torch::Tensor expert_blocks, torch::Tensor n_post_pad,
                   c10::optional<torch::Tensor> topk_weights, long n_pairs,
                   long topk, long block_m, bool scatter);
void gdn_conv_update(torch::Tensor out, torch::Tensor x,
                     torch::Tensor weight, torch::Tensor conv_state,
                     torch::Tensor slots);
void gdn_chunk_prefill(torch::Tensor o, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor g, torch::Tensor beta,
                       torch::Tensor states, double scale);
void gdn_decode(torch::Tensor o, torch::Tensor qn, torch::Tensor kn,
                torch::Tensor v, torch::Tensor g, torch::Tensor beta,
                torch::Tensor state, torch::Tensor slots);
void rmsnorm_gated(torch::Tensor out, torch::Tensor x, torch::Tensor z,
                   torch::Tensor w, double eps);
void apply_repetition_penalty(torch::Tensor logits, torch::Tensor pool,
                              torch::Tensor slots,
                              torch::Tensor penalties);
void topk_topp_filter(torch::Tensor probs, torch::Tensor top_ks,
                      torch::Tensor top_ps,
                      c10::optional<torch::Tensor> min_ps);
std::pair<int64_t, py::bytes> car_alloc(int64_t data_bytes);
std::pair<int64_t, py::bytes> car_alloc_v2(int64_t max_bytes);
void car_all_reduce_v2(torch::Tensor inout, std::vector<int64_t> ptrs,
                       int64_t rank, int64_t world, int64_t max_bytes);
int64_t car_open(py::bytes handle_bytes);
void car_close(int64_t ptr);
void car_free(int64_t ptr);
torch::Tensor car_view_tensor(int64_t ptr, int64_t numel);
void car_all_reduce(torch::Tensor inout, std::vector<int64_t> ptrs,
                    int64_t rank, int64_t world, int64_t epoch);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (gfx950)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "residual += x; x = rmsnorm(residual)");
  m.def("silu_and_mul", &silu_and_mul, "silu(x[:d]) * x[d:]");
  m.def("gelu_and_mul", &gelu_and_mul, "gelu(x[:d]) * x[d:]");
  m.def("rotary_embedding", &rotary_embedding, "in-place RoPE on q,k");
  m.def("reshape_and_cache", &reshape_and_cache,
        "scatter K/V into paged cache");
  m.def("paged_attention_decode", &paged_attention_decode,
        "split-KV paged decode attention");
  m.def("paged_attention_prefill", &paged_attention_prefill,
        "varlen MFMA paged prefill attention");
  m.def("skinny_gemm", &skinny_gemm,
        "decode-regime weight-streaming GEMM (M<=256)");
  m.def("mla_paged_attention", &mla_paged_attention,
        "absorbed-MLA varlen attention over the 576-dim latent cache");
  m.def("cache_latent", &cache_latent,
        "scatter the per-token latent row into the paged cache");
  m.def("moe_align", &moe_align,
        "device-side moe_align_block_size (graph-safe)");
  m.def("moe_gemm", &moe_gemm,
        "grouped MFMA GEMM over sorted (token, expert) pairs");
  m.def("moe_sum", &moe_sum, "sum pair outputs over topk");
  m.def("per_token_group_quant_fp8", &per_token_group_quant_fp8,
        "per (token, 128-group) e4m3 quant + fp32 scales");
  m.def("fp8_skinny_gemm", &fp8_skinny_gemm,
        "block-scale fp8 weight-streaming GEMM (decode)");
  m.def("moe_gemm_fp8", &moe_gemm_fp8,
        "block-scale fp8 grouped MoE GEMM");
  m.def("int4_skinny_gemm", &int4_skinny_gemm,
        "fused-dequant int4 weight-streaming GEMM (decode)");
  m.def("moe_gemm_int4", &moe_gemm_int4,
        "fused-dequant int4 grouped MoE GEMM (w4a16)");
  m.def("gdn_conv_update", &gdn_conv_update,
        "batched causal-conv1d decode step w/ state roll");
  m.def("gdn_chunk_prefill", &gdn_chunk_prefill,
        "fused WY chunk-parallel gated delta rule prefill");
  m.def("gdn_decode", &gdn_decode,
        "batched fused recurrent gated-delta-rule decode step");
  m.def("rmsnorm_gated", &rmsnorm_gated, "rmsnorm(x)*w*silu(z)");
  m.def("apply_repetition_penalty", &apply_repetition_penalty,
        "scaling penalty vs persistent seen-token mask pool");
  m.def("topk_topp_filter", &topk_topp_filter,
        "fused sorting-free top-k/top-p/min-p filter + renormalize");
  m.def("car_alloc", &car_alloc, "alloc hipIpc-shared AR buffer");
  m.def("car_alloc_v2", &car_alloc_v2,
        "alloc v2 AR buffer (device epoch + out slices)");
  m.def("car_all_reduce_v2", &car_all_reduce_v2,
        "two-shot/one-shot xGMI AR, device-epoch (graph-capturable)");
  m.def("car_open", &car_open, "map a peer's AR buffer");
  m.def("car_close", &car_close);
  m.def("car_free", &car_free);
  m.def("car_view_tensor", &car_view_tensor,
        "non-owning bf16 view over raw device memory");
  m.def("car_all_reduce", &car_all_reduce,
        "one-shot xGMI custom all-reduce (epoch advances by 2/call)");
}
