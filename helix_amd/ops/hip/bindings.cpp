// Python bindings for the helix_amd CDNA4 kernel library.
#include <torch/extension.h>

void rms_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps);
void fused_add_rms_norm(torch::Tensor x, torch::Tensor residual,
                        torch::Tensor w, double eps);
void layer_norm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                torch::Tensor b, double eps);
void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor cos_sin,
                      int64_t head_dim);
void silu_and_mul(torch::Tensor out, torch::Tensor x);
void rope_qkv_cache(torch::Tensor positions, torch::Tensor qkv,
                    torch::Tensor q_out,
                    c10::optional<torch::Tensor> k_cache,
                    c10::optional<torch::Tensor> v_cache,
                    c10::optional<torch::Tensor> slot_mapping,
                    c10::optional<torch::Tensor> k_out,
                    c10::optional<torch::Tensor> v_out,
                    torch::Tensor cos_sin, int64_t num_q_heads,
                    int64_t num_kv_heads, int64_t head_dim);
void gelu_tanh(torch::Tensor out, torch::Tensor x);
void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping);
void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_tables, torch::Tensor seq_lens,
                       double scale, torch::Tensor tmp_out,
                       torch::Tensor tmp_ml, int64_t partition_size,
                       int64_t window);
void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, torch::Tensor cu_seqlens,
                  torch::Tensor cu_seqlens_k, int64_t max_seqlen,
                  double scale, bool causal, int64_t window);
void sample_tokens(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperatures, torch::Tensor seeds);
void sample_tokens_ext(torch::Tensor out, torch::Tensor logits,
                       torch::Tensor temperatures, torch::Tensor seeds,
                       torch::Tensor top_p, torch::Tensor top_k,
                       torch::Tensor rep_pen, torch::Tensor pres_pen,
                       torch::Tensor freq_pen,
                       c10::optional<torch::Tensor> counts,
                       c10::optional<torch::Tensor> seen,
                       c10::optional<torch::Tensor> row_map);
void gemm_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               c10::optional<torch::Tensor> bias, int64_t act);
void gemm_skinny_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> bias,
                      c10::optional<torch::Tensor> scratch, int64_t split_k,
                      int64_t act, int64_t swizzle);
void gemm_fp8(torch::Tensor out, torch::Tensor x, torch::Tensor w,
              torch::Tensor x_scale, torch::Tensor w_scale,
              c10::optional<torch::Tensor> bias, int64_t act);
void mfma_probe(torch::Tensor d, torch::Tensor a, torch::Tensor b);
torch::Tensor ar_create(int64_t world, int64_t rank, int64_t capacity);
void ar_open(std::vector<torch::Tensor> handles);
int64_t ar_capacity();
void ar_allreduce(torch::Tensor inp, torch::Tensor out);
void ar_destroy();
void mfma_probe_fp8(torch::Tensor d, torch::Tensor a, torch::Tensor b,
                    int64_t scale_a, int64_t scale_b);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm,
        "Fused residual-add + RMSNorm (in-place)");
  m.def("layer_norm", &layer_norm, "LayerNorm (bf16)");
  m.def("rotary_embedding", &rotary_embedding,
        "Apply rotary embedding to q,k in-place");
  m.def("silu_and_mul", &silu_and_mul, "Fused SiLU-gated multiply");
  m.def("rope_qkv_cache", &rope_qkv_cache,
        "Fused strided-QKV rope + paged-cache write");
  m.def("gelu_tanh", &gelu_tanh, "GELU (tanh approx)");
  m.def("reshape_and_cache", &reshape_and_cache,
        "Scatter K/V rows into paged cache");
  m.def("paged_attn_decode", &paged_attn_decode,
        "Paged decode attention (GQA, flash-decoding partitions)");
  m.def("attn_prefill", &attn_prefill,
        "Varlen causal flash prefill attention (MFMA)");
  m.def("sample_tokens", &sample_tokens, "Greedy/Gumbel token sampling");
  m.def("sample_tokens_ext", &sample_tokens_ext,
        "Fused top-k/top-p/penalty Gumbel sampling (histogram threshold)");
  m.def("gemm_bf16", &gemm_bf16, "MFMA bf16 GEMM: x @ w^T (+bias, act)");
  m.def("gemm_skinny_bf16", &gemm_skinny_bf16,
        "Skinny-M decode GEMM: split-K + XCD-chunked tile swizzle");
  m.def("gemm_fp8", &gemm_fp8,
        "MX-fp8 e4m3 MFMA GEMM with epilogue per-row/col dequant");
  m.def("mfma_probe", &mfma_probe, "16x16x32 MFMA layout probe");
  m.def("ar_create", &ar_create,
        "Allocate one-shot allreduce mailbox; returns IPC handle bytes");
  m.def("ar_open", &ar_open, "Map peer mailboxes from IPC handles");
  m.def("ar_capacity", &ar_capacity, "Opened AR data capacity in bytes");
  m.def("ar_allreduce", &ar_allreduce,
        "One-shot bf16 allreduce over xGMI peer mailboxes");
  m.def("ar_destroy", &ar_destroy, "Tear down the allreduce context");
  m.def("mfma_probe_fp8", &mfma_probe_fp8,
        "16x16x128 MX-fp8 MFMA layout probe (unity e8m0 scales)");
}
