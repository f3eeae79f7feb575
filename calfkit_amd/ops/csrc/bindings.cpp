// Python bindings for the calfkit_amd CDNA4 kernel set (gfx950 only).

#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor inp, torch::Tensor weight, double eps);
void fused_add_rmsnorm(torch::Tensor hidden, torch::Tensor residual,
                       torch::Tensor weight, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gateup);
void rope_qk(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
             torch::Tensor positions);
void rope_qkv_write(torch::Tensor qkv, torch::Tensor kcache,
                    torch::Tensor vcache, torch::Tensor cos_sin,
                    torch::Tensor positions, torch::Tensor slots,
                    int64_t hq, int64_t hkv);
void rope_qkv_write_slab(torch::Tensor qkv, torch::Tensor work,
                         torch::Tensor kcache, torch::Tensor vcache,
                         torch::Tensor cos_sin, torch::Tensor positions,
                         torch::Tensor slots, int64_t hq, int64_t hkv,
                         int64_t Mp, int64_t splitk);
void fused_add_rmsnorm_slab(torch::Tensor hidden, torch::Tensor residual,
                            torch::Tensor work, torch::Tensor weight,
                            double eps, int64_t Mp, int64_t splitk);
void skinny_gemm_slabs(torch::Tensor x, torch::Tensor w, torch::Tensor work,
                       int64_t splitk);
void kv_write(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
              torch::Tensor v, torch::Tensor slots);
void attn_decode(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                 torch::Tensor vcache, torch::Tensor page_table,
                 torch::Tensor seq_lens, double scale);
void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, double scale, int64_t past_len);
void attn_prefill_varlen(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                         torch::Tensor v, torch::Tensor block_map,
                         torch::Tensor q_starts, torch::Tensor kv_starts,
                         torch::Tensor past_lens, double scale);
void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor seed_step, torch::Tensor topk, torch::Tensor cand,
            torch::Tensor row_seeds);
void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor work, int64_t splitk);
void skinny_gemm_silu(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                      torch::Tensor work, int64_t splitk);
void moe_gateup_silu(torch::Tensor act, torch::Tensor x, torch::Tensor w,
                     torch::Tensor tok, torch::Tensor off, torch::Tensor work);
void moe_down(torch::Tensor y, torch::Tensor act, torch::Tensor w,
              torch::Tensor off, torch::Tensor scale);
void moe_route(torch::Tensor tok, torch::Tensor off, torch::Tensor scale,
               torch::Tensor inv, torch::Tensor logits, int64_t topk);
void moe_combine(torch::Tensor out, torch::Tensor y, torch::Tensor inv,
                 int64_t topk);
std::vector<std::tuple<int64_t, double>> hipblaslt_algo_bench(
    torch::Tensor out, torch::Tensor x, torch::Tensor w, int64_t iters,
    int64_t max_algos);
void hipblaslt_gemm_indexed(torch::Tensor out, torch::Tensor x,
                            torch::Tensor w, int64_t algo_index,
                            torch::Tensor workspace);
void tile_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               torch::Tensor work, int64_t bn, int64_t splitk);
void grammar_bias(torch::Tensor logits, torch::Tensor prev,
                  torch::Tensor gstate, torch::Tensor trigger,
                  torch::Tensor names, torch::Tensor pos,
                  torch::Tensor ncalls);
void tile_gemm_small(torch::Tensor out, torch::Tensor x, torch::Tensor w);
void tile_gemm_slabs(torch::Tensor x, torch::Tensor w, torch::Tensor work,
                     int64_t bn, int64_t splitk);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "residual += hidden; hidden = rmsnorm(residual)");
  m.def("silu_mul", &silu_mul, "silu(gate) * up from packed [T,2F]");
  m.def("rope_qk", &rope_qk, "in-place neox RoPE on Q and K");
  m.def("rope_qkv_write", &rope_qkv_write,
        "fused QKV epilogue: in-place RoPE + paged-cache K/V scatter");
  m.def("rope_qkv_write_slab", &rope_qkv_write_slab,
        "split-K reduce + RoPE + cache scatter in one launch");
  m.def("fused_add_rmsnorm_slab", &fused_add_rmsnorm_slab,
        "split-K reduce + residual add + RMSNorm in one launch");
  m.def("skinny_gemm_slabs", &skinny_gemm_slabs,
        "skinny GEMM, slabs only (consumer fuses the reduce)");
  m.def("kv_write", &kv_write, "scatter K/V rows into the paged cache");
  m.def("attn_decode", &attn_decode, "paged GQA decode attention");
  m.def("attn_prefill", &attn_prefill, "causal GQA prefill attention (MFMA)");
  m.def("attn_prefill_varlen", &attn_prefill_varlen,
        "packed varlen causal GQA prefill attention (MFMA)");
  m.def("sample", &sample, "Gumbel-max token sampling");
  m.def("skinny_gemm", &skinny_gemm,
        "decode projection GEMM: out = x @ w^T, w stored [N,K]");
  m.def("skinny_gemm_silu", &skinny_gemm_silu,
        "fused SiLU-MLP gate: out = silu(x@Wg^T) * (x@Wu^T)");
  m.def("moe_gateup_silu", &moe_gateup_silu,
        "grouped MoE gate-up: act[r] = silu/mul of x[tok[r]] @ W[e(r)]^T");
  m.def("moe_down", &moe_down,
        "grouped MoE down: y[r] = scale[r] * act[r] @ W[e(r)]^T");
  m.def("moe_route", &moe_route,
        "fused router: softmax -> top-K -> renorm -> stable expert buckets");
  m.def("moe_combine", &moe_combine,
        "out[t] = sum_j y[inv[t*K+j]] (deterministic scatter-combine)");
  m.def("hipblaslt_algo_bench", &hipblaslt_algo_bench,
        "time every supported hipBLASLt algo for out = x @ w^T");
  m.def("hipblaslt_gemm_indexed", &hipblaslt_gemm_indexed,
        "run out = x @ w^T with a searched hipBLASLt algo index");
  m.def("tile_gemm", &tile_gemm,
        "tiled MFMA GEMM for serving batches (M 65..512+), split-K reduce");
  m.def("tile_gemm_small", &tile_gemm_small,
        "64x128-tile MFMA GEMM, natural fill for M=384/512 projections");
  m.def("tile_gemm_slabs", &tile_gemm_slabs,
        "tiled MFMA GEMM, fp32 slabs only (consumer fuses the reduce)");
  m.def("grammar_bias", &grammar_bias,
        "tool-call grammar constrained-sampling bias (in-place on logits)");
}
