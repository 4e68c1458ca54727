// Python bindings for the aurora_amd gfx950 kernel set.
#include <torch/extension.h>

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor weight, double eps);
std::vector<torch::Tensor> rmsnorm_residual(torch::Tensor x, torch::Tensor res,
                                            torch::Tensor weight, double eps);
torch::Tensor silu_mul(torch::Tensor gu);
void rope_apply_(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                 torch::Tensor cos_sin);
void kv_cache_write(torch::Tensor k, torch::Tensor v, torch::Tensor kc,
                    torch::Tensor vc, torch::Tensor slots);
torch::Tensor paged_attention_decode(torch::Tensor q, torch::Tensor kc,
                                     torch::Tensor vc, torch::Tensor bt,
                                     torch::Tensor ctx, double scale);
torch::Tensor attention_prefill_paged(torch::Tensor q, torch::Tensor kc,
                                      torch::Tensor vc, torch::Tensor bt,
                                      torch::Tensor cu_q, torch::Tensor ctx,
                                      double scale);
torch::Tensor attention_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                torch::Tensor cu, double scale, bool causal);
torch::Tensor embed_pool(torch::Tensor hidden, torch::Tensor cu);
std::vector<torch::Tensor> knn_topk(torch::Tensor query, torch::Tensor docs, long k);
torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w);
torch::Tensor fused_rope_kv_write(torch::Tensor qkv, torch::Tensor positions,
                                  torch::Tensor slots, torch::Tensor cos_sin,
                                  torch::Tensor kc, torch::Tensor vc,
                                  long Hq, long Hkv);
torch::Tensor skinny_gemm_ex(torch::Tensor x, torch::Tensor w, long s_target, long unroll);
torch::Tensor skinny_gemm2_ex(torch::Tensor x, torch::Tensor w, long s_target);
torch::Tensor skinny_gemm3_ex(torch::Tensor x, torch::Tensor w, long s_target);
torch::Tensor moe_grouped_gemm(torch::Tensor xp, torch::Tensor w,
                               torch::Tensor starts, torch::Tensor counts);
std::vector<uint8_t> oneshot_init(long world, long rank, long max_bytes);
void oneshot_connect(std::vector<std::vector<uint8_t>> handles);
bool oneshot_ready();
long oneshot_max_bytes();
void oneshot_allreduce_(torch::Tensor t);
void oneshot_destroy();

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "fused RMSNorm (bf16, gfx950)");
  m.def("rmsnorm_residual", &rmsnorm_residual, "fused residual-add + RMSNorm");
  m.def("silu_mul", &silu_mul, "fused SiLU(gate)*up");
  m.def("rope_apply_", &rope_apply_, "in-place RoPE on q/k");
  m.def("kv_cache_write", &kv_cache_write, "scatter K/V into paged cache");
  m.def("paged_attention_decode", &paged_attention_decode, "paged GQA decode attention");
  m.def("attention_prefill_paged", &attention_prefill_paged,
        "MFMA flash prefill over paged KV");
  m.def("attention_prefill", &attention_prefill, "varlen self-attention (encoder)");
  m.def("embed_pool", &embed_pool, "mean-pool + L2 normalize");
  m.def("knn_topk", &knn_topk, "brute-force kNN top-k");
  m.def("skinny_gemm", &skinny_gemm, "decode-path skinny GEMM (M<=64)");
  m.def("fused_rope_kv_write", &fused_rope_kv_write,
        "fused qkv split + RoPE + paged KV scatter");
  m.def("skinny_gemm_ex", &skinny_gemm_ex, "skinny GEMM with tuning knobs");
  m.def("skinny_gemm2_ex", &skinny_gemm2_ex, "LDS-staged skinny GEMM (M<=64)");
  m.def("skinny_gemm3_ex", &skinny_gemm3_ex, "glds double-buffered projection GEMM (M<=256)");
  m.def("moe_grouped_gemm", &moe_grouped_gemm,
        "packed per-expert grouped GEMM (skips unrouted experts)");
  m.def("oneshot_init", &oneshot_init, "one-shot allreduce: alloc + IPC handle");
  m.def("oneshot_connect", &oneshot_connect, "one-shot allreduce: map peer buffers");
  m.def("oneshot_ready", &oneshot_ready);
  m.def("oneshot_max_bytes", &oneshot_max_bytes);
  m.def("oneshot_allreduce_", &oneshot_allreduce_, "in-place one-shot xGMI all-reduce (bf16)");
  m.def("oneshot_destroy", &oneshot_destroy);
}
