// Encoder-path kernels (gfx950): varlen self-attention without cache
// (bidirectional for the BGE-style embedder), mean-pool + L2 normalize.
//
// Replaces the reference's Weaviate text2vec-transformers sidecar
// (reference: docker-compose.yaml:545, routes/knowledge_base/weaviate_client.py:115)
// with on-GPU embedding.  Encoder sequences are short (<=512) and this is
// not the serving hot path, so the attention kernel favours simplicity:
// one wave per (query token, head), lanes in D/8-lane groups streaming keys
// at 16 B/lane with an online-softmax partial per group, merged in LDS.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

template <int D>
__global__ __launch_bounds__(256) void varlen_attn_kernel(
    const __hip_bfloat16* __restrict__ q,  // [T, Hq, D]
    const __hip_bfloat16* __restrict__ k,  // [T, Hkv, D]
    const __hip_bfloat16* __restrict__ v,
    __hip_bfloat16* __restrict__ out,
    const int* __restrict__ cu,        // [S+1]
    const int* __restrict__ tok_seq,   // [T] seq index of each token
    long n_items, int Hq, int Hkv, float scale, int causal) {
  constexpr int GL = D / 8;
  constexpr int TPW = WAVE / GL;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int grp = lane / GL;
  const int sub = lane % GL;
  const long item = (long)blockIdx.x * (blockDim.x / WAVE) + wave;
  if (item >= n_items) return;
  const int tq = item / Hq;     // global q token index
  const int h = item % Hq;
  const int kvh = h / (Hq / Hkv);
  const int seq = tok_seq[tq];
  const int a = cu[seq], b = cu[seq + 1];
  const int kmax = causal ? (tq + 1) : b;  // keys [a, kmax)

  shortx8 qf = *reinterpret_cast<const shortx8*>(q + ((long)tq * Hq + h) * D + sub * 8);
  float m = -1e30f, l = 0.f, o[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) o[j] = 0.f;

  for (int t = a + grp; t < kmax; t += TPW) {
    const long row = ((long)t * Hkv + kvh) * D + sub * 8;
    const shortx8 kf = *reinterpret_cast<const shortx8*>(k + row);
    const shortx8 vf = *reinterpret_cast<const shortx8*>(v + row);
    float s = group_sum<GL>(dot8_bf16(kf, qf)) * scale;
    const float mn = fmaxf(m, s);
    const float alpha = __expf(m - mn);
    const float p = __expf(s - mn);
    m = mn;
    l = l * alpha + p;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = o[j] * alpha + p * bf16_to_f32((unsigned short)vf[j]);
  }
  // merge the TPW group partials of this wave in LDS
  __shared__ float s_m[4][TPW], s_l[4][TPW], s_o[4][TPW][D];
  if (sub == 0) {
    s_m[wave][grp] = m;
    s_l[wave][grp] = l;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) s_o[wave][grp][sub * 8 + j] = o[j];
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  float M = -1e30f;
#pragma unroll
  for (int i = 0; i < TPW; ++i) M = fmaxf(M, s_m[wave][i]);
  for (int d = lane; d < D; d += WAVE) {
    float L = 0.f, O = 0.f;
#pragma unroll
    for (int i = 0; i < TPW; ++i) {
      const float w = __expf(s_m[wave][i] - M);
      L += w * s_l[wave][i];
      O += w * s_o[wave][i][d];
    }
    *(unsigned short*)(out + ((long)tq * Hq + h) * D + d) = f32_to_bf16(O / L);
  }
}

torch::Tensor attention_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                torch::Tensor cu, double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16, "bf16 GPU only");
  const int T = q.size(0), Hq = q.size(1), D = q.size(2), Hkv = k.size(1);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64/128");
  auto out = torch::empty_like(q);
  if (T == 0) return out;
  // token -> seq map on host (cu is small)
  auto cu_cpu = cu.to(torch::kCPU);
  const int* cup = cu_cpu.data_ptr<int>();
  const int S = cu_cpu.numel() - 1;
  std::vector<int> tok_seq(T);
  for (int s = 0; s < S; ++s)
    for (int t = cup[s]; t < cup[s + 1]; ++t) tok_seq[t] = s;
  auto ts = torch::from_blob(tok_seq.data(), {T}, torch::kInt32).to(q.device());
  const long items = (long)T * Hq;
  dim3 grid((items + 3) / 4, 1), block(256);
  auto stream = CUR_STREAM;
  if (D == 64)
    hipLaunchKernelGGL((varlen_attn_kernel<64>), grid, block, 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(), cu.data_ptr<int>(),
                       ts.data_ptr<int>(), items, Hq, Hkv, (float)scale, causal ? 1 : 0);
  else
    hipLaunchKernelGGL((varlen_attn_kernel<128>), grid, block, 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(), cu.data_ptr<int>(),
                       ts.data_ptr<int>(), items, Hq, Hkv, (float)scale, causal ? 1 : 0);
  HIP_CHECK_KERNEL();
  return out;
}

// ---- mean-pool + L2 normalize per sequence -------------------------------

__global__ void embed_pool_kernel(const __hip_bfloat16* __restrict__ hidden,
                                  __hip_bfloat16* __restrict__ out,
                                  const int* __restrict__ cu, int H) {
  const int seq = blockIdx.x;
  const int a = cu[seq], b = cu[seq + 1];
  const int tid = threadIdx.x;
  __shared__ float s_norm[16];
  __shared__ float s_inv;
  // each thread owns hidden dims tid, tid+256, ...
  float acc[8];  // supports H <= 2048 at 256 threads
  const int nchunk = (H + blockDim.x - 1) / blockDim.x;
  for (int c = 0; c < nchunk; ++c) acc[c] = 0.f;
  for (int t = a; t < b; ++t)
    for (int c = 0; c < nchunk; ++c) {
      const int d = tid + c * blockDim.x;
      if (d < H)
        acc[c] += bf16_to_f32(*(const unsigned short*)(hidden + (long)t * H + d));
    }
  const float inv_n = 1.f / max(b - a, 1);
  float ss = 0.f;
  for (int c = 0; c < nchunk; ++c) {
    acc[c] *= inv_n;
    ss += acc[c] * acc[c];
  }
  ss = wave_sum(ss);
  if (tid % WAVE == 0) s_norm[tid / WAVE] = ss;
  __syncthreads();
  if (tid == 0) {
    float t = 0.f;
    for (int w = 0; w < blockDim.x / WAVE; ++w) t += s_norm[w];
    s_inv = rsqrtf(t + 1e-24f);
  }
  __syncthreads();
  for (int c = 0; c < nchunk; ++c) {
    const int d = tid + c * blockDim.x;
    if (d < H)
      *(unsigned short*)(out + (long)seq * H + d) = f32_to_bf16(acc[c] * s_inv);
  }
}

torch::Tensor embed_pool(torch::Tensor hidden, torch::Tensor cu) {
  TORCH_CHECK(hidden.is_cuda() && hidden.dtype() == torch::kBFloat16, "bf16 GPU only");
  const int H = hidden.size(1);
  TORCH_CHECK(H <= 2048, "embed_pool supports H <= 2048");
  const int S = cu.numel() - 1;
  auto out = torch::empty({S, H}, hidden.options());
  if (S == 0) return out;
  hipLaunchKernelGGL(embed_pool_kernel, dim3(S), dim3(256), 0, CUR_STREAM,
                     (const __hip_bfloat16*)hidden.data_ptr(),
                     (__hip_bfloat16*)out.data_ptr(), cu.data_ptr<int>(), H);
  HIP_CHECK_KERNEL();
  return out;
}

// ---- brute-force kNN over the HBM-resident doc matrix --------------------
// Scores via hipBLASLt GEMM (a plain library GEMM) + torch topk; the win on
// MI355X is that the whole corpus lives in 288 GB HBM.  A fused
// score+select kernel is a later optimisation (profiles will say if the
// [Q, N] score materialisation ever matters).

std::vector<torch::Tensor> knn_topk(torch::Tensor query, torch::Tensor docs, long k) {
  TORCH_CHECK(query.is_cuda(), "GPU only");
  auto scores = torch::mm(query.to(docs.dtype()), docs.t()).to(torch::kFloat32);
  k = std::min<long>(k, docs.size(0));
  auto [vals, idx] = scores.topk(k, -1);
  return {vals, idx};
}
