// Paged-KV decode attention for gfx950 (one new token per sequence).
//
// Memory-bound by the KV read: the kernel streams each (seq, kv_head)'s
// pages exactly once at 16 B/lane and amortises that read over all G
// grouped query heads (GQA), which is the bandwidth win that matters on
// MI355X (8 TB/s HBM3E, 64-wide waves).
//
// Geometry: grid = (num_seqs, n_kv_heads); 256 threads = 4 waves.
//   * a wave's 64 lanes form TPW = 64/GL token-groups of GL = D/8 lanes;
//     each group owns one token per inner step (16 B of K then V per lane);
//   * waves stride the context in round-robin page chunks; each (wave,
//     group) keeps an online-softmax partial (m, l, o[8]/lane) per q head;
//   * partials merge through LDS at the end (flash-decode combine).
//
// Replaces the per-token provider round-trip of the reference agent loop
// (reference: server/chat/backend/agent/llm.py:192) with an in-process
// batched decode step.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

template <int D, int G>
__global__ __launch_bounds__(256) void paged_decode_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B, Hq, D]
    const __hip_bfloat16* __restrict__ kc,  // [NP, Hkv, P, D]
    const __hip_bfloat16* __restrict__ vc,
    __hip_bfloat16* __restrict__ out,       // [B, Hq, D]
    const int* __restrict__ block_tables,   // [B, MP]
    const int* __restrict__ ctx_lens,       // [B]
    int Hkv, int P, int MP, float scale) {
  constexpr int GL = D / 8;        // lanes per token-group
  constexpr int TPW = WAVE / GL;   // token-groups per wave
  constexpr int NW = 4;            // waves per workgroup
  constexpr int NPART = NW * TPW;  // partials to combine

  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int L = ctx_lens[b];
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int grp = lane / GL;   // token-group within the wave
  const int sub = lane % GL;   // 16-B slot within the token row

  // q fragments for the G grouped heads: lane's 8 bf16 of each head's row
  shortx8 qf[G];
#pragma unroll
  for (int g = 0; g < G; ++g)
    qf[g] = *reinterpret_cast<const shortx8*>(
        q + (((long)b * Hkv + kvh) * G + g) * D + sub * 8);

  float m[G], l[G], o[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1e30f;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[g][j] = 0.f;
  }

  const int* bt = block_tables + (long)b * MP;
  // token index walked by this (wave, grp): wave w covers tokens
  // w*TPW+grp, then strides NW*TPW
  for (int t = wave * TPW + grp; t < L; t += NW * TPW) {
    const int page = bt[t / P];
    const long row = (((long)page * Hkv + kvh) * P + (t % P)) * D + sub * 8;
    const shortx8 kfrag = *reinterpret_cast<const shortx8*>(kc + row);
    const shortx8 vfrag = *reinterpret_cast<const shortx8*>(vc + row);
    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      s[g] = dot8_bf16(kfrag, qf[g]);
      s[g] = group_sum<GL>(s[g]) * scale;  // all GL lanes now hold the score
      const float mn = fmaxf(m[g], s[g]);
      const float alpha = __expf(m[g] - mn);
      const float p = __expf(s[g] - mn);
      m[g] = mn;
      l[g] = l[g] * alpha + p;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[g][j] = o[g][j] * alpha + p * bf16_to_f32((unsigned short)vfrag[j]);
    }
  }

  // ---- combine the NPART partials per head through LDS ----
  // layout per head g: m[NPART], l[NPART], o[NPART][D]
  __shared__ float s_m[G][NPART];
  __shared__ float s_l[G][NPART];
  __shared__ float s_o[G][NPART][D];
  const int part = wave * TPW + grp;
#pragma unroll
  for (int g = 0; g < G; ++g) {
    if (sub == 0) {
      s_m[g][part] = m[g];
      s_l[g][part] = l[g];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) s_o[g][part][sub * 8 + j] = o[g][j];
  }
  __syncthreads();
  // 256 threads cover G*D outputs; thread handles (g, d) pairs
  for (int gd = tid; gd < G * D; gd += blockDim.x) {
    const int g = gd / D, d = gd % D;
    float M = -1e30f;
#pragma unroll 4
    for (int i = 0; i < NPART; ++i) M = fmaxf(M, s_m[g][i]);
    float Lsum = 0.f, O = 0.f;
#pragma unroll 4
    for (int i = 0; i < NPART; ++i) {
      const float w = __expf(s_m[g][i] - M);
      Lsum += w * s_l[g][i];
      O += w * s_o[g][i][d];
    }
    *(unsigned short*)(out + (((long)b * Hkv + kvh) * G + g) * D + d) =
        f32_to_bf16(O / Lsum);
  }
}

torch::Tensor paged_attention_decode(torch::Tensor q, torch::Tensor kc,
                                     torch::Tensor vc, torch::Tensor bt,
                                     torch::Tensor ctx, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16, "bf16 GPU only");
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = kc.size(1), P = kc.size(2), MP = bt.size(1);
  const int G = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  auto out = torch::empty_like(q);
  if (B == 0) return out;
  dim3 grid(B, Hkv), block(256);
  auto stream = CUR_STREAM;
#define LAUNCH(DD, GG)                                                        \
  hipLaunchKernelGGL((paged_decode_kernel<DD, GG>), grid, block, 0, stream,   \
                     (const __hip_bfloat16*)q.data_ptr(),                     \
                     (const __hip_bfloat16*)kc.data_ptr(),                    \
                     (const __hip_bfloat16*)vc.data_ptr(),                    \
                     (__hip_bfloat16*)out.data_ptr(), bt.data_ptr<int>(),     \
                     ctx.data_ptr<int>(), Hkv, P, MP, (float)scale)
  if (D == 128) {
    switch (G) {
      case 1: LAUNCH(128, 1); break;
      case 2: LAUNCH(128, 2); break;
      case 4: LAUNCH(128, 4); break;
      case 8: LAUNCH(128, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group size ", G);
    }
  } else if (D == 64) {
    switch (G) {
      case 1: LAUNCH(64, 1); break;
      case 2: LAUNCH(64, 2); break;
      case 4: LAUNCH(64, 4); break;
      case 8: LAUNCH(64, 8); break;
      default: TORCH_CHECK(false, "unsupported GQA group size ", G);
    }
  } else {
    TORCH_CHECK(false, "unsupported head_dim ", D);
  }
#undef LAUNCH
  HIP_CHECK_KERNEL();
  return out;
}
