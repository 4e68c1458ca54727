// Paged-KV decode attention for gfx950 (one new token per sequence).
//
// Memory-bound by the KV read; the kernel streams each (seq, kv_head)'s
// pages once at 16 B/lane and amortises the read over all G grouped query
// heads (GQA).  Two levers keep it at streaming rate:
//  * token-group software pipelining: each D/8-lane group prefetches K/V
//    for 4 tokens before the (serial) online-softmax updates, so the
//    memory system always has 8 x 16 B loads in flight per lane-group
//    (the naive one-token loop is latency-bound through the o-rescale
//    dependency chain: measured 1.07 TB/s at B=16 -> this structure);
//  * flash-decode context splits: grid.z partitions the context into
//    contiguous chunks so small batches still put >=512 workgroups on the
//    256-CU chip; per-chunk (m, l, o) partials combine in a second kernel.
//
// Replaces the per-token provider round-trip of the reference agent loop
// (reference: server/chat/backend/agent/llm.py:192) with an in-process
// batched decode step.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

#ifndef TUNR
#define TUNR 4  // tokens prefetched per group iteration (8 measured neutral)
#endif

template <int D, int G, bool FP8>
__global__ __launch_bounds__(256) void paged_decode_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B, Hq, D]
    const void* __restrict__ kc,  // [NP, Hkv, P, D] bf16 | fp8 e4m3
    const void* __restrict__ vc,
    __hip_bfloat16* __restrict__ out,       // [B, Hq, D] (CS==1)
    float* __restrict__ part,               // [CS, B, Hq, D+2] (CS>1)
    const int* __restrict__ block_tables,   // [B, MP]
    const int* __restrict__ ctx_lens,       // [B]
    int Hkv, int P, int MP, int CS, float scale) {
  constexpr int GL = D / 8;        // lanes per token-group
  constexpr int TPW = WAVE / GL;   // token-groups per wave
  constexpr int NW = 4;            // waves per workgroup
  constexpr int NPART = NW * TPW;  // in-block partials to combine

  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int cs = blockIdx.z;
  const int L = ctx_lens[b];
  // contiguous context chunk for this split
  const int chunk = (L + CS - 1) / CS;
  const int t_lo = cs * chunk;
  const int t_hi = min(L, t_lo + chunk);
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int grp = lane / GL;   // token-group within the wave
  const int sub = lane % GL;   // 16-B slot within the token row

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
  const int stride = NW * TPW;
  // groups stride the chunk; TUNR tokens prefetched per iteration
  for (int t0 = t_lo + (wave * TPW + grp) * TUNR; t0 < t_hi;
       t0 += stride * TUNR) {
    shortx8 kf[TUNR], vf[TUNR];
    int nt = 0;
    if constexpr (FP8) {
      // fp8 cache: half the KV stream bytes (the kernel is KV-bound at
      // ~6.15 TB/s in bf16 — docs/ROUND2.md §3); 8-byte loads, convert
      // to bf16 fragments in-register
      u64b k8[TUNR], v8[TUNR];
#pragma unroll
      for (int u = 0; u < TUNR; ++u) {
        const int t = t0 + u;
        const int tc = min(t, t_hi - 1);
        const long row =
            (((long)bt[tc / P] * Hkv + kvh) * P + (tc % P)) * D + sub * 8;
        k8[u] = *reinterpret_cast<const u64b*>((const unsigned char*)kc + row);
        v8[u] = *reinterpret_cast<const u64b*>((const unsigned char*)vc + row);
      }
#pragma unroll
      for (int u = 0; u < TUNR; ++u) {
        kf[u] = fp8x8_to_bf16x8(k8[u]);
        vf[u] = fp8x8_to_bf16x8(v8[u]);
      }
    } else {
#pragma unroll
      for (int u = 0; u < TUNR; ++u) {
        const int t = t0 + u;
        const int tc = min(t, t_hi - 1);  // clamped prefetch (masked below)
        const long row = (((long)bt[tc / P] * Hkv + kvh) * P + (tc % P)) * D + sub * 8;
        kf[u] = *reinterpret_cast<const shortx8*>((const __hip_bfloat16*)kc + row);
        vf[u] = *reinterpret_cast<const shortx8*>((const __hip_bfloat16*)vc + row);
      }
    }
#pragma unroll
    for (int u = 0; u < TUNR; ++u) {
      if (t0 + u >= t_hi) break;
      float s[G];
#pragma unroll
      for (int g = 0; g < G; ++g) {
        s[g] = group_sum<GL>(dot8_bf16_fast(kf[u], qf[g])) * scale;
        // defer-style rescale: the running max rarely grows, so the O/l
        // rescale (8*G VALU per token) is branch-skipped when it does not
        if (s[g] > m[g]) {
          const float alpha = __expf(m[g] - s[g]);
          m[g] = s[g];
          l[g] *= alpha;
#pragma unroll
          for (int j = 0; j < 8; ++j) o[g][j] *= alpha;
        }
        const float pv = __expf(s[g] - m[g]);
        l[g] += pv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[g][j] += pv * bf16_to_f32((unsigned short)vf[u][j]);
      }
    }
  }

  // ---- combine the NPART in-block partials per head through LDS ----
  __shared__ float s_m[G][NPART];
  __shared__ float s_l[G][NPART];
  __shared__ float s_o[G][NPART][D];
  const int pidx = wave * TPW + grp;
#pragma unroll
  for (int g = 0; g < G; ++g) {
    if (sub == 0) {
      s_m[g][pidx] = m[g];
      s_l[g][pidx] = l[g];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) s_o[g][pidx][sub * 8 + j] = o[g][j];
  }
  __syncthreads();
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
    const long hq = ((long)b * Hkv + kvh) * G + g;
    if (CS == 1) {
      *(unsigned short*)(out + hq * D + d) = f32_to_bf16(O / Lsum);
    } else {
      // emit an un-normalised chunk partial: {o*w-sum, m, l}
      float* dst = part + ((long)cs * gridDim.x * Hkv * G + hq) * (D + 2);
      dst[d] = O;
      if (d == 0) {
        dst[D] = M;
        dst[D + 1] = Lsum;
      }
    }
  }
}

// combine CS chunk partials -> out; one wave per (b, h)
template <int D>
__global__ void decode_combine_kernel(const float* __restrict__ part,
                                      __hip_bfloat16* __restrict__ out,
                                      int CS, long BH) {
  const long bh = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (bh >= BH) return;
  const int lane = threadIdx.x % WAVE;
  float M = -1e30f;
  for (int s = 0; s < CS; ++s)
    M = fmaxf(M, part[((long)s * BH + bh) * (D + 2) + D]);
  float Lsum = 0.f;
  for (int s = 0; s < CS; ++s) {
    const float* p = part + ((long)s * BH + bh) * (D + 2);
    Lsum += __expf(p[D] - M) * p[D + 1];
  }
  for (int d = lane; d < D; d += WAVE) {
    float O = 0.f;
    for (int s = 0; s < CS; ++s) {
      const float* p = part + ((long)s * BH + bh) * (D + 2);
      O += __expf(p[D] - M) * p[d];
    }
    *(unsigned short*)(out + bh * D + d) = f32_to_bf16(O / Lsum);
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
  const bool fp8 = kc.dtype() == torch::kFloat8_e4m3fn;
  auto out = torch::empty_like(q);
  if (B == 0) return out;
  // context splits: fill the chip at small batch (>=512 WGs target)
  // fill target ~1024 WGs: at B=64 the LDS-bound residency is only
  // 2 WGs/CU, so deeper context splits buy memory-level parallelism
  int CS = std::max(1, std::min(8, 1024 / std::max(B * Hkv, 1)));
  torch::Tensor part;
  if (CS > 1)
    part = torch::empty({CS, (long)B * Hq, D + 2},
                        q.options().dtype(torch::kFloat32));
  dim3 grid(B, Hkv, CS), block(256);
  auto stream = CUR_STREAM;
#define LAUNCH1(DD, GG, F8)                                                   \
  hipLaunchKernelGGL((paged_decode_kernel<DD, GG, F8>), grid, block, 0,       \
                     stream, (const __hip_bfloat16*)q.data_ptr(),             \
                     kc.data_ptr(), vc.data_ptr(),                            \
                     (__hip_bfloat16*)out.data_ptr(),                         \
                     CS > 1 ? part.data_ptr<float>() : nullptr,               \
                     bt.data_ptr<int>(), ctx.data_ptr<int>(), Hkv, P, MP, CS, \
                     (float)scale)
#define LAUNCH(DD, GG)                                                        \
  if (fp8) { LAUNCH1(DD, GG, true); } else { LAUNCH1(DD, GG, false); }
#define LAUNCH_D(DD)                                                          \
  switch (G) {                                                                \
    case 1: LAUNCH(DD, 1); break;                                             \
    case 2: LAUNCH(DD, 2); break;                                             \
    case 4: LAUNCH(DD, 4); break;                                             \
    case 8: LAUNCH(DD, 8); break;                                             \
    default: TORCH_CHECK(false, "unsupported GQA group size ", G);            \
  }
  if (D == 128) {
    LAUNCH_D(128)
  } else if (D == 64) {
    LAUNCH_D(64)
  } else {
    TORCH_CHECK(false, "unsupported head_dim ", D);
  }
  if (CS > 1) {
    const long BH = (long)B * Hq;
    dim3 cgrid((BH + 3) / 4), cblock(256);
    if (D == 128)
      hipLaunchKernelGGL((decode_combine_kernel<128>), cgrid, cblock, 0, stream,
                         part.data_ptr<float>(), (__hip_bfloat16*)out.data_ptr(),
                         CS, BH);
    else
      hipLaunchKernelGGL((decode_combine_kernel<64>), cgrid, cblock, 0, stream,
                         part.data_ptr<float>(), (__hip_bfloat16*)out.data_ptr(),
                         CS, BH);
  }
#undef LAUNCH_D
#undef LAUNCH
#undef LAUNCH1
  HIP_CHECK_KERNEL();
  return out;
}
