// Flash-style causal prefill attention over the paged KV cache (gfx950).
//
// MFMA structure (CDNA4, v_mfma_f32_32x32x16_bf16):
//  * swapped QK^T — compute S^T = mfma(A=K-tile, B=Q^T-tile) so each lane
//    owns ONE q column (q = lane&31) and the softmax row-reduce is
//    per-lane over registers + one shfl_xor(32) with its half-wave partner
//    (no cross-lane softmax, no P staging through LDS);
//  * P -> PV A-fragments in-register via bf16 packing + v_permlane32_swap
//    (the cvt_pk/permlane redistribution idiom);
//  * K tile staged in LDS row-major with the ((row&15)<<4) XOR byte-swizzle
//    (ds_read_b128 would otherwise be up-to-16-way bank-conflicted at
//    D=128); V staged TRANSPOSED [D][KVBLK+8] so the PV B-fragment read is
//    a contiguous 16-B ds_read per lane;
//  * O accumulates in registers (col d = lane&31 per d-tile); the epilogue
//    transposes through LDS so global stores are coalesced 16-B rows.
//
// Geometry: 8 waves/WG (512 threads); the WG covers HG = min(G, 4) query
// heads of ONE kv head x (8/HG)*32 query rows, so every staged K/V tile
// feeds 8 waves of MFMA (GQA heads share the stage — staging bandwidth was
// the 156-TF bound of the 4-wave/1-head structure).  KV tiles of 64 keys,
// two 32-key MFMA steps each.
// Covers fresh prefill and append-prefill (chunked agent turns) since K/V
// are read from the paged cache with an absolute-position causal bound.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

typedef __attribute__((ext_vector_type(4))) float f32x4;

DEVINL unsigned int pack2_bf16(float lo, float hi) {
  return ((unsigned int)f32_to_bf16(hi) << 16) | f32_to_bf16(lo);
}

union frag_u {
  unsigned int u[4];
  bf16x8 v;
};

template <int D, int HG, bool FP8>  // HG = query heads per workgroup
__global__ __launch_bounds__(512, 1) void prefill_kernel(
    const __hip_bfloat16* __restrict__ q,   // [T, Hq, D]
    const void* __restrict__ kc,  // [NP, Hkv, P, D] bf16 | fp8 e4m3
    const void* __restrict__ vc,
    __hip_bfloat16* __restrict__ out,       // [T, Hq, D]
    const int* __restrict__ block_seq,      // [NB] seq index of this q-block
    const int* __restrict__ block_qoff,     // [NB] q-row offset within the seq
    const int* __restrict__ cu_q,           // [S+1]
    const int* __restrict__ block_tables,   // [S, MP]
    const int* __restrict__ ctx_lens,       // [S]
    int Hq, int Hkv, int P, int MP, float scale) {
  constexpr int KVB = 64;           // keys staged per tile
  constexpr int NC = D / 16;        // Q^T fragments (MFMA K-chunks of QK^T)
  constexpr int NDT = D / 32;       // 32-wide d-tiles of O
  constexpr int VROW = KVB + 8;     // padded V^T row length (elements)
  // K-tile XOR swizzle mask: a row has D/8 16-B slots, so the swizzle may
  // only permute within that many slots (at D=64, (row&15)<<4 would escape
  // the 128-B row)
  constexpr int SWZ = (D / 8) - 1;
  constexpr int QS = 8 / HG;          // 32-row q sub-blocks per WG
  const int nb = blockIdx.x;
  const int seq = block_seq[nb];
  const int qoff = block_qoff[nb];
  const int q_start = cu_q[seq];
  const int qlen = cu_q[seq + 1] - q_start;
  const int ctx = ctx_lens[seq];
  const int* bt = block_tables + (long)seq * MP;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int col = lane & 31;   // q column owned by this lane (within wave's 32)
  const int half = lane >> 5;  // half-wave
  // wave -> (query head, q sub-block): heads vary fastest so the HG heads
  // of one sub-block sit on consecutive waves
  const int h = blockIdx.y * HG + (wave % HG);
  const int kvh = h / (Hq / Hkv);
  const int qsub = wave / HG;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* kbuf = (short*)smem;                       // [KVB][D] XOR-swizzled
  short* vtbuf = (short*)(smem + KVB * D * 2);      // [D][VROW] transposed
  float* scratch = (float*)(smem + KVB * D * 2 + D * VROW * 2);  // [8][32]
  float* wsc = scratch + wave * 32;

  // ---- Q fragments (B-operand of swapped QK^T): held for the whole block
  const int qrow = qoff + qsub * 32 + col;  // local q row in this seq
  const bool qvalid = qrow < qlen;
  const long qbase = ((long)(q_start + (qvalid ? qrow : 0)) * Hq + h) * D;
  frag_u qf[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    if (qvalid) {
      const shortx8 v =
          *reinterpret_cast<const shortx8*>(q + qbase + c * 16 + half * 8);
      qf[c].u[0] = ((const unsigned int*)&v)[0];
      qf[c].u[1] = ((const unsigned int*)&v)[1];
      qf[c].u[2] = ((const unsigned int*)&v)[2];
      qf[c].u[3] = ((const unsigned int*)&v)[3];
    } else {
      qf[c].u[0] = qf[c].u[1] = qf[c].u[2] = qf[c].u[3] = 0u;
    }
  }

  const int q_abs = ctx - qlen + qrow;  // absolute position of this lane's q
  float m = -1e30f, l = 0.f;
  floatx16 o[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) o[dt][r] = 0.f;

  // keys needed by this q block: up to the last row's absolute position
  const int blk_rows = min(qlen - qoff, QS * 32);
  const int kend = min(ctx, ctx - qlen + qoff + blk_rows);
  const int ntiles = (kend + KVB - 1) / KVB;
  // absolute position of the wave's first (smallest) q row
  const int q_abs_min = ctx - qlen + qoff + qsub * 32;

  for (int tile = 0; tile < ntiles; ++tile) {
    const int kbase = tile * KVB;
    // ---- stage K (swizzled) and V^T cooperatively: 512 threads
    __syncthreads();
#pragma unroll
    for (int i = 0; i < KVB * D / (512 * 8); ++i) {
      const int c = tid + i * 512;
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const int t = kbase + row;
      shortx8 kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      shortx8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (t < ctx) {
        const long base =
            (((long)bt[t / P] * Hkv + kvh) * P + (t % P)) * D + d0;
        if constexpr (FP8) {
          kv8 = fp8x8_to_bf16x8(
              *reinterpret_cast<const u64b*>((const unsigned char*)kc + base));
          vv8 = fp8x8_to_bf16x8(
              *reinterpret_cast<const u64b*>((const unsigned char*)vc + base));
        } else {
          kv8 = *reinterpret_cast<const shortx8*>(
              (const __hip_bfloat16*)kc + base);
          vv8 = *reinterpret_cast<const shortx8*>(
              (const __hip_bfloat16*)vc + base);
        }
      }
      const int kb = (row * D * 2 + ((d0 * 2) ^ ((row & SWZ) << 4))) / 2;
      *reinterpret_cast<shortx8*>(kbuf + kb) = kv8;
#pragma unroll
      for (int j = 0; j < 8; ++j) vtbuf[(d0 + j) * VROW + row] = vv8[j];
    }
    __syncthreads();

    // ---- two 32-key MFMA steps (waves skip steps beyond their causal bound)
    const int wave_kend = min(kend, q_abs_min + 32);
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int ks = kbase + s * 32;  // absolute key base of this step
      if (ks >= wave_kend) break;
      // S^T = K * Q^T : rows k, cols q
      floatx16 st;
#pragma unroll
      for (int r = 0; r < 16; ++r) st[r] = 0.f;
      const int krow = s * 32 + col;  // this lane's A-row in the K tile
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int kb =
            (krow * D * 2 + (((c * 16 + half * 8) * 2) ^ ((krow & SWZ) << 4))) / 2;
        frag_u kf;
        const shortx8 kv = *reinterpret_cast<const shortx8*>(kbuf + kb);
        kf.u[0] = ((const unsigned int*)&kv)[0];
        kf.u[1] = ((const unsigned int*)&kv)[1];
        kf.u[2] = ((const unsigned int*)&kv)[2];
        kf.u[3] = ((const unsigned int*)&kv)[3];
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf.v, qf[c].v, st, 0, 0, 0);
      }
      // ---- mask + scale (S^T reg r holds key row R = (r&3)+8*(r>>2)+4*half)
      float p[16];
      const bool need_mask = (ks + 31 > q_abs_min) || (ks + 31 >= ctx);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int k_abs = ks + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = st[r] * scale;
        if (need_mask && (k_abs > q_abs || k_abs >= ctx)) sv = -3.0e4f;
        p[r] = sv;
      }
      // ---- online softmax (per-lane column view)
      float tmax = p[0];
#pragma unroll
      for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, p[r]);
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
      const float mn = fmaxf(m, tmax);
      const float alpha = __expf(m - mn);
      m = mn;
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf(p[r] - mn);
        psum += p[r];
      }
      psum += __shfl_xor(psum, 32, WAVE);
      l = l * alpha + psum;
      // ---- rescale O by alpha of each O-row's q (broadcast via LDS)
      if (half == 0) wsc[col] = alpha;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const f32x4 av = *reinterpret_cast<const f32x4*>(wsc + 4 * half + 8 * g);
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) {
#pragma unroll
          for (int j = 0; j < 4; ++j) o[dt][4 * g + j] *= av[j];
        }
      }
      // ---- P -> bf16 A-fragments (pack + permlane32_swap)
      frag_u pf[2];
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        unsigned int a0 = pack2_bf16(p[c * 8 + 0], p[c * 8 + 1]);
        unsigned int a1 = pack2_bf16(p[c * 8 + 2], p[c * 8 + 3]);
        unsigned int b0 = pack2_bf16(p[c * 8 + 4], p[c * 8 + 5]);
        unsigned int b1 = pack2_bf16(p[c * 8 + 6], p[c * 8 + 7]);
        {
          auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
          a0 = r0[0];
          b0 = r0[1];
          auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
          a1 = r1[0];
          b1 = r1[1];
        }
        pf[c].u[0] = a0;
        pf[c].u[1] = a1;
        pf[c].u[2] = b0;
        pf[c].u[3] = b1;
      }
      // ---- PV: O[dt] += P * V  (B-frag = 16-B row read of transposed V)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) {
          const int d = dt * 32 + col;
          const int kloc = s * 32 + c * 16 + half * 8;
          frag_u vf;
          const shortx8 vv =
              *reinterpret_cast<const shortx8*>(vtbuf + d * VROW + kloc);
          vf.u[0] = ((const unsigned int*)&vv)[0];
          vf.u[1] = ((const unsigned int*)&vv)[1];
          vf.u[2] = ((const unsigned int*)&vv)[2];
          vf.u[3] = ((const unsigned int*)&vv)[3];
          o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf[c].v, vf.v, o[dt], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: divide by l (per q row), transpose via LDS, store rows
  if (half == 0) wsc[col] = l;
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  float linv[4][4];
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    const f32x4 lv = *reinterpret_cast<const f32x4*>(wsc + 4 * half + 8 * g);
#pragma unroll
    for (int j = 0; j < 4; ++j) linv[g][j] = 1.f / lv[j];
  }
  // everyone done with K/V buffers; reuse as O staging in two rounds of
  // 4 waves (8 waves' staging would not fit the K+V area)
#pragma unroll
  for (int round = 0; round < 2; ++round) {
    __syncthreads();
    if ((wave >> 2) != round) continue;
    short* obuf = (short*)smem + (wave & 3) * 32 * D;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt) {
      const int d = dt * 32 + col;
#pragma unroll
      for (int g = 0; g < 4; ++g) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int r = (j) + 8 * g + 4 * half;  // q row within the wave's 32
          obuf[r * D + d] = (short)f32_to_bf16(o[dt][4 * g + j] * linv[g][j]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    // each lane stores half a row (D/2 elems = D bytes) with 16-B stores
    const int r = lane >> 1;            // 0..31
    const int seg = (lane & 1) * (D / 2);
    const int qr = qoff + qsub * 32 + r;
    if (qr < qlen) {
      const long ob = ((long)(q_start + qr) * Hq + h) * D + seg;
#pragma unroll
      for (int i = 0; i < D / 16; ++i) {
        const shortx8 v =
            *reinterpret_cast<const shortx8*>(obuf + r * D + seg + i * 8);
        *reinterpret_cast<shortx8*>(out + ob + i * 8) = v;
      }
    }
  }
}

torch::Tensor attention_prefill_paged(torch::Tensor q, torch::Tensor kc,
                                      torch::Tensor vc, torch::Tensor bt,
                                      torch::Tensor cu_q, torch::Tensor ctx,
                                      double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16, "bf16 GPU only");
  const int T = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = kc.size(1), P = kc.size(2), MP = bt.size(1);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128, got ", D);
  auto out = torch::empty_like(q);
  if (T == 0) return out;
  // heads per workgroup: the GQA group shares each staged K/V tile
  const int G = Hq / Hkv;
  const int HG = G >= 4 ? 4 : G;  // 1, 2 or 4
  const int QROWS = 32 * (8 / HG);  // q rows covered per workgroup
  // host-side q-block map (cu_q is tiny; engine built it on CPU this step)
  auto cu_cpu = cu_q.to(torch::kCPU);
  const int* cu = cu_cpu.data_ptr<int>();
  const int S = cu_cpu.numel() - 1;
  std::vector<int> bseq, boff;
  for (int s = 0; s < S; ++s) {
    const int qlen = cu[s + 1] - cu[s];
    for (int off = 0; off < qlen; off += QROWS) {
      bseq.push_back(s);
      boff.push_back(off);
    }
  }
  const int NB = bseq.size();
  auto opts = torch::TensorOptions().dtype(torch::kInt32).device(q.device());
  auto bseq_t = torch::from_blob(bseq.data(), {NB}, torch::kInt32).to(opts.device());
  auto boff_t = torch::from_blob(boff.data(), {NB}, torch::kInt32).to(opts.device());

  const int VROW = 64 + 8;
  const size_t lds = 64 * D * 2 + D * VROW * 2 + 8 * 32 * 4;
  const bool fp8 = kc.dtype() == torch::kFloat8_e4m3fn;
  dim3 grid(NB, Hq / HG), block(512);
  auto stream = CUR_STREAM;
#define PLAUNCH1(DD, HGV, F8)                                                 \
  hipLaunchKernelGGL((prefill_kernel<DD, HGV, F8>), grid, block, lds, stream, \
                     (const __hip_bfloat16*)q.data_ptr(),                     \
                     kc.data_ptr(), vc.data_ptr(),                            \
                     (__hip_bfloat16*)out.data_ptr(),                         \
                     bseq_t.data_ptr<int>(), boff_t.data_ptr<int>(),          \
                     cu_q.data_ptr<int>(), bt.data_ptr<int>(),                \
                     ctx.data_ptr<int>(), Hq, Hkv, P, MP, (float)scale)
#define PLAUNCH(DD, HGV)                                                      \
  if (fp8) { PLAUNCH1(DD, HGV, true); } else { PLAUNCH1(DD, HGV, false); }
#define PLAUNCH_D(DD)                                                         \
  switch (HG) {                                                               \
    case 1: PLAUNCH(DD, 1); break;                                            \
    case 2: PLAUNCH(DD, 2); break;                                            \
    default: PLAUNCH(DD, 4); break;                                           \
  }
  if (D == 128) {
    PLAUNCH_D(128)
  } else {
    PLAUNCH_D(64)
  }
#undef PLAUNCH_D
#undef PLAUNCH
#undef PLAUNCH1
  HIP_CHECK_KERNEL();
  return out;
}
