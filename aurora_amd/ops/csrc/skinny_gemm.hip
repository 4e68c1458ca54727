// Skinny GEMM for the decode path: out[M<=64, N] = x[M, K] @ W[N, K]^T.
//
// Decode serving is bound by streaming the weight matrices once per token
// step (llama3-8b: 16 GB/step); hipBLASLt's skinny kernels measured only
// 2.55 TB/s on [16, 6144x4096] (profiles/kernel_perf_r01_v1.json), far off
// the ~6.3 TB/s the chip streams.  This kernel reads W exactly once at
// 16 B/lane as direct MFMA B-fragments (v_mfma_f32_16x16x32_bf16: lane
// holds B[k=(l>>4)*8+j][col=l&15] = 16 contiguous bf16 of one W row), so
// the matrix pipe is free and the kernel runs at the W-streaming limit.
// x ([M,K] <= 0.5 MB) stays L1/L2-resident and is re-read as the A
// fragment.  Split-K keeps >=512 workgroups on the 256-CU / 8-XCD chip;
// f32 partials combine in a second tiny kernel (deterministic, no
// atomics).
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

typedef __attribute__((ext_vector_type(4))) float f32x4;

union sfrag {
  unsigned int u[4];
  bf16x8 v;
};

// MT = number of 16-row M tiles (M <= 16*MT); SPLIT: write f32 partials.
template <int MT, bool SPLIT, int UNR>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    const __hip_bfloat16* __restrict__ x,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    __hip_bfloat16* __restrict__ out,      // [M, N] (SPLIT=false)
    float* __restrict__ part,              // [S, M, N] (SPLIT=true)
    int M, int N, int K, int KS) {
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int n0 = blockIdx.x * 64 + wave * 16;  // this wave's 16 W rows
  if (n0 >= N) return;
  const int split = blockIdx.y;
  const int k_begin = split * KS;
  const int k_end = min(K, k_begin + KS);

  const int col = lane & 15;          // n offset / m row (A)
  const int krow = (lane >> 4) * 8;   // k offset of this lane's 8 elements

  const int n = n0 + col;
  const bool nvalid = n < N;
  // branchless k-loop: clamp out-of-range rows to a safe address and mask
  // at the store instead (a runtime guard inside the loop makes hipcc
  // branch around each load and drain vmcnt per element)
  const long wbase = (long)min(n, N - 1) * K;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) acc[t][r] = 0.f;

  long xbase[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    xbase[t] = (long)min(t * 16 + col, M - 1) * K;  // clamped (masked at store)

  // UNR k-steps per iteration: UNR W loads (+UNR*MT x loads) in flight
  for (int k = k_begin; k < k_end; k += 32 * UNR) {
    shortx8 wv[UNR];
    sfrag bf[UNR], af[UNR][MT];
#pragma unroll
    for (int u = 0; u < UNR; ++u)
      wv[u] = *reinterpret_cast<const shortx8*>(w + wbase + k + u * 32 + krow);
#pragma unroll
    for (int u = 0; u < UNR; ++u)
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        const shortx8 xv =
            *reinterpret_cast<const shortx8*>(x + xbase[t] + k + u * 32 + krow);
        __builtin_memcpy(&af[u][t], &xv, 16);
      }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      __builtin_memcpy(&bf[u], &wv[u], 16);
#pragma unroll
      for (int t = 0; t < MT; ++t)
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u][t].v, bf[u].v,
                                                         acc[t], 0, 0, 0);
    }
  }

  if (!nvalid) return;
  // C layout (16x16): lane holds col j = lane&15 (= n), rows (lane>>4)*4+r
#pragma unroll
  for (int t = 0; t < MT; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = t * 16 + (lane >> 4) * 4 + r;
      if (m >= M) continue;
      if (SPLIT) {
        part[((long)split * M + m) * N + n] = acc[t][r];
      } else {
        *(unsigned short*)(out + (long)m * N + n) = f32_to_bf16(acc[t][r]);
      }
    }
  }
}

__global__ void skinny_combine_kernel(const float* __restrict__ part,
                                      __hip_bfloat16* __restrict__ out,
                                      int S, long MN) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < MN;
       i += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < S; ++s) acc += part[s * MN + i];
    *(unsigned short*)(out + i) = f32_to_bf16(acc);
  }
}

torch::Tensor skinny_gemm_ex(torch::Tensor x, torch::Tensor w, long s_target, long unroll) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "bf16 GPU only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1));
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M <= 64, "skinny_gemm is for M <= 64");
  TORCH_CHECK(K % 64 == 0, "K must be a multiple of 64");
  auto out = torch::empty({M, N}, x.options());
  if (M == 0 || N == 0) return out;
  const int nblocks = (N + 63) / 64;
  const int target = s_target > 0 ? (int)s_target : 1024;
  const int kquant = 32 * (int)unroll;
  int S = std::max(1, std::min(32, target / std::max(nblocks, 1)));
  int KS = ((K / S + kquant - 1) / kquant) * kquant;
  S = (K + KS - 1) / KS;  // actual split count after rounding
  TORCH_CHECK(K % kquant == 0, "K must be a multiple of ", kquant);
  const int MT = (M + 15) / 16;
  auto stream = CUR_STREAM;
  torch::Tensor part;
  if (S > 1)
    part = torch::empty({S, M, N}, x.options().dtype(torch::kFloat32));
  dim3 grid(nblocks, S), block(256);
#define SG_LAUNCH(MTV, SPLITV)                                               \
  if (unroll == 4)                                                           \
    hipLaunchKernelGGL((skinny_gemm_kernel<MTV, SPLITV, 4>), grid, block, 0, \
                     stream, (const __hip_bfloat16*)x.data_ptr(),            \
                     (const __hip_bfloat16*)w.data_ptr(),                    \
                     (__hip_bfloat16*)out.data_ptr(),                        \
                     SPLITV ? part.data_ptr<float>() : nullptr, M, N, K, KS);\
  else                                                                       \
    hipLaunchKernelGGL((skinny_gemm_kernel<MTV, SPLITV, 2>), grid, block, 0, \
                     stream, (const __hip_bfloat16*)x.data_ptr(),            \
                     (const __hip_bfloat16*)w.data_ptr(),                    \
                     (__hip_bfloat16*)out.data_ptr(),                        \
                     SPLITV ? part.data_ptr<float>() : nullptr, M, N, K, KS)
  if (S > 1) {
    switch (MT) {
      case 1: SG_LAUNCH(1, true); break;
      case 2: SG_LAUNCH(2, true); break;
      case 3: SG_LAUNCH(3, true); break;
      default: SG_LAUNCH(4, true); break;
    }
    const long MN = (long)M * N;
    int cblocks = (int)std::min<long>((MN + 255) / 256, 1024);
    hipLaunchKernelGGL(skinny_combine_kernel, dim3(cblocks), dim3(256), 0,
                       stream, part.data_ptr<float>(),
                       (__hip_bfloat16*)out.data_ptr(), S, MN);
  } else {
    switch (MT) {
      case 1: SG_LAUNCH(1, false); break;
      case 2: SG_LAUNCH(2, false); break;
      case 3: SG_LAUNCH(3, false); break;
      default: SG_LAUNCH(4, false); break;
    }
  }
#undef SG_LAUNCH
  HIP_CHECK_KERNEL();
  return out;
}


torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w) {
  return skinny_gemm_ex(x, w, 0, 2);
}
