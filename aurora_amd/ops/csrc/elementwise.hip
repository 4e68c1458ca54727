// Fused element-wise / normalisation kernels (bf16, gfx950).
//
// All memory-bound: the design rule is G13 of the platform guide — bf16
// loads vectorized as shortx8 (16 B/lane), grid-stride, fused passes so a
// tensor is read exactly once.  These replace the reference's CPU-side
// text plumbing with the engine's per-layer hot path (no reference analog;
// SURVEY.md preamble table row 1).
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()

// ---------------------------------------------------------------- rmsnorm

// One workgroup per row; 256 threads; row length H must be a multiple of 8.
// RESIDUAL: also adds `res` first and writes the sum back to res_out.
template <bool RESIDUAL>
__global__ void rmsnorm_kernel(
    const __hip_bfloat16* __restrict__ x,
    const __hip_bfloat16* __restrict__ res,
    __hip_bfloat16* __restrict__ out,
    __hip_bfloat16* __restrict__ res_out,
    const __hip_bfloat16* __restrict__ weight,
    int H, float eps) {
  const int row = blockIdx.x;
  const long base = (long)row * H;
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  __shared__ float s_partial[16];
  __shared__ float s_inv;

  float ss = 0.f;
  // each thread strides over the row in 8-element vectors
  for (int i = tid * 8; i < H; i += nthr * 8) {
    shortx8 v = *reinterpret_cast<const shortx8*>(x + base + i);
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[j] = bf16_to_f32((unsigned short)v[j]);
    if (RESIDUAL) {
      shortx8 r = *reinterpret_cast<const shortx8*>(res + base + i);
      shortx8 sum;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        vals[j] += bf16_to_f32((unsigned short)r[j]);
        sum[j] = (short)f32_to_bf16(vals[j]);
      }
      *reinterpret_cast<shortx8*>(res_out + base + i) = sum;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += vals[j] * vals[j];
  }
  ss = wave_sum(ss);
  const int wid = tid / WAVE, lid = tid % WAVE;
  if (lid == 0) s_partial[wid] = ss;
  __syncthreads();
  if (tid == 0) {
    float t = 0.f;
    for (int w = 0; w < nthr / WAVE; ++w) t += s_partial[w];
    s_inv = rsqrtf(t / H + eps);
  }
  __syncthreads();
  const float inv = s_inv;
  const __hip_bfloat16* src = RESIDUAL ? res_out : x;
  for (int i = tid * 8; i < H; i += nthr * 8) {
    shortx8 v = *reinterpret_cast<const shortx8*>(src + base + i);
    shortx8 w = *reinterpret_cast<const shortx8*>(weight + i);
    shortx8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (short)f32_to_bf16(bf16_to_f32((unsigned short)v[j]) * inv *
                                bf16_to_f32((unsigned short)w[j]));
    *reinterpret_cast<shortx8*>(out + base + i) = o;
  }
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "rmsnorm: bf16 GPU only");
  TORCH_CHECK(x.size(-1) % 8 == 0, "H must be a multiple of 8");
  auto x2 = x.view({-1, x.size(-1)});
  auto out = torch::empty_like(x);
  int T = x2.size(0), H = x2.size(1);
  if (T == 0) return out;
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(rmsnorm_kernel<false>, grid, block, 0,
                     CUR_STREAM,
                     (const __hip_bfloat16*)x.data_ptr(), nullptr,
                     (__hip_bfloat16*)out.data_ptr(), nullptr,
                     (const __hip_bfloat16*)weight.data_ptr(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return out;
}

std::vector<torch::Tensor> rmsnorm_residual(torch::Tensor x, torch::Tensor res,
                                            torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "bf16 GPU only");
  auto out = torch::empty_like(x);
  auto res_out = torch::empty_like(x);
  auto x2 = x.view({-1, x.size(-1)});
  int T = x2.size(0), H = x2.size(1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  if (T == 0) return {out, res_out};
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(rmsnorm_kernel<true>, grid, block, 0,
                     CUR_STREAM,
                     (const __hip_bfloat16*)x.data_ptr(),
                     (const __hip_bfloat16*)res.contiguous().data_ptr(),
                     (__hip_bfloat16*)out.data_ptr(),
                     (__hip_bfloat16*)res_out.data_ptr(),
                     (const __hip_bfloat16*)weight.data_ptr(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {out, res_out};
}

// --------------------------------------------------------------- silu_mul

__global__ void silu_mul_kernel(const __hip_bfloat16* __restrict__ gu,
                                __hip_bfloat16* __restrict__ out,
                                int T, int I) {
  // gu is [T, 2I]; out [T, I]; 8 elements per thread, grid-stride
  const long total = (long)T * I / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / (I / 8);
    const long col = (idx % (I / 8)) * 8;
    shortx8 g = *reinterpret_cast<const shortx8*>(gu + row * 2 * I + col);
    shortx8 u = *reinterpret_cast<const shortx8*>(gu + row * 2 * I + I + col);
    shortx8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32((unsigned short)g[j]);
      float uf = bf16_to_f32((unsigned short)u[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = (short)f32_to_bf16(s * uf);
    }
    *reinterpret_cast<shortx8*>(out + row * I + col) = o;
  }
}

torch::Tensor silu_mul(torch::Tensor gu) {
  TORCH_CHECK(gu.is_cuda() && gu.dtype() == torch::kBFloat16, "bf16 GPU only");
  int I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "intermediate must be a multiple of 8");
  auto sizes = gu.sizes().vec();
  sizes.back() = I2 / 2;
  auto out = torch::empty(sizes, gu.options());
  auto gu2 = gu.view({-1, I2});
  int T = gu2.size(0), I = I2 / 2;
  if (T == 0) return out;
  long total = (long)T * I / 8;
  int blocks = (int)std::min<long>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0,
                     CUR_STREAM,
                     (const __hip_bfloat16*)gu.data_ptr(),
                     (__hip_bfloat16*)out.data_ptr(), T, I);
  HIP_CHECK_KERNEL();
  return out;
}

// ------------------------------------------------------------------- rope

// In-place NeoX-pairing RoPE on q [T, Hq, D] and k [T, Hkv, D].
// cos_sin: [max_pos, D] f32 (first half cos, second half sin).
// One wave per (token, head); lane handles pair (lane mod D/2).
__global__ void rope_kernel(__hip_bfloat16* __restrict__ q,
                            __hip_bfloat16* __restrict__ k,
                            const int* __restrict__ positions,
                            const float* __restrict__ cos_sin,
                            int T, int Hq, int Hkv, int D) {
  const int half = D / 2;
  const int waves_per_block = blockDim.x / WAVE;
  const long wave_id = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long total = (long)T * (Hq + Hkv);
  if (wave_id >= total) return;
  const int t = wave_id / (Hq + Hkv);
  const int h = wave_id % (Hq + Hkv);
  const int pos = positions[t];
  __hip_bfloat16* base = (h < Hq) ? (q + ((long)t * Hq + h) * D)
                                  : (k + ((long)t * Hkv + (h - Hq)) * D);
  for (int i = lane; i < half; i += WAVE) {
    float c = cos_sin[(long)pos * D + i];
    float s = cos_sin[(long)pos * D + half + i];
    float x1 = bf16_to_f32(*(unsigned short*)(base + i));
    float x2 = bf16_to_f32(*(unsigned short*)(base + half + i));
    *(unsigned short*)(base + i) = f32_to_bf16(x1 * c - x2 * s);
    *(unsigned short*)(base + half + i) = f32_to_bf16(x2 * c + x1 * s);
  }
}

void rope_apply_(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                 torch::Tensor cos_sin) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16, "bf16 GPU only");
  int T = q.size(0), Hq = q.size(1), D = q.size(2), Hkv = k.size(1);
  if (T == 0) return;
  auto cs = cos_sin.to(torch::kFloat32).contiguous();
  long waves = (long)T * (Hq + Hkv);
  int wpb = 4;  // 256 threads
  long blocks = (waves + wpb - 1) / wpb;
  hipLaunchKernelGGL(rope_kernel, dim3(blocks), dim3(wpb * WAVE), 0,
                     CUR_STREAM,
                     (__hip_bfloat16*)q.data_ptr(), (__hip_bfloat16*)k.data_ptr(),
                     positions.data_ptr<int>(), cs.data_ptr<float>(), T, Hq, Hkv, D);
  HIP_CHECK_KERNEL();
}

// ---------------------------------------------------------- kv_cache_write

// Scatter new K/V rows into the paged cache.
// k/v: [T, Hkv, D]; caches: [NP, Hkv, P, D]; slot = page*P + off.
template <bool FP8>
__global__ void kv_cache_write_kernel(
    const __hip_bfloat16* __restrict__ k, const __hip_bfloat16* __restrict__ v,
    void* __restrict__ kc, void* __restrict__ vc,
    const int* __restrict__ slots, int T, int Hkv, int P, int D) {
  // one wave per (token, head); lane copies 8 elements at a time
  const int waves_per_block = blockDim.x / WAVE;
  const long wave_id = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (wave_id >= (long)T * Hkv) return;
  const int t = wave_id / Hkv;
  const int h = wave_id % Hkv;
  const int slot = slots[t];
  if (slot < 0) return;
  const int page = slot / P, off = slot % P;
  const long src = ((long)t * Hkv + h) * D;
  const long dst = (((long)page * Hkv + h) * P + off) * D;
  for (int i = lane * 8; i < D; i += WAVE * 8) {
    if constexpr (FP8) {
      *reinterpret_cast<u64b*>((unsigned char*)kc + dst + i) =
          bf16x8_to_fp8x8(*reinterpret_cast<const shortx8*>(k + src + i));
      *reinterpret_cast<u64b*>((unsigned char*)vc + dst + i) =
          bf16x8_to_fp8x8(*reinterpret_cast<const shortx8*>(v + src + i));
    } else {
      *reinterpret_cast<shortx8*>((__hip_bfloat16*)kc + dst + i) =
          *reinterpret_cast<const shortx8*>(k + src + i);
      *reinterpret_cast<shortx8*>((__hip_bfloat16*)vc + dst + i) =
          *reinterpret_cast<const shortx8*>(v + src + i);
    }
  }
}

void kv_cache_write(torch::Tensor k, torch::Tensor v, torch::Tensor kc,
                    torch::Tensor vc, torch::Tensor slots) {
  TORCH_CHECK(k.is_cuda() && k.dtype() == torch::kBFloat16, "bf16 GPU only");
  int T = k.size(0), Hkv = k.size(1), D = k.size(2);
  int P = kc.size(2);
  if (T == 0) return;
  long waves = (long)T * Hkv;
  int wpb = 4;
  const bool fp8 = kc.dtype() == torch::kFloat8_e4m3fn;
  if (fp8)
    hipLaunchKernelGGL(kv_cache_write_kernel<true>,
                       dim3((waves + wpb - 1) / wpb), dim3(wpb * WAVE), 0,
                       CUR_STREAM, (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), kc.data_ptr(),
                       vc.data_ptr(), slots.data_ptr<int>(), T, Hkv, P, D);
  else
    hipLaunchKernelGGL(kv_cache_write_kernel<false>,
                       dim3((waves + wpb - 1) / wpb), dim3(wpb * WAVE), 0,
                       CUR_STREAM, (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(), kc.data_ptr(),
                       vc.data_ptr(), slots.data_ptr<int>(), T, Hkv, P, D);
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------- fused qkv post-process
// One kernel replaces: q/k/v slice copies (the qkv GEMM output is sliced
// non-contiguously), RoPE on q and k, and the paged-cache K/V scatter —
// rocprof showed those as ~5 launches + 3 copies per layer of pure launch
// floor in the decode step (profiles/decode_kernel_stats_r01.csv).
// qkv: [T, (Hq+2*Hkv)*D] -> writes q_out [T, Hq, D] (roped), kc/vc slots.
template <bool FP8>
__global__ void fused_rope_kv_write_kernel(
    const __hip_bfloat16* __restrict__ qkv,
    __hip_bfloat16* __restrict__ q_out,
    void* __restrict__ kc, void* __restrict__ vc,
    const int* __restrict__ positions, const int* __restrict__ slots,
    const float* __restrict__ cos_sin,
    int T, int Hq, int Hkv, int D, int P) {
  const int half = D / 2;
  const int HT = Hq + 2 * Hkv;
  const int waves_per_block = blockDim.x / WAVE;
  const long wave_id = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (wave_id >= (long)T * HT) return;
  const int t = wave_id / HT;
  const int h = wave_id % HT;
  const __hip_bfloat16* src = qkv + ((long)t * HT + h) * D;
  if (h < Hq + Hkv) {
    // roped write (q -> q_out, k -> cache)
    const int pos = positions[t];
    const bool is_q = h < Hq;
    long dst_row;
    if (is_q) {
      dst_row = ((long)t * Hq + h) * D;
    } else {
      const int slot = slots[t];
      if (slot < 0) return;
      dst_row = ((((long)(slot / P) * Hkv + (h - Hq)) * P + slot % P)) * D;
    }
    for (int i = lane; i < half; i += WAVE) {
      const float c = cos_sin[(long)pos * D + i];
      const float s = cos_sin[(long)pos * D + half + i];
      const float x1 = bf16_to_f32(*(const unsigned short*)(src + i));
      const float x2 = bf16_to_f32(*(const unsigned short*)(src + half + i));
      const float r1 = x1 * c - x2 * s;
      const float r2 = x2 * c + x1 * s;
      if (is_q) {
        __hip_bfloat16* dq = q_out + dst_row;
        *(unsigned short*)(dq + i) = f32_to_bf16(r1);
        *(unsigned short*)(dq + half + i) = f32_to_bf16(r2);
      } else if constexpr (FP8) {
        unsigned char* dk = (unsigned char*)kc + dst_row;
        dk[i] = f32_to_fp8(r1);
        dk[half + i] = f32_to_fp8(r2);
      } else {
        __hip_bfloat16* dk = (__hip_bfloat16*)kc + dst_row;
        *(unsigned short*)(dk + i) = f32_to_bf16(r1);
        *(unsigned short*)(dk + half + i) = f32_to_bf16(r2);
      }
    }
  } else {
    const int slot = slots[t];
    if (slot < 0) return;
    const long dst_row =
        ((((long)(slot / P) * Hkv + (h - Hq - Hkv)) * P + slot % P)) * D;
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      if constexpr (FP8) {
        *reinterpret_cast<u64b*>((unsigned char*)vc + dst_row + i) =
            bf16x8_to_fp8x8(*reinterpret_cast<const shortx8*>(src + i));
      } else {
        *reinterpret_cast<shortx8*>((__hip_bfloat16*)vc + dst_row + i) =
            *reinterpret_cast<const shortx8*>(src + i);
      }
    }
  }
}

torch::Tensor fused_rope_kv_write(torch::Tensor qkv, torch::Tensor positions,
                                  torch::Tensor slots, torch::Tensor cos_sin,
                                  torch::Tensor kc, torch::Tensor vc,
                                  long Hq, long Hkv) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16, "bf16 GPU only");
  const int T = qkv.size(0);
  const int P = kc.size(2), D = kc.size(3);
  TORCH_CHECK(qkv.size(1) == (Hq + 2 * Hkv) * D, "qkv width mismatch");
  auto q_out = torch::empty({(long)T, Hq, (long)D}, qkv.options());
  if (T == 0) return q_out;
  const long waves = (long)T * (Hq + 2 * Hkv);
  const int wpb = 4;
  const bool fp8 = kc.dtype() == torch::kFloat8_e4m3fn;
  if (fp8)
    hipLaunchKernelGGL(fused_rope_kv_write_kernel<true>,
                       dim3((waves + wpb - 1) / wpb), dim3(wpb * WAVE), 0,
                       CUR_STREAM, (const __hip_bfloat16*)qkv.data_ptr(),
                       (__hip_bfloat16*)q_out.data_ptr(), kc.data_ptr(),
                       vc.data_ptr(), positions.data_ptr<int>(),
                       slots.data_ptr<int>(), cos_sin.data_ptr<float>(),
                       T, (int)Hq, (int)Hkv, D, P);
  else
    hipLaunchKernelGGL(fused_rope_kv_write_kernel<false>,
                       dim3((waves + wpb - 1) / wpb), dim3(wpb * WAVE), 0,
                       CUR_STREAM, (const __hip_bfloat16*)qkv.data_ptr(),
                       (__hip_bfloat16*)q_out.data_ptr(), kc.data_ptr(),
                       vc.data_ptr(), positions.data_ptr<int>(),
                       slots.data_ptr<int>(), cos_sin.data_ptr<float>(),
                       T, (int)Hq, (int)Hkv, D, P);
  HIP_CHECK_KERNEL();
  return q_out;
}
