// Common helpers for the aurora_amd gfx950 (CDNA4) kernels.
// Wave width is 64 on CDNA4 — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(16))) float floatx16;
typedef __attribute__((ext_vector_type(4))) short shortx4;
typedef __attribute__((ext_vector_type(8))) short shortx8;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

DEVINL float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

DEVINL unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fff + lsb;
  return (unsigned short)(v.i >> 16);
}

// dot product of 8 bf16 pairs (unpacked to f32 FMAs; exact f32 accumulate)
DEVINL float dot8_bf16(const shortx8 a, const shortx8 b) {
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    acc += bf16_to_f32((unsigned short)a[i]) * bf16_to_f32((unsigned short)b[i]);
  return acc;
}

// packed-dot version: 4x v_dot2_f32_bf16 instead of 24 VALU ops — the
// decode-attention score path is VALU-bound without it
DEVINL float dot8_bf16_fast(const shortx8 a, const shortx8 b, float acc = 0.f) {
  const bf16x2* pa = (const bf16x2*)&a;
  const bf16x2* pb = (const bf16x2*)&b;
#pragma unroll
  for (int i = 0; i < 4; ++i)
    acc = __builtin_amdgcn_fdot2_f32_bf16(pa[i], pb[i], acc, false);
  return acc;
}

DEVINL unsigned int pack2_bf16d(float lo, float hi) {
  return ((unsigned int)f32_to_bf16(hi) << 16) | f32_to_bf16(lo);
}

// full-wave (64-lane) reductions
DEVINL float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
DEVINL float wave_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}
// reduction within a contiguous group of W lanes (W power of two <= 64)
template <int W>
DEVINL float group_sum(float x) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
template <int W>
DEVINL float group_max(float x) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// ---- fp8 (OCP e4m3) KV-cache conversion helpers (gfx950 native cvt;
// the chip's fp8 is e4m3fn, NOT MI300X fnuz — platform guide §4) -------
typedef unsigned long long u64b;

// 8 fp8 bytes -> 8 bf16 (used where attention kernels read the cache).
// Every e4m3 value (4-bit exp, 3-bit mantissa) is EXACTLY representable
// in bf16 (8-bit exp, 7-bit mantissa), so the f32->bf16 step is a pure
// high-half truncation — no rounding VALU; ~3 ops per converted pair.
DEVINL shortx8 fp8x8_to_bf16x8(u64b v) {
  shortx8 out;
  unsigned int* o32 = (unsigned int*)&out;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const floatx2 f = __builtin_amdgcn_cvt_pk_f32_fp8(
        (unsigned int)(v >> (16 * i)) & 0xffffu, false);
    union { float f; unsigned int u; } a, b;
    a.f = f[0];
    b.f = f[1];
    o32[i] = (a.u >> 16) | (b.u & 0xffff0000u);
  }
  return out;
}

// 8 bf16 -> 8 fp8 bytes (cache write path; saturates at +-448)
DEVINL u64b bf16x8_to_fp8x8(const shortx8 v) {
  u64b out = 0;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const unsigned int p = __builtin_amdgcn_cvt_pk_fp8_f32(
        bf16_to_f32((unsigned short)v[2 * i]),
        bf16_to_f32((unsigned short)v[2 * i + 1]), 0u, false);
    out |= (u64b)(p & 0xffffu) << (16 * i);
  }
  return out;
}

DEVINL unsigned char f32_to_fp8(float a) {
  return (unsigned char)(__builtin_amdgcn_cvt_pk_fp8_f32(a, 0.f, 0u, false)
                         & 0xffu);
}

DEVINL float fp8_to_f32(unsigned char b) {
  const floatx2 f = __builtin_amdgcn_cvt_pk_f32_fp8((unsigned int)b, false);
  return f[0];
}

#define HIP_CHECK_KERNEL()                                                     \
  do {                                                                         \
    hipError_t e = hipGetLastError();                                          \
    if (e != hipSuccess)                                                       \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e));  \
  } while (0)
