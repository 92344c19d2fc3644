// Common device helpers for the tosem2021_amd gfx950 (CDNA4) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes; all cross-lane reductions are 64-wide shuffles.
//  * bf16 global traffic is always vectorized as short4/short8 reinterprets
//    (16 B per lane) — hipcc does not auto-vectorize scalar bf16 loads (G13).
//  * memory-bound kernels use grid-stride loops capped at ~2048 workgroups
//    (G11: 256 CUs x 8 blocks/CU).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) short short8_t;
typedef __attribute__((ext_vector_type(4))) short short4_t;
typedef __attribute__((ext_vector_type(4))) float float4_t;

// ---- bf16 <-> f32 -----------------------------------------------------------
__device__ __forceinline__ float bf16_to_f32(short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's float->bfloat16 conversion
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  if ((x & 0x7fffffffu) > 0x7f800000u) return (short)0x7fc0;  // NaN
  unsigned int lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;
  return (short)(x >> 16);
}

// ---- 64-wide wave reductions ------------------------------------------------
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// ---- gelu (tanh approximation, matches torch.nn.GELU(approximate="tanh")) ---
// tanh via the hardware exp: tanh(y) = 1 - 2/(exp(2y) + 1).  libm tanhf
// compiles to a ~60-instruction branchy sequence on amdgcn; PMC showed
// bias_gelu_bwd spending ~100 VALU instructions PER ELEMENT through it
// (profiles/r28).  This form is ~6 instructions + one v_exp and is exact
// at the extremes (e -> inf gives 1, e -> 0 gives -1).
__device__ __forceinline__ float fast_tanh(float y) {
  float e = __expf(2.0f * y);
  return 1.0f - 2.0f / (e + 1.0f);
}

__device__ __forceinline__ float gelu_tanh(float x) {
  const float k0 = 0.7978845608028654f;   // sqrt(2/pi)
  const float k1 = 0.044715f;
  float inner = k0 * (x + k1 * x * x * x);
  return 0.5f * x * (1.0f + fast_tanh(inner));
}

__device__ __forceinline__ float gelu_tanh_grad(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float x2 = x * x;
  float inner = k0 * (x + k1 * x * x2);
  float t = fast_tanh(inner);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * k0 * (1.0f + 3.0f * k1 * x2);
}

// Kernel files expose C-linkage launchers returning hipError_t; the torch
// binding layer (ops.cpp) is the only place that includes torch headers.

// Place `group` consecutive block ids on the same XCD (dispatcher maps block
// b -> XCD b % 8; guide T1).  Bijective when (n_blocks / group) % 8 == 0;
// falls back to identity otherwise.  Performance-only (G16: never rely on
// placement for correctness).
__device__ __forceinline__ int xcd_group_remap(int bid, int n_blocks,
                                               int group) {
  int per = n_blocks / group;
  if (n_blocks % group != 0 || per % 8 != 0) return bid;
  int g = bid / group, r = bid % group;
  return g + r * per;
}
