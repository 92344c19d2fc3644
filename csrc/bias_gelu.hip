// Fused bias + GeLU (tanh) forward/backward, bf16, gfx950.
//
// y = gelu(x + b); bwd recomputes the pre-activation from x,b (no saved
// activation: 1 extra read instead of a [N,Dff] bf16 save — HBM3E-friendly).
// dbias uses deterministic per-block LDS partials + colsum (layernorm.hip).
// x: [N, D] bf16 row-major, b: [D] bf16.  Elementwise, memory-bound: 16 B/lane
// vector loads, grid-stride, <= 2048 workgroups (guide G11/G13).

#include "common.h"

#define BG_BLOCK 256

extern "C" {

__global__ void __launch_bounds__(BG_BLOCK)
bias_gelu_fwd_kernel(const short* __restrict__ x, const short* __restrict__ b,
                     short* __restrict__ y, long n_elem, int D) {
  long idx0 = ((long)blockIdx.x * BG_BLOCK + threadIdx.x) * 8;
  long stride = (long)gridDim.x * BG_BLOCK * 8;
  if (stride % D == 0) {
    // fixed column window: hoist the bias load and the 64-bit modulo
    const int col = (int)(idx0 % D);
    short8_t b8 = *(const short8_t*)(b + col);
    float bb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) bb[j] = bf16_to_f32(b8[j]);
    for (long i = idx0; i < n_elem; i += stride) {
      short8_t v = *(const short8_t*)(x + i);
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f32_to_bf16(gelu_tanh(bf16_to_f32(v[j]) + bb[j]));
      *(short8_t*)(y + i) = o;
    }
    return;
  }
  for (long i = idx0; i < n_elem; i += stride) {
    short8_t v = *(const short8_t*)(x + i);
    int col = (int)(i % D);  // D % 8 == 0 so the packet stays in one row
    short8_t b8 = *(const short8_t*)(b + col);
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f32_to_bf16(gelu_tanh(bf16_to_f32(v[j]) + bf16_to_f32(b8[j])));
    *(short8_t*)(y + i) = o;
  }
}

// When the grid stride is a multiple of D (the host picks such a grid), each
// thread's packet stays on ONE fixed 8-column window for its whole loop, so
// the dbias partial accumulates in 8 registers and costs 8 LDS atomics per
// thread total, not 8 per element (was 8.1 ms/step, baseline profile).
__global__ void __launch_bounds__(BG_BLOCK)
bias_gelu_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                     const short* __restrict__ b, short* __restrict__ dx,
                     float* __restrict__ ws_dbias, long n_elem, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sdb = (float*)smem;  // [D]
  for (int i = threadIdx.x; i < D; i += BG_BLOCK) sdb[i] = 0.f;
  __syncthreads();
  long idx0 = ((long)blockIdx.x * BG_BLOCK + threadIdx.x) * 8;
  long stride = (long)gridDim.x * BG_BLOCK * 8;
  if (stride % D == 0) {
    float acc[8] = {0.f};
    const int col = (int)(idx0 % D);
    short8_t bv = *(const short8_t*)(b + col);
    float bb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) bb[j] = bf16_to_f32(bv[j]);
    for (long i = idx0; i < n_elem; i += stride) {
      short8_t vd = *(const short8_t*)(dy + i);
      short8_t vx = *(const short8_t*)(x + i);
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf16_to_f32(vd[j]);
        float pre = bf16_to_f32(vx[j]) + bb[j];
        float dpre = g * gelu_tanh_grad(pre);
        o[j] = f32_to_bf16(dpre);
        acc[j] += dpre;  // dL/db = sum over rows of dy * gelu'(pre)
      }
      *(short8_t*)(dx + i) = o;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) atomicAdd(&sdb[col + j], acc[j]);
  } else {
    for (long i = idx0; i < n_elem; i += stride) {
      short8_t vd = *(const short8_t*)(dy + i);
      short8_t vx = *(const short8_t*)(x + i);
      int col = (int)(i % D);
      short8_t b8 = *(const short8_t*)(b + col);
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf16_to_f32(vd[j]);
        float pre = bf16_to_f32(vx[j]) + bf16_to_f32(b8[j]);
        float dpre = g * gelu_tanh_grad(pre);
        o[j] = f32_to_bf16(dpre);
        atomicAdd(&sdb[col + j], dpre);
      }
      *(short8_t*)(dx + i) = o;
    }
  }
  __syncthreads();
  float* out = ws_dbias + (long)blockIdx.x * D;
  for (int i = threadIdx.x; i < D; i += BG_BLOCK) out[i] = sdb[i];
}

hipError_t bias_gelu_fwd_launch(const void* x, const void* b, void* y,
                                long n_elem, int D, int grid, hipStream_t s) {
  bias_gelu_fwd_kernel<<<grid, BG_BLOCK, 0, s>>>((const short*)x, (const short*)b,
                                                 (short*)y, n_elem, D);
  return hipGetLastError();
}

hipError_t bias_gelu_bwd_launch(const void* dy, const void* x, const void* b,
                                void* dx, void* ws_dbias, long n_elem, int D,
                                int grid, hipStream_t s) {
  size_t shm = (size_t)D * sizeof(float);
  bias_gelu_bwd_kernel<<<grid, BG_BLOCK, shm, s>>>(
      (const short*)dy, (const short*)x, (const short*)b, (short*)dx,
      (float*)ws_dbias, n_elem, D);
  return hipGetLastError();
}

}  // extern "C"
