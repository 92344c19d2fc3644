// Fused bias + GeLU (tanh) forward/backward, bf16, gfx950.
//
// y = gelu(x + b); bwd recomputes the pre-activation from x,b (no saved
// activation: 1 extra read instead of a [N,Dff] bf16 save — HBM3E-friendly).
// dbias uses deterministic per-block LDS partials + colsum (layernorm.hip).
// x: [N, D] bf16 row-major, b: [D] bf16.
//
// Round 2: 4 packets per thread per iteration for memory-level parallelism,
// laid out BLOCK-STRIDED (packet p of thread t at tile + p*BG_BLOCK*8 + t*8)
// so every b128 load instruction stays perfectly dense across the wave —
// the first attempt used 4 CONSECUTIVE packets per thread (64 B lane
// stride), which broke per-instruction coalescing and measured ~15% slower
// than round 1.  Fast path needs the grid stride and the per-packet stride
// (BG_BLOCK*8 = 2048) multiples of D and n % 8 == 0; generic fallback
// otherwise (host rounds the grid accordingly).

#include "common.h"

#define BG_BLOCK 256
#define BG_PK 4                      // packets of 8 bf16 per iteration
#define BG_TILE (BG_BLOCK * 8)       // elements per packet-slab per block

extern "C" {

__global__ void __launch_bounds__(BG_BLOCK)
bias_gelu_fwd_kernel(const short* __restrict__ x, const short* __restrict__ b,
                     short* __restrict__ y, long n_elem, int D) {
  long tile0 = (long)blockIdx.x * (BG_TILE * BG_PK) + threadIdx.x * 8;
  long stride = (long)gridDim.x * (BG_TILE * BG_PK);
  if (stride % D == 0) {
    // per-packet columns are loop-invariant: hoist the bias loads
    float bb[BG_PK * 8];
#pragma unroll
    for (int p = 0; p < BG_PK; ++p) {
      int col = (int)((tile0 + p * BG_TILE) % D);
      short8_t b8 = *(const short8_t*)(b + col);
#pragma unroll
      for (int j = 0; j < 8; ++j) bb[8 * p + j] = bf16_to_f32(b8[j]);
    }
    for (long i = tile0; i < n_elem; i += stride) {
      short8_t v[BG_PK];
      bool ok[BG_PK];
#pragma unroll
      for (int p = 0; p < BG_PK; ++p) {
        ok[p] = i + p * BG_TILE < n_elem;
        v[p] = ok[p] ? *(const short8_t*)(x + i + p * BG_TILE) : short8_t{};
      }
#pragma unroll
      for (int p = 0; p < BG_PK; ++p) {
        if (!ok[p]) continue;
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = f32_to_bf16(gelu_tanh(bf16_to_f32(v[p][j]) + bb[8 * p + j]));
        *(short8_t*)(y + i + p * BG_TILE) = o;
      }
    }
    return;
  }
  // generic path (one packet per iteration, per-packet bias reload)
  long idx0 = ((long)blockIdx.x * BG_BLOCK + threadIdx.x) * 8;
  long gstride = (long)gridDim.x * BG_BLOCK * 8;
  for (long i = idx0; i < n_elem; i += gstride) {
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

// dbias: each thread's per-packet fixed 8-column windows accumulate in
// registers; 32 LDS atomics per thread at the END (not per element), then
// one per-block f32 partial row for the deterministic colsum reduction.
__global__ void __launch_bounds__(BG_BLOCK)
bias_gelu_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                     const short* __restrict__ b, short* __restrict__ dx,
                     float* __restrict__ ws_dbias, long n_elem, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sdb = (float*)smem;  // [D]
  for (int i = threadIdx.x; i < D; i += BG_BLOCK) sdb[i] = 0.f;
  __syncthreads();
  long tile0 = (long)blockIdx.x * (BG_TILE * BG_PK) + threadIdx.x * 8;
  long stride = (long)gridDim.x * (BG_TILE * BG_PK);
  if (stride % D == 0) {
    float acc[BG_PK * 8] = {0.f};
    int cols[BG_PK];
    float bb[BG_PK * 8];
#pragma unroll
    for (int p = 0; p < BG_PK; ++p) {
      cols[p] = (int)((tile0 + p * BG_TILE) % D);
      short8_t b8 = *(const short8_t*)(b + cols[p]);
#pragma unroll
      for (int j = 0; j < 8; ++j) bb[8 * p + j] = bf16_to_f32(b8[j]);
    }
    for (long i = tile0; i < n_elem; i += stride) {
      short8_t vd[BG_PK], vx[BG_PK];
      bool ok[BG_PK];
#pragma unroll
      for (int p = 0; p < BG_PK; ++p) {
        ok[p] = i + p * BG_TILE < n_elem;
        vd[p] = ok[p] ? *(const short8_t*)(dy + i + p * BG_TILE) : short8_t{};
        vx[p] = ok[p] ? *(const short8_t*)(x + i + p * BG_TILE) : short8_t{};
      }
#pragma unroll
      for (int p = 0; p < BG_PK; ++p) {
        if (!ok[p]) continue;
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = bf16_to_f32(vd[p][j]);
          float pre = bf16_to_f32(vx[p][j]) + bb[8 * p + j];
          float dpre = g * gelu_tanh_grad(pre);
          o[j] = f32_to_bf16(dpre);
          acc[8 * p + j] += dpre;
        }
        *(short8_t*)(dx + i + p * BG_TILE) = o;
      }
    }
#pragma unroll
    for (int p = 0; p < BG_PK; ++p)
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&sdb[cols[p] + j], acc[8 * p + j]);
  } else {
    long idx0 = ((long)blockIdx.x * BG_BLOCK + threadIdx.x) * 8;
    long gstride = (long)gridDim.x * BG_BLOCK * 8;
    for (long i = idx0; i < n_elem; i += gstride) {
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
