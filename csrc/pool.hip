// Masked mean-pool over the sequence axis, bf16, gfx950.
//
//   pooled[b, d] = sum_l mask[b, l] * x[b, l, d] / max(count[b], 1)
//
// One 256-thread workgroup per batch row; threads stride D with f32
// accumulators, rows visited once (coalesced 16-B packets).  Backward
// broadcasts dpooled/count into the valid rows.  Replaces the
// mul+reduce+div chain torch emits for the classifier's pooling.

#include "common.h"

#define PL_BLOCK 256

extern "C" __global__ void __launch_bounds__(PL_BLOCK)
masked_pool_fwd_kernel(const short* __restrict__ x,
                       const bool* __restrict__ mask,
                       short* __restrict__ pooled,
                       float* __restrict__ counts, int B, int L, int D) {
  const int b = blockIdx.x;
  const short* xb = x + (long)b * L * D;
  const bool* mb = mask ? mask + (long)b * L : nullptr;
  // count valid rows once (thread 0 lane-parallel would be overkill)
  __shared__ float s_cnt;
  if (threadIdx.x == 0) {
    int c = 0;
    for (int l = 0; l < L; ++l) c += mb ? (int)mb[l] : 1;
    s_cnt = (float)max(c, 1);
    counts[b] = s_cnt;
  }
  __syncthreads();
  const float inv = 1.0f / s_cnt;
  for (int d0 = threadIdx.x * 8; d0 < D; d0 += PL_BLOCK * 8) {
    float acc[8] = {0.f};
    for (int l = 0; l < L; ++l) {
      if (mb && !mb[l]) continue;
      short8_t v = *(const short8_t*)(xb + (long)l * D + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32(v[j]);
    }
    short8_t o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f32_to_bf16(acc[j] * inv);
    *(short8_t*)(pooled + (long)b * D + d0) = o;
  }
}

extern "C" __global__ void __launch_bounds__(PL_BLOCK)
masked_pool_bwd_kernel(const short* __restrict__ dpooled,
                       const bool* __restrict__ mask,
                       const float* __restrict__ counts,
                       short* __restrict__ dx, int B, int L, int D) {
  // grid-stride over [B, L] rows; each block writes whole rows
  long n_rows = (long)B * L;
  for (long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    int b = (int)(row / L);
    int l = (int)(row % L);
    bool valid = mask ? mask[(long)b * L + l] : true;
    const float inv = valid ? 1.0f / counts[b] : 0.f;
    const short* dp = dpooled + (long)b * D;
    short* out = dx + row * D;
    for (int d0 = threadIdx.x * 8; d0 < D; d0 += PL_BLOCK * 8) {
      short8_t o;
      if (valid) {
        short8_t g = *(const short8_t*)(dp + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f32_to_bf16(bf16_to_f32(g[j]) * inv);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = 0;
      }
      *(short8_t*)(out + d0) = o;
    }
  }
}

extern "C" hipError_t masked_pool_fwd_launch(const void* x, const void* mask,
                                             void* pooled, void* counts,
                                             int B, int L, int D,
                                             hipStream_t s) {
  masked_pool_fwd_kernel<<<B, PL_BLOCK, 0, s>>>(
      (const short*)x, (const bool*)mask, (short*)pooled, (float*)counts,
      B, L, D);
  return hipGetLastError();
}

extern "C" hipError_t masked_pool_bwd_launch(const void* dpooled,
                                             const void* mask,
                                             const void* counts, void* dx,
                                             int B, int L, int D,
                                             hipStream_t s) {
  long n_rows = (long)B * L;
  int grid = (int)(n_rows < 2048 ? n_rows : 2048);
  masked_pool_bwd_kernel<<<grid, PL_BLOCK, 0, s>>>(
      (const short*)dpooled, (const bool*)mask, (const float*)counts,
      (short*)dx, B, L, D);
  return hipGetLastError();
}
