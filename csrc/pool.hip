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

// Round 2: two-stage forward.  The round-1 kernel used one workgroup per
// batch row (B=128 blocks on a 256-CU chip, serial 512-row loop: 452 us
// for ~130 MB).  Stage 1 splits the sequence into PL_SPLITS chunks per
// batch (B*PL_SPLITS workgroups) accumulating f32 partials; stage 2
// reduces the splits in fixed order (deterministic) and scales.
#define PL_SPLITS 16

extern "C" __global__ void __launch_bounds__(PL_BLOCK)
masked_pool_fwd_stage1_kernel(const short* __restrict__ x,
                              const bool* __restrict__ mask,
                              float* __restrict__ ws,  // [B, S, D]
                              int B, int L, int D) {
  const int b = blockIdx.x / PL_SPLITS;
  const int s = blockIdx.x % PL_SPLITS;
  const int chunk = (L + PL_SPLITS - 1) / PL_SPLITS;
  const int l0 = s * chunk;
  const int l1 = min(l0 + chunk, L);
  const short* xb = x + (long)b * L * D;
  const bool* mb = mask ? mask + (long)b * L : nullptr;
  float* out = ws + ((long)b * PL_SPLITS + s) * D;
  for (int d0 = threadIdx.x * 8; d0 < D; d0 += PL_BLOCK * 8) {
    float acc[8] = {0.f};
    for (int l = l0; l < l1; ++l) {
      if (mb && !mb[l]) continue;
      short8_t v = *(const short8_t*)(xb + (long)l * D + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32(v[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) out[d0 + j] = acc[j];
  }
}

extern "C" __global__ void __launch_bounds__(PL_BLOCK)
masked_pool_fwd_stage2_kernel(const float* __restrict__ ws,
                              const bool* __restrict__ mask,
                              short* __restrict__ pooled,
                              float* __restrict__ counts,
                              int B, int L, int D) {
  const int b = blockIdx.x;
  __shared__ float s_cnt;
  if (threadIdx.x == 0) {
    int c = 0;
    const bool* mb = mask ? mask + (long)b * L : nullptr;
    for (int l = 0; l < L; ++l) c += mb ? (int)mb[l] : 1;
    s_cnt = (float)max(c, 1);
    counts[b] = s_cnt;
  }
  __syncthreads();
  const float inv = 1.0f / s_cnt;
  const float* wb = ws + (long)b * PL_SPLITS * D;
  for (int d = threadIdx.x; d < D; d += PL_BLOCK) {
    float acc = 0.f;
#pragma unroll 4
    for (int s = 0; s < PL_SPLITS; ++s) acc += wb[(long)s * D + d];
    pooled[(long)b * D + d] = f32_to_bf16(acc * inv);
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
                                             void* ws, int B, int L, int D,
                                             hipStream_t s) {
  masked_pool_fwd_stage1_kernel<<<B * PL_SPLITS, PL_BLOCK, 0, s>>>(
      (const short*)x, (const bool*)mask, (float*)ws, B, L, D);
  masked_pool_fwd_stage2_kernel<<<B, PL_BLOCK, 0, s>>>(
      (const float*)ws, (const bool*)mask, (short*)pooled, (float*)counts,
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
