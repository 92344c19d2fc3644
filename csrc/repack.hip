// Attention layout repack kernels, gfx950.
//
// qkv_repack_fwd: [B, L, 3, H, dh] bf16 (the qkv Linear's natural output)
//                 -> [3, B, H, L, dh] contiguous (bmm-ready q/k/v)
// qkv_repack_bwd: the inverse gather (grad of the repack).
// out_repack_fwd: [B, H, L, dh] -> [B, L, H*dh]   (attention output merge)
// out_repack_bwd: inverse.
//
// These replace torch's generic permute-copy + CatArrayBatchedCopy pairs
// (3.5 + 4.2 ms/step in profiles/r01_kernel_stats_baseline.md follow-ups).
// dh is a multiple of 8: one short8 (16 B) packet per lane-iteration; both
// source and destination visits are coalesced along dh.

#include "common.h"

#define RP_BLOCK 256

// Flat view: element packets of 8 bf16.  total_pkts = B*L*3*H*dh/8.
// fwd: dst[s][b][h][l][d] = src[b][l][s][h][d]
__global__ void __launch_bounds__(RP_BLOCK)
qkv_repack_fwd_kernel(const short* __restrict__ src, short* __restrict__ dst,
                      int B, int L, int H, int dh_pkts, long total_pkts) {
  long i0 = (long)blockIdx.x * RP_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = i0; i < total_pkts; i += stride) {
    // decompose destination-ordered index: (((s*B + b)*H + h)*L + l)*dh_pkts + d
    long t = i;
    int d = (int)(t % dh_pkts); t /= dh_pkts;
    int l = (int)(t % L); t /= L;
    int h = (int)(t % H); t /= H;
    int b = (int)(t % B); t /= B;
    int s = (int)t;
    long src_idx = ((((long)b * L + l) * 3 + s) * H + h) * dh_pkts + d;
    *(short8_t*)(dst + i * 8) = *(const short8_t*)(src + src_idx * 8);
  }
}

__global__ void __launch_bounds__(RP_BLOCK)
qkv_repack_bwd_kernel(const short* __restrict__ dgrad, short* __restrict__ dsrc,
                      int B, int L, int H, int dh_pkts, long total_pkts) {
  long i0 = (long)blockIdx.x * RP_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = i0; i < total_pkts; i += stride) {
    // i indexes the SOURCE layout [B, L, 3, H, dh] (coalesced writes)
    long t = i;
    int d = (int)(t % dh_pkts); t /= dh_pkts;
    int h = (int)(t % H); t /= H;
    int s = (int)(t % 3); t /= 3;
    int l = (int)(t % L); t /= L;
    int b = (int)t;
    long g_idx = ((((long)s * B + b) * H + h) * L + l) * dh_pkts + d;
    *(short8_t*)(dsrc + i * 8) = *(const short8_t*)(dgrad + g_idx * 8);
  }
}

// out: [B, H, L, dh] -> [B, L, H*dh]
__global__ void __launch_bounds__(RP_BLOCK)
out_repack_fwd_kernel(const short* __restrict__ src, short* __restrict__ dst,
                      int B, int L, int H, int dh_pkts, long total_pkts) {
  long i0 = (long)blockIdx.x * RP_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = i0; i < total_pkts; i += stride) {
    // i indexes destination [B, L, H, dh]
    long t = i;
    int d = (int)(t % dh_pkts); t /= dh_pkts;
    int h = (int)(t % H); t /= H;
    int l = (int)(t % L); t /= L;
    int b = (int)t;
    long src_idx = ((((long)b * H + h) * L + l)) * dh_pkts + d;
    *(short8_t*)(dst + i * 8) = *(const short8_t*)(src + src_idx * 8);
  }
}

__global__ void __launch_bounds__(RP_BLOCK)
out_repack_bwd_kernel(const short* __restrict__ dgrad, short* __restrict__ dsrc,
                      int B, int L, int H, int dh_pkts, long total_pkts) {
  long i0 = (long)blockIdx.x * RP_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = i0; i < total_pkts; i += stride) {
    // i indexes [B, H, L, dh] (coalesced writes)
    long t = i;
    int d = (int)(t % dh_pkts); t /= dh_pkts;
    int l = (int)(t % L); t /= L;
    int h = (int)(t % H); t /= H;
    int b = (int)t;
    long g_idx = (((long)b * L + l) * H + h) * dh_pkts + d;
    *(short8_t*)(dsrc + i * 8) = *(const short8_t*)(dgrad + g_idx * 8);
  }
}

__global__ void __launch_bounds__(RP_BLOCK)
qkv_repack_bwd3_kernel(const short* __restrict__ dq, const short* __restrict__ dk,
                       const short* __restrict__ dv, short* __restrict__ dsrc,
                       int B, int L, int H, int dh_pkts, long total_pkts) {
  long i0 = (long)blockIdx.x * RP_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = i0; i < total_pkts; i += stride) {
    // i indexes destination [B, L, 3, H, dh] (coalesced writes)
    long t = i;
    int d = (int)(t % dh_pkts); t /= dh_pkts;
    int h = (int)(t % H); t /= H;
    int s = (int)(t % 3); t /= 3;
    int l = (int)(t % L); t /= L;
    int b = (int)t;
    const short* g = s == 0 ? dq : (s == 1 ? dk : dv);
    long g_idx = (((long)b * H + h) * L + l) * dh_pkts + d;
    *(short8_t*)(dsrc + i * 8) = *(const short8_t*)(g + g_idx * 8);
  }
}

extern "C" {

hipError_t qkv_repack_bwd3_launch(const void* dq, const void* dk,
                                  const void* dv, void* dsrc, int B, int L,
                                  int H, int dh, hipStream_t stream) {
  int dh_pkts = dh / 8;
  long total = (long)B * L * 3 * H * dh_pkts;
  int grid = (int)((total + RP_BLOCK - 1) / RP_BLOCK);
  if (grid > 2048) grid = 2048;
  qkv_repack_bwd3_kernel<<<grid, RP_BLOCK, 0, stream>>>(
      (const short*)dq, (const short*)dk, (const short*)dv, (short*)dsrc,
      B, L, H, dh_pkts, total);
  return hipGetLastError();
}


hipError_t qkv_repack_launch(const void* src, void* dst, int B, int L, int H,
                             int dh, int backward, hipStream_t stream) {
  int dh_pkts = dh / 8;
  long total = (long)B * L * 3 * H * dh_pkts;
  int grid = (int)((total + RP_BLOCK - 1) / RP_BLOCK);
  if (grid > 2048) grid = 2048;
  if (backward)
    qkv_repack_bwd_kernel<<<grid, RP_BLOCK, 0, stream>>>(
        (const short*)src, (short*)dst, B, L, H, dh_pkts, total);
  else
    qkv_repack_fwd_kernel<<<grid, RP_BLOCK, 0, stream>>>(
        (const short*)src, (short*)dst, B, L, H, dh_pkts, total);
  return hipGetLastError();
}

hipError_t out_repack_launch(const void* src, void* dst, int B, int L, int H,
                             int dh, int backward, hipStream_t stream) {
  int dh_pkts = dh / 8;
  long total = (long)B * L * H * dh_pkts;
  int grid = (int)((total + RP_BLOCK - 1) / RP_BLOCK);
  if (grid > 2048) grid = 2048;
  if (backward)
    out_repack_bwd_kernel<<<grid, RP_BLOCK, 0, stream>>>(
        (const short*)src, (short*)dst, B, L, H, dh_pkts, total);
  else
    out_repack_fwd_kernel<<<grid, RP_BLOCK, 0, stream>>>(
        (const short*)src, (short*)dst, B, L, H, dh_pkts, total);
  return hipGetLastError();
}

}  // extern "C"
