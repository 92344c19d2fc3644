// Fused scaled masked row-softmax forward/backward for attention scores,
// bf16 in/out with f32 math, gfx950.
//
// P = softmax(scale * S + mask_bias) over the last dim.
//   scores: [B, H, Lq, Lk] bf16 (contiguous), mask_bias: [B, Lk] f32
//   (0 for valid keys, -inf/-1e9 for padding) or nullptr.
// Backward: dS = scale * P * (dP - rowsum(dP * P)).
//
// One 64-lane wave owns one row; Lk <= 64*8*MAX_SM_PKT held in registers.
// Classifier uses Lk in {128, 512}; general loop covers larger Lk.

#include "common.h"

#define SM_WPB 4
#define SM_BLOCK (SM_WPB * WAVE)
#define MAX_SM_PKT 4  // register path for Lk <= 2048

extern "C" {

__global__ void __launch_bounds__(SM_BLOCK)
softmax_fwd_kernel(const short* __restrict__ s_in, const float* __restrict__ mask,
                   short* __restrict__ p_out, long n_rows, int Lk, int H_Lq,
                   float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int pkts = Lk / (WAVE * 8);
  for (long row = blockIdx.x * SM_WPB + wid; row < n_rows;
       row += (long)gridDim.x * SM_WPB) {
    const short* sr = s_in + row * Lk;
    short* pr = p_out + row * Lk;
    const float* mrow = mask ? mask + (row / H_Lq) * Lk : nullptr;
    if (pkts >= 1 && pkts <= MAX_SM_PKT && Lk == pkts * WAVE * 8) {
      float v[MAX_SM_PKT * 8];
      float m = -3.0e38f;
#pragma unroll
      for (int p = 0; p < MAX_SM_PKT; ++p) {
        if (p >= pkts) break;
        int base = (p * WAVE + lane) * 8;
        short8_t x = *(const short8_t*)(sr + base);
        float4_t m0, m1;
        if (mrow) {
          m0 = *(const float4_t*)(mrow + base);
          m1 = *(const float4_t*)(mrow + base + 4);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32(x[j]) * scale;
          if (mrow) f += j < 4 ? m0[j] : m1[j - 4];
          v[p * 8 + j] = f;
          m = fmaxf(m, f);
        }
      }
      m = wave_max(m);
      float sum = 0.f;
#pragma unroll
      for (int p = 0; p < MAX_SM_PKT; ++p) {
        if (p >= pkts) break;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float e = __expf(v[p * 8 + j] - m);
          v[p * 8 + j] = e;
          sum += e;
        }
      }
      sum = wave_sum(sum);
      float inv = 1.0f / sum;
#pragma unroll
      for (int p = 0; p < MAX_SM_PKT; ++p) {
        if (p >= pkts) break;
        int base = (p * WAVE + lane) * 8;
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f32_to_bf16(v[p * 8 + j] * inv);
        *(short8_t*)(pr + base) = o;
      }
    } else {
      // general path (Lk % 8 == 0): 3 passes
      float m = -3.0e38f;
      for (int i = lane * 8; i < Lk; i += WAVE * 8) {
        short8_t x = *(const short8_t*)(sr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32(x[j]) * scale;
          if (mrow) f += mrow[i + j];
          m = fmaxf(m, f);
        }
      }
      m = wave_max(m);
      float sum = 0.f;
      for (int i = lane * 8; i < Lk; i += WAVE * 8) {
        short8_t x = *(const short8_t*)(sr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32(x[j]) * scale;
          if (mrow) f += mrow[i + j];
          sum += __expf(f - m);
        }
      }
      sum = wave_sum(sum);
      float inv = 1.0f / sum;
      for (int i = lane * 8; i < Lk; i += WAVE * 8) {
        short8_t x = *(const short8_t*)(sr + i);
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32(x[j]) * scale;
          if (mrow) f += mrow[i + j];
          o[j] = f32_to_bf16(__expf(f - m) * inv);
        }
        *(short8_t*)(pr + i) = o;
      }
    }
  }
}

__global__ void __launch_bounds__(SM_BLOCK)
softmax_bwd_kernel(const short* __restrict__ dp, const short* __restrict__ p_in,
                   short* __restrict__ ds, long n_rows, int Lk, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int pkts = Lk / (WAVE * 8);
  for (long row = blockIdx.x * SM_WPB + wid; row < n_rows;
       row += (long)gridDim.x * SM_WPB) {
    const short* dpr = dp + row * Lk;
    const short* pr = p_in + row * Lk;
    short* dsr = ds + row * Lk;
    if (pkts >= 1 && pkts <= MAX_SM_PKT && Lk == pkts * WAVE * 8) {
      float dv[MAX_SM_PKT * 8], pv[MAX_SM_PKT * 8];
      float dot = 0.f;
#pragma unroll
      for (int p = 0; p < MAX_SM_PKT; ++p) {
        if (p >= pkts) break;
        int base = (p * WAVE + lane) * 8;
        short8_t a = *(const short8_t*)(dpr + base);
        short8_t b = *(const short8_t*)(pr + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float d = bf16_to_f32(a[j]), q = bf16_to_f32(b[j]);
          dv[p * 8 + j] = d; pv[p * 8 + j] = q;
          dot += d * q;
        }
      }
      dot = wave_sum(dot);
#pragma unroll
      for (int p = 0; p < MAX_SM_PKT; ++p) {
        if (p >= pkts) break;
        int base = (p * WAVE + lane) * 8;
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int k = p * 8 + j;
          o[j] = f32_to_bf16(scale * pv[k] * (dv[k] - dot));
        }
        *(short8_t*)(dsr + base) = o;
      }
    } else {
      float dot = 0.f;
      for (int i = lane * 8; i < Lk; i += WAVE * 8) {
        short8_t a = *(const short8_t*)(dpr + i);
        short8_t b = *(const short8_t*)(pr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) dot += bf16_to_f32(a[j]) * bf16_to_f32(b[j]);
      }
      dot = wave_sum(dot);
      for (int i = lane * 8; i < Lk; i += WAVE * 8) {
        short8_t a = *(const short8_t*)(dpr + i);
        short8_t b = *(const short8_t*)(pr + i);
        short8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = f32_to_bf16(scale * bf16_to_f32(b[j]) *
                             (bf16_to_f32(a[j]) - dot));
        *(short8_t*)(dsr + i) = o;
      }
    }
  }
}

hipError_t softmax_fwd_launch(const void* s, const void* mask, void* p,
                              long n_rows, int Lk, int H_Lq, float scale,
                              int grid, hipStream_t stream) {
  softmax_fwd_kernel<<<grid, SM_BLOCK, 0, stream>>>(
      (const short*)s, (const float*)mask, (short*)p, n_rows, Lk, H_Lq, scale);
  return hipGetLastError();
}

hipError_t softmax_bwd_launch(const void* dp, const void* p, void* ds,
                              long n_rows, int Lk, float scale, int grid,
                              hipStream_t stream) {
  softmax_bwd_kernel<<<grid, SM_BLOCK, 0, stream>>>(
      (const short*)dp, (const short*)p, (short*)ds, n_rows, Lk, scale);
  return hipGetLastError();
}

}  // extern "C"

// P = exp(scale * S + mask_bias - lse_row): elementwise probability
// recompute from the flash forward's logsumexp (flash backward path).
extern "C" __global__ void __launch_bounds__(SM_BLOCK)
p_from_lse_kernel(const short* __restrict__ s_in, const float* __restrict__ mask,
                  const float* __restrict__ lse, short* __restrict__ p_out,
                  long n_rows, int Lk, int H_Lq, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  for (long row = blockIdx.x * SM_WPB + wid; row < n_rows;
       row += (long)gridDim.x * SM_WPB) {
    const short* sr = s_in + row * Lk;
    short* pr = p_out + row * Lk;
    const float* mrow = mask ? mask + (row / H_Lq) * Lk : nullptr;
    const float l = lse[row];
    for (int i = lane * 8; i < Lk; i += WAVE * 8) {
      short8_t x = *(const short8_t*)(sr + i);
      float4_t m0, m1;
      if (mrow) {
        m0 = *(const float4_t*)(mrow + i);
        m1 = *(const float4_t*)(mrow + i + 4);
      }
      short8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32(x[j]) * scale - l;
        if (mrow) f += j < 4 ? m0[j] : m1[j - 4];
        o[j] = f32_to_bf16(__expf(f));
      }
      *(short8_t*)(pr + i) = o;
    }
  }
}

extern "C" hipError_t p_from_lse_launch(const void* s, const void* mask,
                                        const void* lse, void* p, long n_rows,
                                        int Lk, int H_Lq, float scale,
                                        int grid, hipStream_t stream) {
  p_from_lse_kernel<<<grid, SM_BLOCK, 0, stream>>>(
      (const short*)s, (const float*)mask, (const float*)lse, (short*)p,
      n_rows, Lk, H_Lq, scale);
  return hipGetLastError();
}
