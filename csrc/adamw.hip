// Fused AdamW step over FLAT parameter storage, gfx950.
//
// The trainer flattens every parameter into one contiguous bf16 tensor with a
// matching f32 master copy and f32 m/v states (tosem2021_amd/train.py), so the
// whole optimizer step is ONE bandwidth-bound kernel pass: read g(bf16),
// m,v,master(f32), write m,v,master(f32) + p(bf16) = 26 B/elem read +
// 22 B/elem write.  Decoupled weight decay (AdamW), bias-corrected.
//
// grad may be bf16 (local step) or f32 (set by the DDP reduce path).

#include "common.h"

#define AD_BLOCK 256

template <bool GRAD_F32>
__global__ void __launch_bounds__(AD_BLOCK)
adamw_kernel(short* __restrict__ p_bf16, const void* __restrict__ grad,
             float* __restrict__ m, float* __restrict__ v,
             float* __restrict__ master, long n, float lr, float beta1,
             float beta2, float eps, float wd, float bc1, float bc2,
             float grad_scale) {
  long i0 = ((long)blockIdx.x * AD_BLOCK + threadIdx.x) * 4;
  long stride = (long)gridDim.x * AD_BLOCK * 4;
  for (long i = i0; i < n; i += stride) {
    // 4-wide: 16 B f32 / 8 B bf16 per lane per tensor
    float4_t gm, mm, vv, mw;
    mm = *(float4_t*)(m + i);
    vv = *(float4_t*)(v + i);
    mw = *(float4_t*)(master + i);
    if (GRAD_F32) {
      gm = *(const float4_t*)((const float*)grad + i);
    } else {
      short4_t gs = *(const short4_t*)((const short*)grad + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) gm[j] = bf16_to_f32(gs[j]);
    }
    short4_t po;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = gm[j] * grad_scale;
      float m_ = beta1 * mm[j] + (1.f - beta1) * g;
      float v_ = beta2 * vv[j] + (1.f - beta2) * g * g;
      float mhat = m_ * bc1;
      float vhat = v_ * bc2;
      float w = mw[j];
      w -= lr * (mhat / (sqrtf(vhat) + eps) + wd * w);
      mm[j] = m_; vv[j] = v_; mw[j] = w;
      po[j] = f32_to_bf16(w);
    }
    *(float4_t*)(m + i) = mm;
    *(float4_t*)(v + i) = vv;
    *(float4_t*)(master + i) = mw;
    *(short4_t*)(p_bf16 + i) = po;
  }
}

extern "C" hipError_t adamw_launch(void* p_bf16, const void* grad, int grad_is_f32,
                        void* m, void* v, void* master, long n, float lr,
                        float beta1, float beta2, float eps, float wd,
                        int step, float grad_scale, int grid, hipStream_t s) {
  float bc1 = 1.0f / (1.0f - powf(beta1, (float)step));
  float bc2 = 1.0f / (1.0f - powf(beta2, (float)step));
  if (grad_is_f32)
    adamw_kernel<true><<<grid, AD_BLOCK, 0, s>>>(
        (short*)p_bf16, grad, (float*)m, (float*)v, (float*)master, n, lr,
        beta1, beta2, eps, wd, bc1, bc2, grad_scale);
  else
    adamw_kernel<false><<<grid, AD_BLOCK, 0, s>>>(
        (short*)p_bf16, grad, (float*)m, (float*)v, (float*)master, n, lr,
        beta1, beta2, eps, wd, bc1, bc2, grad_scale);
  return hipGetLastError();
}
