// Flash-attention forward for gfx950 (CDNA4) — bf16 I/O, f32 online softmax,
// MFMA (v_mfma_f32_32x32x16_bf16) for QK^T and PV, head_dim = 64.
//
// Never materializes the [L, L] score matrix: per 32-row Q block (one wave),
// iterate 32-key KV tiles with the running-max/denominator recurrence and
// accumulate O in registers.  Saves the per-row logsumexp for the recompute
// backward (tosem2021_amd/ops: bwd = bmm-recompute + softmax_bwd).
//
// Structure (guide §B "fused attention prefill" adapted):
//  * swapped QK^T — compute mfma(A=K, B=Q^T) so the score column for one
//    q-row is LANE-LOCAL (16 regs + the partner lane at lane^32): softmax is
//    15 in-lane max/sum ops + one __shfl_xor(32) exchange, no LDS round trip.
//  * P -> PV A-fragments via v_cvt_pk_bf16_f32 pairs + permlane32_swap
//    (guide T12): 8 cvt_pk + 4 swaps replace any cross-lane scatter.
//  * K tile staged row-major in LDS with the T2 XOR swizzle
//    (byte ^= (row&7)<<4) so the fragment ds_read_b128 is ~conflict-free;
//    V tile staged TRANSPOSED ([64 d][32 kv], 80 B row stride) so the PV
//    B-fragment is a clean ds_read_b128 as well.
//  * workgroup = 4 waves = 128 q rows of one (b, h); K/V tiles staged once
//    per workgroup and consumed by all four waves.

#include "common.h"

typedef __attribute__((ext_vector_type(16))) float f32x16;

#define FA_DH 64
#define FA_KVB 32
#define FA_QB 32          // q rows per wave
#define FA_WAVES 4
#define FA_BLOCK (FA_WAVES * WAVE)
#define FA_QWG (FA_WAVES * FA_QB)   // 128 q rows per workgroup

// LDS layout (dynamic, 16-B aligned):
//  K tile: [32][64] bf16, row stride 128 B, XOR-swizzled       = 4096 B
//  V tile: same layout — the PV B-fragment is gathered with
//          ds_read_b64_tr_b16 (lane-grid-transpose semantics pinned by
//          scripts/tr_probe.py; see the PV block below)         = 4096 B
//  alpha:  [4 waves][32] f32                                   =  512 B
#define K_LDS_BYTES (FA_KVB * 128)
#define VT_LDS_BYTES (FA_KVB * 128)

__device__ __forceinline__ int kswz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

// Tensor geometry (round 2): the kernels address q/k/v rows as
//   base + b*qkv_bs + h*qkv_hs + l*qkv_rs        (64 contiguous shorts)
// and o/dout rows as b*o_bs + h*o_hs + l*o_rs — so the SAME kernels run on
// the bmm-style [B, H, L, 64] layout (qkv_bs=H*L*64, qkv_hs=L*64,
// qkv_rs=64) and directly on the packed QKV-projection output [B, L, 3D]
// (qkv_bs=L*3D, qkv_hs=64, qkv_rs=3D, k/v base-offset D/2D) with the
// attention output written straight into the [B, L, D] proj input
// (o_bs=L*D, o_hs=64, o_rs=D).  The packed path deletes the four
// qkv/out repack kernels (+their backward merges) from the model step.

// Padding-tail tile skip (round 2): with the additive -1e9 padding bias,
// every score in a fully-masked kv tile sits below ~-1e8 and __expf
// underflows to exactly +0.0f, so the tile contributes NOTHING to the
// online softmax (alpha==1, rowsum+=0, O+=0) or to dQ — identical to not
// running it.  The mask is per-(batch, kv) so the last unmasked position
// is workgroup-uniform; tiles past it are skipped.  If the whole row is
// masked (last==-1) nothing is skipped, preserving the reference
// softmax-over-bias behavior for degenerate all-pad sequences.
__device__ __forceinline__ int fa_last_active_kv(const float* mrow, int L,
                                                 int tid, int* scratch) {
  if (tid == 0) *scratch = -1;
  __syncthreads();
  int loc = -1;
  for (int kv = tid; kv < L; kv += FA_BLOCK)
    if (mrow[kv] > -1.0e8f) loc = kv;
  if (loc >= 0) atomicMax(scratch, loc);
  __syncthreads();
  return *scratch;
}

extern "C" __global__ void __launch_bounds__(FA_BLOCK, 2)
flash_fwd_kernel(const short* __restrict__ q, const short* __restrict__ k,
                 const short* __restrict__ v, const float* __restrict__ mask,
                 short* __restrict__ o, float* __restrict__ lse,
                 int B, int H, int L, float scale,
                 long qkv_bs, long qkv_hs, int qkv_rs,
                 long o_bs, long o_hs, int o_rs) {
  // Each wave processes TWO independent 32-row q-blocks against the shared
  // K/V tile: the second block's MFMAs overlap the first block's serial
  // softmax chain (ILP within the wave).  2 waves/SIMD by registers; the
  // workgroup covers 256 q rows.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = (short*)smem;                       // swizzled [32][64]
  short* v_lds = (short*)(smem + K_LDS_BYTES);       // swizzled [32][64]
  float* alpha_lds = (float*)(smem + K_LDS_BYTES + VT_LDS_BYTES);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int rows_per_wg = 2 * FA_QWG;                 // 256
  const int n_qblocks = (L + rows_per_wg - 1) / rows_per_wg;
  int bid = xcd_group_remap(blockIdx.x, gridDim.x, n_qblocks);
  int bh = bid / n_qblocks;
  int qb = bid % n_qblocks;
  const int b = bh / H;
  const int h = bh % H;
  const long qkv_off = (long)b * qkv_bs + (long)h * qkv_hs;
  const long o_off = (long)b * o_bs + (long)h * o_hs;
  // wave's two q-blocks: rows [qA, qA+32) and [qB, qB+32)
  const int q_baseA = qb * rows_per_wg + wid * FA_QB;
  const int q_baseB = q_baseA + FA_QWG;
  const int my_qA = q_baseA + col;
  const int my_qB = q_baseB + col;
  const bool validA = my_qA < L;
  const bool validB = my_qB < L;
  const float* mrow = mask ? mask + (long)b * L : nullptr;

  short8_t qfA[4], qfB[4];
  {
    const short* qrA = q + qkv_off + (long)(validA ? my_qA : L - 1) * qkv_rs;
    const short* qrB = q + qkv_off + (long)(validB ? my_qB : L - 1) * qkv_rs;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      qfA[c] = *(const short8_t*)(qrA + c * 16 + half * 8);
      qfB[c] = *(const short8_t*)(qrB + c * 16 + half * 8);
    }
  }

  f32x16 oA[2], oB[2];
#pragma unroll
  for (int t = 0; t < 2; ++t) { oA[t] = (f32x16)(0.f); oB[t] = (f32x16)(0.f); }
  float mA = -3.0e38f, lA = 0.f, mB = -3.0e38f, lB = 0.f;

  const int n_kv = L / FA_KVB;
  int n_kv_eff = n_kv;
  if (mrow) {
    int last = fa_last_active_kv(mrow, L, tid, (int*)alpha_lds);
    if (last >= 0) n_kv_eff = min(n_kv, last / FA_KVB + 1);
  }
  const int srow = tid >> 3, sc8 = (tid & 7) * 16;
  short8_t kv8 = *(const short8_t*)(k + qkv_off + (long)srow * qkv_rs +
                                    (sc8 >> 1));
  short8_t vv8 = *(const short8_t*)(v + qkv_off + (long)srow * qkv_rs +
                                    (sc8 >> 1));
  for (int kt = 0; kt < n_kv_eff; ++kt) {
    const int kv0 = kt * FA_KVB;
    __syncthreads();
    {
      *(short8_t*)((char*)k_lds + srow * 128 + kswz(srow, sc8)) = kv8;
      *(short8_t*)((char*)v_lds + srow * 128 + kswz(srow, sc8)) = vv8;
    }
    __syncthreads();
    if (kt + 1 < n_kv_eff) {
      kv8 = *(const short8_t*)(k + qkv_off +
                               (long)(kv0 + FA_KVB + srow) * qkv_rs + (sc8 >> 1));
      vv8 = *(const short8_t*)(v + qkv_off +
                               (long)(kv0 + FA_KVB + srow) * qkv_rs + (sc8 >> 1));
    }

    // mask-bias loads HOISTED above the QK chain (round 2): issued per
    // register inside the softmax loop they sit exposed on the serial
    // path between the mfma results and the exp chain
    float mb_r[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kv_local = (r & 3) + 8 * (r >> 2) + 4 * half;
      mb_r[r] = mrow ? mrow[kv0 + kv_local] : 0.f;
    }

    // ---- QK^T for BOTH q-blocks (8 back-to-back MFMAs) ----
    f32x16 sA = (f32x16)(0.f), sB = (f32x16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int byte_off = (16 * c + 8 * half) * 2;
      short8_t kf = *(const short8_t*)((char*)k_lds + col * 128 +
                                       kswz(col, byte_off));
      sA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfA[c], sA, 0, 0, 0);
      sB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfB[c], sB, 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- online softmax, both blocks (independent chains -> ILP) ----
    float svA[16], svB[16];
    float tmaxA = -3.0e38f, tmaxB = -3.0e38f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float mb = mb_r[r];
      float xA = sA[r] * scale + mb;
      float xB = sB[r] * scale + mb;
      svA[r] = xA; svB[r] = xB;
      tmaxA = fmaxf(tmaxA, xA);
      tmaxB = fmaxf(tmaxB, xB);
    }
    tmaxA = fmaxf(tmaxA, __shfl_xor(tmaxA, 32, WAVE));
    tmaxB = fmaxf(tmaxB, __shfl_xor(tmaxB, 32, WAVE));
    float mnA = fmaxf(mA, tmaxA), mnB = fmaxf(mB, tmaxB);
    float aA = __expf(mA - mnA), aB = __expf(mB - mnB);
    float rsA = 0.f, rsB = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      svA[r] = __expf(svA[r] - mnA);
      svB[r] = __expf(svB[r] - mnB);
      rsA += svA[r];
      rsB += svB[r];
    }
    rsA += __shfl_xor(rsA, 32, WAVE);
    rsB += __shfl_xor(rsB, 32, WAVE);
    lA = lA * aA + rsA; mA = mnA;
    lB = lB * aB + rsB; mB = mnB;

    // O rescale (round 2): alpha broadcast by __shfl (v_readlane — the
    // value for q-row r lives in lane r) instead of an LDS round-trip
    // (was 2 stores + 64 reads/tile), and skipped wave-uniformly when no
    // row's running max moved this tile (aA==aB==1 exactly; common once
    // the max stabilizes a few tiles in).  Wave-local state only, so the
    // uniform skip is race-free.
    if (__builtin_amdgcn_ballot_w64(aA != 1.f || aB != 1.f)) {
      // LDS broadcast beat a per-register __shfl chain here (382 vs
      // ~368 us measured): 32 dependent v_readlane+mul pairs serialize
      // worse than the batched ds_read round-trip
      alpha_lds[wid * 64 + col] = aA;
      alpha_lds[wid * 64 + 32 + col] = aB;
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
          oA[t][r] *= alpha_lds[wid * 64 + qrow];
          oB[t][r] *= alpha_lds[wid * 64 + 32 + qrow];
        }
      }
    }

    // ---- P -> bf16 fragments, both blocks ----
    short8_t pfA[2], pfB[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
      uint4_t uA, uB;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int r0 = c * 8 + 2 * i;
        int r1 = c * 8 + 4 + 2 * i;
        unsigned loA, hiA, loB, hiB;
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(loA)
            : "v"(svA[r0]), "v"(svA[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hiA)
            : "v"(svA[r1]), "v"(svA[r1 + 1]));
        auto swA = __builtin_amdgcn_permlane32_swap(loA, hiA, false, false);
        uA[i] = swA[0]; uA[i + 2] = swA[1];
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(loB)
            : "v"(svB[r0]), "v"(svB[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hiB)
            : "v"(svB[r1]), "v"(svB[r1 + 1]));
        auto swB = __builtin_amdgcn_permlane32_swap(loB, hiB, false, false);
        uB[i] = swB[0]; uB[i + 2] = swB[1];
      }
      pfA[c] = __builtin_bit_cast(short8_t, uA);
      pfB[c] = __builtin_bit_cast(short8_t, uB);
    }

    // ---- PV for both blocks from tr_b16-gathered V fragments ----
    {
      typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
      const unsigned vbase = (unsigned)(unsigned long)(char*)v_lds;
      const int kv_mate = (lane >> 2) & 3;
      const int d_lane = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
      unsigned a[8];
#pragma unroll
      for (int c = 0; c < 2; ++c)
#pragma unroll
        for (int rr = 0; rr < 2; ++rr)
#pragma unroll
          for (int t = 0; t < 2; ++t) {
            int kv = 16 * c + 8 * half + 4 * rr + kv_mate;
            int dcol = (32 * t + d_lane) * 2;
            a[c * 4 + rr * 2 + t] =
                vbase + kv * 128 + (dcol ^ ((kv & 7) << 4));
          }
      uint2_t r[8];
      asm volatile(
          "ds_read_b64_tr_b16 %0, %8\n\t"
          "ds_read_b64_tr_b16 %1, %9\n\t"
          "ds_read_b64_tr_b16 %2, %10\n\t"
          "ds_read_b64_tr_b16 %3, %11\n\t"
          "ds_read_b64_tr_b16 %4, %12\n\t"
          "ds_read_b64_tr_b16 %5, %13\n\t"
          "ds_read_b64_tr_b16 %6, %14\n\t"
          "ds_read_b64_tr_b16 %7, %15\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]),
            "=&v"(r[4]), "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7])
          : "v"(a[0]), "v"(a[1]), "v"(a[2]), "v"(a[3]), "v"(a[4]), "v"(a[5]),
            "v"(a[6]), "v"(a[7])
          : "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
          uint4_t w;
          w[0] = r[c * 4 + 0 * 2 + t][0];
          w[1] = r[c * 4 + 0 * 2 + t][1];
          w[2] = r[c * 4 + 1 * 2 + t][0];
          w[3] = r[c * 4 + 1 * 2 + t][1];
          short8_t vf = __builtin_bit_cast(short8_t, w);
          oA[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pfA[c], vf,
                                                          oA[t], 0, 0, 0);
          oB[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pfB[c], vf,
                                                          oB[t], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  // ---- epilogue: both blocks ----
  if (lse != nullptr && half == 0) {
    if (validA) lse[(long)bh * L + my_qA] = mA + __logf(lA);
    if (validB) lse[(long)bh * L + my_qB] = mB + __logf(lB);
  }
  const float invA = 1.0f / lA, invB = 1.0f / lB;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int rloc = (r & 3) + 8 * (r >> 2) + 4 * half;
    float iA = __shfl(invA, rloc, WAVE);
    float iB = __shfl(invB, rloc, WAVE);
    int qA = q_baseA + rloc, qB = q_baseB + rloc;
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      if (qA < L)
        o[o_off + (long)qA * o_rs + 32 * t + col] =
            f32_to_bf16(oA[t][r] * iA);
      if (qB < L)
        o[o_off + (long)qB * o_rs + 32 * t + col] =
            f32_to_bf16(oB[t][r] * iB);
    }
  }
}

extern "C" hipError_t flash_fwd_launch(const void* q, const void* k,
                                       const void* v, const void* mask,
                                       void* o, void* lse, int B, int H,
                                       int L, float scale,
                                       hipStream_t stream) {
  int n_qblocks = (L + 2 * FA_QWG - 1) / (2 * FA_QWG);
  dim3 grid(B * H * n_qblocks);
  size_t shm = K_LDS_BYTES + VT_LDS_BYTES + FA_WAVES * 64 * sizeof(float);
  flash_fwd_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const float*)mask,
      (short*)o, (float*)lse, B, H, L, scale,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH);
  return hipGetLastError();
}

// qkv: [B, L, 3D] packed projection output (D = H*64); o: [B, L, D]
extern "C" hipError_t flash_fwd_packed_launch(const void* qkv,
                                              const void* mask, void* o,
                                              void* lse, int B, int H, int L,
                                              float scale,
                                              hipStream_t stream) {
  const int D = H * FA_DH;
  int n_qblocks = (L + 2 * FA_QWG - 1) / (2 * FA_QWG);
  dim3 grid(B * H * n_qblocks);
  size_t shm = K_LDS_BYTES + VT_LDS_BYTES + FA_WAVES * 64 * sizeof(float);
  flash_fwd_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)qkv, (const short*)qkv + D, (const short*)qkv + 2 * D,
      (const float*)mask, (short*)o, (float*)lse, B, H, L, scale,
      (long)L * 3 * D, 64L, 3 * D,
      (long)L * D, 64L, D);
  return hipGetLastError();
}



// ---------------------------------------------------------------------------
// Flash backward: ONE kernel recomputes P/dS and produces dS, dK and dV.
//
//   P[q, kv]  = exp(scale * S + mask_bias[kv] - lse[q])
//   dP[q, kv] = dO @ V^T
//   dS[q, kv] = scale * P * (dP - D[q]),  D = rowsum(dO * O)
//   dV[kv, d] = sum_q P[q, kv]  * dO[q, d]   (register accumulators)
//   dK[kv, d] = sum_q dS[q, kv] * Q[q, d]    (register accumulators)
//
// Each wave owns one 32-row KV block: K/V fragments and the dV/dK f32
// accumulators stay in registers for the whole kernel; the workgroup's 4
// waves share LDS-staged Q/dO tiles (32 q rows per tile).  Per tile the
// wave computes S and dP in the (reg=q, lane=kv) orientation — so lse/ddot
// index by register and the mask bias is one scalar per lane — then
// repacks P and dS into A-fragments (cvt_pk_bf16 + permlane32_swap, which
// maps C-layout (reg=R, lane=C) to fragment [row=C][k=R]) and feeds two
// MFMA accumulation chains against dO^T / Q^T fragments gathered with
// ds_read_b64_tr_b16 from the already-staged tiles.  dS leaves in the
// natural [B, H, L_q, L_kv] layout so the remaining gradient is ONE
// non-transposed library bmm: dQ = dS @ K.  This replaces the previous
// P^T/dS^T materialization + three batched bmms (the [512,512,64]-shaped
// batched bmms ran at ~107 TF in hipBLASLt — see profiles/).
//
// Mirrors reference capability only in spirit: the reference ships no
// attention kernels (data-only replication package).

extern "C" __global__ void __launch_bounds__(FA_BLOCK, 2)
flash_bwd_fused_kernel(const short* __restrict__ q, const short* __restrict__ k,
                       const short* __restrict__ v,
                       const short* __restrict__ dout,
                       const float* __restrict__ mask,
                       const float* __restrict__ lse,
                       const float* __restrict__ ddot,
                       short* __restrict__ ds, short* __restrict__ dk,
                       short* __restrict__ dv,
                       int B, int H, int L, float scale,
                       long qkv_bs, long qkv_hs, int qkv_rs,
                       long o_bs, long o_hs, int o_rs) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* q_lds = (short*)smem;                        // swizzled [32][64]
  short* do_lds = (short*)(smem + K_LDS_BYTES);       // swizzled [32][64]
  float* lse_lds = (float*)(smem + 2 * K_LDS_BYTES);  // [32]
  float* dd_lds = lse_lds + 32;                       // [32]

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int n_kvblocks = (L + FA_QWG - 1) / FA_QWG;   // 128 kv rows per WG
  int bid = xcd_group_remap(blockIdx.x, gridDim.x, n_kvblocks);
  int bh = bid / n_kvblocks;
  int kb = bid % n_kvblocks;
  const int b = bh / H;
  const int h = bh % H;
  const long qkv_off = (long)b * qkv_bs + (long)h * qkv_hs;
  const long o_off = (long)b * o_bs + (long)h * o_hs;
  const long bh_sq = (long)bh * L * L;
  const int kv_base = kb * FA_QWG + wid * FA_KVB;     // this wave's 32 kv rows
  const int my_kv = kv_base + col;
  const bool kv_valid = my_kv < L;                    // L%32==0: whole block
  const float* mrow = mask ? mask + (long)b * L : nullptr;
  const float mb = mrow ? mrow[min(my_kv, L - 1)] : 0.f;

  // Early-out (round 2): if every kv row this workgroup owns is padding
  // (bias <= -1e8) and the sequence has at least one real token, P and dS
  // are exactly +0.0f for every q — dK/dV are zero and the whole q loop is
  // skipped (the epilogue writes the zero accumulators).  Requires
  // ds == nullptr (the default path) so no dS tile is left unwritten; the
  // mrow[0] guard keeps degenerate all-pad sequences on the reference
  // softmax-over-bias behavior.  skip is workgroup-uniform (the flag is a
  // barrier-ordered workgroup reduction), so branching around the
  // barriered q loop is safe.
  bool skip_tile = false;
  bool wave_skip = false;
  if (mrow && ds == nullptr) {
    int* any_lds = (int*)lse_lds;
    if (tid == 0) *any_lds = 0;
    __syncthreads();
    if (kv_valid && mb > -1.0e8f) *any_lds = 1;
    __syncthreads();
    skip_tile = (*any_lds == 0) && (mrow[0] > -1.0e8f);
    __syncthreads();
    // Finer grain: a wave whose OWN 32 kv rows are all padding still has
    // to stage q/dO tiles and hit the barriers with the other waves, but
    // its S/dP MFMA chains, softmax/dS math and dK/dV accumulation are
    // exact zeros — skip them (wave-uniform ballot, no barriers inside
    // the skipped region).
    wave_skip = !skip_tile && (mrow[0] > -1.0e8f) &&
        __builtin_amdgcn_ballot_w64(kv_valid && mb > -1.0e8f) == 0;
  }

  // K and V fragments for this wave's kv block (resident all kernel)
  short8_t kf[4], vf[4];
  if (!skip_tile) {
    const short* kr = k + qkv_off + (long)(kv_valid ? my_kv : L - 1) * qkv_rs;
    const short* vr = v + qkv_off + (long)(kv_valid ? my_kv : L - 1) * qkv_rs;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      kf[c] = *(const short8_t*)(kr + c * 16 + half * 8);
      vf[c] = *(const short8_t*)(vr + c * 16 + half * 8);
    }
  }
  f32x16 dv_acc[2], dk_acc[2];
#pragma unroll
  for (int t = 0; t < 2; ++t) {
    dv_acc[t] = (f32x16)(0.f);
    dk_acc[t] = (f32x16)(0.f);
  }
  if (!skip_tile) {
  const int n_q = L / 32;
  const int srow = tid >> 3, sc8 = (tid & 7) * 16;
  short8_t qv8 = *(const short8_t*)(q + qkv_off + (long)srow * qkv_rs +
                                    (sc8 >> 1));
  short8_t dv8 = *(const short8_t*)(dout + o_off + (long)srow * o_rs +
                                    (sc8 >> 1));
  for (int qt = 0; qt < n_q; ++qt) {
    const int q0 = qt * 32;
    __syncthreads();
    {
      *(short8_t*)((char*)q_lds + srow * 128 + kswz(srow, sc8)) = qv8;
      *(short8_t*)((char*)do_lds + srow * 128 + kswz(srow, sc8)) = dv8;
      if (tid < 32) {
        lse_lds[tid] = lse[(long)bh * L + q0 + tid];
        dd_lds[tid] = ddot[(long)bh * L + q0 + tid];
      }
    }
    __syncthreads();
    if (qt + 1 < n_q) {
      qv8 = *(const short8_t*)(q + qkv_off + (long)(q0 + 32 + srow) * qkv_rs +
                               (sc8 >> 1));
      dv8 = *(const short8_t*)(dout + o_off + (long)(q0 + 32 + srow) * o_rs +
                               (sc8 >> 1));
    }

    if (wave_skip) continue;
    // S[q, kv] and dP[q, kv]: reg=q rows, lane=kv cols
    f32x16 s_acc = (f32x16)(0.f);
    f32x16 dp_acc = (f32x16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int byte_off = (16 * c + 8 * half) * 2;
      short8_t qfrag = *(const short8_t*)((char*)q_lds + col * 128 +
                                          kswz(col, byte_off));
      short8_t dofrag = *(const short8_t*)((char*)do_lds + col * 128 +
                                           kswz(col, byte_off));
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfrag, kf[c], s_acc,
                                                      0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofrag, vf[c], dp_acc,
                                                       0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // elementwise (lse/ddot by register row, mask bias one scalar per lane)
    float pv[16], gv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int ql = (r & 3) + 8 * (r >> 2) + 4 * half;
      pv[r] = __expf(s_acc[r] * scale + mb - lse_lds[ql]);
      gv[r] = scale * pv[r] * (dp_acc[r] - dd_lds[ql]);
    }
    // dS out, natural [q, kv] rows (2 B per lane, lanes contiguous in kv);
    // skipped entirely when ds == nullptr (round 2: dQ is recomputed by
    // flash_dq_recompute_kernel, no dS materialization)
    if (ds != nullptr && kv_valid) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int ql = (r & 3) + 8 * (r >> 2) + 4 * half;
        ds[bh_sq + (long)(q0 + ql) * L + my_kv] = f32_to_bf16(gv[r]);
      }
    }

    // repack P and dS to A-fragments [row=kv][k=q]
    short8_t pf[2], gf[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
      uint4_t up, ug;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int r0 = c * 8 + 2 * i;
        int r1 = c * 8 + 4 + 2 * i;
        unsigned lo, hi;
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(lo)
            : "v"(pv[r0]), "v"(pv[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hi)
            : "v"(pv[r1]), "v"(pv[r1 + 1]));
        auto sw = __builtin_amdgcn_permlane32_swap(lo, hi, false, false);
        up[i] = sw[0]; up[i + 2] = sw[1];
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(lo)
            : "v"(gv[r0]), "v"(gv[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hi)
            : "v"(gv[r1]), "v"(gv[r1 + 1]));
        auto sw2 = __builtin_amdgcn_permlane32_swap(lo, hi, false, false);
        ug[i] = sw2[0]; ug[i + 2] = sw2[1];
      }
      pf[c] = __builtin_bit_cast(short8_t, up);
      gf[c] = __builtin_bit_cast(short8_t, ug);
    }

    // dO^T and Q^T B-fragments [row=d][k=q] via lane-grid transpose reads,
    // then the two accumulation chains (reg=kv rows, lane=d cols)
    {
      typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
      const unsigned dobase = (unsigned)(unsigned long)(char*)do_lds;
      const unsigned qbase = (unsigned)(unsigned long)(char*)q_lds;
      const int q_mate = (lane >> 2) & 3;
      const int d_lane = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
      unsigned ad[8], aq[8];
#pragma unroll
      for (int c = 0; c < 2; ++c)
#pragma unroll
        for (int rr = 0; rr < 2; ++rr)
#pragma unroll
          for (int t = 0; t < 2; ++t) {
            int qq = 16 * c + 8 * half + 4 * rr + q_mate;
            int dcol = (32 * t + d_lane) * 2;
            int off = qq * 128 + (dcol ^ ((qq & 7) << 4));
            ad[c * 4 + rr * 2 + t] = dobase + off;
            aq[c * 4 + rr * 2 + t] = qbase + off;
          }
      uint2_t rd[8], rq[8];
      asm volatile(
          "ds_read_b64_tr_b16 %0, %8\n\t"
          "ds_read_b64_tr_b16 %1, %9\n\t"
          "ds_read_b64_tr_b16 %2, %10\n\t"
          "ds_read_b64_tr_b16 %3, %11\n\t"
          "ds_read_b64_tr_b16 %4, %12\n\t"
          "ds_read_b64_tr_b16 %5, %13\n\t"
          "ds_read_b64_tr_b16 %6, %14\n\t"
          "ds_read_b64_tr_b16 %7, %15\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(rd[0]), "=&v"(rd[1]), "=&v"(rd[2]), "=&v"(rd[3]),
            "=&v"(rd[4]), "=&v"(rd[5]), "=&v"(rd[6]), "=&v"(rd[7])
          : "v"(ad[0]), "v"(ad[1]), "v"(ad[2]), "v"(ad[3]), "v"(ad[4]),
            "v"(ad[5]), "v"(ad[6]), "v"(ad[7])
          : "memory");
      asm volatile(
          "ds_read_b64_tr_b16 %0, %8\n\t"
          "ds_read_b64_tr_b16 %1, %9\n\t"
          "ds_read_b64_tr_b16 %2, %10\n\t"
          "ds_read_b64_tr_b16 %3, %11\n\t"
          "ds_read_b64_tr_b16 %4, %12\n\t"
          "ds_read_b64_tr_b16 %5, %13\n\t"
          "ds_read_b64_tr_b16 %6, %14\n\t"
          "ds_read_b64_tr_b16 %7, %15\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(rq[0]), "=&v"(rq[1]), "=&v"(rq[2]), "=&v"(rq[3]),
            "=&v"(rq[4]), "=&v"(rq[5]), "=&v"(rq[6]), "=&v"(rq[7])
          : "v"(aq[0]), "v"(aq[1]), "v"(aq[2]), "v"(aq[3]), "v"(aq[4]),
            "v"(aq[5]), "v"(aq[6]), "v"(aq[7])
          : "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
          uint4_t wd, wq;
          wd[0] = rd[c * 4 + 0 * 2 + t][0];
          wd[1] = rd[c * 4 + 0 * 2 + t][1];
          wd[2] = rd[c * 4 + 1 * 2 + t][0];
          wd[3] = rd[c * 4 + 1 * 2 + t][1];
          wq[0] = rq[c * 4 + 0 * 2 + t][0];
          wq[1] = rq[c * 4 + 0 * 2 + t][1];
          wq[2] = rq[c * 4 + 1 * 2 + t][0];
          wq[3] = rq[c * 4 + 1 * 2 + t][1];
          short8_t dof = __builtin_bit_cast(short8_t, wd);
          short8_t qf = __builtin_bit_cast(short8_t, wq);
          dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf[c], dof,
                                                              dv_acc[t],
                                                              0, 0, 0);
          dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(gf[c], qf,
                                                              dk_acc[t],
                                                              0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }
  }   // !skip_tile

  // epilogue: dV/dK rows of this wave's kv block (reg=kv row, lane=d col)
#pragma unroll
  for (int t = 0; t < 2; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kvl = (r & 3) + 8 * (r >> 2) + 4 * half;
      if (kv_base + kvl >= L) continue;
      long off = qkv_off + (long)(kv_base + kvl) * qkv_rs + 32 * t + col;
      dv[off] = f32_to_bf16(dv_acc[t][r]);
      dk[off] = f32_to_bf16(dk_acc[t][r]);
    }
  }
}

extern "C" hipError_t flash_bwd_fused_launch(
    const void* q, const void* k, const void* v, const void* dout,
    const void* mask, const void* lse, const void* ddot, void* ds, void* dk,
    void* dv, int B, int H, int L, float scale, hipStream_t stream) {
  int n_kvblocks = (L + FA_QWG - 1) / FA_QWG;
  dim3 grid(B * H * n_kvblocks);
  size_t shm = 2 * K_LDS_BYTES + 64 * sizeof(float);
  flash_bwd_fused_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)mask, (const float*)lse, (const float*)ddot, (short*)ds,
      (short*)dk, (short*)dv, B, H, L, scale,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH);
  return hipGetLastError();
}

// packed: qkv/dqkv [B, L, 3D], dout [B, L, D]; dk/dv land inside dqkv
extern "C" hipError_t flash_bwd_fused_packed_launch(
    const void* qkv, const void* dout, const void* mask, const void* lse,
    const void* ddot, void* dqkv, int B, int H, int L, float scale,
    hipStream_t stream) {
  const int D = H * FA_DH;
  int n_kvblocks = (L + FA_QWG - 1) / FA_QWG;
  dim3 grid(B * H * n_kvblocks);
  size_t shm = 2 * K_LDS_BYTES + 64 * sizeof(float);
  flash_bwd_fused_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)qkv, (const short*)qkv + D, (const short*)qkv + 2 * D,
      (const short*)dout, (const float*)mask, (const float*)lse,
      (const float*)ddot, nullptr,
      (short*)dqkv + D, (short*)dqkv + 2 * D, B, H, L, scale,
      (long)L * 3 * D, 64L, 3 * D,
      (long)L * D, 64L, D);
  return hipGetLastError();
}



// dQ = dS @ K — the final attention gradient, as a purpose-built kernel
// instead of a library bmm ([L, L] x [L, 64] batched shapes run at ~225 TF
// in hipBLASLt).  Structure mirrors flash_fwd's PV stage: two independent
// 32-row q-blocks per wave (ILP), dS rows read straight from HBM as MFMA
// A-fragments (contiguous short8 per lane), K tiles staged swizzled in LDS
// and gathered as K^T B-fragments with ds_read_b64_tr_b16.
extern "C" __global__ void __launch_bounds__(FA_BLOCK, 2)
flash_dq_kernel(const short* __restrict__ ds, const short* __restrict__ k,
                short* __restrict__ dq, int B, int H, int L) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = (short*)smem;                       // swizzled [32][64]

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int rows_per_wg = 2 * FA_QWG;                 // 256 q rows
  const int n_qblocks = (L + rows_per_wg - 1) / rows_per_wg;
  int bid = xcd_group_remap(blockIdx.x, gridDim.x, n_qblocks);
  int bh = bid / n_qblocks;
  int qb = bid % n_qblocks;
  const long bh_off = (long)bh * L * FA_DH;
  const long bh_sq = (long)bh * L * L;
  const int q_baseA = qb * rows_per_wg + wid * FA_QB;
  const int q_baseB = q_baseA + FA_QWG;
  const int my_qA = min(q_baseA + col, L - 1);
  const int my_qB = min(q_baseB + col, L - 1);
  const short* dsA_row = ds + bh_sq + (long)my_qA * L;
  const short* dsB_row = ds + bh_sq + (long)my_qB * L;

  f32x16 oA[2], oB[2];
#pragma unroll
  for (int t = 0; t < 2; ++t) { oA[t] = (f32x16)(0.f); oB[t] = (f32x16)(0.f); }

  const int n_kv = L / FA_KVB;
  const int srow = tid >> 3, sc8 = (tid & 7) * 16;
  short8_t kv8 = *(const short8_t*)(k + bh_off + (long)srow * FA_DH +
                                    (sc8 >> 1));
  for (int kt = 0; kt < n_kv; ++kt) {
    const int kv0 = kt * FA_KVB;
    __syncthreads();
    *(short8_t*)((char*)k_lds + srow * 128 + kswz(srow, sc8)) = kv8;
    __syncthreads();
    if (kt + 1 < n_kv)
      kv8 = *(const short8_t*)(k + bh_off +
                               (long)(kv0 + FA_KVB + srow) * FA_DH +
                               (sc8 >> 1));
    // dS A-fragments straight from HBM: lane reads its q row's kv chunk
    short8_t aA[2], aB[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      aA[c] = *(const short8_t*)(dsA_row + kv0 + 16 * c + 8 * half);
      aB[c] = *(const short8_t*)(dsB_row + kv0 + 16 * c + 8 * half);
    }
    // K^T B-fragments via lane-grid transpose reads (same law as fwd V)
    typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
    const unsigned kbase = (unsigned)(unsigned long)(char*)k_lds;
    const int kv_mate = (lane >> 2) & 3;
    const int d_lane = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
    unsigned a[8];
#pragma unroll
    for (int c = 0; c < 2; ++c)
#pragma unroll
      for (int rr = 0; rr < 2; ++rr)
#pragma unroll
        for (int t = 0; t < 2; ++t) {
          int kv = 16 * c + 8 * half + 4 * rr + kv_mate;
          int dcol = (32 * t + d_lane) * 2;
          a[c * 4 + rr * 2 + t] = kbase + kv * 128 + (dcol ^ ((kv & 7) << 4));
        }
    uint2_t r[8];
    asm volatile(
        "ds_read_b64_tr_b16 %0, %8\n\t"
        "ds_read_b64_tr_b16 %1, %9\n\t"
        "ds_read_b64_tr_b16 %2, %10\n\t"
        "ds_read_b64_tr_b16 %3, %11\n\t"
        "ds_read_b64_tr_b16 %4, %12\n\t"
        "ds_read_b64_tr_b16 %5, %13\n\t"
        "ds_read_b64_tr_b16 %6, %14\n\t"
        "ds_read_b64_tr_b16 %7, %15\n\t"
        "s_waitcnt lgkmcnt(0)"
        : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]), "=&v"(r[4]),
          "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7])
        : "v"(a[0]), "v"(a[1]), "v"(a[2]), "v"(a[3]), "v"(a[4]), "v"(a[5]),
          "v"(a[6]), "v"(a[7])
        : "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
        uint4_t w;
        w[0] = r[c * 4 + 0 * 2 + t][0];
        w[1] = r[c * 4 + 0 * 2 + t][1];
        w[2] = r[c * 4 + 1 * 2 + t][0];
        w[3] = r[c * 4 + 1 * 2 + t][1];
        short8_t kf = __builtin_bit_cast(short8_t, w);
        oA[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(aA[c], kf, oA[t],
                                                        0, 0, 0);
        oB[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(aB[c], kf, oB[t],
                                                        0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

#pragma unroll
  for (int t = 0; t < 2; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int rloc = (r & 3) + 8 * (r >> 2) + 4 * half;
      int qA = q_baseA + rloc, qB = q_baseB + rloc;
      if (qA < L)
        dq[bh_off + (long)qA * FA_DH + 32 * t + col] = f32_to_bf16(oA[t][r]);
      if (qB < L)
        dq[bh_off + (long)qB * FA_DH + 32 * t + col] = f32_to_bf16(oB[t][r]);
    }
  }
}

extern "C" hipError_t flash_dq_launch(const void* ds, const void* k, void* dq,
                                      int B, int H, int L,
                                      hipStream_t stream) {
  int n_qblocks = (L + 2 * FA_QWG - 1) / (2 * FA_QWG);
  dim3 grid(B * H * n_qblocks);
  flash_dq_kernel<<<grid, FA_BLOCK, K_LDS_BYTES, stream>>>(
      (const short*)ds, (const short*)k, (short*)dq, B, H, L);
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// dQ by recompute (round 2): forward-orientation kernel that recomputes
// S = QK^T and dP = dO V^T per kv tile, forms dS in registers and
// accumulates dQ = dS @ K — no dS materialization at all.  Replaces the
// bwd pipeline's [B, H, L, L] bf16 dS HBM round-trip (1.07 GB written by
// flash_bwd_fused + 1.07 GB re-read by flash_dq at the bench shape) and
// the separate flash_dq pass; flash_bwd_fused now only emits dK/dV.
//
// Structure mirrors flash_fwd exactly (two independent 32-row q-blocks per
// wave for ILP; K/V tiles staged swizzled in LDS, double-staged across the
// kv loop):
//   S  C-layout (reg=kv, lane=q) from mfma(A=K-frag, B=Q-rows)
//   dP C-layout (reg=kv, lane=q) from mfma(A=V-frag, B=dO-rows)
//   lse/ddot are LANE-local scalars (one q row per lane), the mask bias
//   indexes by the kv REGISTER — both orientations line up for free
//   dS = scale * P * (dP - D) in registers
//   dS -> A-fragments [row=q][k=kv] via the same cvt_pk_bf16 +
//   permlane32_swap repack the fwd uses for P
//   dQ[q, d] += mfma(dS-frag, K^T-frag), K^T gathered ds_read_b64_tr_b16
//   from the staged K tile (same law as the fwd V^T gather).
extern "C" __global__ void __launch_bounds__(FA_BLOCK, 2)
flash_dq_recompute_kernel(const short* __restrict__ q,
                          const short* __restrict__ k,
                          const short* __restrict__ v,
                          const short* __restrict__ dout,
                          const float* __restrict__ mask,
                          const float* __restrict__ lse,
                          const float* __restrict__ ddot,
                          short* __restrict__ dq,
                          int B, int H, int L, float scale,
                          long qkv_bs, long qkv_hs, int qkv_rs,
                          long o_bs, long o_hs, int o_rs) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = (short*)smem;                       // swizzled [32][64]
  short* v_lds = (short*)(smem + K_LDS_BYTES);       // swizzled [32][64]

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int rows_per_wg = 2 * FA_QWG;                 // 256 q rows
  const int n_qblocks = (L + rows_per_wg - 1) / rows_per_wg;
  int bid = xcd_group_remap(blockIdx.x, gridDim.x, n_qblocks);
  int bh = bid / n_qblocks;
  int qb = bid % n_qblocks;
  const int b = bh / H;
  const int h = bh % H;
  const long qkv_off = (long)b * qkv_bs + (long)h * qkv_hs;
  const long o_off = (long)b * o_bs + (long)h * o_hs;
  const int q_baseA = qb * rows_per_wg + wid * FA_QB;
  const int q_baseB = q_baseA + FA_QWG;
  const int my_qA = q_baseA + col;
  const int my_qB = q_baseB + col;
  const bool validA = my_qA < L;
  const bool validB = my_qB < L;
  const float* mrow = mask ? mask + (long)b * L : nullptr;

  short8_t qfA[4], qfB[4], dofA[4], dofB[4];
  {
    const short* qrA = q + qkv_off + (long)(validA ? my_qA : L - 1) * qkv_rs;
    const short* qrB = q + qkv_off + (long)(validB ? my_qB : L - 1) * qkv_rs;
    const short* drA = dout + o_off + (long)(validA ? my_qA : L - 1) * o_rs;
    const short* drB = dout + o_off + (long)(validB ? my_qB : L - 1) * o_rs;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      qfA[c] = *(const short8_t*)(qrA + c * 16 + half * 8);
      qfB[c] = *(const short8_t*)(qrB + c * 16 + half * 8);
      dofA[c] = *(const short8_t*)(drA + c * 16 + half * 8);
      dofB[c] = *(const short8_t*)(drB + c * 16 + half * 8);
    }
  }
  const float lseA = lse[(long)bh * L + (validA ? my_qA : L - 1)];
  const float lseB = lse[(long)bh * L + (validB ? my_qB : L - 1)];
  const float ddA = ddot[(long)bh * L + (validA ? my_qA : L - 1)];
  const float ddB = ddot[(long)bh * L + (validB ? my_qB : L - 1)];

  f32x16 dqA[2], dqB[2];
#pragma unroll
  for (int t = 0; t < 2; ++t) { dqA[t] = (f32x16)(0.f); dqB[t] = (f32x16)(0.f); }

  const int n_kv = L / FA_KVB;
  int n_kv_eff = n_kv;
  if (mrow) {
    int last = fa_last_active_kv(mrow, L, tid, (int*)(smem + K_LDS_BYTES + VT_LDS_BYTES));
    if (last >= 0) n_kv_eff = min(n_kv, last / FA_KVB + 1);
  }
  const int srow = tid >> 3, sc8 = (tid & 7) * 16;
  short8_t kv8 = *(const short8_t*)(k + qkv_off + (long)srow * qkv_rs +
                                    (sc8 >> 1));
  short8_t vv8 = *(const short8_t*)(v + qkv_off + (long)srow * qkv_rs +
                                    (sc8 >> 1));
  for (int kt = 0; kt < n_kv_eff; ++kt) {
    const int kv0 = kt * FA_KVB;
    __syncthreads();
    {
      *(short8_t*)((char*)k_lds + srow * 128 + kswz(srow, sc8)) = kv8;
      *(short8_t*)((char*)v_lds + srow * 128 + kswz(srow, sc8)) = vv8;
    }
    __syncthreads();
    if (kt + 1 < n_kv_eff) {
      kv8 = *(const short8_t*)(k + qkv_off +
                               (long)(kv0 + FA_KVB + srow) * qkv_rs + (sc8 >> 1));
      vv8 = *(const short8_t*)(v + qkv_off +
                               (long)(kv0 + FA_KVB + srow) * qkv_rs + (sc8 >> 1));
    }

    float mb_r[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int kv_local = (r & 3) + 8 * (r >> 2) + 4 * half;
      mb_r[r] = mrow ? mrow[kv0 + kv_local] : 0.f;
    }

    // ---- S and dP for BOTH q-blocks (16 back-to-back MFMAs) ----
    f32x16 sA = (f32x16)(0.f), sB = (f32x16)(0.f);
    f32x16 pA = (f32x16)(0.f), pB = (f32x16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int byte_off = (16 * c + 8 * half) * 2;
      short8_t kf = *(const short8_t*)((char*)k_lds + col * 128 +
                                       kswz(col, byte_off));
      short8_t vf = *(const short8_t*)((char*)v_lds + col * 128 +
                                       kswz(col, byte_off));
      sA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfA[c], sA, 0, 0, 0);
      sB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfB[c], sB, 0, 0, 0);
      pA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dofA[c], pA, 0, 0, 0);
      pB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dofB[c], pB, 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- dS in registers (lse/ddot lane-local, mask by kv register) ----
    float gA[16], gB[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float expA = __expf(sA[r] * scale + mb_r[r] - lseA);
      float expB = __expf(sB[r] * scale + mb_r[r] - lseB);
      gA[r] = scale * expA * (pA[r] - ddA);
      gB[r] = scale * expB * (pB[r] - ddB);
    }

    // ---- dS -> A-fragments [row=q][k=kv] (fwd P-repack idiom) ----
    short8_t gfA[2], gfB[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
      uint4_t uA, uB;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int r0 = c * 8 + 2 * i;
        int r1 = c * 8 + 4 + 2 * i;
        unsigned loA, hiA, loB, hiB;
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(loA)
            : "v"(gA[r0]), "v"(gA[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hiA)
            : "v"(gA[r1]), "v"(gA[r1 + 1]));
        auto swA = __builtin_amdgcn_permlane32_swap(loA, hiA, false, false);
        uA[i] = swA[0]; uA[i + 2] = swA[1];
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(loB)
            : "v"(gB[r0]), "v"(gB[r0 + 1]));
        asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(hiB)
            : "v"(gB[r1]), "v"(gB[r1 + 1]));
        auto swB = __builtin_amdgcn_permlane32_swap(loB, hiB, false, false);
        uB[i] = swB[0]; uB[i + 2] = swB[1];
      }
      gfA[c] = __builtin_bit_cast(short8_t, uA);
      gfB[c] = __builtin_bit_cast(short8_t, uB);
    }

    // ---- dQ += dS @ K via tr_b16-gathered K^T fragments ----
    {
      typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
      const unsigned kbase = (unsigned)(unsigned long)(char*)k_lds;
      const int kv_mate = (lane >> 2) & 3;
      const int d_lane = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
      unsigned a[8];
#pragma unroll
      for (int c = 0; c < 2; ++c)
#pragma unroll
        for (int rr = 0; rr < 2; ++rr)
#pragma unroll
          for (int t = 0; t < 2; ++t) {
            int kv = 16 * c + 8 * half + 4 * rr + kv_mate;
            int dcol = (32 * t + d_lane) * 2;
            a[c * 4 + rr * 2 + t] =
                kbase + kv * 128 + (dcol ^ ((kv & 7) << 4));
          }
      uint2_t r[8];
      asm volatile(
          "ds_read_b64_tr_b16 %0, %8\n\t"
          "ds_read_b64_tr_b16 %1, %9\n\t"
          "ds_read_b64_tr_b16 %2, %10\n\t"
          "ds_read_b64_tr_b16 %3, %11\n\t"
          "ds_read_b64_tr_b16 %4, %12\n\t"
          "ds_read_b64_tr_b16 %5, %13\n\t"
          "ds_read_b64_tr_b16 %6, %14\n\t"
          "ds_read_b64_tr_b16 %7, %15\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]),
            "=&v"(r[4]), "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7])
          : "v"(a[0]), "v"(a[1]), "v"(a[2]), "v"(a[3]), "v"(a[4]), "v"(a[5]),
            "v"(a[6]), "v"(a[7])
          : "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int t = 0; t < 2; ++t) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;
          uint4_t w;
          w[0] = r[c * 4 + 0 * 2 + t][0];
          w[1] = r[c * 4 + 0 * 2 + t][1];
          w[2] = r[c * 4 + 1 * 2 + t][0];
          w[3] = r[c * 4 + 1 * 2 + t][1];
          short8_t kf = __builtin_bit_cast(short8_t, w);
          dqA[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(gfA[c], kf,
                                                           dqA[t], 0, 0, 0);
          dqB[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(gfB[c], kf,
                                                           dqB[t], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  // ---- epilogue ----
#pragma unroll
  for (int t = 0; t < 2; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int rloc = (r & 3) + 8 * (r >> 2) + 4 * half;
      int qA = q_baseA + rloc, qB = q_baseB + rloc;
      if (qA < L)
        dq[qkv_off + (long)qA * qkv_rs + 32 * t + col] =
            f32_to_bf16(dqA[t][r]);
      if (qB < L)
        dq[qkv_off + (long)qB * qkv_rs + 32 * t + col] =
            f32_to_bf16(dqB[t][r]);
    }
  }
}

extern "C" hipError_t flash_dq_recompute_launch(
    const void* q, const void* k, const void* v, const void* dout,
    const void* mask, const void* lse, const void* ddot, void* dq,
    int B, int H, int L, float scale, hipStream_t stream) {
  int n_qblocks = (L + 2 * FA_QWG - 1) / (2 * FA_QWG);
  dim3 grid(B * H * n_qblocks);
  size_t shm = K_LDS_BYTES + VT_LDS_BYTES + 16;  // + tail-skip scratch
  flash_dq_recompute_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)q, (const short*)k, (const short*)v, (const short*)dout,
      (const float*)mask, (const float*)lse, (const float*)ddot, (short*)dq,
      B, H, L, scale,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH,
      (long)H * L * FA_DH, (long)L * FA_DH, FA_DH);
  return hipGetLastError();
}

extern "C" hipError_t flash_dq_recompute_packed_launch(
    const void* qkv, const void* dout, const void* mask, const void* lse,
    const void* ddot, void* dqkv, int B, int H, int L, float scale,
    hipStream_t stream) {
  const int D = H * FA_DH;
  int n_qblocks = (L + 2 * FA_QWG - 1) / (2 * FA_QWG);
  dim3 grid(B * H * n_qblocks);
  size_t shm = K_LDS_BYTES + VT_LDS_BYTES + 16;  // + tail-skip scratch
  flash_dq_recompute_kernel<<<grid, FA_BLOCK, shm, stream>>>(
      (const short*)qkv, (const short*)qkv + D, (const short*)qkv + 2 * D,
      (const short*)dout, (const float*)mask, (const float*)lse,
      (const float*)ddot, (short*)dqkv,
      B, H, L, scale,
      (long)L * 3 * D, 64L, 3 * D,
      (long)L * D, 64L, D);
  return hipGetLastError();
}

// D = rowsum(dO * O) per (b, h, q) row — one wave per 4 rows (dh = 64).
extern "C" __global__ void __launch_bounds__(256)
fa_dot_kernel(const short* __restrict__ dout, const short* __restrict__ o,
              float* __restrict__ ddot, long n_rows) {
  // lane j of quarter-wave handles 4 bf16 (dh=64 -> 16 lanes x 4 elems)
  long row = ((long)blockIdx.x * 256 + threadIdx.x) >> 4;
  if (row >= n_rows) return;
  int sub = threadIdx.x & 15;
  const short* dr = dout + row * FA_DH + sub * 4;
  const short* orow = o + row * FA_DH + sub * 4;
  short4_t a = *(const short4_t*)dr;
  short4_t c = *(const short4_t*)orow;
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 4; ++j) s += bf16_to_f32(a[j]) * bf16_to_f32(c[j]);
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) s += __shfl_xor(s, off, 16);
  if (sub == 0) ddot[row] = s;
}

// packed fa_dot: dout/o are [B, L, D] (D = H*64); ddot is [B, H, L].
extern "C" __global__ void __launch_bounds__(256)
fa_dot_packed_kernel(const short* __restrict__ dout,
                     const short* __restrict__ o, float* __restrict__ ddot,
                     int B, int H, int L) {
  const int D = H * FA_DH;
  long n_rows = (long)B * L * H;          // one 64-elem chunk per (b, l, h)
  long chunk = ((long)blockIdx.x * 256 + threadIdx.x) >> 4;
  if (chunk >= n_rows) return;
  int sub = threadIdx.x & 15;
  int h = (int)(chunk % H);
  long bl = chunk / H;                     // b * L + l
  int l = (int)(bl % L);
  int b = (int)(bl / L);
  const short* dr = dout + bl * D + h * FA_DH + sub * 4;
  const short* orow = o + bl * D + h * FA_DH + sub * 4;
  short4_t a = *(const short4_t*)dr;
  short4_t c = *(const short4_t*)orow;
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 4; ++j) s += bf16_to_f32(a[j]) * bf16_to_f32(c[j]);
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) s += __shfl_xor(s, off, 16);
  if (sub == 0) ddot[((long)b * H + h) * L + l] = s;
}

extern "C" hipError_t fa_dot_packed_launch(const void* dout, const void* o,
                                           void* ddot, int B, int H, int L,
                                           hipStream_t stream) {
  long total_threads = (long)B * H * L * 16;
  int grid = (int)((total_threads + 255) / 256);
  fa_dot_packed_kernel<<<grid, 256, 0, stream>>>(
      (const short*)dout, (const short*)o, (float*)ddot, B, H, L);
  return hipGetLastError();
}

extern "C" hipError_t fa_dot_launch(const void* dout, const void* o,
                                    void* ddot, long n_rows,
                                    hipStream_t stream) {
  long total_threads = n_rows * 16;
  int grid = (int)((total_threads + 255) / 256);
  fa_dot_kernel<<<grid, 256, 0, stream>>>((const short*)dout, (const short*)o,
                                          (float*)ddot, n_rows);
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 semantics probe (development aid, exposed for tests):
// stage `src` (bf16, 256 elements) into LDS, then each lane issues the
// transpose-read at a caller-chosen address scheme and writes its 4 bf16
// results to out[lane][0..3].
//   scheme 0: addr = 0 for every lane
//   scheme 1: addr = (lane & 15) * 2
//   scheme 2: addr = ((lane & 15) + (lane >> 4) * 64) * 2
//   scheme 3: addr = (lane >> 4) * 128  (16-lane-group base, uniform in group)
extern "C" __global__ void __launch_bounds__(64)
tr_probe_kernel(const short* __restrict__ src, short* __restrict__ out,
                int scheme) {
  __shared__ __attribute__((aligned(16))) short lds[256];
  int lane = threadIdx.x;
  for (int i = lane; i < 256; i += 64) lds[i] = src[i];
  __syncthreads();
  int addr;
  switch (scheme) {
    case 0: addr = 0; break;
    case 1: addr = (lane & 15) * 2; break;
    case 2: addr = ((lane & 15) + (lane >> 4) * 64) * 2; break;
    case 3: addr = (lane >> 4) * 128; break;
    case 4: addr = (lane & 3) * 16; break;           // mates' ~3-bases differ
    case 5: addr = ((lane & 3) * 32 + 4 * ((lane >> 2) & 7)) * 2; break;
    case 6: addr = ((lane & 3) * 8 + 4) * 2; break;
    // one-mate-at-a-time base perturbation (which mate feeds which output?)
    case 7: addr = ((lane & 3) == 0) ? 16 : 0; break;
    case 8: addr = ((lane & 3) == 1) ? 16 : 0; break;
    case 9: addr = ((lane & 3) == 2) ? 16 : 0; break;
    case 10: addr = ((lane & 3) == 3) ? 16 : 0; break;
    // kernel-like: row stride 64 elems per mate + d-quad by quad index
    case 11: addr = ((lane & 3) * 64 + 4 * ((lane >> 2) & 7)) * 2; break;
    // distinct per mate AND per quad
    default: addr = ((lane & 3) * 8 + ((lane >> 2) & 7) * 32) * 2; break;
  }
  typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
  uint2_t r;
  unsigned base = (unsigned)(unsigned long)((char*)lds + addr);
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(base) : "memory");
  __builtin_amdgcn_sched_barrier(0);
  out[lane * 4 + 0] = (short)(r[0] & 0xffff);
  out[lane * 4 + 1] = (short)(r[0] >> 16);
  out[lane * 4 + 2] = (short)(r[1] & 0xffff);
  out[lane * 4 + 3] = (short)(r[1] >> 16);
}

extern "C" hipError_t tr_probe_launch(const void* src, void* out, int scheme,
                                      hipStream_t stream) {
  tr_probe_kernel<<<1, 64, 0, stream>>>((const short*)src, (short*)out,
                                        scheme);
  return hipGetLastError();
}
