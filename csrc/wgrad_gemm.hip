// Split-K weight-gradient GEMM for gfx950:  C[M, N] = A[K, M]^T @ B[K, N]
// (A = dy, B = x, both bf16 row-major with K the leading/token axis — the
// natural layouts of the training wgrads dW = dy^T @ x at K = tokens).
//
// Motivation (round 2): the four wgrad shapes (K=65536, M/N in 1024..4096)
// run at 0.43–0.94 PF/s through hipBLASLt/rocBLAS even after full
// TunableOp search — 2.5–3x off the bf16 dense peak — while the forward
// GEMMs reach ~1.7 PF/s; chunked-K accumulation through the library is
// SLOWER (scripts/wgrad_probe.py).  This kernel applies the guide's 256²
// multi-wave tiling with the flash kernels' proven staging idioms:
//
//   * 256x256 output tile per 512-thread workgroup (8 waves, 2Mx4N grid;
//     each wave owns a 128x64 sub-tile = 4x2 mfma_f32_32x32x16_bf16
//     fragments = 128 f32 accumulators);
//   * K-loop stages BK=32 K-rows of A and B into LDS as four/four
//     [32][64] column-blocks in the flash tile format (128-B rows, XOR
//     swizzle byte ^= (k&7)<<4), staged by perfectly-coalesced b128 loads
//     (global rows are M/N-contiguous) with one-tile register prefetch;
//   * A^T / B fragments gathered with ds_read_b64_tr_b16 (the lane-grid
//     transpose law pinned by scripts/tr_probe.py) — no transpose pass,
//     no pre-swizzled global layout;
//   * split-K over blockIdx.z: each split accumulates K/S in registers
//     and writes an f32 partial [S, M, N]; a deterministic reduce kernel
//     folds the splits to bf16 (no atomics — the determinism lane
//     covers this kernel too).
//
// Reference parity note: the reference ships no GEMMs (data-only package,
// SURVEY.md §2.1); this backs the MLTC classifier lane's training step.

#include "common.h"
#include <cstdlib>

typedef __attribute__((ext_vector_type(16))) float f32x16;

#define WG_BM 256
#define WG_BN 256
#define WG_BK 32
#define WG_THREADS 512          // 8 waves: 2 (M) x 4 (N)
#define WG_WAVE_M 128           // per-wave output rows
#define WG_WAVE_N 64            // per-wave output cols
// LDS: A tile 4 blocks of [32][64] bf16 (4 KB each) + B tile same = 32 KB
#define WG_BLK_BYTES 4096
#define WG_TILE_BYTES (4 * WG_BLK_BYTES)

__device__ __forceinline__ int wg_swz(int k, int byte_off) {
  return byte_off ^ ((k & 7) << 4);
}

// XCD-aware supertile decode (round 2): HBM traffic is the wall — each
// 256-col A panel is re-read by tiles_n tile-columns and each B panel by
// tiles_m rows (~3.2 GB vs the 0.54 GB ideal at the qkv shape).  Group
// output tiles into 2(M) x 4(N) supertiles and give all 8 members ids
// congruent mod 8 so the dispatcher (XCD = id % 8) co-locates them on one
// XCD: the shared A/B panel k-windows then hit that XCD's L2 (A traffic
// /4, B /2).  Grid x = SP_padded * 8 with SP padded to a multiple of 8
// (members sit at x = sp + j*SP_padded); pad blocks exit immediately.
// Requires tiles_m % 2 == 0 and tiles_n % 4 == 0 (host guarantees via the
// 256-multiple shape check and falls back to plain ids otherwise).
// MEASURED NEGATIVE (round 2): a 2x4 supertile mapping with all 8 members
// id-congruent mod 8 (same XCD, shared panel k-windows in its L2)
// regressed EVERY shape — outp 174 -> 566 us, qkv 488 -> 590 — XCD
// co-location serializes the members on one XCD's CUs and the padded
// grid wastes dispatch slots; plain row-major tile order keeps panel
// reuse in the chip-level L2 well enough.  Kept as the plain decode.
__device__ __forceinline__ bool wg_decode(int x, int tiles_m, int tiles_n,
                                          int* tm, int* tn) {
  if (x >= tiles_m * tiles_n) return false;
  *tm = x / tiles_n;
  *tn = x % tiles_n;
  return true;
}

// Gather the four [32 k] x [64 col] fragments of one LDS column-block via
// ds_read_b64_tr_b16 (c = k-chunk 0/1, t = col-half 0/1), exactly the
// flash kernels' V^T/K^T idiom.
typedef __attribute__((ext_vector_type(2))) unsigned uint2_t;
typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;

__device__ __forceinline__ void wg_tr_gather(unsigned base, int lane,
                                             short8_t frag[2][2]) {
  const int mate = (lane >> 2) & 3;
  const int half = lane >> 5;
  const int d_lane = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
  unsigned a[8];
#pragma unroll
  for (int c = 0; c < 2; ++c)
#pragma unroll
    for (int rr = 0; rr < 2; ++rr)
#pragma unroll
      for (int t = 0; t < 2; ++t) {
        int k = 16 * c + 8 * half + 4 * rr + mate;
        int dcol = (32 * t + d_lane) * 2;
        a[c * 4 + rr * 2 + t] = base + k * 128 + wg_swz(k, dcol);
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
#pragma unroll
  for (int c = 0; c < 2; ++c)
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      uint4_t w;
      w[0] = r[c * 4 + 0 * 2 + t][0];
      w[1] = r[c * 4 + 0 * 2 + t][1];
      w[2] = r[c * 4 + 1 * 2 + t][0];
      w[3] = r[c * 4 + 1 * 2 + t][1];
      frag[c][t] = __builtin_bit_cast(short8_t, w);
    }
}

extern "C" __global__ void __launch_bounds__(WG_THREADS, 1)
wgrad_gemm_kernel(const short* __restrict__ a,   // [K, M] bf16
                  const short* __restrict__ b,   // [K, N] bf16
                  float* __restrict__ ws,        // [S, M, N] f32 partials
                  int M, int N, long K, int S) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;                        // 4 x [32][64] blocks
  char* b_lds = smem + WG_TILE_BYTES;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;

  const int tiles_m = (M + WG_BM - 1) / WG_BM;
  const int tiles_n = (N + WG_BN - 1) / WG_BN;
  int tm, tn;
  if (!wg_decode((int)blockIdx.x, tiles_m, tiles_n, &tm, &tn)) return;
  const int split = blockIdx.y;
  const long k_per = (K / S / WG_BK) * WG_BK;   // host guarantees exact
  const long k0 = (long)split * k_per;
  const long k1 = (split == S - 1) ? K : k0 + k_per;

  const int m_base = tm * WG_BM;
  const int n_base = tn * WG_BN;
  // wave sub-tile
  const int wr = wid >> 2;                    // 0..1  (M)
  const int wc = wid & 3;                     // 0..3  (N)
  const int wm = m_base + wr * WG_WAVE_M;     // wave's first output row
  const int wn = n_base + wc * WG_WAVE_N;     // wave's first output col

  f32x16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  // staging: 512 threads x 16 B covers 8 KB; each tile is 16 KB -> each
  // thread stages TWO b128 packets per tile (slots tid and tid+512).
  // slot s (0..1023): k = s >> 5, m16 = s & 31 (16-B group along M/N).
  const int s0 = tid, s1 = tid + WG_THREADS;
  const int sk0 = s0 >> 5, sm0 = s0 & 31;
  const int sk1 = s1 >> 5, sm1 = s1 & 31;
  // LDS target: block = m16 >> 3, within-block 16B col = m16 & 7
  const int a_off0 = (sm0 >> 3) * WG_BLK_BYTES + sk0 * 128 +
                     wg_swz(sk0, (sm0 & 7) * 16);
  const int a_off1 = (sm1 >> 3) * WG_BLK_BYTES + sk1 * 128 +
                     wg_swz(sk1, (sm1 & 7) * 16);

  short8_t pa0, pa1, pb0, pb1;
  {
    const long ka0 = k0 + sk0, ka1 = k0 + sk1;
    pa0 = *(const short8_t*)(a + ka0 * M + m_base + sm0 * 8);
    pa1 = *(const short8_t*)(a + ka1 * M + m_base + sm1 * 8);
    pb0 = *(const short8_t*)(b + ka0 * N + n_base + sm0 * 8);
    pb1 = *(const short8_t*)(b + ka1 * N + n_base + sm1 * 8);
  }

  for (long kt = k0; kt < k1; kt += WG_BK) {
    __syncthreads();
    *(short8_t*)(a_lds + a_off0) = pa0;
    *(short8_t*)(a_lds + a_off1) = pa1;
    *(short8_t*)(b_lds + a_off0) = pb0;
    *(short8_t*)(b_lds + a_off1) = pb1;
    __syncthreads();
    if (kt + WG_BK < k1) {
      const long ka0 = kt + WG_BK + sk0, ka1 = kt + WG_BK + sk1;
      pa0 = *(const short8_t*)(a + ka0 * M + m_base + sm0 * 8);
      pa1 = *(const short8_t*)(a + ka1 * M + m_base + sm1 * 8);
      pb0 = *(const short8_t*)(b + ka0 * N + n_base + sm0 * 8);
      pb1 = *(const short8_t*)(b + ka1 * N + n_base + sm1 * 8);
    }

    // fragments: wave's A rows = blocks [wr*2, wr*2+1] (2 x 64 cols),
    // B cols = block [wc] (1 x 64)
    short8_t fa[2][2][2];          // [m 64-block][k-chunk][col-half]
    short8_t fb[2][2];             // [k-chunk][col-half]
    wg_tr_gather((unsigned)(unsigned long)(a_lds +
                                           (wr * 2 + 0) * WG_BLK_BYTES),
                 lane, fa[0]);
    wg_tr_gather((unsigned)(unsigned long)(a_lds +
                                           (wr * 2 + 1) * WG_BLK_BYTES),
                 lane, fa[1]);
    wg_tr_gather((unsigned)(unsigned long)(b_lds + wc * WG_BLK_BYTES),
                 lane, fb);
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 2; ++c) {          // k-chunk of 16
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {     // wave m-fragments (32 rows)
        short8_t af = fa[mf >> 1][c][mf & 1];
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {   // wave n-fragments (32 cols)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, fb[c][nf], acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: acc[mf][nf] C-layout reg = A-row (m), lane = B-col (n):
  // register r -> m-local (r&3)+8*(r>>2)+4*half, lane col -> n-local.
  const int half = lane >> 5;
  float* wsp = ws + (long)split * M * N;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int ml = (r & 3) + 8 * (r >> 2) + 4 * half;
        long row = wm + mf * 32 + ml;
        long cc = wn + nf * 32 + col;
        if (row < M && cc < N) wsp[row * N + cc] = acc[mf][nf][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v2 (round 2): same tiling/compute, but staging via global_load_lds
// (direct-to-LDS DMA) with DOUBLE-buffered tiles, raw s_barriers and
// counted vmcnt waits — the guide's pipeline structure that the
// register-staged 2-barrier loop cannot reach (its drain-everything
// barrier caps the structure ~0.9 PF).  The XOR tile swizzle survives
// glds' linear wave-order writes by PRE-SWIZZLING THE PER-LANE GLOBAL
// ADDRESS (the swizzle is an XOR permutation of 16-B groups, so each
// lane simply fetches the group that belongs at its linear LDS slot);
// the tr-read compute path is unchanged and stays conflict-free.
// vmcnt discipline: 4 glds per wave per tile (2 A + 2 B), one tile
// prefetched ahead -> steady-state s_waitcnt vmcnt(4), drain 0 on last.

__device__ __forceinline__ void wg_issue_glds(
    const short* __restrict__ g, char* lds_tile, int wid, int lane,
    long ktop, long ld, int base_col, int width) {
  // wave w stages LDS bytes [w*2048, w*2048+2048) of the 16 KB tile as
  // two 1024-B glds chunks; per chunk each lane contributes 16 B at
  // lds + chunk_base + lane*16, fetching the swizzle-compensated group.
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int o = wid * 2048 + c * 1024 + lane * 16;
    int block = o >> 12;
    int ob = o & 4095;
    int k = ob >> 7;
    int s16 = (ob & 127) >> 4;
    int m16 = s16 ^ (k & 7);
    const short* gp = g + (ktop + k) * ld + base_col + (block * 8 + m16) * 8;
    // issue via inline asm (M0 = wave-uniform LDS base; the instruction
    // adds lane*16): hipcc's builtin path inserts a conservative
    // s_waitcnt vmcnt(0) before every glds batch, draining the prefetch
    // pipeline — the asm form keeps the compiler out of the loop.
    unsigned lds_base = (unsigned)(unsigned long)
        (lds_tile + wid * 2048 + c * 1024);
    unsigned m0v = __builtin_amdgcn_readfirstlane(lds_base);
    asm volatile(
        "s_mov_b32 m0, %0\n\t"
        "global_load_lds_dwordx4 %1, off"
        :: "s"(m0v), "v"(gp));
  }
}

extern "C" __global__ void __launch_bounds__(WG_THREADS, 1)
wgrad_gemm_glds_kernel(const short* __restrict__ a,   // [K, M] bf16
                       const short* __restrict__ b,   // [K, N] bf16
                       float* __restrict__ ws,        // [S, M, N] f32
                       int M, int N, long K, int S) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered: [2][A 16KB][B 16KB]
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int col = lane & 31;

  const int tiles_m = (M + WG_BM - 1) / WG_BM;
  const int tiles_n = (N + WG_BN - 1) / WG_BN;
  int tm, tn;
  if (!wg_decode((int)blockIdx.x, tiles_m, tiles_n, &tm, &tn)) return;
  const int split = blockIdx.y;
  const long k_per = (K / S / WG_BK) * WG_BK;
  const long k0 = (long)split * k_per;
  const long k1 = (split == S - 1) ? K : k0 + k_per;
  const int m_base = tm * WG_BM;
  const int n_base = tn * WG_BN;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int wm = m_base + wr * WG_WAVE_M;
  const int wn = n_base + wc * WG_WAVE_N;

  f32x16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  const long nt = (k1 - k0) / WG_BK;
  wg_issue_glds(a, smem, wid, lane, k0, M, m_base, WG_BM);
  wg_issue_glds(b, smem + WG_TILE_BYTES, wid, lane, k0, N, n_base, WG_BN);
  for (long t = 0; t < nt; ++t) {
    char* buf = smem + (t & 1) * (2 * WG_TILE_BYTES);
    if (t + 1 < nt) {
      char* nbuf = smem + ((t + 1) & 1) * (2 * WG_TILE_BYTES);
      long ktop = k0 + (t + 1) * WG_BK;
      wg_issue_glds(a, nbuf, wid, lane, ktop, M, m_base, WG_BM);
      wg_issue_glds(b, nbuf + WG_TILE_BYTES, wid, lane, ktop, N, n_base,
                    WG_BN);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    short8_t fa[2][2][2];
    short8_t fb[2][2];
    wg_tr_gather((unsigned)(unsigned long)(buf + (wr * 2 + 0) * WG_BLK_BYTES),
                 lane, fa[0]);
    wg_tr_gather((unsigned)(unsigned long)(buf + (wr * 2 + 1) * WG_BLK_BYTES),
                 lane, fa[1]);
    wg_tr_gather((unsigned)(unsigned long)(buf + WG_TILE_BYTES +
                                           wc * WG_BLK_BYTES),
                 lane, fb);
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        short8_t af = fa[mf >> 1][c][mf & 1];
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, fb[c][nf], acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  const int half = lane >> 5;
  float* wsp = ws + (long)split * M * N;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int ml = (r & 3) + 8 * (r >> 2) + 4 * half;
        long row = wm + mf * 32 + ml;
        long cc = wn + nf * 32 + col;
        if (row < M && cc < N) wsp[row * N + cc] = acc[mf][nf][r];
      }
    }
  }
}

// Deterministic split reduce: out[MN] bf16 = sum_s ws[s][MN]
extern "C" __global__ void __launch_bounds__(256)
wgrad_reduce_kernel(const float* __restrict__ ws, short* __restrict__ out,
                    long mn, int S) {
  long i = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i >= mn) return;
  float4_t acc = *(const float4_t*)(ws + i);
  for (int s = 1; s < S; ++s) {
    float4_t v = *(const float4_t*)(ws + (long)s * mn + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[j] += v[j];
  }
  short4_t o;
#pragma unroll
  for (int j = 0; j < 4; ++j) o[j] = f32_to_bf16(acc[j]);
  *(short4_t*)(out + i) = o;
}

extern "C" hipError_t wgrad_gemm_launch(const void* a, const void* b,
                                        void* ws, void* out, int M, int N,
                                        long K, int S, hipStream_t stream) {
  int tiles_m = (M + WG_BM - 1) / WG_BM;
  int tiles_n = (N + WG_BN - 1) / WG_BN;
  dim3 grid(tiles_m * tiles_n, S);
  const char* v = getenv("TOSEM_WGRAD_V1");
  if (v && v[0] == '1') {
    size_t shm = 2 * WG_TILE_BYTES;
    wgrad_gemm_kernel<<<grid, WG_THREADS, shm, stream>>>(
        (const short*)a, (const short*)b, (float*)ws, M, N, K, S);
  } else {
    size_t shm = 4 * WG_TILE_BYTES;    // double-buffered glds variant
    wgrad_gemm_glds_kernel<<<grid, WG_THREADS, shm, stream>>>(
        (const short*)a, (const short*)b, (float*)ws, M, N, K, S);
  }
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return e;
  long mn = (long)M * N;
  int rgrid = (int)((mn / 4 + 255) / 256);
  wgrad_reduce_kernel<<<rgrid, 256, 0, stream>>>((const float*)ws,
                                                 (short*)out, mn, S);
  return hipGetLastError();
}
