// Torch bindings for the tosem2021_amd gfx950 HIP kernels.
//
// Host-only translation unit: device code lives in the *.hip files, exposed
// through C-linkage launchers.  Uses the HIP-native ATen stream API directly
// (c10::hip) — no CUDA names, no hipify translation of this file is needed.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <algorithm>
#include <numeric>

extern "C" {
hipError_t ln_fwd_launch(const void*, const void*, const void*, const void*,
                         void*, void*, void*, void*, int, int, float, int,
                         hipStream_t);
hipError_t ln_bwd_launch(const void*, const void*, const void*, const void*,
                         const void*, const void*, void*, void*, void*, int,
                         int, int, hipStream_t);
hipError_t colsum_launch(const void*, void*, void*, int, int, hipStream_t);
hipError_t bias_gelu_fwd_launch(const void*, const void*, void*, long, int,
                                int, hipStream_t);
hipError_t bias_gelu_bwd_launch(const void*, const void*, const void*, void*,
                                void*, long, int, int, hipStream_t);
hipError_t softmax_fwd_launch(const void*, const void*, void*, long, int, int,
                              float, int, hipStream_t);
hipError_t softmax_bwd_launch(const void*, const void*, void*, long, int,
                              float, int, hipStream_t);
hipError_t adamw_launch(void*, const void*, int, void*, void*, void*, long,
                        float, float, float, float, float, int, float, int,
                        hipStream_t);
hipError_t qkv_repack_launch(const void*, void*, int, int, int, int, int,
                             hipStream_t);
hipError_t out_repack_launch(const void*, void*, int, int, int, int, int,
                             hipStream_t);
hipError_t qkv_repack_bwd3_launch(const void*, const void*, const void*,
                                  void*, int, int, int, int, hipStream_t);
hipError_t flash_fwd_launch(const void*, const void*, const void*, const void*,
                            void*, void*, int, int, int, float, hipStream_t);
hipError_t p_from_lse_launch(const void*, const void*, const void*, void*,
                             long, int, int, float, int, hipStream_t);
hipError_t flash_bwd_fused_launch(const void*, const void*, const void*,
                                  const void*, const void*, const void*,
                                  const void*, void*, void*, void*, int, int,
                                  int, float, hipStream_t);
hipError_t fa_dot_launch(const void*, const void*, void*, long, hipStream_t);
hipError_t fa_dot_packed_launch(const void*, const void*, void*, int, int,
                                int, hipStream_t);
hipError_t flash_fwd_packed_launch(const void*, const void*, void*, void*,
                                   int, int, int, float, hipStream_t);
hipError_t flash_bwd_fused_packed_launch(const void*, const void*,
                                         const void*, const void*,
                                         const void*, void*, int, int, int,
                                         float, hipStream_t);
hipError_t flash_dq_recompute_packed_launch(const void*, const void*,
                                            const void*, const void*,
                                            const void*, void*, int, int,
                                            int, float, hipStream_t);
hipError_t flash_dq_recompute_launch(const void*, const void*, const void*,
                                     const void*, const void*, const void*,
                                     const void*, void*, int, int, int, float,
                                     hipStream_t);
hipError_t flash_dq_launch(const void*, const void*, void*, int, int, int,
                           hipStream_t);
hipError_t masked_pool_fwd_launch(const void*, const void*, void*, void*,
                                  void*, int, int, int, hipStream_t);
hipError_t masked_pool_bwd_launch(const void*, const void*, const void*,
                                  void*, int, int, int, hipStream_t);
hipError_t tr_probe_launch(const void*, void*, int, hipStream_t);
hipError_t colsum_bf16_launch(const void*, void*, void*, long, int,
                              hipStream_t);
hipError_t wgrad_gemm_launch(const void*, const void*, void*, void*, int,
                             int, long, int, hipStream_t);
}

torch::Tensor lt_linear_gelu_bias(torch::Tensor, torch::Tensor,
                                  torch::Tensor);
bool lt_probe_epilogue(long, long, long, long);
std::vector<double> lt_bench_algos(long, long, long, bool, bool, long);

namespace {

#define CHECK_HIP(call)                                                   \
  do {                                                                    \
    hipError_t e_ = (call);                                               \
    TORCH_CHECK(e_ == hipSuccess, "HIP error: ", hipGetErrorString(e_));  \
  } while (0)

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.device().is_cuda(), name, " must be on GPU");
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be f32");
  TORCH_CHECK(t.device().is_cuda(), name, " must be on GPU");
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x,
                                         c10::optional<torch::Tensor> residual,
                                         torch::Tensor gamma,
                                         torch::Tensor beta, double eps) {
  check_bf16(x, "x"); check_bf16(gamma, "gamma"); check_bf16(beta, "beta");
  const int D = (int)x.size(-1);
  const long N = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8, got ", D);
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({N}, opts);
  auto rstd = torch::empty({N}, opts);
  int grid = (int)std::min<long>((N + 3) / 4, 4096);
  const void* res_ptr = nullptr;
  torch::Tensor s_out;
  void* s_ptr = nullptr;
  if (residual.has_value()) {
    check_bf16(*residual, "residual");
    TORCH_CHECK(residual->sizes() == x.sizes(), "residual shape mismatch");
    res_ptr = residual->data_ptr();
    s_out = torch::empty_like(x);
    s_ptr = s_out.data_ptr();
  } else {
    s_out = x;  // residual stream unchanged
  }
  CHECK_HIP(ln_fwd_launch(x.data_ptr(), res_ptr, gamma.data_ptr(),
                          beta.data_ptr(), y.data_ptr(), s_ptr,
                          mean.data_ptr(), rstd.data_ptr(), (int)N, D,
                          (float)eps, grid, cur_stream()));
  return {y, s_out, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor gamma, torch::Tensor mean,
                                         torch::Tensor rstd,
                                         c10::optional<torch::Tensor> ds_extra) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(gamma, "gamma");
  const int D = (int)x.size(-1);
  const long N = x.numel() / D;
  auto dx = torch::empty_like(x);
  // bwd grid stays modest: the [rows, D] dgamma/dbeta workspace and its
  // column-sum scale linearly with the grid
  int grid = (int)std::min<long>((N + 3) / 4, 1024);
  // the template path (D in {512, 1024}) writes ONE ws row per wave
  // (deterministic, no atomics); the general path writes one per block
  const bool templ = (D == 512 || D == 1024);
  long ws_rows = templ ? (long)grid * 4 : (long)grid;
  auto opts = x.options().dtype(torch::kFloat32);
  auto ws_dg = torch::empty({ws_rows, D}, opts);
  auto ws_db = torch::empty({ws_rows, D}, opts);
  const void* de = nullptr;
  if (ds_extra.has_value()) {
    check_bf16(*ds_extra, "ds_extra");
    TORCH_CHECK(ds_extra->numel() == x.numel(), "ds_extra shape mismatch");
    de = ds_extra->data_ptr();
  }
  CHECK_HIP(ln_bwd_launch(dy.data_ptr(), x.data_ptr(), gamma.data_ptr(),
                          mean.data_ptr(), rstd.data_ptr(), de, dx.data_ptr(),
                          ws_dg.data_ptr(), ws_db.data_ptr(), (int)N, D, grid,
                          cur_stream()));
  auto dgamma = torch::empty({D}, opts);
  auto dbeta = torch::empty({D}, opts);
  auto scratch = torch::empty({256 + 16, D}, opts);
  CHECK_HIP(colsum_launch(ws_dg.data_ptr(), scratch.data_ptr(),
                          dgamma.data_ptr(), (int)ws_rows, D, cur_stream()));
  CHECK_HIP(colsum_launch(ws_db.data_ptr(), scratch.data_ptr(),
                          dbeta.data_ptr(), (int)ws_rows, D, cur_stream()));
  return {dx, dgamma, dbeta};
}

torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor b) {
  check_bf16(x, "x"); check_bf16(b, "b");
  const int D = (int)x.size(-1);
  const long n = x.numel();
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8");
  auto y = torch::empty_like(x);
  // 32 elements per thread per iteration (BG_PK=4 in bias_gelu.hip); round
  // the grid so the stride grid*256*32 is a multiple of D (fast path)
  int grid = (int)std::min<long>((n / 32 + 255) / 256, 4096);
  long g0f = D / std::__gcd((long)D, 8192L);
  if (g0f <= 2048) grid = (int)((grid + g0f - 1) / g0f * g0f);
  CHECK_HIP(bias_gelu_fwd_launch(x.data_ptr(), b.data_ptr(), y.data_ptr(), n,
                                 D, grid, cur_stream()));
  return y;
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor b) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(b, "b");
  const int D = (int)x.size(-1);
  const long n = x.numel();
  auto dx = torch::empty_like(x);
  int grid = (int)std::min<long>((n / 32 + 255) / 256, 1024);
  // round the grid up to a multiple of D/gcd(D, 256*32) so the kernel's
  // grid stride is a multiple of D -> fixed per-thread column window
  long g = std::__gcd((long)D, 8192L);
  long g0 = D / g;
  if (g0 <= 2048) grid = (int)((grid + g0 - 1) / g0 * g0);
  auto ws = torch::empty({grid, D}, x.options().dtype(torch::kFloat32));
  CHECK_HIP(bias_gelu_bwd_launch(dy.data_ptr(), x.data_ptr(), b.data_ptr(),
                                 dx.data_ptr(), ws.data_ptr(), n, D, grid,
                                 cur_stream()));
  auto dbias = torch::empty({D}, x.options().dtype(torch::kFloat32));
  auto scratch = torch::empty({256 + 16, D},
                             x.options().dtype(torch::kFloat32));
  CHECK_HIP(colsum_launch(ws.data_ptr(), scratch.data_ptr(), dbias.data_ptr(),
                          grid, D, cur_stream()));
  return {dx, dbias};
}

torch::Tensor softmax_fwd(torch::Tensor scores, c10::optional<torch::Tensor> mask,
                          double scale) {
  check_bf16(scores, "scores");
  const int Lk = (int)scores.size(-1);
  const long n_rows = scores.numel() / Lk;
  TORCH_CHECK(Lk % 8 == 0, "Lk must be a multiple of 8");
  int H_Lq = 1;
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    TORCH_CHECK(scores.dim() == 4, "masked softmax expects [B,H,Lq,Lk]");
    TORCH_CHECK(mask->size(0) == scores.size(0) && mask->size(-1) == Lk,
                "mask must be [B,Lk]");
    H_Lq = (int)(scores.size(1) * scores.size(2));
    mptr = mask->data_ptr();
  }
  auto p = torch::empty_like(scores);
  int grid = (int)std::min<long>((n_rows + 3) / 4, 2048);
  CHECK_HIP(softmax_fwd_launch(scores.data_ptr(), mptr, p.data_ptr(), n_rows,
                               Lk, H_Lq, (float)scale, grid, cur_stream()));
  return p;
}

torch::Tensor softmax_bwd(torch::Tensor dp, torch::Tensor p, double scale) {
  check_bf16(dp, "dp"); check_bf16(p, "p");
  const int Lk = (int)p.size(-1);
  const long n_rows = p.numel() / Lk;
  auto ds = torch::empty_like(p);
  int grid = (int)std::min<long>((n_rows + 3) / 4, 2048);
  CHECK_HIP(softmax_bwd_launch(dp.data_ptr(), p.data_ptr(), ds.data_ptr(),
                               n_rows, Lk, (float)scale, grid, cur_stream()));
  return ds;
}

void adamw_step(torch::Tensor p, torch::Tensor grad, torch::Tensor m,
                torch::Tensor v, torch::Tensor master, double lr, double beta1,
                double beta2, double eps, double wd, long step,
                double grad_scale) {
  check_bf16(p, "p"); check_f32(m, "m"); check_f32(v, "v");
  check_f32(master, "master");
  TORCH_CHECK(grad.is_contiguous() && grad.device().is_cuda(), "bad grad");
  int gf32 = grad.scalar_type() == torch::kFloat32;
  TORCH_CHECK(gf32 || grad.scalar_type() == torch::kBFloat16,
              "grad must be f32 or bf16");
  const long n = p.numel();
  TORCH_CHECK(n % 4 == 0, "flat parameter buffer must be padded to 4 elems");
  TORCH_CHECK(grad.numel() == n && m.numel() == n && v.numel() == n &&
              master.numel() == n, "size mismatch");
  int grid = (int)std::min<long>((n / 4 + 255) / 256, 2048);
  CHECK_HIP(adamw_launch(p.data_ptr(), grad.data_ptr(), gf32, m.data_ptr(),
                         v.data_ptr(), master.data_ptr(), n, (float)lr,
                         (float)beta1, (float)beta2, (float)eps, (float)wd,
                         (int)step, (float)grad_scale, grid, cur_stream()));
}

torch::Tensor qkv_repack(torch::Tensor qkv, long H, bool backward) {
  check_bf16(qkv, "qkv");
  long B, L, dh;
  torch::Tensor out;
  if (!backward) {
    // [B, L, 3*H*dh] or [B, L, 3, H, dh] -> [3, B, H, L, dh]
    B = qkv.size(0); L = qkv.size(1);
    dh = qkv.numel() / (B * L * 3 * H);
    out = torch::empty({3, B, H, L, dh}, qkv.options());
  } else {
    // grad [3, B, H, L, dh] -> [B, L, 3*H*dh]
    B = qkv.size(1); L = qkv.size(3);
    dh = qkv.size(4);
    out = torch::empty({B, L, 3 * H * dh}, qkv.options());
  }
  TORCH_CHECK(dh % 8 == 0, "head_dim must be a multiple of 8");
  CHECK_HIP(qkv_repack_launch(qkv.data_ptr(), out.data_ptr(), (int)B, (int)L,
                              (int)H, (int)dh, backward ? 1 : 0,
                              cur_stream()));
  return out;
}

torch::Tensor qkv_repack_bwd3(torch::Tensor dq, torch::Tensor dk,
                              torch::Tensor dv) {
  check_bf16(dq, "dq"); check_bf16(dk, "dk"); check_bf16(dv, "dv");
  long B = dq.size(0), H = dq.size(1), L = dq.size(2), dh = dq.size(3);
  auto out = torch::empty({B, L, 3 * H * dh}, dq.options());
  CHECK_HIP(qkv_repack_bwd3_launch(dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                                   out.data_ptr(), (int)B, (int)L, (int)H,
                                   (int)dh, cur_stream()));
  return out;
}

torch::Tensor out_repack(torch::Tensor x, bool backward) {
  check_bf16(x, "x");
  long B, L, H, dh;
  torch::Tensor out;
  if (!backward) {
    // [B, H, L, dh] -> [B, L, H*dh]
    B = x.size(0); H = x.size(1); L = x.size(2); dh = x.size(3);
    out = torch::empty({B, L, H * dh}, x.options());
  } else {
    TORCH_CHECK(false, "pass the 4-D grad with explicit dims via out_repack_bwd");
  }
  TORCH_CHECK(dh % 8 == 0, "head_dim must be a multiple of 8");
  CHECK_HIP(out_repack_launch(x.data_ptr(), out.data_ptr(), (int)B, (int)L,
                              (int)H, (int)dh, 0, cur_stream()));
  return out;
}

torch::Tensor out_repack_bwd(torch::Tensor dgrad, long H) {
  check_bf16(dgrad, "dgrad");
  long B = dgrad.size(0), L = dgrad.size(1);
  long dh = dgrad.numel() / (B * L * H);
  auto out = torch::empty({B, H, L, dh}, dgrad.options());
  CHECK_HIP(out_repack_launch(dgrad.data_ptr(), out.data_ptr(), (int)B,
                              (int)L, (int)H, (int)dh, 1, cur_stream()));
  return out;
}

std::vector<torch::Tensor> flash_fwd(torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v,
                                     c10::optional<torch::Tensor> mask,
                                     double scale) {
  check_bf16(q, "q"); check_bf16(k, "k"); check_bf16(v, "v");
  TORCH_CHECK(q.dim() == 4, "q must be [B, H, L, dh]");
  const long B = q.size(0), H = q.size(1), L = q.size(2), dh = q.size(3);
  TORCH_CHECK(dh == 64, "flash_fwd is specialized for head_dim 64");
  TORCH_CHECK(L % 32 == 0, "flash_fwd needs L % 32 == 0");
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes());
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    TORCH_CHECK(mask->size(0) == B && mask->size(-1) == L, "mask is [B, L]");
    mptr = mask->data_ptr();
  }
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, L}, q.options().dtype(torch::kFloat32));
  CHECK_HIP(flash_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), mptr,
                             o.data_ptr(), lse.data_ptr(), (int)B, (int)H,
                             (int)L, (float)scale, cur_stream()));
  return {o, lse};
}

torch::Tensor p_from_lse(torch::Tensor scores, c10::optional<torch::Tensor> mask,
                         torch::Tensor lse, double scale) {
  check_bf16(scores, "scores"); check_f32(lse, "lse");
  const int Lk = (int)scores.size(-1);
  const long n_rows = scores.numel() / Lk;
  TORCH_CHECK(lse.numel() == n_rows, "lse size mismatch");
  int H_Lq = 1;
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    H_Lq = (int)(scores.size(1) * scores.size(2));
    mptr = mask->data_ptr();
  }
  auto p = torch::empty_like(scores);
  int grid = (int)std::min<long>((n_rows + 3) / 4, 2048);
  CHECK_HIP(p_from_lse_launch(scores.data_ptr(), mptr, lse.data_ptr(),
                              p.data_ptr(), n_rows, Lk, H_Lq, (float)scale,
                              grid, cur_stream()));
  return p;
}

torch::Tensor fa_dot(torch::Tensor dout, torch::Tensor o) {
  check_bf16(dout, "dout"); check_bf16(o, "o");
  const long n_rows = dout.numel() / dout.size(-1);
  TORCH_CHECK(dout.size(-1) == 64, "fa_dot expects head_dim 64");
  auto d = torch::empty({n_rows}, dout.options().dtype(torch::kFloat32));
  CHECK_HIP(fa_dot_launch(dout.data_ptr(), o.data_ptr(), d.data_ptr(), n_rows,
                          cur_stream()));
  return d.view({dout.size(0), dout.size(1), dout.size(2)});
}

std::vector<torch::Tensor> flash_bwd_fused(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, torch::Tensor dout,
                                           c10::optional<torch::Tensor> mask,
                                           torch::Tensor lse,
                                           torch::Tensor ddot, double scale,
                                           bool emit_ds) {
  check_bf16(q, "q"); check_bf16(k, "k"); check_bf16(v, "v");
  check_bf16(dout, "dout"); check_f32(lse, "lse"); check_f32(ddot, "ddot");
  const long B = q.size(0), H = q.size(1), L = q.size(2);
  TORCH_CHECK(q.size(3) == 64 && L % 32 == 0, "dh=64 and L%32==0 required");
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    mptr = mask->data_ptr();
  }
  // emit_ds=false (the default path since round 2): no [B, H, L, L] dS
  // materialization — dQ comes from flash_dq_recompute instead.
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  torch::Tensor ds;
  void* ds_ptr = nullptr;
  if (emit_ds) {
    ds = torch::empty({B, H, L, L}, q.options());
    ds_ptr = ds.data_ptr();
  }
  CHECK_HIP(flash_bwd_fused_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                                   dout.data_ptr(), mptr, lse.data_ptr(),
                                   ddot.data_ptr(), ds_ptr,
                                   dk.data_ptr(), dv.data_ptr(), (int)B,
                                   (int)H, (int)L, (float)scale,
                                   cur_stream()));
  if (emit_ds) return {ds, dk, dv};
  return {dk, dv};
}

torch::Tensor wgrad_gemm(torch::Tensor dy, torch::Tensor x,
                         long splits) {
  // dW[M, N] = dy[K, M]^T @ x[K, N]; custom split-K MFMA kernel
  check_bf16(dy, "dy"); check_bf16(x, "x");
  TORCH_CHECK(dy.dim() == 2 && x.dim() == 2, "2-D operands required");
  const long K = dy.size(0), M = dy.size(1), N = x.size(1);
  TORCH_CHECK(x.size(0) == K, "K mismatch");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0, "M, N must be 256-multiples");
  TORCH_CHECK(K % 32 == 0, "K must be a 32-multiple");
  int S = (int)splits;
  if (S <= 0) {   // auto: enough workgroups to fill 256 CUs
    long tiles = (M / 256) * (N / 256);
    S = 1;
    while (S < 32 && tiles * S < 512 && (K / (2L * S)) % 32 == 0) S *= 2;
  }
  TORCH_CHECK((K / S) % 32 == 0, "K/S must be a 32-multiple");
  auto ws = torch::empty({S, M, N}, dy.options().dtype(torch::kFloat32));
  auto out = torch::empty({M, N}, dy.options());
  CHECK_HIP(wgrad_gemm_launch(dy.data_ptr(), x.data_ptr(), ws.data_ptr(),
                              out.data_ptr(), (int)M, (int)N, K, S,
                              cur_stream()));
  return out;
}

torch::Tensor bias_grad(torch::Tensor dy) {
  check_bf16(dy, "dy");
  const int D = (int)dy.size(-1);
  const long N = dy.numel() / D;
  TORCH_CHECK(D % 8 == 0, "D % 8 == 0 required");
  auto scratch = torch::empty({1024 + 16, D},
                              dy.options().dtype(torch::kFloat32));
  auto out = torch::empty({D}, dy.options());
  CHECK_HIP(colsum_bf16_launch(dy.data_ptr(), scratch.data_ptr(),
                               out.data_ptr(), N, D, cur_stream()));
  return out;
}

std::vector<torch::Tensor> flash_fwd_packed(torch::Tensor qkv, long H,
                                            c10::optional<torch::Tensor> mask,
                                            double scale) {
  // qkv: [B, L, 3D] packed projection output, D = H*64
  check_bf16(qkv, "qkv");
  TORCH_CHECK(qkv.dim() == 3, "qkv must be [B, L, 3D]");
  const long B = qkv.size(0), L = qkv.size(1), D3 = qkv.size(2);
  TORCH_CHECK(D3 == 3 * H * 64, "qkv last dim must be 3*H*64");
  TORCH_CHECK(L % 32 == 0, "L % 32 == 0 required");
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    mptr = mask->data_ptr();
  }
  auto o = torch::empty({B, L, D3 / 3}, qkv.options());
  auto lse = torch::empty({B, H, L}, qkv.options().dtype(torch::kFloat32));
  CHECK_HIP(flash_fwd_packed_launch(qkv.data_ptr(), mptr, o.data_ptr(),
                                    lse.data_ptr(), (int)B, (int)H, (int)L,
                                    (float)scale, cur_stream()));
  return {o, lse};
}

torch::Tensor flash_bwd_packed(torch::Tensor qkv, torch::Tensor o,
                               torch::Tensor dout,
                               c10::optional<torch::Tensor> mask,
                               torch::Tensor lse, long H, double scale) {
  // full packed backward: ddot + dK/dV + dQ, all into one [B, L, 3D] grad
  check_bf16(qkv, "qkv"); check_bf16(o, "o"); check_bf16(dout, "dout");
  check_f32(lse, "lse");
  const long B = qkv.size(0), L = qkv.size(1), D3 = qkv.size(2);
  TORCH_CHECK(D3 == 3 * H * 64 && L % 32 == 0, "bad packed shapes");
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    mptr = mask->data_ptr();
  }
  auto ddot = torch::empty({B, H, L}, qkv.options().dtype(torch::kFloat32));
  CHECK_HIP(fa_dot_packed_launch(dout.data_ptr(), o.data_ptr(),
                                 ddot.data_ptr(), (int)B, (int)H, (int)L,
                                 cur_stream()));
  auto dqkv = torch::empty_like(qkv);
  CHECK_HIP(flash_bwd_fused_packed_launch(
      qkv.data_ptr(), dout.data_ptr(), mptr, lse.data_ptr(), ddot.data_ptr(),
      dqkv.data_ptr(), (int)B, (int)H, (int)L, (float)scale, cur_stream()));
  CHECK_HIP(flash_dq_recompute_packed_launch(
      qkv.data_ptr(), dout.data_ptr(), mptr, lse.data_ptr(), ddot.data_ptr(),
      dqkv.data_ptr(), (int)B, (int)H, (int)L, (float)scale, cur_stream()));
  return dqkv;
}

torch::Tensor flash_dq_recompute(torch::Tensor q, torch::Tensor k,
                                 torch::Tensor v, torch::Tensor dout,
                                 c10::optional<torch::Tensor> mask,
                                 torch::Tensor lse, torch::Tensor ddot,
                                 double scale) {
  check_bf16(q, "q"); check_bf16(k, "k"); check_bf16(v, "v");
  check_bf16(dout, "dout"); check_f32(lse, "lse"); check_f32(ddot, "ddot");
  const long B = q.size(0), H = q.size(1), L = q.size(2);
  TORCH_CHECK(q.size(3) == 64 && L % 32 == 0, "dh=64 and L%32==0 required");
  const void* mptr = nullptr;
  if (mask.has_value()) {
    check_f32(*mask, "mask");
    mptr = mask->data_ptr();
  }
  auto dq = torch::empty_like(q);
  CHECK_HIP(flash_dq_recompute_launch(q.data_ptr(), k.data_ptr(),
                                      v.data_ptr(), dout.data_ptr(), mptr,
                                      lse.data_ptr(), ddot.data_ptr(),
                                      dq.data_ptr(), (int)B, (int)H, (int)L,
                                      (float)scale, cur_stream()));
  return dq;
}

torch::Tensor flash_dq(torch::Tensor ds, torch::Tensor k) {
  check_bf16(ds, "ds"); check_bf16(k, "k");
  const long B = k.size(0), H = k.size(1), L = k.size(2);
  TORCH_CHECK(k.size(3) == 64 && L % 32 == 0, "dh=64 and L%32==0 required");
  TORCH_CHECK(ds.size(2) == L && ds.size(3) == L, "ds must be [B,H,L,L]");
  auto dq = torch::empty_like(k);
  CHECK_HIP(flash_dq_launch(ds.data_ptr(), k.data_ptr(), dq.data_ptr(),
                            (int)B, (int)H, (int)L, cur_stream()));
  return dq;
}

std::vector<torch::Tensor> masked_pool_fwd(torch::Tensor x,
                                           c10::optional<torch::Tensor> mask) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 3, "x must be [B, L, D]");
  long B = x.size(0), L = x.size(1), D = x.size(2);
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8");
  const void* mptr = nullptr;
  if (mask.has_value()) {
    TORCH_CHECK(mask->scalar_type() == torch::kBool && mask->is_contiguous(),
                "mask must be contiguous bool");
    mptr = mask->data_ptr();
  }
  auto pooled = torch::empty({B, D}, x.options());
  auto counts = torch::empty({B}, x.options().dtype(torch::kFloat32));
  auto ws = torch::empty({B, 16, D}, x.options().dtype(torch::kFloat32));
  CHECK_HIP(masked_pool_fwd_launch(x.data_ptr(), mptr, pooled.data_ptr(),
                                   counts.data_ptr(), ws.data_ptr(), (int)B,
                                   (int)L, (int)D, cur_stream()));
  return {pooled, counts};
}

torch::Tensor masked_pool_bwd(torch::Tensor dpooled,
                              c10::optional<torch::Tensor> mask,
                              torch::Tensor counts, long L) {
  check_bf16(dpooled, "dpooled");
  long B = dpooled.size(0), D = dpooled.size(1);
  const void* mptr = nullptr;
  if (mask.has_value()) mptr = mask->data_ptr();
  auto dx = torch::empty({B, L, D}, dpooled.options());
  CHECK_HIP(masked_pool_bwd_launch(dpooled.data_ptr(), mptr,
                                   counts.data_ptr(), dx.data_ptr(), (int)B,
                                   (int)L, (int)D, cur_stream()));
  return dx;
}

torch::Tensor tr_probe(torch::Tensor src, long scheme) {
  check_bf16(src, "src");
  TORCH_CHECK(src.numel() == 256);
  auto out = torch::zeros({64, 4}, src.options());
  CHECK_HIP(tr_probe_launch(src.data_ptr(), out.data_ptr(), (int)scheme,
                            cur_stream()));
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("tr_probe", &tr_probe, "ds_read_b64_tr_b16 semantics probe");
  m.def("masked_pool_fwd", &masked_pool_fwd, "masked mean-pool fwd");
  m.def("masked_pool_bwd", &masked_pool_bwd, "masked mean-pool bwd");
  m.def("flash_fwd", &flash_fwd, "flash attention fwd (gfx950 MFMA, dh=64)");
  m.def("flash_bwd_fused", &flash_bwd_fused,
        "fused flash bwd: register-accumulated dK/dV (+ optional dS)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("dout"),
        py::arg("mask"), py::arg("lse"), py::arg("ddot"), py::arg("scale"),
        py::arg("emit_ds") = false);
  m.def("flash_dq_recompute", &flash_dq_recompute,
        "dQ by recompute: S/dP/dS in-register, no dS materialization");
  m.def("wgrad_gemm", &wgrad_gemm,
        "split-K wgrad GEMM: dW = dy^T @ x (custom MFMA)",
        py::arg("dy"), py::arg("x"), py::arg("splits") = 0);
  m.def("flash_fwd_packed", &flash_fwd_packed,
        "flash fwd on the packed [B,L,3D] QKV projection output");
  m.def("flash_bwd_packed", &flash_bwd_packed,
        "full packed flash bwd: ddot + dK/dV + dQ into one [B,L,3D] grad");
  m.def("fa_dot", &fa_dot, "rowsum(dO*O) per attention row");
  m.def("flash_dq", &flash_dq, "dQ = dS @ K (MFMA, tr_b16 K^T fragments)");
  m.def("bias_grad", &bias_grad, "bf16 column-sum for linear bias grads");
  m.def("p_from_lse", &p_from_lse, "probabilities from saved logsumexp");
  m.def("qkv_repack", &qkv_repack, "qkv layout repack (fwd/bwd)");
  m.def("qkv_repack_bwd3", &qkv_repack_bwd3, "qkv repack bwd from dq,dk,dv");
  m.def("out_repack", &out_repack, "attention output merge");
  m.def("out_repack_bwd", &out_repack_bwd, "attention output merge bwd");
  m.def("layernorm_fwd", &layernorm_fwd, "fused LayerNorm fwd (bf16, gfx950)");
  m.def("layernorm_bwd", &layernorm_bwd, "fused LayerNorm bwd");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "fused bias+GeLU fwd");
  m.def("lt_linear_gelu_bias", &lt_linear_gelu_bias,
        "hipBLASLt GEMM with fused GELU_BIAS epilogue (inference)");
  m.def("lt_bench_algos", &lt_bench_algos,
        "time the heuristic's top algos for a bf16 GEMM config");
  m.def("lt_probe_epilogue", &lt_probe_epilogue,
        "does this hipBLASLt have kernels for (m, n, k, epilogue)?");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "fused bias+GeLU bwd");
  m.def("softmax_fwd", &softmax_fwd, "fused scaled masked softmax fwd");
  m.def("softmax_bwd", &softmax_bwd, "fused softmax bwd");
  m.def("adamw_step", &adamw_step, "fused AdamW over flat params");
}
