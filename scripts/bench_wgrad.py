"""Numerics + perf of the custom split-K wgrad GEMM vs torch/hipBLASLt."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from tosem2021_amd import ops


def timeit(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = ops.hip_ops()
    torch.manual_seed(0)
    # numerics first, small shape
    K, M, N = 2048, 256, 512
    dy = torch.randn(K, M, device="cuda", dtype=torch.bfloat16) * 0.5
    x = torch.randn(K, N, device="cuda", dtype=torch.bfloat16) * 0.5
    dw = ext.wgrad_gemm(dy, x, 4)
    ref = torch.mm(dy.t().float(), x.float())
    err = (dw.float() - ref).abs()
    rel = err.max() / ref.abs().max()
    print(f"numerics K={K} M={M} N={N}: max_abs={err.max():.3f} "
          f"rel={rel:.5f} ref_scale={ref.abs().max():.1f}")
    assert rel < 2e-2, "NUMERICS FAIL"
    d1 = ext.wgrad_gemm(dy, x, 4)
    d2 = ext.wgrad_gemm(dy, x, 4)
    print("deterministic:", bool(torch.equal(d1, d2)))

    K = 65536
    for name, M, N in [("qkv", 3072, 1024), ("outp", 1024, 1024),
                       ("ffn-up", 4096, 1024), ("ffn-dn", 1024, 4096)]:
        dy = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        gf = 2.0 * M * N * K / 1e9
        dt_t = timeit(lambda: torch.mm(dy.t(), x))
        for S in (0, 8, 16, 32):
            try:
                dt = timeit(lambda: ext.wgrad_gemm(dy, x, S))
            except RuntimeError as e:
                print(f"  S={S}: {e}")
                continue
            print(f"{name:7s} S={S:2d}: {dt*1e6:7.1f} us  {gf/dt/1e3:5.2f} PF"
                  f"   (torch: {dt_t*1e6:7.1f} us {gf/dt_t/1e3:5.2f} PF)")
        # spot numerics at full shape
        dw = ext.wgrad_gemm(dy[:4096], x[:4096], 4)
        ref = torch.mm(dy[:4096].t().float(), x[:4096].float())
        rel = (dw.float() - ref).abs().max() / ref.abs().max()
        print(f"        numerics(K=4096): rel={rel:.5f}")
        del dy, x
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
