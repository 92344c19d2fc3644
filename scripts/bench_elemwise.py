"""Microbench the memory-bound kernels (LN, bias+GeLU) at bench.py shapes.

Reports achieved GB/s against the ~8 TB/s HBM3E roofline.
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from tosem2021_amd import ops


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = ops.hip_ops()
    N, D = 128 * 512, 1024
    Dff = 4096
    x = torch.randn(N, D, device="cuda").to(torch.bfloat16)
    res = torch.randn_like(x)
    g = torch.randn(D, device="cuda").to(torch.bfloat16)
    b = torch.randn(D, device="cuda").to(torch.bfloat16)
    dy = torch.randn_like(x)

    dt = timeit(lambda: ext.layernorm_fwd(x, None, g, b, 1e-5))
    gb = N * D * 2 * 2 / dt / 1e9
    print(f"ln_fwd           {dt*1e6:7.1f} us  {gb:6.0f} GB/s")
    dt = timeit(lambda: ext.layernorm_fwd(x, res, g, b, 1e-5))
    gb = N * D * 2 * 4 / dt / 1e9
    print(f"ln_fwd+res       {dt*1e6:7.1f} us  {gb:6.0f} GB/s")
    out = ext.layernorm_fwd(x, None, g, b, 1e-5)
    mean, rstd = out[-2], out[-1]
    dt = timeit(lambda: ext.layernorm_bwd(dy, x, g, mean, rstd, None))
    gb = N * D * 2 * 3 / dt / 1e9
    print(f"ln_bwd           {dt*1e6:7.1f} us  {gb:6.0f} GB/s")
    dt = timeit(lambda: ext.layernorm_bwd(dy, x, g, mean, rstd, res))
    gb = N * D * 2 * 4 / dt / 1e9
    print(f"ln_bwd+dsx       {dt*1e6:7.1f} us  {gb:6.0f} GB/s")

    xf = torch.randn(N, Dff, device="cuda").to(torch.bfloat16)
    bf = torch.randn(Dff, device="cuda").to(torch.bfloat16)
    dyf = torch.randn_like(xf)
    dt = timeit(lambda: ext.bias_gelu_fwd(xf, bf))
    gb = N * Dff * 2 * 2 / dt / 1e9
    print(f"bias_gelu_fwd    {dt*1e6:7.1f} us  {gb:6.0f} GB/s")
    dt = timeit(lambda: ext.bias_gelu_bwd(dyf, xf, bf))
    gb = N * Dff * 2 * 3 / dt / 1e9
    print(f"bias_gelu_bwd    {dt*1e6:7.1f} us  {gb:6.0f} GB/s")


if __name__ == "__main__":
    main()
