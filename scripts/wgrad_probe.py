"""Probe: wgrad GEMMs (K=65536) vs chunked-K accumulation via hipBLASLt.

The four wgrad shapes run at 0.65–0.94 PF/s — 2.5–3× off the dense peak —
while the forward GEMMs reach ~1.7 PF/s.  Hypothesis: the K=65536 panel
stream exceeds L2 reuse; splitting K into library-sized chunks (the
library still split-Ks WITHIN a chunk) may dispatch better.  Measures
torch.mm dW = dy^T @ x at K=65536 single-shot vs 4/8/16 chunks with
bf16 beta-accumulation, plus an f32-out variant where supported.
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch


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
    torch.manual_seed(0)
    K = 65536
    shapes = [("qkv", 3072, 1024), ("outp", 1024, 1024),
              ("ffn-up", 4096, 1024), ("ffn-dn", 1024, 4096)]
    for name, M, N in shapes:
        dy = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        gf = 2.0 * M * N * K / 1e9

        dt0 = timeit(lambda: torch.mm(dy.t(), x))
        print(f"{name:7s} single K=65536: {dt0*1e6:7.1f} us  "
              f"{gf/dt0/1e3:5.2f} PF")
        for S in (4, 8, 16):
            ck = K // S
            dw = torch.zeros(M, N, device="cuda", dtype=torch.bfloat16)

            def chunked():
                dw.zero_()
                for s in range(S):
                    dw.addmm_(dy[s * ck:(s + 1) * ck].t(),
                              x[s * ck:(s + 1) * ck], beta=1.0, alpha=1.0)
            dt = timeit(chunked)
            print(f"        {S:2d} x K={ck}: {dt*1e6:7.1f} us  "
                  f"{gf/dt/1e3:5.2f} PF")
        del dy, x
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
