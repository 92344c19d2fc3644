"""Diagnose the dq bmm: warmed timings for layout variants (GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch


def t(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    B, H, L, dh = 128, 16, 512, 64
    ds = torch.randn(B, H, L, L, device="cuda").to(torch.bfloat16)
    k = torch.randn(B, H, L, dh, device="cuda").to(torch.bfloat16)
    print("dq = ds @ k            :", round(t(lambda: torch.matmul(ds, k)), 1), "us")
    dst = ds.transpose(-1, -2).contiguous()
    print("dq = ds_t^T @ k (view) :", round(t(lambda: torch.matmul(dst.transpose(-1, -2), k)), 1), "us")
    d3 = ds.reshape(B * H, L, L)
    k3 = k.reshape(B * H, L, dh)
    print("bmm 3d                 :", round(t(lambda: torch.bmm(d3, k3)), 1), "us")
    out = torch.empty(B * H, L, dh, device="cuda", dtype=torch.bfloat16)
    print("bmm 3d out=            :", round(t(lambda: torch.bmm(d3, k3, out=out)), 1), "us")
    # 2d flattened with strided A is impossible; try f32 accum path
    print("matmul fp32 k          :", round(t(lambda: torch.matmul(ds, k.float().to(torch.bfloat16))), 1), "us")


if __name__ == "__main__":
    main()
