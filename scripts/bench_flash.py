"""Standalone flash-attention kernel microbench (for rocprofv3 PMC runs)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from tosem2021_amd import ops


def main():
    B, H, L, dh = 128, 16, 512, 64
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    torch.manual_seed(0)
    q = torch.randn(B, H, L, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    mask = torch.zeros(B, L, device="cuda")
    mask[:, 400:] = -1e9
    mask = mask.contiguous()
    scale = 0.125
    ext = ops.hip_ops()
    for _ in range(5):
        o, lse = ext.flash_fwd(q, k, v, mask, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        o, lse = ext.flash_fwd(q, k, v, mask, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    flops = 4 * B * H * L * L * dh
    print(f"flash_fwd {dt*1e6:.1f} us/call  {flops/dt/1e12:.1f} TF")
    # bwd stage (round-2 default: dk/dv only, no dS materialization)
    do = torch.randn_like(q)
    ddot = ext.fa_dot(do, o)
    for _ in range(3):
        dk, dv = ext.flash_bwd_fused(q, k, v, do, mask, lse, ddot, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        dk, dv = ext.flash_bwd_fused(q, k, v, do, mask, lse, ddot, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"flash_bwd_fused {dt*1e6:.1f} us/call  {3*flops/dt/1e12:.1f} TF-eq")
    for _ in range(3):
        dq = ext.flash_dq_recompute(q, k, v, do, mask, lse, ddot, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        dq = ext.flash_dq_recompute(q, k, v, do, mask, lse, ddot, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"flash_dq_recompute {dt*1e6:.1f} us/call  {3*flops/dt/1e12:.1f} TF-eq")
    # round-1 chain for comparison (dS round-trip + dq-from-ds)
    ds, dk, dv = ext.flash_bwd_fused(q, k, v, do, mask, lse, ddot, scale,
                                     emit_ds=True)
    for _ in range(3):
        dq = ext.flash_dq(ds, k)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ds, dk, dv = ext.flash_bwd_fused(q, k, v, do, mask, lse, ddot, scale,
                                         emit_ds=True)
        dq = ext.flash_dq(ds, k)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"r1 bwd chain (fused+emit_ds + dq) {dt*1e6:.1f} us")
    del ds
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        dk, dv = ext.flash_bwd_fused(q, k, v, do, mask, lse, ddot, scale)
        dq = ext.flash_dq_recompute(q, k, v, do, mask, lse, ddot, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"r2 bwd chain (fused + dq_recompute) {dt*1e6:.1f} us")


if __name__ == "__main__":
    main()
