"""Diagnose bias_gelu numerics vs the fp32 reference (run on GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tosem2021_amd import ops
from tosem2021_amd.ops import reference as ref


def main():
    torch.manual_seed(2)
    for N, D in [(128, 4096), (63, 256)]:
        x = torch.randn(N, D)
        b = torch.randn(D)
        xg = x.to(torch.bfloat16).cuda().contiguous()
        bg = b.to(torch.bfloat16).cuda().contiguous()
        y = ops.hip_ops().bias_gelu_fwd(xg, bg)
        ye = ref.bias_gelu_fwd(xg.float().cpu(), bg.float().cpu())
        d = (y.float().cpu() - ye).abs()
        rel = d / ye.abs().clamp(min=1e-3)
        idx = d.argmax()
        print(f"[{N}x{D}] fwd max abs {d.max():.5f} max rel {rel.max():.5f} "
              f"worst at {idx}: got {y.float().cpu().flatten()[idx]:.6f} "
              f"want {ye.flatten()[idx]:.6f} "
              f"x={xg.float().cpu().flatten()[idx]:.6f} "
              f"b={bg.float().cpu().flatten()[idx % D]:.6f}")
        dyg = torch.randn(N, D).to(torch.bfloat16).cuda().contiguous()
        dx, dbias = ops.hip_ops().bias_gelu_bwd(dyg, xg, bg)
        dxe, dbe = ref.bias_gelu_bwd(dyg.float().cpu(), xg.float().cpu(),
                                     bg.float().cpu())
        d1 = (dx.float().cpu() - dxe).abs()
        d2 = (dbias.cpu() - dbe).abs()
        print(f"[{N}x{D}] bwd dx max abs {d1.max():.5f}  dbias max abs "
              f"{d2.max():.5f} (dbias scale {dbe.abs().max():.3f})")


if __name__ == "__main__":
    main()
