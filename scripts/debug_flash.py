"""Dump the error structure of flash_fwd vs the fp32 reference (GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tosem2021_amd import ops
from tosem2021_amd.ops import reference as ref


def main():
    torch.manual_seed(10)
    B, H, L = 2, 3, 128
    q = torch.randn(B, H, L, 64).to(torch.bfloat16).cuda()
    k = torch.randn(B, H, L, 64).to(torch.bfloat16).cuda()
    v = torch.randn(B, H, L, 64).to(torch.bfloat16).cuda()
    mask = torch.zeros(B, L)
    mask[:, L - L // 4:] = -1e9
    mg = mask.cuda().contiguous()
    scale = 0.125
    o, lse = ops.hip_ops().flash_fwd(q, k, v, mg, scale)
    oe, le = ref.flash_attention_fwd(q.float().cpu(), k.float().cpu(),
                                     v.float().cpu(), mask, scale)
    d = (o.float().cpu() - oe.float()).abs()
    print("o max diff", float(d.max()), "mean", float(d.mean()))
    dl = (lse.cpu() - le).abs()
    print("lse max diff", float(dl.max()))
    # which q rows are wrong (bh 0)?
    bad = (d[0, 0] > 0.05).any(dim=-1)
    print("bad q rows bh(0,0):", bad.nonzero().flatten().tolist()[:40])
    # which d columns?
    badc = (d[0, 0] > 0.05).any(dim=0)
    print("bad d cols bh(0,0):", badc.nonzero().flatten().tolist()[:40])
    # second call — deterministic?
    o2, _ = ops.hip_ops().flash_fwd(q, k, v, mg, scale)
    print("run-to-run identical:", bool(torch.equal(o, o2)))


if __name__ == "__main__":
    main()
