"""Empirically determine the ds_read_b64_tr_b16 lane mapping (GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tosem2021_amd import ops

def main():
    # distinct bf16-exact values: 0..255
    src = torch.arange(256, dtype=torch.float32).to(torch.bfloat16).cuda()
    for scheme in [7, 8, 9, 10, 11, 12]:
        out = ops.hip_ops().tr_probe(src, scheme).float().cpu().int()
        print(f"-- scheme {scheme}")
        for l in range(16):
            print(f"  lane {l:2d}: {out[l].tolist()}")

if __name__ == "__main__":
    main()
