import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
torch.zeros(1, device="cuda")
import tosem2021_amd._hip_ops as m
N = 65536
# (label, m, n, k, ta, tb) in col-major terms of the model's GEMMs
shapes = [
    ("qkv fwd      (M3072 K1024)", 3072, N, 1024, True, False),
    ("qkv dgrad    (M1024 K3072)", 1024, N, 3072, False, False),
    ("qkv wgrad    (K=65536)", 1024, 3072, N, False, True),
    ("outp wgrad   (K=65536)", 1024, 1024, N, False, True),
    ("ffn-up wgrad (K=65536)", 1024, 4096, N, False, True),
    ("ffn-up fwd   (M4096 K1024)", 4096, N, 1024, True, False),
    ("ffn-dn fwd   (M1024 K4096)", 1024, N, 4096, True, False),
    ("ffn-dn dgrad (M4096 K1024)", 4096, N, 1024, False, False),
]
for label, mm, nn, kk, ta, tb in shapes:
    ts = m.lt_bench_algos(mm, nn, kk, ta, tb, 10)
    ok = [t for t in ts if t > 0]
    gf = 2.0 * mm * nn * kk / 1e9
    print(f"{label}: first={ts[0]*1e3:7.1f}us ({gf/ts[0]:6.0f} GF/ms={gf/ts[0]/1e3:5.2f} PF) "
          f"best={min(ok)*1e3:7.1f}us ({gf/min(ok)/1e3:5.2f} PF) "
          f"idx_best={ts.index(min(ok))} n_algos={len(ok)}")
