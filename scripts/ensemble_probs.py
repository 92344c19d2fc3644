#!/usr/bin/env python3
"""Shared-split seed-ensemble evaluation from dumped sigmoid probs.

Each member is produced by
    python -m tosem2021_amd.cli train --taxonomy artifacts/mltc_train_repo_s256.pt \
        --steps <K> --batch 64 --dropout 0.1 --focal-gamma-property 2 \
        --seed <S> --split-seed 0 --dump-probs <file.pt>
`--split-seed 0` pins one val partition across all seeds, so the dumped
val probs are directly averageable (docs/ROADMAP.md round-2 item:
"k-fold ensembles with a SHARED split").

Usage: python scripts/ensemble_probs.py probs1.pt probs2.pt [...]
Prints per-member and ensemble micro-F1 for both multilabel heads.
"""
import sys

import torch


def f1(pred, gold):
    tp = (pred & gold).sum().item()
    fp = (pred & ~gold).sum().item()
    fn = (~pred & gold).sum().item()
    return 2 * tp / max(2 * tp + fp + fn, 1)


def main(paths):
    dumps = [torch.load(p, weights_only=True) for p in paths]
    gold = {h: dumps[0]["val_gold"][h].bool()
            for h in ("strategy", "property")}
    for d in dumps[1:]:
        for h in gold:
            assert torch.equal(d["val_gold"][h].bool(), gold[h]), \
                "val splits differ — members must share --split-seed"
    out = {}
    for h in ("strategy", "property"):
        for p, d in zip(paths, dumps):
            print(f"{h:9s} {p:32s} @0.5 "
                  f"{f1(d['val_probs'][h] > .5, gold[h]):.4f}")
        ens = sum(d["val_probs"][h] for d in dumps) / len(dumps)
        out[h] = {t: round(f1(ens > t, gold[h]), 4) for t in (0.3, 0.4, 0.5)}
        print(f"{h:9s} ENSEMBLE x{len(dumps)}  @0.5 {out[h][0.5]:.4f}  "
              f"@0.4 {out[h][0.4]:.4f}  @0.3 {out[h][0.3]:.4f}")
    return out


if __name__ == "__main__":
    if len(sys.argv) < 3:
        sys.exit(__doc__)
    main(sys.argv[1:])
