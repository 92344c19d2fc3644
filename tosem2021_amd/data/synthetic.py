"""Deterministic synthetic batches shaped like the taxonomy training data.

bench.py and the GPU smoke use these (there is no network on the build or GPU
boxes, so the benchmark is specified on synthetic data of the flagship shape —
random token ids + random taxonomy labels, random-init weights).
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from tosem2021_amd.models.classifier import MLTCConfig, MULTILABEL_HEADS


def synthetic_batch(
    cfg: MLTCConfig, batch: int, seq: Optional[int] = None,
    device: str | torch.device = "cpu", seed: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor, Dict[str, torch.Tensor]]:
    seq = seq or cfg.max_seq
    g = torch.Generator(device="cpu").manual_seed(seed)
    tokens = torch.randint(0, cfg.vocab_size, (batch, seq), generator=g)
    lengths = torch.randint(seq // 2, seq + 1, (batch,), generator=g)
    mask = torch.arange(seq)[None, :] < lengths[:, None]
    labels: Dict[str, torch.Tensor] = {}
    for name, n in cfg.heads.items():
        if name in MULTILABEL_HEADS:
            labels[name] = (torch.rand(batch, n, generator=g) < 0.15).float()
        else:
            labels[name] = torch.randint(0, n, (batch,), generator=g)
    dev = torch.device(device)
    tokens = tokens.to(dev)
    mask = mask.to(dev)
    labels = {k: v.to(dev) for k, v in labels.items()}
    return tokens, mask, labels
