"""hipGraph capture for launch-bound inference (torch.cuda.CUDAGraph on
ROCm records a hipGraph).

Small-batch classification is launch-bound: an mltc-base forward issues
~90 kernels, and at batch 8 most are a few µs each — per-launch CPU
overhead dominates.  `CapturedForward` records the whole forward once
into a hipGraph and replays it with a single launch; inputs are copied
into static buffers, outputs read from static tensors.

Constraints (standard graph-capture rules): fixed shapes, no
host-synchronizing ops inside the captured region, model in eval/no-grad.
The MLTC inference path qualifies (flash fwd, GELU_BIAS epilogue GEMMs,
LN, pooling — all shape-static).  Measured: see scripts/bench_infer.py
--graph and docs/PERF.md.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch


class CapturedForward:
    """Capture `model(tokens, mask)` for a fixed (batch, seq) shape."""

    def __init__(self, model, batch: int, seq: int,
                 device: Optional[torch.device] = None, warmup: int = 3):
        assert torch.cuda.is_available(), "graph capture needs a GPU"
        self.device = device or torch.device("cuda")
        self.model = model.eval()
        self.static_tokens = torch.zeros(batch, seq, dtype=torch.long,
                                         device=self.device)
        self.static_mask = torch.ones(batch, seq, dtype=torch.bool,
                                      device=self.device)
        # warm up on a side stream (allocator + library handles), then record
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup):
                self.model(self.static_tokens, self.static_mask)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph), torch.no_grad():
            self.static_out = self.model(self.static_tokens,
                                         self.static_mask)

    @torch.no_grad()
    def __call__(self, tokens: torch.Tensor,
                 mask: Optional[torch.Tensor] = None
                 ) -> Dict[str, torch.Tensor]:
        self.static_tokens.copy_(tokens, non_blocking=True)
        if mask is None:
            self.static_mask.fill_(True)
        else:
            self.static_mask.copy_(mask, non_blocking=True)
        self.graph.replay()
        return self.static_out
