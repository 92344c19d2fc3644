"""Trainer for the MLTC classifier: flat bf16 parameters, fused AdamW,
bucketed-allreduce DP, checkpoint/resume, metrics and tracing.

Memory layout (MI355X-first): every parameter is a view into ONE contiguous
bf16 buffer; gradients accumulate into a matching flat bf16 buffer; the
optimizer keeps flat f32 master/m/v.  One fused kernel performs the whole
AdamW step (csrc/adamw.hip); DP reduces the flat grad buffer in large
buckets over RCCL/xGMI (parallel/ddp.py).

Capability parity: checkpoint/resume mirrors the corpus patterns the study
measured (DeepSpeech util/checkpoints.py:126,140; nni recoverable.py) —
atomic save, load-or-init, resume from step.
"""
from __future__ import annotations

import math
import os
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from tosem2021_amd import ops
from tosem2021_amd.models import MLTCConfig, build_model
from tosem2021_amd.parallel.ddp import BucketedAllReduce
from tosem2021_amd.utils.metrics import get_metrics
from tosem2021_amd.utils.trace import trace

PAD_ELEMS = 512  # flat buffers padded so the fused AdamW's 4-wide loop is exact


class FlatParams:
    """Flatten a model's parameters into contiguous training state."""

    def __init__(self, model: torch.nn.Module):
        params = [p for p in model.parameters() if p.requires_grad]
        total = sum(p.numel() for p in params)
        padded = (total + PAD_ELEMS - 1) // PAD_ELEMS * PAD_ELEMS
        device = params[0].device
        dtype = params[0].dtype
        self.flat = torch.zeros(padded, dtype=dtype, device=device)
        self.grad_flat = torch.zeros(padded, dtype=dtype, device=device)
        self.offsets: List[int] = []
        off = 0
        for p in params:
            n = p.numel()
            self.flat[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat[off:off + n].view(p.shape)
            p.grad = self.grad_flat[off:off + n].view(p.shape)
            self.offsets.append(off)
            off += n
        self.params = params
        self.numel = total
        self.padded = padded
        # f32 optimizer state
        self.master = self.flat.detach().clone().float()
        self.m = torch.zeros(padded, dtype=torch.float32, device=device)
        self.v = torch.zeros(padded, dtype=torch.float32, device=device)

    def zero_grad(self):
        self.grad_flat.zero_()


@dataclass
class TrainConfig:
    model: str = "mltc-base"
    lr: float = 3e-4
    beta1: float = 0.9
    beta2: float = 0.999
    eps: float = 1e-8
    weight_decay: float = 0.01
    warmup_steps: int = 100
    total_steps: int = 10000
    bucket_mb: int = 64
    dtype: str = "bf16"  # "bf16" | "f32" (f32: CPU/gloo tests)
    ckpt_dir: Optional[str] = None
    ckpt_every: int = 0          # 0 = only on explicit save()
    ckpt_keep: int = 3           # retain the newest k checkpoints (0 = all)


class Trainer:
    def __init__(self, cfg: TrainConfig, device: Optional[torch.device] = None,
                 model_cfg: Optional[MLTCConfig] = None):
        self.cfg = cfg
        self.device = device or (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
        dtype = torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
        model = build_model(model_cfg if model_cfg is not None else cfg.model,
                            dtype=dtype)
        self.model = model.to(self.device)
        self.flat = FlatParams(self.model)
        self.ddp = BucketedAllReduce(self.flat.params, self.flat.grad_flat,
                                     self.flat.offsets,
                                     bucket_bytes=cfg.bucket_mb << 20)
        if self.ddp.enabled:
            # rank-0 init everywhere: broadcast the flat params once
            dist.broadcast(self.flat.flat, src=0)
            self.flat.master.copy_(self.flat.flat.float())
        self.step_num = 0
        self.metrics: Dict[str, float] = {}

    # ---- schedule -----------------------------------------------------------
    def _lr(self) -> float:
        s, c = self.step_num, self.cfg
        if s < c.warmup_steps:
            return c.lr * (s + 1) / max(c.warmup_steps, 1)
        t = (s - c.warmup_steps) / max(c.total_steps - c.warmup_steps, 1)
        return c.lr * 0.5 * (1.0 + math.cos(math.pi * min(t, 1.0)))

    # ---- one optimization step ----------------------------------------------
    def step_accum(self, micro_batches) -> float:
        """One optimizer step over several micro-batches (gradient
        accumulation): collectives fire only on the last micro-backward,
        and the 1/n_micro mean folds into the AdamW grad scale."""
        micro_batches = list(micro_batches)
        n = len(micro_batches)
        with trace("train_step_accum", step=self.step_num + 1, micro=n):
            self.flat.zero_grad()
            total_loss = 0.0
            for i, (tokens, attn_mask, labels) in enumerate(micro_batches):
                self.ddp.sync = (i == n - 1)
                logits = self.model(tokens, attn_mask)
                loss = self.model.loss(logits, labels)
                loss.backward()
                total_loss += float(loss.detach())
            self.ddp.sync = True
            self.ddp.finalize()
            self.step_num += 1
            lr = self._lr()
            ops.adamw_step(
                self.flat.flat, self.flat.grad_flat, self.flat.m, self.flat.v,
                self.flat.master, lr=lr, beta1=self.cfg.beta1,
                beta2=self.cfg.beta2, eps=self.cfg.eps,
                wd=self.cfg.weight_decay, step=self.step_num,
                grad_scale=self.ddp.grad_scale / n)
        get_metrics().observe_step(self.step_num, loss=total_loss / n, lr=lr)
        return total_loss / n

    def step(self, tokens: torch.Tensor, attn_mask: Optional[torch.Tensor],
             labels: Dict[str, torch.Tensor]) -> float:
        with trace("train_step", step=self.step_num + 1):
            self.flat.zero_grad()
            logits = self.model(tokens, attn_mask)
            loss = self.model.loss(logits, labels)
            loss.backward()
            self.ddp.finalize()
            self.step_num += 1
            lr = self._lr()
            ops.adamw_step(
                self.flat.flat, self.flat.grad_flat, self.flat.m, self.flat.v,
                self.flat.master, lr=lr, beta1=self.cfg.beta1,
                beta2=self.cfg.beta2, eps=self.cfg.eps, wd=self.cfg.weight_decay,
                step=self.step_num, grad_scale=self.ddp.grad_scale)
        get_metrics().observe_step(self.step_num, loss=float(loss.detach()),
                                   lr=lr)
        if self.cfg.ckpt_every and self.cfg.ckpt_dir and \
                self.step_num % self.cfg.ckpt_every == 0:
            self.save()
        return float(loss.detach())

    # ---- checkpoint / resume -------------------------------------------------
    def state_dict(self) -> dict:
        return {
            "step": self.step_num,
            "flat": self.flat.flat,
            "master": self.flat.master,
            "m": self.flat.m,
            "v": self.flat.v,
            "model": self.cfg.model,
        }

    def save(self, path: Optional[str] = None) -> str:
        assert path or self.cfg.ckpt_dir, "no checkpoint path configured"
        if path is None:
            path = os.path.join(self.cfg.ckpt_dir, f"ckpt_{self.step_num:08d}.pt")
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        tmp = path + ".tmp"
        torch.save(self.state_dict(), tmp)
        os.replace(tmp, path)  # atomic publish (DeepSpeech-style durable ckpt)
        self._prune(os.path.dirname(path))
        return path

    def _prune(self, ckpt_dir: str):
        """Keep the newest cfg.ckpt_keep checkpoints (tune checkpoint_manager
        style retention)."""
        k = self.cfg.ckpt_keep
        if not k or not ckpt_dir:
            return
        cks = sorted(f for f in os.listdir(ckpt_dir)
                     if f.startswith("ckpt_") and f.endswith(".pt"))
        for f in cks[:-k]:
            try:
                os.remove(os.path.join(ckpt_dir, f))
            except OSError:
                pass

    def load(self, path: str):
        sd = torch.load(path, map_location=self.device, weights_only=True)
        self.flat.flat.copy_(sd["flat"])
        self.flat.master.copy_(sd["master"])
        self.flat.m.copy_(sd["m"])
        self.flat.v.copy_(sd["v"])
        self.step_num = sd["step"]

    @staticmethod
    def latest_checkpoint(ckpt_dir: str) -> Optional[str]:
        if not os.path.isdir(ckpt_dir):
            return None
        cks = sorted(f for f in os.listdir(ckpt_dir)
                     if f.startswith("ckpt_") and f.endswith(".pt"))
        return os.path.join(ckpt_dir, cks[-1]) if cks else None

    def load_or_init(self) -> bool:
        """nni-recoverable-style resume: load the newest checkpoint if any."""
        if not self.cfg.ckpt_dir:
            return False
        path = self.latest_checkpoint(self.cfg.ckpt_dir)
        if path:
            self.load(path)
            return True
        return False
