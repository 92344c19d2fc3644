"""MLTC — the learned test-case classifier (the framework's flagship model).

A bidirectional transformer encoder over tokenized test code / assertion text
that predicts the reference study's taxonomy labels (RQs/taxonomy_test2.csv:1):

  * 19 test strategies   (RQ1_tests.csv rows)      — multi-label
  * 21 quality properties (tests_prop_rq3.csv:1)   — multi-label
  * 9 ML workflow stages  (RQ1_tests.csv columns)  — single-label
  * 4 test methods        (tests_methods.csv:2-5)  — single-label

MI355X-first design: all parameters bf16 (flat-packed by the trainer), GEMMs
through hipBLASLt (torch.matmul / F.linear on ROCm), and the non-GEMM hot ops
(LayerNorm, bias+GeLU, masked softmax) are hand-written gfx950 HIP kernels
(csrc/*.hip) behind tosem2021_amd.ops.
"""
from __future__ import annotations

import math
import os
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from tosem2021_amd import ops

# Label spaces (mirrors tosem2021_amd.extract.schema; duplicated sizes only)
N_STRATEGIES = 19
N_PROPERTIES = 21
N_STAGES = 9
N_METHODS = 4

HEAD_SIZES = {
    "strategy": N_STRATEGIES,   # multi-label (BCE)
    "property": N_PROPERTIES,   # multi-label (BCE)
    "stage": N_STAGES,          # single-label (CE)
    "method": N_METHODS,        # single-label (CE)
}
MULTILABEL_HEADS = ("strategy", "property")


@dataclass
class MLTCConfig:
    vocab_size: int = 32768
    d_model: int = 1024
    n_heads: int = 16
    n_layers: int = 12
    d_ff: int = 4096
    max_seq: int = 512
    dropout: float = 0.0
    layer_norm_eps: float = 1e-5
    heads: Dict[str, int] = field(default_factory=lambda: dict(HEAD_SIZES))

    @property
    def head_dim(self) -> int:
        assert self.d_model % self.n_heads == 0
        return self.d_model // self.n_heads


CONFIGS: Dict[str, MLTCConfig] = {
    # flagship: the bench.py / BASELINE config
    "mltc-base": MLTCConfig(),
    # small config for fast CPU tests / smoke
    "mltc-tiny": MLTCConfig(vocab_size=512, d_model=128, n_heads=4, n_layers=2,
                            d_ff=256, max_seq=64),
    # scale-up config (1.28B params): exercises the same kernels at d=2048
    # (LN template PKTS=4, 32 flash heads) — sized for 288 GB HBM3E
    "mltc-large": MLTCConfig(d_model=2048, n_heads=32, n_layers=24,
                             d_ff=8192),
}


class FusedLayerNorm(nn.Module):
    def __init__(self, d: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(d))
        self.bias = nn.Parameter(torch.zeros(d))
        self.eps = eps

    def forward(self, x):
        return ops.fused_layernorm(x, self.weight, self.bias, self.eps)


# A/B knob (TOSEM_FUSED_LINEAR=1): route the bias-carrying projections
# through ops.fused_linear — torch addmm forward (same epilogue GEMM) but
# backward computes db with the colsum bias_grad kernel (~20 us) instead of
# at::native::reduce (~58 us).  RE-MEASURED round 2 with the committed
# TunableOp table active: still a 13.5 ms/step REGRESSION (97.6 vs 84.1 ms
# back-to-back on one box) — the explicit dgrad/wgrad matmul layouts are
# not in the tuned table and dispatch worse than addmm's backward, exactly
# as in round 1.  Default stays OFF; the ~1.4 ms of at::native::reduce
# bias grads is the price of addmm's better GEMM dispatch.
_USE_FUSED_LINEAR = os.environ.get("TOSEM_FUSED_LINEAR", "0") == "1"
# TOSEM_WGRAD=1: weight grads of the big projections through the custom
# split-K MFMA wgrad kernel (csrc/wgrad_gemm.hip) + colsum bias grad;
# forward addmm and dgrad dispatch unchanged.
_USE_WGRAD = os.environ.get("TOSEM_WGRAD", "0") == "1"


def _linear(mod: nn.Linear, x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda and torch.is_grad_enabled():
        if _USE_WGRAD and ops.wgrad_linear_supported(
                mod.out_features, mod.in_features,
                x.numel() // x.shape[-1]):
            return ops.wgrad_linear(x, mod.weight, mod.bias)
        if _USE_FUSED_LINEAR and mod.bias is not None:
            return ops.fused_linear(x, mod.weight, mod.bias)
    return mod(x)


class Attention(nn.Module):
    """Multi-head bidirectional self-attention.

    QK^T and PV run as batched hipBLASLt GEMMs (MFMA); the scaled masked
    softmax between them is the fused gfx950 kernel (csrc/softmax.hip).
    """

    def __init__(self, cfg: MLTCConfig):
        super().__init__()
        self.n_heads = cfg.n_heads
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        self.qkv = nn.Linear(cfg.d_model, 3 * cfg.d_model, bias=True)
        self.proj = nn.Linear(cfg.d_model, cfg.d_model, bias=True)

    def forward(self, x, mask_bias: Optional[torch.Tensor]):
        B, L, D = x.shape
        if ops.flash_supported(self.head_dim, L):
            # MFMA flash kernels run straight on the packed [B, L, 3D]
            # projection output and write attention out (and, in backward,
            # every gradient) in the packed layouts — no repack kernels,
            # no [L, L] score tensor, O(L) attention memory
            out = ops.flash_attention_packed(_linear(self.qkv, x), self.n_heads,
                                             mask_bias, self.scale)
            return _linear(self.proj, out)
        # general-shape fallback: gather kernel to bmm-ready
        # [3, B, H, L, dh] (csrc/repack.hip), bmm + fused softmax
        q, k, v = ops.qkv_repack(self.qkv(x), self.n_heads)  # [B, H, L, dh]
        scores = torch.matmul(q, k.transpose(-1, -2))  # [B, H, L, L]
        p = ops.fused_softmax(scores, mask_bias, self.scale)
        out = torch.matmul(p, v)                       # [B, H, L, dh]
        return self.proj(ops.out_repack(out))


class FFN(nn.Module):
    """d_model -> d_ff (bias+GeLU fused into the gfx950 kernel) -> d_model."""

    def __init__(self, cfg: MLTCConfig):
        super().__init__()
        self.up = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.up_bias = nn.Parameter(torch.zeros(cfg.d_ff))
        self.down = nn.Linear(cfg.d_ff, cfg.d_model, bias=True)

    def forward(self, x):
        if x.is_cuda and not torch.is_grad_enabled():
            # inference: bias+GeLU fused into the up GEMM's epilogue
            B, L, D = x.shape
            a = ops.lt_linear_gelu_bias(x.reshape(-1, D).contiguous(),
                                        self.up.weight, self.up_bias)
            return self.down(a.view(B, L, -1))
        h = _linear(self.up, x)
        h = ops.fused_bias_gelu(h, self.up_bias)
        return _linear(self.down, h)


class EncoderBlock(nn.Module):
    """Pre-LN block carrying an un-added (stream, delta) residual pair.

    The residual add is fused into the next LayerNorm's first read
    (ops.fused_add_layernorm), so the block never launches a separate
    elementwise add: it receives the previous sublayer's (x, delta), and
    returns its own.
    """

    def __init__(self, cfg: MLTCConfig):
        super().__init__()
        self.ln1 = FusedLayerNorm(cfg.d_model, cfg.layer_norm_eps)
        self.attn = Attention(cfg)
        self.ln2 = FusedLayerNorm(cfg.d_model, cfg.layer_norm_eps)
        self.ffn = FFN(cfg)
        self.dropout = cfg.dropout

    def _drop(self, h):
        return F.dropout(h, self.dropout, self.training) if self.dropout else h

    def forward(self, x, delta, mask_bias):
        if delta is None:
            y1 = ops.fused_layernorm(x, self.ln1.weight, self.ln1.bias,
                                     self.ln1.eps)
            s1 = x
        else:
            y1, s1 = ops.fused_add_layernorm(x, delta, self.ln1.weight,
                                             self.ln1.bias, self.ln1.eps)
        h = self._drop(self.attn(y1, mask_bias))
        y2, s2 = ops.fused_add_layernorm(s1, h, self.ln2.weight,
                                         self.ln2.bias, self.ln2.eps)
        h2 = self._drop(self.ffn(y2))
        return s2, h2


class MLTC(nn.Module):
    def __init__(self, cfg: MLTCConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.pos_emb = nn.Embedding(cfg.max_seq, cfg.d_model)
        self.blocks = nn.ModuleList(EncoderBlock(cfg) for _ in range(cfg.n_layers))
        self.ln_f = FusedLayerNorm(cfg.d_model, cfg.layer_norm_eps)
        self.heads = nn.ModuleDict(
            {name: nn.Linear(cfg.d_model, n) for name, n in cfg.heads.items()}
        )
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, tokens: torch.Tensor,
                attn_mask: Optional[torch.Tensor] = None) -> Dict[str, torch.Tensor]:
        """tokens: [B, L] int64; attn_mask: [B, L] bool (True = valid)."""
        B, L = tokens.shape
        pos = torch.arange(L, device=tokens.device)
        x = self.tok_emb(tokens) + self.pos_emb(pos)[None, :, :]
        mask_bias = None
        if attn_mask is not None:
            mask_bias = torch.where(
                attn_mask, 0.0, -1e9
            ).to(torch.float32).contiguous()
        delta = None
        for blk in self.blocks:
            x, delta = blk(x, delta, mask_bias)
        if delta is None:
            x = self.ln_f(x)
        else:
            x, _ = ops.fused_add_layernorm(x, delta, self.ln_f.weight,
                                           self.ln_f.bias, self.ln_f.eps)
        # masked mean-pool over valid tokens (fused kernel: csrc/pool.hip)
        pooled = ops.masked_mean_pool(x, attn_mask)
        return {name: head(pooled) for name, head in self.heads.items()}

    def set_pos_weights(self, weights: Dict[str, torch.Tensor]) -> None:
        """Per-class positive weights for the multi-label heads (label
        frequencies in the taxonomy are heavily imbalanced)."""
        for name, w in weights.items():
            self.register_buffer(f"_pw_{name}", w.float(), persistent=False)

    def set_loss_options(self, focal_gamma: Dict[str, float] = None,
                         label_smoothing: float = 0.0) -> None:
        """Per-head focal-BCE gamma (multilabel heads) + label smoothing —
        round-2 levers for the rare-class property head.
        """
        self._focal_gamma = dict(focal_gamma or {})
        self._label_smoothing = float(label_smoothing)

    def loss(self, logits: Dict[str, torch.Tensor],
             labels: Dict[str, torch.Tensor]) -> torch.Tensor:
        total = None
        gammas = getattr(self, "_focal_gamma", {})
        smooth = getattr(self, "_label_smoothing", 0.0)
        for name, lg in logits.items():
            if name not in labels:
                continue
            lg = lg.float()
            if name in MULTILABEL_HEADS:
                pw = getattr(self, f"_pw_{name}", None)
                y = labels[name].float()
                if smooth > 0:
                    y = y * (1.0 - smooth) + 0.5 * smooth
                gamma = gammas.get(name, 0.0)
                if gamma > 0:
                    # focal BCE: down-weight easy negatives/positives
                    bce = F.binary_cross_entropy_with_logits(
                        lg, y, pos_weight=pw, reduction="none")
                    p = torch.sigmoid(lg)
                    p_t = p * y + (1 - p) * (1 - y)
                    li = ((1 - p_t).clamp(min=1e-4) ** gamma * bce).mean()
                else:
                    li = F.binary_cross_entropy_with_logits(
                        lg, y, pos_weight=pw)
            else:
                li = F.cross_entropy(lg, labels[name],
                                     label_smoothing=smooth)
            total = li if total is None else total + li
        assert total is not None, "no labels matched any head"
        return total


def build_model(name_or_cfg, dtype=torch.bfloat16) -> MLTC:
    cfg = CONFIGS[name_or_cfg] if isinstance(name_or_cfg, str) else name_or_cfg
    model = MLTC(cfg)
    return model.to(dtype)
