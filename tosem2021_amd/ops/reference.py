"""Plain-PyTorch fp32 reference implementations of the fused gfx950 ops.

These are the numerical ground truth for the HIP kernels (tests compare the
kernel output on GPU against these run in fp32) and the CPU execution path of
the model, so the full framework runs — slowly — on a GPU-less box.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def layernorm_fwd(
    x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (y in x.dtype, mean f32, rstd f32); stats over the last dim."""
    xf = x.float()
    mean = xf.mean(dim=-1)
    var = xf.var(dim=-1, unbiased=False)
    rstd = torch.rsqrt(var + eps)
    xhat = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    y = xhat * gamma.float() + beta.float()
    return y.to(x.dtype), mean.reshape(-1), rstd.reshape(-1)


def add_layernorm_fwd(x, residual, gamma, beta, eps):
    """s = x + residual (rounded to x.dtype); y = LN(s). Returns (y, s, mean, rstd)."""
    s = (x.float() + residual.float()).to(x.dtype)
    y, mean, rstd = layernorm_fwd(s, gamma, beta, eps)
    return y, s, mean, rstd


def layernorm_bwd(
    dy: torch.Tensor,
    x: torch.Tensor,
    gamma: torch.Tensor,
    mean: torch.Tensor,
    rstd: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    D = x.shape[-1]
    xf = x.float().reshape(-1, D)
    dyf = dy.float().reshape(-1, D)
    mean = mean.reshape(-1, 1)
    rstd = rstd.reshape(-1, 1)
    xhat = (xf - mean) * rstd
    dyg = dyf * gamma.float()
    c1 = dyg.mean(dim=-1, keepdim=True)
    c2 = (dyg * xhat).mean(dim=-1, keepdim=True)
    dx = rstd * (dyg - c1 - xhat * c2)
    dgamma = (dyf * xhat).sum(dim=0)
    dbeta = dyf.sum(dim=0)
    return dx.reshape(x.shape).to(x.dtype), dgamma, dbeta


def bias_gelu_fwd(x: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    pre = x.float() + b.float()
    return torch.nn.functional.gelu(pre, approximate="tanh").to(x.dtype)


def bias_gelu_bwd(
    dy: torch.Tensor, x: torch.Tensor, b: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    pre = (x.float() + b.float()).requires_grad_(True)
    with torch.enable_grad():
        y = torch.nn.functional.gelu(pre, approximate="tanh")
    (dpre,) = torch.autograd.grad(y, pre, dy.float())
    dbias = dpre.reshape(-1, x.shape[-1]).sum(dim=0)
    return dpre.to(x.dtype), dbias


def softmax_fwd(
    scores: torch.Tensor, mask: Optional[torch.Tensor], scale: float
) -> torch.Tensor:
    s = scores.float() * scale
    if mask is not None:
        # mask: [B, Lk] additive bias broadcast over heads and query positions
        s = s + mask.float().view(mask.shape[0], 1, 1, mask.shape[-1])
    return torch.softmax(s, dim=-1).to(scores.dtype)


def softmax_bwd(dp: torch.Tensor, p: torch.Tensor, scale: float) -> torch.Tensor:
    dpf = dp.float()
    pf = p.float()
    dot = (dpf * pf).sum(dim=-1, keepdim=True)
    return (scale * pf * (dpf - dot)).to(p.dtype)


def flash_attention_fwd(q, k, v, mask, scale):
    """fp32 reference of the flash forward: returns (o in q.dtype, lse f32)."""
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if mask is not None:
        s = s + mask.float().view(mask.shape[0], 1, 1, -1)
    lse = torch.logsumexp(s, dim=-1)
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, v.float())
    return o.to(q.dtype), lse


def p_from_lse(scores, mask, lse, scale):
    s = scores.float() * scale
    if mask is not None:
        s = s + mask.float().view(mask.shape[0], 1, 1, -1)
    return torch.exp(s - lse.unsqueeze(-1)).to(scores.dtype)


def adamw_step(
    p: torch.Tensor,
    grad: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    master: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    wd: float,
    step: int,
    grad_scale: float = 1.0,
) -> None:
    g = grad.float() * grad_scale
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1.0 / (1.0 - beta1**step)
    bc2 = 1.0 / (1.0 - beta2**step)
    mhat = m * bc1
    vhat = v * bc2
    master.add_(-(lr * (mhat / (vhat.sqrt() + eps) + wd * master)))
    p.copy_(master.to(p.dtype))
