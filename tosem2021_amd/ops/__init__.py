"""Fused gfx950 ops: autograd wrappers dispatching to the in-tree HIP
extension on GPU and to the fp32 torch references on CPU.

Policy: on a GPU box the HIP extension is REQUIRED — a missing extension
raises instead of silently falling back to eager PyTorch, so a GPU test can
never pass on a non-native path.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import reference

_EXT = None
_EXT_ERR: Optional[str] = None


def hip_ops():
    """Return the compiled HIP extension module; raise loudly if missing."""
    global _EXT, _EXT_ERR
    if _EXT is None:
        try:
            from tosem2021_amd import _hip_ops  # type: ignore

            _EXT = _hip_ops
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = str(e)
            raise RuntimeError(
                "tosem2021_amd._hip_ops is not built. On a GPU box this is a "
                "hard error (no eager fallback). Build it in-tree with: "
                "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace "
                f"(import error: {e})"
            ) from e
    return _EXT


def hip_available() -> bool:
    try:
        hip_ops()
        return True
    except RuntimeError:
        return False


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        if x.is_cuda:
            y, _, mean, rstd = hip_ops().layernorm_fwd(x, None, gamma, beta, eps)
        else:
            y, mean, rstd = reference.layernorm_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            dx, dgamma, dbeta = hip_ops().layernorm_bwd(dy, x, gamma, mean,
                                                        rstd, None)
        else:
            dx, dgamma, dbeta = reference.layernorm_bwd(dy, x, gamma, mean, rstd)
        return dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None


def fused_layernorm(x, gamma, beta, eps: float = 1e-5):
    return _LayerNormFn.apply(x.contiguous(), gamma, beta, eps)


class _AddLayerNormFn(torch.autograd.Function):
    """(y, s) = (LN(x + residual), x + residual) — the residual-stream add is
    fused into the LN kernel's first read (one kernel, no separate add)."""

    @staticmethod
    def forward(ctx, x, residual, gamma, beta, eps):
        if x.is_cuda:
            y, s, mean, rstd = hip_ops().layernorm_fwd(x, residual, gamma,
                                                       beta, eps)
        else:
            y, s, mean, rstd = reference.add_layernorm_fwd(x, residual, gamma,
                                                           beta, eps)
        ctx.save_for_backward(s, gamma, mean, rstd)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, gamma, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if s.is_cuda:
            # the downstream residual grad ds is added INSIDE the kernel
            de = ds.contiguous() if ds is not None else None
            dx, dgamma, dbeta = hip_ops().layernorm_bwd(dy, s, gamma, mean,
                                                        rstd, de)
        else:
            dx, dgamma, dbeta = reference.layernorm_bwd(dy, s, gamma, mean, rstd)
            if ds is not None:
                dx = dx + ds
        return dx, dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None


def fused_add_layernorm(x, residual, gamma, beta, eps: float = 1e-5):
    """Returns (y, s): y = LN(x+residual), s = the new residual stream."""
    return _AddLayerNormFn.apply(x.contiguous(), residual.contiguous(),
                                 gamma, beta, eps)


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, b):
        ctx.save_for_backward(x, b)
        if x.is_cuda:
            return hip_ops().bias_gelu_fwd(x, b)
        return reference.bias_gelu_fwd(x, b)

    @staticmethod
    def backward(ctx, dy):
        x, b = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            dx, dbias = hip_ops().bias_gelu_bwd(dy, x, b)
        else:
            dx, dbias = reference.bias_gelu_bwd(dy, x, b)
        return dx, dbias.to(b.dtype)


def fused_bias_gelu(x, b):
    """y = gelu_tanh(x + b), bf16, bias grad fused."""
    return _BiasGeluFn.apply(x.contiguous(), b)


class _SoftmaxFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, mask, scale):
        if scores.is_cuda:
            p = hip_ops().softmax_fwd(scores, mask, scale)
        else:
            p = reference.softmax_fwd(scores, mask, scale)
        ctx.save_for_backward(p)
        ctx.scale = scale
        return p

    @staticmethod
    def backward(ctx, dp):
        (p,) = ctx.saved_tensors
        dp = dp.contiguous()
        if p.is_cuda:
            ds = hip_ops().softmax_bwd(dp, p, ctx.scale)
        else:
            ds = reference.softmax_bwd(dp, p, ctx.scale)
        return ds, None, None


def fused_softmax(scores, mask=None, scale: float = 1.0):
    """P = softmax(scale * scores + mask_bias) over the last dim.

    scores: [B, H, Lq, Lk] bf16; mask: optional [B, Lk] f32 additive bias.
    """
    return _SoftmaxFn.apply(scores.contiguous(), mask, scale)


class _QkvRepackFn(torch.autograd.Function):
    """[B, L, 3*H*dh] -> three contiguous [B, H, L, dh] (bmm-ready q/k/v).

    Backward gathers (dq, dk, dv) straight back to the Linear's layout in one
    kernel — no stack/cat materialization."""

    @staticmethod
    def forward(ctx, qkv, n_heads):
        ctx.n_heads = n_heads
        if qkv.is_cuda:
            stacked = hip_ops().qkv_repack(qkv, n_heads, False)
        else:
            B, L, _ = qkv.shape
            dh = qkv.shape[-1] // (3 * n_heads)
            stacked = (qkv.view(B, L, 3, n_heads, dh).permute(2, 0, 3, 1, 4)
                       .contiguous())
        q, k, v = stacked.unbind(0)
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        dq = dq.contiguous(); dk = dk.contiguous(); dv = dv.contiguous()
        if dq.is_cuda:
            return hip_ops().qkv_repack_bwd3(dq, dk, dv), None
        B, H, L, dh = dq.shape
        g = torch.stack([dq, dk, dv], dim=0)
        return g.permute(1, 3, 0, 2, 4).reshape(B, L, 3 * H * dh), None


def qkv_repack(qkv, n_heads: int):
    return _QkvRepackFn.apply(qkv.contiguous(), n_heads)


class _OutRepackFn(torch.autograd.Function):
    """[B, H, L, dh] -> [B, L, H*dh] (attention output merge)."""

    @staticmethod
    def forward(ctx, x):
        ctx.n_heads = x.shape[1]
        if x.is_cuda:
            return hip_ops().out_repack(x, False)
        B, H, L, dh = x.shape
        return x.permute(0, 2, 1, 3).reshape(B, L, H * dh)

    @staticmethod
    def backward(ctx, g):
        g = g.contiguous()
        if g.is_cuda:
            return hip_ops().out_repack_bwd(g, ctx.n_heads)
        B, L, D = g.shape
        H = ctx.n_heads
        return g.view(B, L, H, D // H).permute(0, 2, 1, 3).contiguous()


def out_repack(x):
    return _OutRepackFn.apply(x.contiguous())


class _LinearFn(torch.autograd.Function):
    """F.linear with the bias gradient computed by the two-stage bf16
    column-sum kernel (csrc/layernorm.hip colsum_bf16).  MEASURED NOTE: not
    used by the model — routing the projections through this Function cost
    13.7 ms/step because the explicit dgrad/wgrad matmuls here dispatch to
    slower hipBLASLt algos than torch's addmm backward (which reaches the
    tuned split-K Custom_*SK3 kernels); the ~38 us/call bias-grad saving
    cannot pay for that.  Kept as a standalone op (serving/other shapes)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return torch.addmm(b, x, w.t())

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.matmul(dy, w)
        dw = torch.matmul(dy.transpose(0, 1), x)
        db = hip_ops().bias_grad(dy) if dy.is_cuda else dy.sum(0)
        return dx, dw, db


def fused_linear(x, w, b):
    """x @ w^T + b with the custom bias-grad path (CUDA bf16 only)."""
    lead = x.shape[:-1]
    y = _LinearFn.apply(x.reshape(-1, x.shape[-1]).contiguous(), w, b)
    return y.view(*lead, -1)


class _LinearWgradFn(torch.autograd.Function):
    """Linear whose WEIGHT gradient runs through the custom split-K MFMA
    wgrad kernel (csrc/wgrad_gemm.hip) and the bias gradient through the
    colsum bias_grad kernel; forward is the same fused addmm GEMM, dgrad
    the same torch matmul dispatch."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        if b is None:
            return torch.matmul(x, w.t())
        return torch.addmm(b, x, w.t())

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.matmul(dy, w)
        dw = hip_ops().wgrad_gemm(dy, x, 0)
        db = hip_ops().bias_grad(dy) if ctx.has_bias else None
        return dx, dw, db


def wgrad_linear_supported(out_features: int, in_features: int,
                           tokens: int) -> bool:
    return out_features % 256 == 0 and in_features % 256 == 0 and \
        tokens % 32 == 0


def wgrad_linear(x, w, b=None):
    """Linear with the custom wgrad kernel (bf16 CUDA; 256-multiple
    features, token count % 32)."""
    lead = x.shape[:-1]
    y = _LinearWgradFn.apply(x.reshape(-1, x.shape[-1]).contiguous(), w, b)
    return y.view(*lead, -1)


def lt_linear_gelu_bias(x, w1, b1):
    """GELU(x @ w1^T + b1) in one hipBLASLt GEMM (GELU_BIAS epilogue) —
    inference only: this hipBLASLt has no aux epilogues (no pre-activation
    out), so training uses csrc/bias_gelu.hip instead."""
    return hip_ops().lt_linear_gelu_bias(x, w1, b1)


class _FlashAttentionFn(torch.autograd.Function):
    """Flash attention: MFMA forward (csrc/flash_attn.hip, O(L) memory, saves
    logsumexp); backward = two recompute kernels, O(L) memory end to end:
    flash_bwd_fused (register-accumulated dK/dV) + flash_dq_recompute
    (in-register S/dP/dS, accumulates dQ) — no [L, L] dS materialization."""

    @staticmethod
    def forward(ctx, q, k, v, mask, scale):
        if q.is_cuda:
            o, lse = hip_ops().flash_fwd(q, k, v, mask, scale)
        else:
            o, lse = reference.flash_attention_fwd(q, k, v, mask, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.mask = mask
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        mask, scale = ctx.mask, ctx.scale
        do = do.contiguous()
        if q.is_cuda:
            # two recompute MFMA kernels; dS never touches HBM (round 2 —
            # the round-1 pipeline wrote + re-read a [B, H, L, L] bf16 dS)
            ddot = hip_ops().fa_dot(do, o)
            dk, dv = hip_ops().flash_bwd_fused(q, k, v, do, mask, lse,
                                               ddot, scale)
            dq = hip_ops().flash_dq_recompute(q, k, v, do, mask, lse,
                                              ddot, scale)
            return dq, dk, dv, None, None
        s = torch.matmul(q, k.transpose(-1, -2))
        p = reference.p_from_lse(s, mask, lse, scale)
        dp = torch.matmul(do, v.transpose(-1, -2))
        dsc = reference.softmax_bwd(dp, p, scale)
        dq = torch.matmul(dsc, k)
        dk = torch.matmul(dsc.transpose(-1, -2), q)
        dv = torch.matmul(p.transpose(-1, -2), do)
        return dq, dk, dv, None, None


def flash_attention(q, k, v, mask=None, scale: float = 1.0):
    """q/k/v: [B, H, L, 64] bf16 contiguous, L % 32 == 0; mask: [B, L] f32."""
    return _FlashAttentionFn.apply(q.contiguous(), k.contiguous(),
                                   v.contiguous(), mask, scale)


class _FlashAttentionPackedFn(torch.autograd.Function):
    """Flash attention straight on the packed QKV projection output
    [B, L, 3D] -> attention output [B, L, D] (round 2).  The strided-
    geometry kernels read q/k/v from and write every gradient back into
    the packed layouts, eliminating the qkv_repack / out_repack kernel
    quartet (fwd gather, bwd3 merge, out fwd/bwd) from the step."""

    @staticmethod
    def forward(ctx, qkv, n_heads, mask, scale):
        if qkv.is_cuda:
            o, lse = hip_ops().flash_fwd_packed(qkv, n_heads, mask, scale)
        else:
            B, L, D3 = qkv.shape
            d = D3 // 3
            q, k, v = (t.view(B, L, n_heads, 64).transpose(1, 2).contiguous()
                       for t in qkv.split(d, dim=-1))
            o4, lse = reference.flash_attention_fwd(q, k, v, mask, scale)
            o = o4.transpose(1, 2).reshape(B, L, d)
        ctx.save_for_backward(qkv, o, lse)
        ctx.n_heads = n_heads
        ctx.mask = mask
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        n_heads, mask, scale = ctx.n_heads, ctx.mask, ctx.scale
        do = do.contiguous()
        if qkv.is_cuda:
            dqkv = hip_ops().flash_bwd_packed(qkv, o, do, mask, lse,
                                              n_heads, scale)
            return dqkv, None, None, None
        B, L, D3 = qkv.shape
        d = D3 // 3
        q, k, v = (t.view(B, L, n_heads, 64).transpose(1, 2).contiguous()
                   for t in qkv.split(d, dim=-1))
        do4 = do.view(B, L, n_heads, 64).transpose(1, 2).contiguous()
        s = torch.matmul(q, k.transpose(-1, -2))
        p = reference.p_from_lse(s, mask, lse, scale)
        dp = torch.matmul(do4, v.transpose(-1, -2))
        dsc = reference.softmax_bwd(dp, p, scale)
        dq = torch.matmul(dsc, k)
        dk = torch.matmul(dsc.transpose(-1, -2), q)
        dv = torch.matmul(p.transpose(-1, -2), do4)
        dqkv = torch.cat(
            [t.transpose(1, 2).reshape(B, L, d) for t in (dq, dk, dv)],
            dim=-1)
        return dqkv, None, None, None


def flash_attention_packed(qkv, n_heads: int, mask=None, scale: float = 1.0):
    """qkv: [B, L, 3*n_heads*64] bf16 (the fused projection output);
    returns [B, L, n_heads*64] ready for the output projection."""
    return _FlashAttentionPackedFn.apply(qkv.contiguous(), n_heads, mask,
                                         scale)


def flash_supported(head_dim: int, L: int) -> bool:
    return head_dim == 64 and L % 32 == 0


class _MaskedPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, mask):
        if x.is_cuda:
            pooled, counts = hip_ops().masked_pool_fwd(x, mask)
        else:
            if mask is not None:
                w = mask.to(x.dtype).unsqueeze(-1)
                counts = mask.sum(-1).clamp(min=1).float()
                pooled = ((x.float() * w.float()).sum(1) /
                          counts.unsqueeze(-1)).to(x.dtype)
            else:
                counts = torch.full((x.shape[0],), x.shape[1],
                                    dtype=torch.float32)
                pooled = x.float().mean(1).to(x.dtype)
        ctx.save_for_backward(counts)
        ctx.mask = mask
        ctx.L = x.shape[1]
        return pooled

    @staticmethod
    def backward(ctx, dpooled):
        (counts,) = ctx.saved_tensors
        mask = ctx.mask
        dpooled = dpooled.contiguous()
        if dpooled.is_cuda:
            dx = hip_ops().masked_pool_bwd(dpooled, mask, counts, ctx.L)
        else:
            g = (dpooled.float() / counts.unsqueeze(-1)).unsqueeze(1)
            dx = g.expand(-1, ctx.L, -1)
            if mask is not None:
                dx = dx * mask.unsqueeze(-1).float()
            dx = dx.to(dpooled.dtype)
        return dx, None


def masked_mean_pool(x, mask=None):
    """pooled[b] = mean over valid rows of x[b]; x [B,L,D] bf16, mask [B,L]."""
    return _MaskedPoolFn.apply(x.contiguous(),
                               mask.contiguous() if mask is not None else None)


def adamw_step(p, grad, m, v, master, *, lr, beta1=0.9, beta2=0.999, eps=1e-8,
               wd=0.01, step, grad_scale=1.0):
    """Fused AdamW over the flat parameter buffer (see train.py)."""
    if p.is_cuda:
        hip_ops().adamw_step(p, grad, m, v, master, lr, beta1, beta2, eps, wd,
                             step, grad_scale)
    else:
        reference.adamw_step(p, grad, m, v, master, lr, beta1, beta2, eps, wd,
                             step, grad_scale)
