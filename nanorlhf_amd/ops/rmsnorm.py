"""RMSNorm forward/backward.

Replaces transformers' Qwen2RMSNorm used inside every reference model call
(SURVEY.md §2.2 row "RMSNorm, RoPE, SwiGLU").  GPU path is a fused HIP
kernel (one pass, bf16x8 vector loads, fp32 accumulation)."""
from __future__ import annotations

import torch

from . import ext


def _rms_norm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv).to(dtype) * weight


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        y, invrms = ext().rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms = ctx.saved_tensors
        dx, dw = ext().rmsnorm_bwd(dy.contiguous(), x, weight, invrms)
        return dx, dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        return _RMSNormFn.apply(x.contiguous(), weight, eps)
    return _rms_norm_ref(x, weight, eps)


class _AddRMSNormFn(torch.autograd.Function):
    """h = x + res; y = rmsnorm(h)*w — fused (one kernel, h register-resident
    for H<=2048).  Backward: dh = rmsnorm_bwd(dy, h) and the grads of x and
    res are both dh (+ whatever later consumers of h contributed)."""

    @staticmethod
    def forward(ctx, x, res, weight, eps):
        y, h, invrms = ext().rmsnorm_addres_fwd(x, res, weight, eps)
        ctx.save_for_backward(h, weight, invrms)
        return y, h

    @staticmethod
    def backward(ctx, dy, dh_out):
        h, weight, invrms = ctx.saved_tensors
        dh, dw = ext().rmsnorm_bwd(dy.contiguous(), h, weight, invrms)
        if dh_out is not None:
            dh = dh + dh_out
        return dh, dh, dw, None


def add_rms_norm(x: torch.Tensor, res: torch.Tensor | None, weight: torch.Tensor,
                 eps: float = 1e-6):
    """Fused residual-add + RMSNorm: returns (normed, h=x+res).  res=None →
    plain rms_norm (h = x)."""
    if res is None:
        return rms_norm(x, weight, eps), x
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _AddRMSNormFn.apply(x.contiguous(), res.contiguous(), weight, eps)
    h = x + res
    return rms_norm(h, weight, eps), h
