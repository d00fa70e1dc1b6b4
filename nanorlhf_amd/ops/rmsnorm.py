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
