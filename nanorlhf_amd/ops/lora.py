"""Fused LoRA linear: y = x·Wᵀ + (α/r)·(x·Aᵀ)·Bᵀ + bias in ONE MFMA GEMM.

Replaces the 3-launch F.linear chain (round-1 models/lora.py:60-65;
reference peft call site GRPO/grpo.py:226-243).  The adapter contribution
rides as one extra K-tile of the base GEMM (csrc/lora.hip), so the fused
kernel does ≈(H+64)/H of the base GEMM's work in a single launch.

Backward (W frozen under LoRA — no dW GEMM at all):
    du = s·(dy·B)            [M,r]   (skinny, hipBLASLt)
    dx = dy·W + du·A         = the SAME kernel on (dy, Wᵀ) + (du, Aᵀ)
    dA = duᵀ·x               [r,H]   (skinny, hipBLASLt)
    dB = dyᵀ·u               [N,r]   (skinny, hipBLASLt)
Wᵀ is cached once per linear (W never changes while LoRA trains).

Constraints for the fused path: K % 64 == 0, rank == 64 (the reference
default, grpo.py:92); anything else falls back to the chained form.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import ext


def lora_gemm_ref(x, w, u=None, b=None, bias=None):
    """CPU/fp32 oracle of csrc/lora.hip::lora_gemm."""
    y = x.float() @ w.float().t()
    if u is not None:
        y = y + u.float() @ b.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


# Per-shape strategy cache: (N, K) -> "fused" | "epilogue".
# "fused"    = the single wide-K MFMA kernel (wins when our GEMM rate is
#              within the fusion savings of hipBLASLt — e.g. N-heavy shapes);
# "epilogue" = hipBLASLt base GEMM + the custom rank-64 lora_add_ MFMA pass
#              (wins on deep-K shapes where the library GEMM is far ahead).
# Measured once per weight shape on first touch (events, ~1 ms).
_TUNE: dict[tuple, str] = {}


def _time_cuda(fn, reps: int = 3) -> float:
    fn()  # warm
    s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    e.synchronize()
    return s.elapsed_time(e) / reps


def _choose_strategy(x, w, u, b, bias) -> str:
    key = (int(w.size(0)), int(w.size(1)))
    st = _TUNE.get(key)
    if st is None:
        t_fused = _time_cuda(lambda: ext().lora_gemm(x, w, u, b, bias))

        def ep():
            y = torch.nn.functional.linear(x, w, bias)
            ext().lora_add_(y, u, b)
        t_ep = _time_cuda(ep)
        st = "fused" if t_fused <= t_ep else "epilogue"
        _TUNE[key] = st
    return st


def _dispatch_gemm(x, w, u, b, bias):
    """y = x·wᵀ + u·bᵀ (+bias) via the per-shape-tuned strategy."""
    st = _choose_strategy(x, w, u, b, bias)
    if st == "fused":
        return ext().lora_gemm(x, w, u, b, bias)
    y = torch.nn.functional.linear(x, w, bias)
    if not y.is_contiguous():
        y = y.contiguous()
    ext().lora_add_(y, u, b)
    return y


class _FusedLoRAFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, lora_A, lora_B, scaling, w_t_cache):
        # u = s·x·Aᵀ (skinny GEMM, r=64)
        u = (x @ lora_A.t()) * scaling
        u = u.to(x.dtype).contiguous()
        y = _dispatch_gemm(x.contiguous(), w, u, lora_B.contiguous(),
                           bias if bias is not None else None)
        ctx.save_for_backward(x, u, lora_A, lora_B, w_t_cache)
        ctx.scaling = scaling
        return y

    @staticmethod
    def backward(ctx, dy):
        x, u, lora_A, lora_B, w_t = ctx.saved_tensors
        scaling = ctx.scaling
        dy = dy.contiguous()
        # du = s·dy·B  [M, r]
        du = (dy @ lora_B) * scaling
        du = du.to(dy.dtype).contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            # dx = dy·W + du·A → same dispatch on (dy, Wᵀ) + (du, Aᵀ)
            a_t = lora_A.t().contiguous()
            dx = _dispatch_gemm(dy, w_t, du, a_t, None)
        dA = du.t() @ x if ctx.needs_input_grad[3] else None
        dB = dy.t() @ u if ctx.needs_input_grad[4] else None
        return dx, None, None, dA, dB, None, None


def fused_lora_linear(x: torch.Tensor, w: torch.Tensor,
                      bias: Optional[torch.Tensor],
                      lora_A: torch.Tensor, lora_B: torch.Tensor,
                      scaling: float, w_t_cache: torch.Tensor) -> torch.Tensor:
    """x: [T, K]; w: [N, K] frozen; lora_A: [r, K]; lora_B: [N, r]."""
    return _FusedLoRAFn.apply(x, w, bias, lora_A, lora_B, scaling, w_t_cache)


def fused_path_ok(w: torch.Tensor, r: int) -> bool:
    return (w.is_cuda and w.dtype == torch.bfloat16 and r == 64
            and w.size(1) % 64 == 0)
