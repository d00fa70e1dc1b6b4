"""Varlen causal flash attention (forward + backward) for training/scoring.

Replaces the reference's flash_attention_2 dependency (model load with
attn_implementation="flash_attention_2", GRPO/grpo.py:219,223).  Packed
(varlen) layout: q/k/v are [T_total, H, D] with cu_seqlens boundaries — no
pad tokens ever enter the kernel, which is the MI355X-first version of the
reference's de-padding/bucketing tricks (grpo_r1_trainer.py:571-579).

GPU kernels: MFMA 16x16x32 bf16 QK^T / PV with online softmax, LDS-staged
K/V tiles (XOR-swizzled, cdna_hip_programming.md §6 G4), fp32 accumulators;
backward recomputes P from saved (o, lse).  GQA handled by head mapping.
"""
from __future__ import annotations

import math

import torch

from . import ext


def _sdpa_ref(q, k, v, cu_seqlens, scale, causal=True):
    # q: [T, Hq, D], k/v: [T, Hkv, D]
    T, Hq, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    out = torch.zeros_like(q, dtype=torch.float32)
    for i in range(len(cu_seqlens) - 1):
        s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        if e <= s:
            continue
        qi = q[s:e].float().transpose(0, 1)              # [Hq, L, D]
        ki = k[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        vi = v[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        att = torch.matmul(qi, ki.transpose(-1, -2)) * scale
        if causal:
            L = e - s
            mask = torch.triu(torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1)
            att = att.masked_fill(mask, float("-inf"))
        att = torch.softmax(att, dim=-1)
        out[s:e] = torch.matmul(att, vi).transpose(0, 1)
    return out.to(q.dtype)


class _FlashAttnVarlenFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, cu_seqlens, max_seqlen, scale, causal):
        o, lse = ext().fa_fwd_varlen(q, k, v, cu_seqlens, int(max_seqlen), float(scale),
                                     bool(causal))
        ctx.save_for_backward(q, k, v, o, lse, cu_seqlens)
        ctx.scale = scale
        ctx.max_seqlen = max_seqlen
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse, cu_seqlens = ctx.saved_tensors
        dq, dk, dv = ext().fa_bwd_varlen(
            do.contiguous(), q, k, v, o, lse, cu_seqlens, int(ctx.max_seqlen),
            float(ctx.scale), bool(ctx.causal)
        )
        return dq, dk, dv, None, None, None, None


def flash_attn_varlen(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      cu_seqlens: torch.Tensor, max_seqlen: int,
                      scale: float | None = None, causal: bool = True) -> torch.Tensor:
    """Varlen attention.  q [T,Hq,D] bf16, k/v [T,Hkv,D] bf16,
    cu_seqlens [B+1] int32 on the same device."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _FlashAttnVarlenFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                                        cu_seqlens, max_seqlen, scale, causal)
    if q.requires_grad or k.requires_grad or v.requires_grad:
        return _sdpa_ref_autograd(q, k, v, cu_seqlens, scale, causal)
    return _sdpa_ref(q, k, v, cu_seqlens, scale, causal)


def _sdpa_ref_autograd(q, k, v, cu_seqlens, scale, causal=True):
    # same math as _sdpa_ref but differentiable (no in-place writes)
    T, Hq, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    outs = []
    for i in range(len(cu_seqlens) - 1):
        s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        qi = q[s:e].float().transpose(0, 1)
        ki = k[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        vi = v[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        att = torch.matmul(qi, ki.transpose(-1, -2)) * scale
        if causal:
            L = e - s
            mask = torch.triu(torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1)
            att = att.masked_fill(mask, float("-inf"))
        att = torch.softmax(att, dim=-1)
        outs.append(torch.matmul(att, vi).transpose(0, 1))
    return torch.cat(outs, dim=0).to(q.dtype)
