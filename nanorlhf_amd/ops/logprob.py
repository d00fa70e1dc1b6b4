"""Fused token-logprob + entropy over the vocab — the reference's memory
pain point (full [mb, T, 151k] log_softmax at GRPO/grpo_trainer.py:548-549,
653-656, entropy :678-679).

Design: never materialize log-softmax.  The caller supplies hidden states
and the lm_head weight; we chunk rows, run the plain GEMM (rocBLAS/hipBLASLt
— a library GEMM, per the MI355X design rules), then a hand-written HIP
row-reduction computes logprob-of-label + entropy + logsumexp in one pass
with fp32 accumulation (bf16 logits over 151k vocab need fp32 lse — SURVEY
§7 hard part (c)).  Backward recomputes each chunk's logits and forms
dlogits = g * (softmax - onehot) in-kernel, then chunk GEMMs for
dhidden / dweight.  Peak extra memory = one chunk of logits.

Reference quirk preserved: scoring divides logits by (temperature + 1e-7)
(grpo_trainer.py:547) — pass `temperature` for that behavior.
"""
from __future__ import annotations

import torch

from . import ext

_DEF_CHUNK = 16384


def _ref_rowstats(logits: torch.Tensor, labels: torch.Tensor):
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    lp = lf.gather(-1, labels.unsqueeze(-1)).squeeze(-1) - lse
    p = torch.softmax(lf, dim=-1)
    ent = lse - (p * lf).sum(-1)
    return lp, ent, lse


class _TokenLogprobFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, hidden, weight, labels, inv_temp, chunk):
        # hidden [N, H] bf16, weight [V, H] bf16, labels [N] long
        N = hidden.shape[0]
        lp = torch.empty(N, dtype=torch.float32, device=hidden.device)
        ent = torch.empty_like(lp)
        lse = torch.empty_like(lp)
        for s in range(0, N, chunk):
            e = min(s + chunk, N)
            logits = torch.mm(hidden[s:e], weight.t())  # bf16 GEMM, [n, V]
            ext().ce_rowstats(logits, labels[s:e], float(inv_temp),
                              lp[s:e], ent[s:e], lse[s:e])
        ctx.save_for_backward(hidden, weight, labels, lse)
        ctx.inv_temp = inv_temp
        ctx.chunk = chunk
        return lp, ent

    @staticmethod
    def backward(ctx, g_lp, g_ent):
        hidden, weight, labels, lse = ctx.saved_tensors
        if g_ent is not None and g_ent.abs().sum() > 0:
            raise RuntimeError("entropy output of token_logprob_entropy is not differentiable")
        N = hidden.shape[0]
        dh = torch.zeros_like(hidden)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        chunk = ctx.chunk
        for s in range(0, N, chunk):
            e = min(s + chunk, N)
            logits = torch.mm(hidden[s:e], weight.t())
            # in-place: logits <- bf16( g * (softmax(logits*inv_t) - onehot) * inv_t )
            ext().ce_backward_dlogits(logits, labels[s:e], lse[s:e],
                                      g_lp[s:e].contiguous().float(), float(ctx.inv_temp))
            dh[s:e] = torch.mm(logits, weight)
            # bf16 GEMM (rocBLAS accumulates fp32 internally) + fp32 running
            # sum: an explicit fp32 GEMM here ran at the 157 TF f32 ceiling
            # (12.5 ms/chunk in the profile) for no numerics benefit
            dw.add_(torch.mm(logits.t(), hidden[s:e]).float())
        return dh, dw.to(weight.dtype), None, None, None


def token_logprob_entropy(hidden: torch.Tensor, weight: torch.Tensor,
                          labels: torch.Tensor, temperature: float = 1.0,
                          chunk: int = _DEF_CHUNK):
    """Returns (logprob [N] fp32 — differentiable, entropy [N] fp32 — detached).

    logits are implicitly hidden @ weight.T / (temperature + 1e-7)."""
    inv_temp = 1.0 / (temperature + 1e-7)
    if hidden.is_cuda:
        return _TokenLogprobFn.apply(hidden, weight, labels, inv_temp, chunk)
    logits = (hidden.float() @ weight.float().t()) * inv_temp
    lp, ent, _ = _ref_rowstats(logits, labels)
    return lp, ent.detach()
