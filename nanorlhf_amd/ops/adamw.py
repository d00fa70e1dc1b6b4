"""Fused AdamW for MI355X.

Replaces torch AdamW + the reference's blocking optimizer-state shuttles
(GRPO/grpo_trainer.py:168-172 state_to_device, :475,:625).  States are fp32,
params may be bf16 or fp32; the HIP kernel fuses the whole update (decoupled
weight decay, bias correction) in one pass per tensor with vectorized loads.
Host-offload of states is a policy of utils.offload, not baked in here."""
from __future__ import annotations

import torch

from . import ext


def _adamw_ref(p, g, m, v, lr, b1, b2, eps, wd, step):
    pf = p.float()
    gf = g.float()
    m.mul_(b1).add_(gf, alpha=1 - b1)
    v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
    bc1 = 1 - b1**step
    bc2 = 1 - b2**step
    denom = (v / bc2).sqrt().add_(eps)
    pf = pf * (1 - lr * wd) - lr * (m / bc1) / denom
    p.copy_(pf.to(p.dtype))


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-4, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                m, v = state["exp_avg"], state["exp_avg_sq"]
                if p.is_cuda:
                    ext().adamw_step(p.data, p.grad, m, v, group["lr"], b1, b2,
                                     group["eps"], group["weight_decay"], state["step"])
                else:
                    _adamw_ref(p.data, p.grad, m, v, group["lr"], b1, b2,
                               group["eps"], group["weight_decay"], state["step"])
        return loss
