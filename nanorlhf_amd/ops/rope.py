"""Rotary position embedding (NeoX rotate-half layout, Qwen2 semantics).

cos/sin tables are precomputed on host once per model (fp32, [max_pos, D/2])
— on-device trig would turn this memory-bound op VALU-bound
(cdna_hip_programming.md Appendix B).  The HIP kernel applies RoPE in place
to q and k with a sign flag so backward reuses the same kernel with sin
negated (rotation transpose)."""
from __future__ import annotations

import torch

from . import ext


def build_rope_cache(head_dim: int, max_pos: int, theta: float = 1e6,
                     device="cpu") -> torch.Tensor:
    """Returns [max_pos, head_dim] fp32 table: first half cos, second half sin."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim))
    t = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)  # [max_pos, D/2]
    table = torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()
    return table.to(device)


def _rope_ref(x: torch.Tensor, table: torch.Tensor, positions: torch.Tensor,
              sign: float = 1.0) -> torch.Tensor:
    # x: [T, H, D]; positions: [T]
    D = x.shape[-1]
    cs = table[positions]  # [T, D]
    cos = cs[:, : D // 2].unsqueeze(1)  # [T, 1, D/2]
    sin = cs[:, D // 2:].unsqueeze(1) * sign
    x1 = x[..., : D // 2].float()
    x2 = x[..., D // 2:].float()
    o1 = x1 * cos - x2 * sin
    o2 = x2 * cos + x1 * sin
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, table, positions):
        ctx.save_for_backward(table, positions)
        return ext().rope_fwd(x.contiguous(), table, positions, 1.0)

    @staticmethod
    def backward(ctx, dy):
        table, positions = ctx.saved_tensors
        dx = ext().rope_fwd(dy.contiguous(), table, positions, -1.0)
        return dx, None, None


def rope_apply(x: torch.Tensor, table: torch.Tensor, positions: torch.Tensor) -> torch.Tensor:
    """x: [T, H, D] (packed tokens), positions: [T] int32/int64."""
    if x.is_cuda:
        return _RopeFn.apply(x, table, positions)
    if x.requires_grad:
        # autograd-composable CPU path
        return _rope_ref(x, table, positions)
    return _rope_ref(x, table, positions)
