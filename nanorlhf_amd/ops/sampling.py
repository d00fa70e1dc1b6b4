"""Temperature / top-p token sampling.

Replaces vLLM's sampler (SamplingParams(temperature, top_p=0.95, seed=...)
at GRPO/grpo_trainer.py:127).  One HIP kernel per decode step: per row it
(1) finds max and exp-sum at temperature, (2) locates the top-p probability
threshold with an LDS histogram over exp-space (two refinement rounds — no
full 151k sort), (3) draws inverse-CDF over the kept set with a counter-based
philox-like hash of (seed, step, row), so replays are deterministic and the
per-update reseed of the reference (random vLLM seed per update,
grpo_trainer.py:127) is preserved at engine level.

temperature == 0 → greedy argmax (ReMax baseline pass, remax_trainer.py:166-185).
"""
from __future__ import annotations

import torch

from . import ext


def _hash_uniform(seed: int, step: int, n: int, device) -> torch.Tensor:
    # CPU reference RNG (the GPU kernel owns the real splitmix64 counter hash;
    # CPU vs GPU samples are distribution-equal, not bit-equal)
    g = torch.Generator(device="cpu").manual_seed((seed * 1000003 + step) % (2**31 - 1))
    return torch.rand(n, generator=g).to(device)


def _sample_ref(logits: torch.Tensor, temperature: float, top_p: float,
                seed: int, step: int) -> torch.Tensor:
    if temperature == 0.0:
        return logits.argmax(dim=-1)
    probs = torch.softmax(logits.float() / temperature, dim=-1)
    sorted_probs, sorted_idx = torch.sort(probs, descending=True, dim=-1)
    cum = torch.cumsum(sorted_probs, dim=-1)
    # keep smallest prefix with cumulative >= top_p (always keep the first)
    keep = (cum - sorted_probs) < top_p
    sorted_probs = sorted_probs * keep
    sorted_probs = sorted_probs / sorted_probs.sum(dim=-1, keepdim=True)
    u = _hash_uniform(seed, step, logits.shape[0], logits.device)
    cdf = torch.cumsum(sorted_probs, dim=-1)
    choice = (cdf < u.unsqueeze(1)).sum(dim=-1).clamp(max=logits.shape[1] - 1)
    return sorted_idx.gather(1, choice.unsqueeze(1)).squeeze(1)


def sample_tokens(logits: torch.Tensor, temperature: float, top_p: float,
                  seed: int, step: int) -> torch.Tensor:
    """logits: [B, V] bf16/fp32 → tokens [B] int64."""
    if logits.is_cuda:
        return ext().sample_topp(logits.contiguous(), float(temperature), float(top_p),
                                 int(seed), int(step))
    return _sample_ref(logits, temperature, top_p, seed, step)
