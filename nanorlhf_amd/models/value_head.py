"""Scalar-head models: critic (per-token values) and sequence scorer
(reward model).

Replaces AutoModelForSequenceClassification(num_labels=1) value/reward
loading in the reference (PPO/ppo.py:280-287; GRPO/grpo.py:162-198 DeBERTa
reward).  The reward family here is `rm-large` (models/config.py): a
deberta-v3-large-SHAPED bidirectional encoder — same depth/width/compute
shape — with a scalar head read at each sequence's last token (trl
get_reward semantics, ppo_trainer.py:630-634)."""
from __future__ import annotations

import torch
import torch.nn as nn

from .config import ModelConfig, get_config
from .qwen2 import AttnContext, Transformer, make_positions


class ScalarHeadModel(nn.Module):
    """Backbone + Linear(hidden, 1).  causal=True → critic usable per-token;
    bidirectional configs (cfg.bidirectional) → encoder reward model."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.model = Transformer(cfg)
        self.score = nn.Linear(cfg.hidden_size, max(cfg.num_labels, 1), bias=False)
        self.gradient_checkpointing = False
        nn.init.normal_(self.score.weight, std=1.0 / (cfg.hidden_size + 1) ** 0.5)

    def forward(self, input_ids: torch.Tensor, ctx: AttnContext) -> torch.Tensor:
        """Returns per-token scores [T, num_labels]."""
        ctx.causal = not self.cfg.bidirectional
        hidden = self.model(input_ids, ctx, checkpoint=self.gradient_checkpointing)
        return self.score(hidden)

    def sequence_scores(self, input_ids: torch.Tensor, cu_seqlens: torch.Tensor,
                        max_seqlen: int) -> torch.Tensor:
        """Score of each packed sequence at its LAST token → [B]."""
        positions = make_positions(cu_seqlens)
        ctx = AttnContext(mode="train", positions=positions, cu_seqlens=cu_seqlens,
                          max_seqlen=max_seqlen, causal=not self.cfg.bidirectional)
        per_token = self.forward(input_ids, ctx)  # [T, L]
        last_idx = (cu_seqlens[1:].long() - 1).clamp(min=0)
        return per_token[last_idx, 0]

    def token_values(self, input_ids: torch.Tensor, cu_seqlens: torch.Tensor,
                     max_seqlen: int) -> torch.Tensor:
        """Per-token value estimates [T] (critic path, causal)."""
        positions = make_positions(cu_seqlens)
        ctx = AttnContext(mode="train", positions=positions, cu_seqlens=cu_seqlens,
                          max_seqlen=max_seqlen, causal=True)
        return self.forward(input_ids, ctx)[:, 0]

    @classmethod
    def from_preset(cls, name: str, **overrides) -> "ScalarHeadModel":
        cfg = get_config(name, **overrides)
        if cfg.num_labels == 0:
            cfg.num_labels = 1
        return cls(cfg)
