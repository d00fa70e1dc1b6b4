"""RAFT — best-of-K rejection-sampling fine-tuning.

Semantics of RAFT/raft_trainer.py: generate K samples (:480), compute the
KL-shaped sequence reward (:564-585), pick the BEST per prompt (torch.max
:586) and SFT on it with plain NLL (:636 — no clip/ratio/advantage).

Reference quirk (raft_trainer.py:588): the snapshot OVERWRITES the argmax
with torch.randint, making best-of-K selection dead code.  We default to the
intended argmax; set cfg.random_keep_quirk=True to reproduce the reference
exactly (SURVEY.md "quirks to preserve vs fix")."""
from __future__ import annotations

from dataclasses import dataclass

import torch

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class RAFTConfig(RLHFConfig):
    sample_n: int = 4            # raft_sample_K (raft.py:105)
    random_keep_quirk: bool = False


class RAFT(AlgoSpec):
    name = "raft"
    needs_ref = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg = trainer.cfg
        n = ro.sample_n
        lp, ref_lp, ent, mask, _ = trainer.score_rows(
            ro.prompts, ro.responses, with_ref=True,
            rollout_lp=trainer.rollout_lp_for(ro, list(range(ro.num_rows))))
        kl_seq = ((lp - ref_lp) * mask).sum(1)
        rlhf_reward = ro.scores.to(trainer.device) - cfg.kl_coef * kl_seq
        g = rlhf_reward.view(-1, n)
        if getattr(cfg, "random_keep_quirk", False):
            pick = torch.randint(0, n, (g.shape[0],), generator=trainer._keep_gen)
        else:
            pick = g.argmax(dim=1).cpu()
        rows = [int(b * n + pick[b]) for b in range(g.shape[0])]
        sel = torch.tensor(rows, dtype=torch.long, device=trainer.device)
        Lsel = max(1, int(mask[sel].sum(1).max().item()))
        kl_old = F.masked_mean(lp - ref_lp, mask)
        return TrainData(rows=rows,
                         prompts=[ro.prompts[i] for i in rows],
                         responses=[ro.responses[i] for i in rows],
                         old_logprobs=lp[sel, :Lsel], ref_logprobs=ref_lp[sel, :Lsel],
                         mask=mask[sel, :Lsel],
                         advantages=torch.zeros(len(rows), device=trainer.device),
                         sequence_level=True,
                         stats={"kl_old": float(kl_old),
                                "entropy_old": float(F.masked_mean(ent, mask))})

    def loss(self, trainer, td, mb, new_logprobs, vpred):
        loss = F.raft_nll_loss(new_logprobs, mb["mask"])
        with torch.no_grad():
            diff = (new_logprobs - mb["old_logprobs"]) * mb["mask"]
            st = {"approxkl": 0.5 * F.masked_mean(diff**2, mb["mask"]),
                  "pg_clipfrac": torch.tensor(0.0),
                  "ratio": torch.exp(diff.sum(1)).detach()}
        return loss, st


def make_trainer(cfg, policy, ref_policy, reward_fn, train_prompts, **kw) -> RLHFTrainer:
    return RLHFTrainer(cfg, RAFT(), policy, ref_policy, reward_fn, train_prompts, **kw)
