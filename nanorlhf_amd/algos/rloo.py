"""RLOO — REINFORCE leave-one-out.

Semantics of RLOO/rloo_trainer.py: logprobs are scored for ALL N samples
(:512 repeat_interleave), KL is folded into a sequence-level rlhf_reward
(:571-573), leave-one-out baseline per group (:597-599), then a random
1-of-N per prompt is kept for the update (:603-613, "abandon some examples
to save time"), and the loss uses a SEQUENCE-level ratio
(sum_t logprobs, :660-661) with PPO clip averaged over sequences (:669)."""
from __future__ import annotations

from dataclasses import dataclass

import torch

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class RLOOConfig(RLHFConfig):
    sample_n: int = 4            # rloo_sample_N (rloo.py:107)
    keep_one_of_n: bool = True


class RLOO(AlgoSpec):
    name = "rloo"
    needs_ref = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg = trainer.cfg
        n = ro.sample_n
        # score ALL rows (KL enters the reward)
        lp, ref_lp, ent, mask, _ = trainer.score_rows(
            ro.prompts, ro.responses, with_ref=True,
            rollout_lp=trainer.rollout_lp_for(ro, list(range(ro.num_rows))))
        kl_seq = ((lp - ref_lp) * mask).sum(1)            # [B*n]
        scores = ro.scores.to(trainer.device)
        rlhf_reward = scores - cfg.kl_coef * kl_seq       # sequence-level (:571-573)
        adv_seq = F.rloo_baseline_advantage(rlhf_reward, n)
        rows = list(range(ro.num_rows))
        if getattr(cfg, "keep_one_of_n", True) and n > 1:
            rows = F.random_keep_one_per_group(ro.num_rows // n, n,
                                               generator=trainer._keep_gen).tolist()
        sel = torch.tensor(rows, dtype=torch.long, device=trainer.device)
        kl_old = F.masked_mean(lp - ref_lp, mask)
        Lsel = int(mask[sel].sum(1).max().item())
        return TrainData(rows=rows,
                         prompts=[ro.prompts[i] for i in rows],
                         responses=[ro.responses[i] for i in rows],
                         old_logprobs=lp[sel, :Lsel], ref_logprobs=ref_lp[sel, :Lsel],
                         mask=mask[sel, :Lsel],
                         advantages=adv_seq[sel],          # [R] sequence-level
                         sequence_level=True,
                         stats={"kl_old": float(kl_old),
                                "entropy_old": float(F.masked_mean(ent, mask))})

    def loss(self, trainer, td, mb, new_logprobs, vpred):
        loss, st = F.rloo_sequence_loss(new_logprobs, mb["old_logprobs"],
                                        mb["advantages"], mb["mask"], trainer.cfg.cliprange)
        return loss, st


def make_trainer(cfg, policy, ref_policy, reward_fn, train_prompts, **kw) -> RLHFTrainer:
    return RLHFTrainer(cfg, RLOO(), policy, ref_policy, reward_fn, train_prompts, **kw)
