"""ReMax — greedy-baseline REINFORCE.

Semantics of ReMax/remax_trainer.py: a second GREEDY generation pass
(temperature=0, :166-185 — cheap with the in-process sampler, no re-boot),
sequence advantage = sampled reward − greedy reward (:506-513), KL folded
into the token reward stream (:587-589), reward-to-go, token-level PPO-clip
loss with no explicit KL term in the loss."""
from __future__ import annotations

from dataclasses import dataclass

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class RemaxConfig(RLHFConfig):
    sample_n: int = 1  # n=1 + greedy baseline (SURVEY §2.1 ReMax row)


class ReMax(AlgoSpec):
    name = "remax"
    needs_ref = True
    greedy_baseline = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg = trainer.cfg
        assert greedy_scores is not None
        rows = list(range(ro.num_rows))
        lp, ref_lp, ent, mask, _ = trainer.score_rows(
            ro.prompts, ro.responses, with_ref=True,
            rollout_lp=trainer.rollout_lp_for(ro, rows))
        adv_scores = F.remax_advantage(ro.scores, greedy_scores).to(trainer.device)
        eos_idx = mask.sum(1).long() - 1
        rewards = F.kl_shaped_rewards(adv_scores, lp, ref_lp, mask, eos_idx, cfg.kl_coef)
        if cfg.whiten_rewards:
            rewards = F.masked_whiten(rewards, mask, shift_mean=True) * mask
        adv = F.reward_to_go(rewards, gamma=cfg.gamma) * mask
        if cfg.advantage_whiten:
            adv = F.masked_whiten(adv, mask) * mask
        kl_old = F.masked_mean(lp - ref_lp, mask)
        return TrainData(rows=rows, prompts=ro.prompts, responses=ro.responses,
                         old_logprobs=lp, ref_logprobs=ref_lp, mask=mask, advantages=adv,
                         stats={"kl_old": float(kl_old),
                                "entropy_old": float(F.masked_mean(ent, mask))})


def make_trainer(cfg, policy, ref_policy, reward_fn, train_prompts, **kw) -> RLHFTrainer:
    return RLHFTrainer(cfg, ReMax(), policy, ref_policy, reward_fn, train_prompts, **kw)
