"""REINFORCE with whitened reward-to-go and a PPO-clip surrogate (the clip
enables the off-policy minibatch reuse, REINFORCE/reinforce_trainer.py:639-640).

Pipeline (reinforce_trainer.py:568-591): KL-shaped token rewards (score at
EOS), discounted reward-to-go (gamma), whitening (default ON —
reinforce.py:103 "reinforce without any baseline will fail"), token clip loss."""
from __future__ import annotations

from dataclasses import dataclass

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class ReinforceConfig(RLHFConfig):
    sample_n: int = 1
    advantage_whiten: bool = True


class REINFORCE(AlgoSpec):
    name = "reinforce"
    needs_ref = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg = trainer.cfg
        rows = list(range(ro.num_rows))
        lp, ref_lp, ent, mask, _ = trainer.score_rows(
            ro.prompts, ro.responses, with_ref=True,
            rollout_lp=trainer.rollout_lp_for(ro, rows))
        eos_idx = mask.sum(1).long() - 1
        scores = ro.scores.to(trainer.device)
        rewards = F.kl_shaped_rewards(scores, lp, ref_lp, mask, eos_idx, cfg.kl_coef)
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
    return RLHFTrainer(cfg, REINFORCE(), policy, ref_policy, reward_fn, train_prompts, **kw)
