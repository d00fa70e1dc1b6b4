"""PPO with a learned critic.

Semantics of PPO/ppo_trainer.py: critic = ScalarHeadModel run alongside the
policy (PolicyAndValueWrapper :87-99 → here just a second model), value
estimates per response token (:630-634), KL folded into the reward stream
(:672-674), GAE(gamma=1.0, lam=0.95) (:688-697), returns = adv + values
(:697), token PPO-clip + clipped value MSE joint loss pg + vf_coef*vf
(:742-756).  Checkpoints add value_model/ (:413-416) — handled by
RLHFTrainer.save.  Separate policy/value LRs (ppo.py:118-119) via
value_learning_rate."""
from __future__ import annotations

from dataclasses import dataclass

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class PPOConfig(RLHFConfig):
    sample_n: int = 1
    value_learning_rate: float = 3e-6
    whiten_rewards: bool = False
    whiten_advantages: bool = False  # reference default: PPO/ppo.py:166 advantage_whiten=False
    # value-LoRA (reference PPO/ppo.py:141-159: value_use_lora=True, r=64)
    value_use_lora: bool = True
    value_lora_r: int = 64
    value_lora_alpha: int = 16
    value_lora_dropout: float = 0.0


class PPO(AlgoSpec):
    name = "ppo"
    needs_ref = True
    needs_value = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg = trainer.cfg
        rows = list(range(ro.num_rows))
        lp, ref_lp, ent, mask, values = trainer.score_rows(
            ro.prompts, ro.responses, with_ref=True, with_values=True,
            rollout_lp=trainer.rollout_lp_for(ro, list(range(ro.num_rows))))
        values = values * mask
        eos_idx = mask.sum(1).long() - 1
        scores = ro.scores.to(trainer.device)
        rewards = F.kl_shaped_rewards(scores, lp, ref_lp, mask, eos_idx, cfg.kl_coef)
        if cfg.whiten_rewards:
            rewards = F.masked_whiten(rewards, mask, shift_mean=True) * mask
        adv, returns = F.gae(rewards, values, cfg.gamma, cfg.lam)
        adv = adv * mask
        returns = returns * mask
        if getattr(cfg, "whiten_advantages", False):
            adv = F.masked_whiten(adv, mask) * mask
        kl_old = F.masked_mean(lp - ref_lp, mask)
        return TrainData(rows=rows, prompts=ro.prompts, responses=ro.responses,
                         old_logprobs=lp, ref_logprobs=ref_lp, mask=mask,
                         advantages=adv, values=values, returns=returns,
                         stats={"kl_old": float(kl_old),
                                "entropy_old": float(F.masked_mean(ent, mask))})

    def loss(self, trainer, td, mb, new_logprobs, vpred):
        cfg = trainer.cfg
        pg_loss, st = F.ppo_clip_token_loss(new_logprobs, mb["old_logprobs"],
                                            mb["advantages"], mb["mask"], cfg.cliprange)
        vf_loss, vf_clipfrac = F.value_clip_loss(vpred, mb["values"], mb["returns"],
                                                 mb["mask"], cfg.cliprange_value)
        st["vf_loss"] = vf_loss.detach()
        st["vf_clipfrac"] = vf_clipfrac
        return pg_loss + cfg.vf_coef * vf_loss, st


def make_trainer(cfg, policy, ref_policy, reward_fn, train_prompts,
                 value_model=None, **kw) -> RLHFTrainer:
    # value-LoRA (ppo.py:141-159): adapters on the critic backbone, score
    # head + embeddings fully trained (value_modules_to_save)
    if value_model is not None and getattr(cfg, "value_use_lora", False):
        from ..models.lora import LoraConfig, apply_lora
        apply_lora(value_model, LoraConfig(
            r=cfg.value_lora_r, alpha=cfg.value_lora_alpha,
            dropout=cfg.value_lora_dropout,
            modules_to_save=("embed_tokens", "lm_head", "score")))
    t = RLHFTrainer(cfg, PPO(), policy, ref_policy, reward_fn, train_prompts,
                    value_model=value_model, **kw)
    # separate value LR (ppo.py:118-119) on the value param groups
    # (policy/value × decay/no-decay — ppo_trainer.py:341-402)
    if hasattr(cfg, "value_learning_rate"):
        for g in t.optimizer.param_groups:
            if g.get("name", "").startswith("value"):
                g["lr"] = cfg.value_learning_rate
                g["initial_lr"] = cfg.value_learning_rate
    return t
