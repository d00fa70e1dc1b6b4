"""GRPO — group-relative policy optimization.

Semantics of GRPO/grpo_trainer.py:406-778 (see each functional for exact
line cites): group advantage over N samples per prompt (nan→0 guard),
random keep-1-of-N, sparse group-score at EOS → undiscounted reward-to-go
token advantages, PPO-clip token loss with the k3 KL-to-reference penalty
added in-loss.  Sparse-GRPO ("r1", grpo_r1_trainer.py) = cfg.sparse_filter
(drop score==0 rows) + token-budget train buckets (cfg.train_token_budget).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from ..config import RLHFConfig
from . import functional as F
from .trainer import AlgoSpec, Rollout, RLHFTrainer, TrainData


@dataclass
class GRPOConfig(RLHFConfig):
    sample_n: int = 4            # grpo_sample_N (grpo.py:106)
    advantage_whiten: bool = False  # grpo.py:104
    keep_one_of_n: bool = True   # random 1-of-N keep (grpo_trainer.py:513-520)


class GRPO(AlgoSpec):
    name = "grpo"
    needs_ref = True

    def make_train_data(self, trainer: RLHFTrainer, ro: Rollout, greedy_scores=None) -> TrainData:
        cfg: GRPOConfig = trainer.cfg
        n = ro.sample_n
        adv_seq = F.grpo_group_advantage(ro.scores, n)      # [B*n]
        # one host transfer for the whole selection (the per-element
        # float(adv_seq[i]) loop was one sync per row on GPU scores)
        adv_cpu = adv_seq.detach().cpu()
        if cfg.sparse_filter:
            # sparse GRPO: drop zero-advantage samples (grpo_r1_trainer.py:565-568)
            rows = torch.nonzero(adv_cpu != 0.0, as_tuple=False).squeeze(1).tolist()
            if not rows:
                rows = [0]
        else:
            rows = list(range(ro.num_rows))
        if getattr(cfg, "keep_one_of_n", True) and n > 1:
            by_group: dict[int, list[int]] = {}
            for i in rows:
                by_group.setdefault(i // n, []).append(i)
            keep = []
            for gidx in sorted(by_group):
                members = by_group[gidx]
                pick = int(torch.randint(0, len(members), (1,), generator=trainer._keep_gen))
                keep.append(members[pick])
            rows = keep
        prompts = [ro.prompts[i] for i in rows]
        responses = [ro.responses[i] for i in rows]
        lp, ref_lp, ent, mask, _ = trainer.score_rows(
            prompts, responses, with_ref=True,
            rollout_lp=trainer.rollout_lp_for(ro, rows))
        adv_seq_kept = adv_seq[rows].to(trainer.device)
        eos_idx = mask.sum(1).long() - 1
        rewards = F.sparse_reward_at_eos(adv_seq_kept, mask, eos_idx)
        if cfg.whiten_rewards:
            rewards = F.masked_whiten(rewards, mask, shift_mean=True) * mask
        adv_tok = F.reward_to_go(rewards, gamma=1.0) * mask
        if cfg.advantage_whiten:
            adv_tok = F.masked_whiten(adv_tok, mask) * mask
        kl_old = F.masked_mean(lp - ref_lp, mask)
        return TrainData(rows=rows, prompts=prompts, responses=responses,
                         old_logprobs=lp, ref_logprobs=ref_lp, mask=mask,
                         advantages=adv_tok,
                         stats={"kl_old": float(kl_old),
                                "entropy_old": float(F.masked_mean(ent, mask))})

    def loss(self, trainer, td, mb, new_logprobs, vpred):
        cfg = trainer.cfg
        mask = mb["mask"]
        diff = new_logprobs - mb["old_logprobs"]
        ratio = torch.exp(diff)
        pg1 = -mb["advantages"] * ratio
        pg2 = -mb["advantages"] * torch.clamp(ratio, 1 - cfg.cliprange, 1 + cfg.cliprange)
        per_tok = torch.max(pg1, pg2)
        # k3 KL penalty to the reference policy, in-loss (grpo_trainer.py:667-670)
        per_tok = per_tok + F.k3_kl_penalty(new_logprobs, mb["ref_logprobs"], cfg.kl_coef)
        loss = F.masked_mean(per_tok, mask)
        with torch.no_grad():
            st = {"pg_clipfrac": F.masked_mean((pg2 > pg1).float(), mask),
                  "approxkl": 0.5 * F.masked_mean(diff**2, mask),
                  "ratio": ratio.detach()[mask > 0]}
        return loss, st


def make_trainer(cfg: GRPOConfig, policy, ref_policy, reward_fn, train_prompts,
                 **kw) -> RLHFTrainer:
    return RLHFTrainer(cfg, GRPO(), policy, ref_policy, reward_fn, train_prompts, **kw)
