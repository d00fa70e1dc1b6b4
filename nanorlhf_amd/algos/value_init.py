"""PPO value-model initializer.

Re-implements PPO/value_initializer.py:69-388: before PPO starts, generate
ONE rollout batch, compute KL-shaped reward-to-go returns from the frozen
policy/ref logprobs, then regression-finetune the critic's score head on
those returns (masked MSE over response tokens), early-stopping on a held
split's eval loss."""
from __future__ import annotations

import torch

from ..sampler.engine import SamplingParams
from ..utils.seed import rank_seed
from . import functional as F


def finetune_value_model(trainer, num_prompts: int = 500, epochs: int = 8,
                         lr: float = 1e-5, eval_frac: float = 0.1,
                         patience: int = 2, minibatch_rows: int = 16) -> dict:
    """Mutates trainer.value_model in place; returns stats.  `trainer` is a
    PPO RLHFTrainer (policy, ref, reward_fn, sampler already wired)."""
    cfg = trainer.cfg
    vm = trainer.value_model
    assert vm is not None
    device = trainer.device
    prompts = [trainer.train_prompts[i % len(trainer.train_prompts)]
               for i in range(num_prompts)]
    params = SamplingParams(n=1, temperature=cfg.temperature, top_p=cfg.top_p,
                            max_tokens=cfg.response_length,
                            seed=rank_seed(cfg.seed, trainer.rank) + 424243,
                            stop_token_id=cfg.stop_token_id)
    resp_pad = trainer.sampler.generate(prompts, params, pad_token_id=cfg.pad_token_id)
    responses = []
    for row in resp_pad.tolist():
        if cfg.stop_token_id is not None and cfg.stop_token_id in row:
            row = row[: row.index(cfg.stop_token_id) + 1]
        else:
            while row and row[-1] == cfg.pad_token_id:
                row.pop()
        responses.append(row)
    scores = trainer._call_reward([list(p) for p in prompts], responses)
    lp, ref_lp, _, mask, _ = trainer.score_rows(prompts, responses, with_ref=True)
    eos_idx = mask.sum(1).long() - 1
    rewards = F.kl_shaped_rewards(scores.to(device), lp, ref_lp, mask, eos_idx, cfg.kl_coef)
    returns = F.reward_to_go(rewards, gamma=cfg.gamma) * mask  # :352-359

    n_eval = max(1, int(num_prompts * eval_frac))
    train_rows = list(range(n_eval, num_prompts))
    eval_rows = list(range(n_eval))
    opt = torch.optim.AdamW([p for p in vm.parameters() if p.requires_grad], lr=lr)

    def run_rows(rows, train: bool):
        total, count = 0.0, 0
        for s in range(0, len(rows), minibatch_rows):
            chunk = rows[s: s + minibatch_rows]
            lens = []
            ids_list = []
            for i in chunk:
                ids_list.append(list(prompts[i]) + responses[i])
                lens.append(len(ids_list[-1]))
            ids = torch.tensor([t for seq in ids_list for t in seq], dtype=torch.long,
                               device=device)
            cu = torch.zeros(len(chunk) + 1, dtype=torch.int32, device=device)
            cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=device), 0)
            v = vm.token_values(ids, cu, max(lens))
            # value predictions at response-token positions
            loss_terms = []
            off = 0
            for j, i in enumerate(chunk):
                pl, rl = len(prompts[i]), len(responses[i])
                vi = v[off + pl - 1: off + pl - 1 + rl]
                tgt = returns[i, :rl]
                loss_terms.append(((vi - tgt) ** 2).sum())
                off += pl + rl
            ntok = int(mask[chunk].sum())
            loss = torch.stack(loss_terms).sum() / max(ntok, 1)
            if train:
                opt.zero_grad(set_to_none=True)
                loss.backward()
                opt.step()
            total += float(loss.detach())
            count += 1
        return total / max(count, 1)

    best, bad, history = float("inf"), 0, []
    for ep in range(epochs):
        tr = run_rows(train_rows, train=True)
        with torch.no_grad():
            ev = run_rows(eval_rows, train=False)
        history.append((tr, ev))
        if ev < best - 1e-6:
            best, bad = ev, 0
        else:
            bad += 1
            if bad >= patience:
                break
    return {"epochs_ran": len(history), "best_eval_loss": best, "history": history}
