"""End-to-end CPU training tests — one per algorithm, tiny model.

test_reinforce_tiny_constant_reward is BASELINE.json config #1 (the
permanent plumbing integration test: 2-layer 128-dim model, CPU eager
generate, constant reward)."""
import os

import pytest
import torch

from nanorlhf_amd.algos import (GRPOConfig, PPOConfig, RAFTConfig, ReinforceConfig,
                                RemaxConfig, RLOOConfig, finetune_value_model)
from nanorlhf_amd.algos import grpo, ppo, raft, reinforce, remax, rloo
from nanorlhf_amd.data import hh_shaped_prompts
from nanorlhf_amd.models import CausalLM, ScalarHeadModel
from nanorlhf_amd.rewards import constant_reward


def _mk(cfg_cls, tmp_path, **kw):
    base = dict(model_preset="tiny", dtype="float32", use_lora=True, lora_r=4,
                lora_alpha=8, per_device_train_batch_size=2,
                gradient_accumulation_steps=2, num_mini_batches=2,
                total_episodes=16, response_length=6, temperature=1.0,
                stop_token_id=1, output_dir=str(tmp_path),
                gradient_checkpointing=False, score_token_budget=512,
                missing_eos_penalty=1.0, save_steps=0)
    base.update(kw)
    return cfg_cls(**base)


def _models(seed=0):
    torch.manual_seed(seed)
    policy = CausalLM.from_preset("tiny")
    ref = CausalLM.from_preset("tiny")
    ref.load_state_dict(policy.state_dict())
    return policy, ref


def _varied_reward(seqs):
    # content-dependent reward so within-group advantages are non-degenerate
    return torch.tensor([(sum(s) + len(s)) % 5 - 2.0 for s in seqs])


def test_reinforce_tiny_constant_reward(tmp_path):
    cfg = _mk(ReinforceConfig, tmp_path, save_steps=1)
    policy, ref = _models()
    prompts = hh_shaped_prompts(32, 1024, min_len=4, max_len=12)
    tr = reinforce.make_trainer(cfg, policy, ref, lambda s: constant_reward(s), prompts)
    tr.train(num_updates=2)
    assert tr.episode == 16
    assert os.path.isdir(os.path.join(tmp_path, "checkpoint-2"))
    assert os.path.exists(os.path.join(tmp_path, "checkpoint-2", "trainer_state.json"))
    assert os.path.exists(os.path.join(tmp_path, "metrics.jsonl"))


def test_grpo_tiny(tmp_path):
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2)
    policy, ref = _models(1)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=1)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    before = {n: p.clone() for n, p in tr.policy.named_parameters() if p.requires_grad}
    tr.train(num_updates=1)
    changed = any(not torch.equal(before[n], p)
                  for n, p in tr.policy.named_parameters() if n in before)
    assert changed, "update must move trainable params"


def test_grpo_sparse_filter(tmp_path):
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2, sparse_filter=True,
              train_token_budget=64)
    policy, ref = _models(2)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=2)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_rloo_tiny(tmp_path):
    cfg = _mk(RLOOConfig, tmp_path, sample_n=2)
    policy, ref = _models(3)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=3)
    tr = rloo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_remax_tiny(tmp_path):
    cfg = _mk(RemaxConfig, tmp_path)
    policy, ref = _models(4)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=4)
    tr = remax.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_raft_tiny(tmp_path):
    cfg = _mk(RAFTConfig, tmp_path, sample_n=2)
    policy, ref = _models(5)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=5)
    tr = raft.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_ppo_tiny_with_value_init(tmp_path):
    cfg = _mk(PPOConfig, tmp_path, save_steps=1)
    policy, ref = _models(6)
    torch.manual_seed(6)
    vm = ScalarHeadModel.from_preset("tiny", num_labels=1, bidirectional=False)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=6)
    tr = ppo.make_trainer(cfg, policy, ref, _varied_reward, prompts, value_model=vm)
    stats = finetune_value_model(tr, num_prompts=8, epochs=2, minibatch_rows=4)
    assert stats["epochs_ran"] >= 1
    tr.train(num_updates=1)
    # PPO checkpoints include value_model/ (ppo_trainer.py:413-416)
    assert os.path.isdir(os.path.join(tmp_path, "checkpoint-1", "value_model"))


def test_model_reward_path(tmp_path):
    from nanorlhf_amd.rewards import ModelReward
    torch.manual_seed(7)
    rm = ScalarHeadModel.from_preset("rm-tiny")
    reward = ModelReward(rm, device="cpu", token_budget=128)
    seqs = [list(range(2, 12)), list(range(2, 30))]
    s = reward(seqs)
    assert s.shape == (2,)
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2)
    policy, ref = _models(7)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=8, seed=7)
    tr = grpo.make_trainer(cfg, policy, ref, reward, prompts)
    tr.train(num_updates=1)


def test_step_every_microbatch_quirk_mode(tmp_path):
    """Reference quirk flag (optimizer.step per micro-batch inside
    accumulate(), grpo_trainer.py:692) must run end to end."""
    cfg = _mk(ReinforceConfig, tmp_path, step_every_microbatch=True)
    policy, ref = _models(11)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=11)
    tr = reinforce.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_score_rows_matches_direct_forward(tmp_path):
    """score_rows' bucketed packed scoring must equal a direct per-row
    forward (logprob of each response token)."""
    import torch.nn.functional as TF
    from nanorlhf_amd.models import pack_sequences
    cfg = _mk(GRPOConfig, tmp_path, sample_n=1, score_token_budget=64)
    policy, ref = _models(12)
    prompts = hh_shaped_prompts(6, 1024, min_len=4, max_len=9, seed=12)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    rows_p = [p for p in prompts[:4]]
    rows_r = [[5, 9, 2], [7, 1], [3, 3, 3, 3], [8]]
    lp, ref_lp, ent, mask, _ = tr.score_rows(rows_p, rows_r, with_ref=True)
    for i, (p, r) in enumerate(zip(rows_p, rows_r)):
        ids, cu, mx, pos = pack_sequences([torch.tensor(list(p) + r)])
        h = tr.policy(ids, tr.policy.train_ctx(cu, mx, pos))
        logits = tr.policy.logits(h).float() / (cfg.temperature + 1e-7)
        logp = TF.log_softmax(logits, dim=-1)
        for t, tok in enumerate(r):
            want = float(logp[len(p) - 1 + t, tok])
            assert abs(float(lp[i, t]) - want) < 1e-3, (i, t)
        assert mask[i, : len(r)].sum() == len(r)
        assert mask[i, len(r):].sum() == 0


def test_rollout_logprobs_match_scoring(tmp_path):
    """The sampler's per-token logprobs (use_rollout_logprobs shortcut) must
    agree with the recomputed scoring pass on an fp32 model."""
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2, use_lora=False,
              temperature=0.9)
    policy, ref = _models(21)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=8, seed=21)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    ro, _ = tr._rollout(1)
    assert ro.logprobs is not None and len(ro.logprobs) == ro.num_rows
    lp, _, _, mask, _ = tr.score_rows(ro.prompts, ro.responses, with_ref=False)
    for i in range(ro.num_rows):
        n = len(ro.responses[i])
        for t in range(n):
            assert abs(ro.logprobs[i][t] - float(lp[i, t])) < 5e-3, (i, t)


def test_grpo_with_rollout_logprobs(tmp_path):
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2, use_rollout_logprobs=True)
    policy, ref = _models(22)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=22)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_metric_names_logged(tmp_path):
    """The reference's stable metric names must appear in metrics.jsonl
    (SURVEY §5 observability; dashboards transfer)."""
    import json
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2)
    policy, ref = _models(31)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=8, seed=31)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)
    rec = json.loads(open(os.path.join(tmp_path, "metrics.jsonl")).readline())
    for name in ("objective/kl_old", "objective/entropy_old",
                 "eval_objective/rlhf_reward_old", "policy/approxkl_avg_new",
                 "policy/clipfrac_avg_new", "loss/policy_avg_new",
                 "policy/entropy_avg_new", "val/ratio_new", "val/ratio_var_new",
                 "val/num_eos_tokens_old", "lr", "episode"):
        assert name in rec, name


def test_multi_epoch_and_whiten_rewards(tmp_path):
    """num_ppo_epochs=2 reuses the batch off-policy (clip active);
    whiten_rewards normalizes the reward stream."""
    cfg = _mk(GRPOConfig, tmp_path, sample_n=2, num_ppo_epochs=2,
              whiten_rewards=True)
    policy, ref = _models(32)
    prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=10, seed=32)
    tr = grpo.make_trainer(cfg, policy, ref, _varied_reward, prompts)
    tr.train(num_updates=1)


def test_lr_schedule_progression(tmp_path):
    cfg = _mk(ReinforceConfig, tmp_path, total_episodes=16,
              lr_scheduler_type="cosine_with_min_lr", min_lr_ratio=0.5,
              learning_rate=1e-4)
    policy, ref = _models(33)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=8, seed=33)
    tr = reinforce.make_trainer(cfg, policy, ref,
                                lambda s: constant_reward(s), prompts)
    lr0 = tr._lr()
    tr.train(num_updates=2)
    assert tr.lr_step > 0
    assert tr._lr() < lr0  # cosine decays
    assert tr._lr() >= cfg.learning_rate * cfg.min_lr_ratio - 1e-12


def test_callback_stop_and_oversize_prompt_error(tmp_path):
    from nanorlhf_amd.utils.callbacks import TrainerCallback

    class StopNow(TrainerCallback):
        def __init__(self):
            self.calls = 0

        def on_update_end(self, trainer, metrics):
            self.calls += 1
            return True

    cfg = _mk(ReinforceConfig, tmp_path)
    policy, ref = _models(41)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=8, seed=41)
    cb = StopNow()
    tr = reinforce.make_trainer(cfg, policy, ref,
                                lambda s: constant_reward(s), prompts,
                                callbacks=[cb])
    tr.train(num_updates=5)
    assert cb.calls == 1 and tr.global_step == 1

    # a sequence that can never fit the pool must raise, not hang
    from nanorlhf_amd.sampler import SamplerEngine, SamplingParams
    eng = SamplerEngine(tr.policy, kv_pool_tokens=64, page_size=16)
    with pytest.raises(RuntimeError, match="KV pool"):
        eng.generate([[2] * 200], SamplingParams(n=1, max_tokens=50))


def test_ppo_optimizer_group_parity(tmp_path):
    """Reference PPO optimizer parity (PPO/ppo_trainer.py:341-402): 4 param
    groups policy/value × decay/no-decay; biases + norm weights never decay;
    value groups run at value_learning_rate; value-LoRA applied
    (ppo.py:141-159)."""
    cfg = _mk(PPOConfig, tmp_path, weight_decay=0.01, value_learning_rate=1e-5,
              value_use_lora=True, value_lora_r=4, value_lora_alpha=8)
    policy, ref = _models(7)
    torch.manual_seed(7)
    vm = ScalarHeadModel.from_preset("tiny", num_labels=1, bidirectional=False)
    prompts = hh_shaped_prompts(8, 1024, min_len=4, max_len=10, seed=7)
    tr = ppo.make_trainer(cfg, policy, ref, _varied_reward, prompts, value_model=vm)

    names = [g.get("name") for g in tr.optimizer.param_groups]
    assert names == ["policy_decay", "policy_nodecay", "value_decay", "value_nodecay"]
    for g in tr.optimizer.param_groups:
        if g["name"].endswith("_nodecay"):
            assert g["weight_decay"] == 0.0
        else:
            assert g["weight_decay"] == 0.01
        if g["name"].startswith("value"):
            assert g["lr"] == 1e-5
    # value-LoRA engaged: adapters exist, backbone frozen, score head trained
    from nanorlhf_amd.models.lora import LoRALinear
    assert any(isinstance(m, LoRALinear) for m in vm.modules())
    assert vm.score.weight.requires_grad
    assert not vm.model.layers[0].self_attn.qkv_proj.base.weight.requires_grad
    # no-decay groups contain only biases/norm weights (by construction all
    # 1-D here)
    for g in tr.optimizer.param_groups:
        if g["name"].endswith("_nodecay"):
            assert all(p.dim() == 1 for p in g["params"])
    tr.train(num_updates=1)
