"""Buckets, checkpoint rotation, math verifier, synthetic data, batch algebra."""
import os

import pytest
import torch

from nanorlhf_amd.config import RLHFConfig
from nanorlhf_amd.data import create_batches, hh_shaped_prompts
from nanorlhf_amd.rewards import MathRuleReward
from nanorlhf_amd.rewards.mathcheck import answers_equal, extract_boxed, normalize_answer
from nanorlhf_amd.utils.checkpoint import CheckpointManager


def test_create_batches_budget_invariant():
    lengths = [5, 100, 17, 33, 2, 64, 80, 9]
    budget = 128
    buckets = create_batches(lengths, budget)
    seen = sorted(i for b in buckets for i in b)
    assert seen == list(range(len(lengths)))
    for b in buckets:
        mx = max(lengths[i] for i in b)
        assert mx * len(b) <= budget or len(b) == 1


def test_create_batches_sorted_packing():
    buckets = create_batches([1, 2, 3, 4], 4)
    # sorted by length; [1,2]-> max2*2=4 ok; adding 3 -> 9 > 4
    assert buckets[0] == [0, 1]


def test_batch_algebra_reference_default():
    cfg = RLHFConfig(per_device_train_batch_size=4, gradient_accumulation_steps=8,
                     num_mini_batches=16, total_episodes=1_000_000)
    s = cfg.batch_sizes(world_size=1)
    assert s["local_batch_size"] == 512  # grpo default (grpo_trainer.py:216-247)
    assert s["local_mini_batch_size"] == 32
    s8 = cfg.batch_sizes(world_size=8)
    assert s8["batch_size"] == 4096


def test_extract_boxed():
    assert extract_boxed(r"so the answer is \boxed{42}.") == "42"
    assert extract_boxed(r"\boxed{\frac{1}{2}} and \boxed{x+1}") == "x+1"
    assert extract_boxed("no box here") is None
    assert extract_boxed(r"\boxed{a{b}c}") == "a{b}c"


def test_normalize_and_equal():
    assert normalize_answer(r"\frac{1}{2}") == "((1)/(2))"
    assert answers_equal("42", "42.0")
    assert answers_equal(r"\frac{1}{2}", "0.5")
    assert answers_equal("1,234", "1234")
    assert not answers_equal("41", "42")
    assert answers_equal("x + x", "2x")  # sympy stage


def test_math_rule_reward():
    rw = MathRuleReward(["42", "7"])
    s = rw([r"the answer is \boxed{42}", r"\boxed{8}"])
    assert s.tolist() == [1.0, 0.0]


def test_checkpoint_rotation_and_best(tmp_path):
    cm = CheckpointManager(str(tmp_path), save_total_limit=2,
                           metric_for_best="eval_objective/rlhf_reward_old")
    state = {"w": torch.zeros(2)}
    for step, metric in [(1, 1.0), (2, 5.0), (3, 0.5), (4, 0.1)]:
        cm.save(step, step * 10, state, {"m": 1}, None, None, None,
                {"eval_objective/rlhf_reward_old": metric})
    dirs = sorted(os.listdir(tmp_path))
    # keeps 2 newest + the best; *_old metric at step2 describes checkpoint-1
    assert "checkpoint-4" in dirs and "checkpoint-3" in dirs
    assert os.path.basename(cm.best_checkpoint) == "checkpoint-1"
    loaded = CheckpointManager.load_policy_state(os.path.join(tmp_path, "checkpoint-4"))
    assert "w" in loaded


def test_hh_shaped_prompts_reproducible():
    a = hh_shaped_prompts(5, 1000, seed=3)
    b = hh_shaped_prompts(5, 1000, seed=3)
    assert a == b
    assert all(2 <= t < 1000 for p in a for t in p)
