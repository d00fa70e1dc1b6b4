"""Unit tests of the pure algorithm math against hand-computed tensors
(SURVEY.md §4 test strategy)."""
import math

import pytest
import torch

from nanorlhf_amd.algos import functional as F


def test_masked_mean_whiten():
    v = torch.tensor([[1.0, 2.0, 3.0, 99.0]])
    m = torch.tensor([[1.0, 1.0, 1.0, 0.0]])
    assert float(F.masked_mean(v, m)) == pytest.approx(2.0)
    # trl semantics: shift_mean=True -> zero-centered, False -> mean preserved
    w = F.masked_whiten(v, m, shift_mean=True)
    mean = float(F.masked_mean(w, m))
    assert mean == pytest.approx(0.0, abs=1e-4)
    w2 = F.masked_whiten(v, m, shift_mean=False)
    assert float(F.masked_mean(w2, m)) == pytest.approx(2.0, abs=1e-4)


def test_first_true_and_truncate():
    b = torch.tensor([[False, True, False], [False, False, False]])
    idx = F.first_true_indices(b)
    assert idx.tolist() == [1, 3]
    resp = torch.tensor([[5, 7, 9, 9], [5, 5, 5, 5]])
    out = F.truncate_response(stop_token_id=7, pad_token_id=0, responses=resp)
    assert out.tolist() == [[5, 7, 0, 0], [5, 5, 5, 5]]


def test_grpo_group_advantage_hand():
    # group [1, 3]: mean 2, std sqrt(2) -> adv = ∓1/sqrt(2)... std unbiased: sqrt(2)
    scores = torch.tensor([1.0, 3.0, 5.0, 5.0])
    adv = F.grpo_group_advantage(scores, n=2)
    s = math.sqrt(2.0)
    assert adv[0] == pytest.approx(-1.0 / s * 1.0, rel=1e-5)
    assert adv[1] == pytest.approx(1.0 / s * 1.0, rel=1e-5)
    # degenerate group (std=0) -> nan -> 0 (reference nan guard :508-512)
    assert adv[2] == 0.0 and adv[3] == 0.0


def test_rloo_baseline_hand():
    r = torch.tensor([1.0, 2.0, 3.0])
    adv = F.rloo_baseline_advantage(r, n=3)
    # baselines: (2+3)/2=2.5, (1+3)/2=2, (1+2)/2=1.5
    assert adv.tolist() == pytest.approx([-1.5, 0.0, 1.5])


def test_reward_to_go_matches_loop_oracle():
    torch.manual_seed(0)
    r = torch.randn(3, 7)
    for gamma in (1.0, 0.9):
        out = F.reward_to_go(r, gamma)
        # reference-style python loop (grpo_trainer.py:611-617)
        oracle = torch.zeros_like(r)
        acc = torch.zeros(3)
        for t in range(6, -1, -1):
            acc = r[:, t] + gamma * acc
            oracle[:, t] = acc
        assert torch.allclose(out, oracle, atol=1e-6)


def test_gae_hand():
    rewards = torch.tensor([[1.0, 0.0]])
    values = torch.tensor([[0.5, 0.25]])
    adv, ret = F.gae(rewards, values, gamma=1.0, lam=0.95)
    # t=1: delta = 0 + 0 - 0.25 = -0.25 ; adv1 = -0.25
    # t=0: delta = 1 + 0.25 - 0.5 = 0.75 ; adv0 = 0.75 + 0.95*(-0.25) = 0.5125
    assert adv[0, 1] == pytest.approx(-0.25)
    assert adv[0, 0] == pytest.approx(0.5125)
    assert torch.allclose(ret, adv + values)


def test_sparse_reward_and_kl_shaped():
    mask = torch.ones(2, 4)
    scores = torch.tensor([2.0, -1.0])
    eos = torch.tensor([3, 1])
    r = F.sparse_reward_at_eos(scores, mask, eos)
    assert r[0].tolist() == [0, 0, 0, 2.0]
    assert r[1].tolist() == [0, -1.0, 0, 0]
    lp = torch.zeros(2, 4)
    ref = torch.full((2, 4), 0.5)
    shaped = F.kl_shaped_rewards(scores, lp, ref, mask, eos, kl_coef=0.1)
    # -0.1*(0-0.5) = 0.05 per token, plus score at eos
    assert shaped[0, 0] == pytest.approx(0.05)
    assert shaped[0, 3] == pytest.approx(2.05)


def test_ppo_clip_loss_hand():
    old = torch.zeros(1, 2)
    new = torch.log(torch.tensor([[1.5, 0.5]]))  # ratios 1.5, 0.5
    adv = torch.ones(1, 2)
    mask = torch.ones(1, 2)
    loss, st = F.ppo_clip_token_loss(new, old, adv, mask, cliprange=0.2)
    # token0: max(-1.5, -1.2) = -1.2 (clipped); token1: max(-0.5,-0.8) = -0.5
    assert float(loss) == pytest.approx((-1.2 - 0.5) / 2, rel=1e-5)
    assert float(st["pg_clipfrac"]) == pytest.approx(0.5)


def test_rloo_sequence_loss_hand():
    mask = torch.ones(1, 2)
    old = torch.zeros(1, 2)
    new = torch.log(torch.tensor([[1.1, 1.1]]))  # seq ratio 1.21
    adv = torch.tensor([1.0])
    loss, st = F.rloo_sequence_loss(new, old, adv, mask, cliprange=0.2)
    assert float(loss) == pytest.approx(-1.2, rel=1e-5)  # clipped at 1.2


def test_k3_kl_nonnegative():
    torch.manual_seed(1)
    new, ref = torch.randn(100), torch.randn(100)
    k3 = F.k3_kl_penalty(new, ref, kl_coef=1.0)
    assert (k3 >= -1e-6).all()
    assert float(F.k3_kl_penalty(torch.ones(3), torch.ones(3), 1.0).sum()) == pytest.approx(0.0)


def test_value_clip_loss():
    vpred = torch.tensor([[1.0]])
    old = torch.tensor([[0.0]])
    ret = torch.tensor([[2.0]])
    mask = torch.ones(1, 1)
    loss, frac = F.value_clip_loss(vpred, old, ret, mask, cliprange_value=0.2)
    # clipped pred = 0.2 -> (0.2-2)^2 = 3.24 > (1-2)^2=1 -> max -> 0.5*3.24
    assert float(loss) == pytest.approx(1.62)
    assert float(frac) == 1.0


def test_entropy_from_logits():
    logits = torch.tensor([[0.0, 0.0, 0.0, 0.0]])
    assert float(F.entropy_from_logits(logits)) == pytest.approx(math.log(4), rel=1e-5)


def test_missing_eos_penalty():
    scores = torch.tensor([1.0, 1.0])
    resp = torch.tensor([[3, 7], [3, 4]])
    out = F.missing_eos_penalty(scores, resp, eos_token_id=7, penalty=0.4)
    assert out.tolist() == pytest.approx([1.0, 0.6])


def test_remax_advantage():
    assert F.remax_advantage(torch.tensor([2.0]), torch.tensor([0.5])).tolist() == [1.5]


def test_raft_nll():
    lp = torch.tensor([[-1.0, -2.0]])
    assert float(F.raft_nll_loss(lp, torch.ones(1, 2))) == pytest.approx(3.0)


def test_exact_div():
    assert F.exact_div(12, 3) == 4
    with pytest.raises(ValueError):
        F.exact_div(7, 2)


def test_random_keep_one_per_group():
    g = torch.Generator().manual_seed(0)
    idx = F.random_keep_one_per_group(5, 4, generator=g)
    assert len(idx) == 5
    for b, i in enumerate(idx.tolist()):
        assert b * 4 <= i < (b + 1) * 4
