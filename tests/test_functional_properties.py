"""Property-based tests of the pure algorithm math (hypothesis)."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from nanorlhf_amd.algos import functional as F


@settings(max_examples=50, deadline=None)
@given(st.integers(2, 6), st.integers(2, 5), st.integers(0, 2**31 - 1))
def test_grpo_group_advantage_zero_mean(b, n, seed):
    g = torch.Generator().manual_seed(seed)
    scores = torch.randn(b * n, generator=g)
    adv = F.grpo_group_advantage(scores, n).view(b, n)
    # each non-degenerate group has ~zero mean after normalization
    for i in range(b):
        if float(scores.view(b, n)[i].std()) > 1e-6:
            assert abs(float(adv[i].mean())) < 1e-5


@settings(max_examples=50, deadline=None)
@given(st.integers(2, 6), st.integers(0, 2**31 - 1))
def test_rloo_advantage_sums_to_zero(n, seed):
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(3 * n, generator=g)
    adv = F.rloo_baseline_advantage(r, n).view(3, n)
    # leave-one-out advantages sum to zero within each group
    assert torch.allclose(adv.sum(1), torch.zeros(3), atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 5), st.integers(1, 12), st.floats(0.5, 1.0),
       st.integers(0, 2**31 - 1))
def test_reward_to_go_matches_loop(b, t, gamma, seed):
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(b, t, generator=g)
    out = F.reward_to_go(r, gamma)
    oracle = torch.zeros_like(r)
    acc = torch.zeros(b)
    for i in range(t - 1, -1, -1):
        acc = r[:, i] + gamma * acc
        oracle[:, i] = acc
    assert torch.allclose(out, oracle, atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 4), st.integers(2, 10), st.integers(0, 2**31 - 1))
def test_gae_lambda1_gamma1_telescopes(b, t, seed):
    """At gamma=lam=1, GAE telescopes: adv = reward-to-go - values."""
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(b, t, generator=g)
    v = torch.randn(b, t, generator=g)
    adv, ret = F.gae(r, v, gamma=1.0, lam=1.0)
    rtg = F.reward_to_go(r, 1.0)
    assert torch.allclose(adv, rtg - v, atol=1e-4)
    assert torch.allclose(ret, rtg, atol=1e-4)


@settings(max_examples=30, deadline=None)
@given(st.integers(2, 6), st.integers(3, 12), st.integers(0, 2**31 - 1))
def test_masked_whiten_moments(b, t, seed):
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(b, t, generator=g) * 3 + 1
    mask = (torch.rand(b, t, generator=g) > 0.3).float()
    if float(mask.sum()) < 2 or float(F.masked_var(v, mask)) < 1e-6:
        return
    w = F.masked_whiten(v, mask, shift_mean=True)
    assert abs(float(F.masked_mean(w, mask))) < 1e-3
    assert abs(float(F.masked_var(w, mask)) - 1.0) < 0.05


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 5), st.integers(0, 2**31 - 1))
def test_k3_nonnegative_and_zero_at_equal(n, seed):
    g = torch.Generator().manual_seed(seed)
    a = torch.randn(n, generator=g)
    b_ = torch.randn(n, generator=g)
    assert (F.k3_kl_penalty(a, b_, 1.0) >= -1e-6).all()
    assert torch.allclose(F.k3_kl_penalty(a, a, 1.0), torch.zeros(n), atol=1e-6)


@settings(max_examples=20, deadline=None)
@given(st.lists(st.integers(1, 200), min_size=1, max_size=40),
       st.integers(16, 400))
def test_create_batches_properties(lengths, budget):
    from nanorlhf_amd.data import create_batches
    buckets = create_batches(lengths, budget)
    seen = sorted(i for b in buckets for i in b)
    assert seen == list(range(len(lengths)))
    for b in buckets:
        assert len(b) == 1 or max(lengths[i] for i in b) * len(b) <= budget
