"""GPU end-to-end: sampler engine on the HIP path vs full-forward oracle,
and a short GRPO training run on a small bf16 model."""
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda"


def _small_model(seed=0, layers=2, vocab=4096):
    from nanorlhf_amd.models import CausalLM, get_config
    torch.manual_seed(seed)
    cfg = get_config("qwen2.5-1.5b", num_layers=layers, vocab_size=vocab)
    return CausalLM(cfg).to(DEV).to(torch.bfloat16)


def test_sampler_gpu_greedy_matches_oracle():
    from nanorlhf_amd.models import CausalLM, pack_sequences
    from nanorlhf_amd.sampler import SamplerEngine, SamplingParams
    m = _small_model().eval()
    eng = SamplerEngine(m, kv_pool_tokens=16384, page_size=16)
    prompts = [torch.randint(2, 4096, (n,)).tolist() for n in (5, 21, 40)]
    params = SamplingParams(n=1, temperature=0.0, top_p=1.0, max_tokens=12, seed=1)
    out = eng.generate(prompts, params)

    def oracle_follow(prompt, got, steps):
        """Replay the ENGINE's tokens through full recompute; at each step
        report the oracle argmax and the margin between the engine's token
        and the best token.  A bookkeeping bug shows up as disagreement at a
        CONFIDENT step; bf16 tie-flips only at near-zero margins."""
        toks = list(prompt)
        report = []
        for j in range(steps):
            ids, cu, mx, pos = pack_sequences([torch.tensor(toks)], device=DEV)
            h = m(ids, CausalLM.train_ctx(cu, mx, pos))
            logits = m.logits(h[-1:]).float()[0]
            best = int(logits.argmax())
            margin = float(logits[best] - logits[got[j]])
            report.append((got[j], best, margin))
            toks.append(got[j])  # follow the engine's trajectory
        return report

    for i, p in enumerate(prompts):
        got = out[i].tolist()
        rep = oracle_follow(p, got, 12)
        for j, (g, best, margin) in enumerate(rep):
            # engine's pick must be the oracle's argmax, or within bf16 noise
            assert g == best or margin < 0.08, (i, j, rep)


def test_grpo_gpu_short_training():
    from nanorlhf_amd.algos import grpo
    from nanorlhf_amd.algos.grpo import GRPOConfig
    from nanorlhf_amd.data import hh_shaped_prompts
    from nanorlhf_amd.models import CausalLM, get_config

    torch.manual_seed(0)
    cfg_m = get_config("qwen2.5-1.5b", num_layers=2, vocab_size=4096)
    policy = CausalLM(cfg_m)
    ref = CausalLM(cfg_m)
    ref.load_state_dict(policy.state_dict())
    cfg = GRPOConfig(model_preset="custom", dtype="bfloat16", use_lora=True,
                     lora_r=8, lora_alpha=16, per_device_train_batch_size=2,
                     gradient_accumulation_steps=2, num_mini_batches=2,
                     total_episodes=16, sample_n=2, response_length=16,
                     temperature=1.0, stop_token_id=1,
                     output_dir="/tmp/nanorlhf_gpu_grpo",
                     gradient_checkpointing=True, score_token_budget=4096,
                     missing_eos_penalty=1.0)
    prompts = hh_shaped_prompts(16, 4096, min_len=8, max_len=24)

    def reward(seqs):
        return torch.tensor([(sum(s) + len(s)) % 5 - 2.0 for s in seqs])

    tr = grpo.make_trainer(cfg, policy, ref, reward, prompts)
    before = {n: p.clone() for n, p in tr.policy.named_parameters() if p.requires_grad}
    tr.train(num_updates=1)
    moved = any(not torch.equal(before[n], p)
                for n, p in tr.policy.named_parameters() if n in before)
    assert moved
    m = tr._last_metrics
    assert all(torch.isfinite(torch.tensor(float(v))) for k, v in m.items()
               if isinstance(v, (int, float))), m


def test_offload_engine_roundtrip():
    """Pinned-host offload → restore must preserve forward results
    (the ref/reward-model shuttle path, utils/offload.py)."""
    import torch
    from nanorlhf_amd.models import CausalLM, pack_sequences
    from nanorlhf_amd.utils.offload import OffloadEngine
    m = _small_model(seed=3).eval()
    ids, cu, mx, pos = pack_sequences([torch.randint(2, 4096, (24,))], device=DEV)
    ctx = type(m).train_ctx(cu, mx, pos)
    with torch.no_grad():
        h0 = m(ids, ctx).clone()
    eng = OffloadEngine(torch.device(DEV), enabled=True)
    eng.model_to_host(m)
    eng.synchronize()
    assert all(p.device.type == "cpu" for p in m.parameters())
    eng.model_to_device(m)
    eng.synchronize()
    assert all(p.device.type == "cuda" for p in m.parameters())
    with torch.no_grad():
        h1 = m(ids, ctx)
    assert torch.equal(h0, h1)


def test_offload_orders_after_pending_compute():
    """model_to_host must see the RESULT of kernels still in flight on the
    compute stream (round-1 race: the side-stream copy launched without
    waiting).  Queue a long dependent-chain write into the params, then
    offload immediately with no manual sync."""
    from nanorlhf_amd.utils.offload import OffloadEngine
    m = torch.nn.Linear(4096, 4096, bias=False).to(DEV)
    eng = OffloadEngine(torch.device(DEV), enabled=True)
    with torch.no_grad():
        # long chain on the default stream that finally overwrites the weight
        x = torch.randn(4096, 4096, device=DEV)
        acc = torch.eye(4096, device=DEV)
        for _ in range(30):
            acc = acc @ x * 1e-3
        m.weight.copy_(acc)
        expect = acc.detach().cpu().clone()
    eng.model_to_host(m)   # no torch.cuda.synchronize() first — by design
    eng.synchronize()
    assert torch.equal(m.weight.detach(), expect)


def test_optimizer_state_offload_across_update():
    """cfg.offload_optimizer path at 7B geometry: the
    host↔device shuttle of optimizer state must be a BITWISE round trip
    (same run — cross-run bitwise equality is not guaranteed because some
    backward kernels accumulate with atomics), and training must continue
    finite afterwards."""
    from nanorlhf_amd import ops
    from nanorlhf_amd.models import CausalLM, get_config, pack_sequences
    from nanorlhf_amd.utils.offload import OffloadEngine

    torch.manual_seed(0)
    cfg = get_config("qwen2.5-7b", num_layers=2, vocab_size=8192)
    m = CausalLM(cfg).to(DEV).to(torch.bfloat16)
    opt = ops.FusedAdamW(m.parameters(), lr=1e-3)
    eng = OffloadEngine(torch.device(DEV), enabled=True)

    def one_step():
        ids, cu, mx, pos = pack_sequences([torch.randint(2, 8192, (48,))], device=DEV)
        h = m(ids, CausalLM.train_ctx(cu, mx, pos))
        lp, _ = ops.token_logprob_entropy(h, m.lm_head_weight,
                                          torch.roll(ids, -1), 1.0)
        (-lp.mean()).backward()
        opt.step()
        opt.zero_grad(set_to_none=True)

    for rt in range(2):
        one_step()
        before = {(id(p), k): v.detach().clone()
                  for p, st in opt.state.items()
                  for k, v in st.items() if torch.is_tensor(v)}
        eng.optimizer_state_to(opt, "cpu")
        eng.synchronize()
        assert all(v.device.type == "cpu"
                   for st in opt.state.values()
                   for v in st.values() if torch.is_tensor(v))
        eng.optimizer_state_to(opt, DEV)
        eng.join_compute()
        torch.cuda.synchronize()
        after = {(id(p), k): v
                 for p, st in opt.state.items()
                 for k, v in st.items() if torch.is_tensor(v)}
        assert set(before) == set(after)
        for key in before:
            assert after[key].device.type == "cuda"
            assert torch.equal(before[key], after[key].to(before[key].device)), key
    one_step()
    for p in m.parameters():
        assert torch.isfinite(p.detach().float()).all()


def test_trainer_offload_optimizer_knob():
    """offload_optimizer=True wired through a real trainer update (the
    round-1 dead knob)."""
    from nanorlhf_amd.algos import grpo
    from nanorlhf_amd.algos.grpo import GRPOConfig
    from nanorlhf_amd.data import hh_shaped_prompts
    from nanorlhf_amd.models import CausalLM, get_config

    torch.manual_seed(0)
    cfg_m = get_config("qwen2.5-1.5b", num_layers=2, vocab_size=4096)
    policy = CausalLM(cfg_m)
    ref = CausalLM(cfg_m)
    ref.load_state_dict(policy.state_dict())
    cfg = GRPOConfig(model_preset="custom", dtype="bfloat16", use_lora=True,
                     lora_r=8, per_device_train_batch_size=2,
                     gradient_accumulation_steps=2, num_mini_batches=2,
                     total_episodes=16, sample_n=2, response_length=16,
                     temperature=1.0, stop_token_id=1,
                     output_dir="/tmp/nanorlhf_gpu_offopt",
                     score_token_budget=4096, offload_optimizer=True)
    prompts = hh_shaped_prompts(16, 4096, min_len=8, max_len=24)
    tr = grpo.make_trainer(cfg, policy, ref,
                           lambda seqs: torch.tensor([float(len(s) % 3) for s in seqs]),
                           prompts)
    tr.train(num_updates=2)
    assert tr.global_step == 2
    m = tr._last_metrics
    assert all(torch.isfinite(torch.tensor(float(v))) for k, v in m.items()
               if isinstance(v, (int, float))), m


def test_7b_geometry_small_depth():
    """Qwen2.5-7B geometry (28 q heads / 4 kv heads -> GQA G=7, untied
    lm_head) through rollout + fwd/bwd at reduced depth."""
    import torch
    from nanorlhf_amd import ops
    from nanorlhf_amd.models import CausalLM, get_config, pack_sequences
    from nanorlhf_amd.sampler import SamplerEngine, SamplingParams
    torch.manual_seed(0)
    cfg = get_config("qwen2.5-7b", num_layers=2, vocab_size=8192)
    m = CausalLM(cfg).to(DEV).to(torch.bfloat16)
    eng = SamplerEngine(m, kv_pool_tokens=16384)
    prompts = [torch.randint(2, 8192, (n,)).tolist() for n in (7, 30)]
    out = eng.generate(prompts, SamplingParams(n=2, temperature=0.8, top_p=0.95,
                                               max_tokens=8, seed=3))
    assert out.shape == (4, 8) and (out >= 0).all()
    ids, cu, mx, pos = pack_sequences([torch.randint(2, 8192, (40,))], device=DEV)
    h = m(ids, CausalLM.train_ctx(cu, mx, pos))
    lp, _ = ops.token_logprob_entropy(h, m.lm_head_weight, torch.roll(ids, -1), 1.0)
    (-lp.mean()).backward()
    g = m.model.layers[0].self_attn.qkv_proj.weight.grad
    assert g is not None and torch.isfinite(g.float()).all()


def test_deberta_gpu_matches_cpu():
    """DeBERTa-v3 RM on GPU bf16 vs the CPU fp32 reference (the real-RM
    scoring path exercised on device)."""
    from nanorlhf_amd.models.deberta import DebertaConfig, DebertaV3Reward
    torch.manual_seed(0)
    cfg = DebertaConfig(vocab_size=2048, hidden_size=128, num_layers=4,
                        num_heads=8, intermediate_size=256,
                        max_position_embeddings=128, position_buckets=32,
                        position_biased_input=False, num_labels=1)
    m = DebertaV3Reward(cfg).eval()
    ids = torch.randint(0, 2048, (4, 96))
    mask = torch.ones(4, 96, dtype=torch.long)
    mask[1, 60:] = 0
    with torch.no_grad():
        want = m(ids, mask)
        got = m.to(DEV).to(torch.bfloat16)(ids.to(DEV), mask.to(DEV))
    err = float((got.float().cpu() - want).abs().max() / (want.abs().max() + 1e-6))
    assert err < 0.08, err
