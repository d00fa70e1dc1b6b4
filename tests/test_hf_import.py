"""HF-checkpoint weight import + tokenizer seam (VERDICT round-1 item #1).

Round-trip tests on synthetic safetensors files: build an HF-layout Qwen2
checkpoint from random tensors, import onto the fused layout, and verify
(a) every projection landed in the right fused rows, (b) forward logits of
the imported model equal a hand-composed per-projection forward, and
(c) export → import is the identity.  Tokenizer tests exercise the
[PAD]-addition, chat-scaffold prompt prep and the StringReward decode
boundary with a locally-built byte-level BPE tokenizer (no network).
"""
import json
import os

import pytest
import torch

from nanorlhf_amd.models import hf_import
from nanorlhf_amd.models.config import ModelConfig
from nanorlhf_amd.models.qwen2 import CausalLM, pack_sequences


def tiny_hf_cfg(tmp_path, tie=True):
    cfg = {
        "architectures": ["Qwen2ForCausalLM"],
        "model_type": "qwen2",
        "vocab_size": 256,
        "hidden_size": 64,
        "num_hidden_layers": 2,
        "num_attention_heads": 4,
        "num_key_value_heads": 2,
        "head_dim": 16,
        "intermediate_size": 128,
        "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0,
        "max_position_embeddings": 512,
        "tie_word_embeddings": tie,
        "torch_dtype": "float32",
    }
    with open(os.path.join(tmp_path, "config.json"), "w") as f:
        json.dump(cfg, f)
    return cfg


def make_hf_state(cfg, seed=0):
    g = torch.Generator().manual_seed(seed)
    H, I = cfg["hidden_size"], cfg["intermediate_size"]
    D = cfg["head_dim"]
    nq, nkv = cfg["num_attention_heads"], cfg["num_key_value_heads"]
    V = cfg["vocab_size"]

    def r(*shape):
        return torch.randn(*shape, generator=g) * 0.05

    sd = {"model.embed_tokens.weight": r(V, H), "model.norm.weight": 1 + 0.01 * r(H)}
    for i in range(cfg["num_hidden_layers"]):
        p = f"model.layers.{i}"
        sd[f"{p}.input_layernorm.weight"] = 1 + 0.01 * r(H)
        sd[f"{p}.post_attention_layernorm.weight"] = 1 + 0.01 * r(H)
        sd[f"{p}.self_attn.q_proj.weight"] = r(nq * D, H)
        sd[f"{p}.self_attn.q_proj.bias"] = r(nq * D)
        sd[f"{p}.self_attn.k_proj.weight"] = r(nkv * D, H)
        sd[f"{p}.self_attn.k_proj.bias"] = r(nkv * D)
        sd[f"{p}.self_attn.v_proj.weight"] = r(nkv * D, H)
        sd[f"{p}.self_attn.v_proj.bias"] = r(nkv * D)
        sd[f"{p}.self_attn.o_proj.weight"] = r(H, nq * D)
        sd[f"{p}.mlp.gate_proj.weight"] = r(I, H)
        sd[f"{p}.mlp.up_proj.weight"] = r(I, H)
        sd[f"{p}.mlp.down_proj.weight"] = r(H, I)
    if not cfg["tie_word_embeddings"]:
        sd["lm_head.weight"] = r(V, H)
    return sd


def write_checkpoint(tmp_path, tie=True, shards=1, seed=0):
    from safetensors.torch import save_file
    cfg = tiny_hf_cfg(tmp_path, tie=tie)
    sd = make_hf_state(cfg, seed=seed)
    if shards == 1:
        save_file(sd, os.path.join(tmp_path, "model.safetensors"))
    else:
        keys = sorted(sd)
        per = (len(keys) + shards - 1) // shards
        weight_map = {}
        for s in range(shards):
            part = {k: sd[k] for k in keys[s * per:(s + 1) * per]}
            name = f"model-{s + 1:05d}-of-{shards:05d}.safetensors"
            save_file(part, os.path.join(tmp_path, name))
            weight_map.update({k: name for k in part})
        with open(os.path.join(tmp_path, "model.safetensors.index.json"), "w") as f:
            json.dump({"weight_map": weight_map}, f)
    return cfg, sd


def test_config_from_hf(tmp_path):
    tiny_hf_cfg(tmp_path)
    cfg = hf_import.config_from_hf(str(tmp_path))
    assert cfg.hidden_size == 64 and cfg.num_kv_heads == 2 and cfg.head_dim == 16
    assert cfg.tie_word_embeddings and cfg.qkv_bias


@pytest.mark.parametrize("tie,shards", [(True, 1), (False, 3)])
def test_fused_rows_land_correctly(tmp_path, tie, shards):
    cfg_d, hf_sd = write_checkpoint(tmp_path, tie=tie, shards=shards)
    model = hf_import.load_pretrained(str(tmp_path))
    sd = model.state_dict()
    D, nq, nkv = 16, 4, 2
    qkv = sd["model.layers.0.self_attn.qkv_proj.weight"]
    assert torch.equal(qkv[: nq * D], hf_sd["model.layers.0.self_attn.q_proj.weight"])
    assert torch.equal(qkv[nq * D: nq * D + nkv * D],
                       hf_sd["model.layers.0.self_attn.k_proj.weight"])
    assert torch.equal(qkv[nq * D + nkv * D:],
                       hf_sd["model.layers.0.self_attn.v_proj.weight"])
    gu = sd["model.layers.1.mlp.gate_up_proj.weight"]
    assert torch.equal(gu[:128], hf_sd["model.layers.1.mlp.gate_proj.weight"])
    assert torch.equal(gu[128:], hf_sd["model.layers.1.mlp.up_proj.weight"])
    if tie:
        assert torch.equal(model.lm_head_weight, hf_sd["model.embed_tokens.weight"])
    else:
        assert torch.equal(model.lm_head_weight, hf_sd["lm_head.weight"])


def test_imported_forward_matches_reference_composition(tmp_path):
    """Logits of the imported fused model == a straight per-projection
    torch composition of the same HF tensors (fp32, CPU)."""
    cfg_d, hf = write_checkpoint(tmp_path, tie=True, seed=3)
    model = hf_import.load_pretrained(str(tmp_path)).eval()
    torch.manual_seed(0)
    seqs = [torch.randint(0, 256, (n,)) for n in (7, 12)]
    ids, cu, mx, pos = pack_sequences(seqs)
    ctx = CausalLM.train_ctx(cu, mx, pos)
    with torch.no_grad():
        ours = model.logits(model(ids, ctx))

    # independent composition from the raw HF tensors
    import torch.nn.functional as TF
    from nanorlhf_amd.ops import build_rope_cache
    from nanorlhf_amd.ops.rope import _rope_ref

    def rms(x, w, eps=1e-6):
        v = x.float()
        return (v * torch.rsqrt(v.pow(2).mean(-1, keepdim=True) + eps) * w).to(x.dtype)

    table = build_rope_cache(16, 512, 1e4)
    outs = []
    for s in seqs:
        x = hf["model.embed_tokens.weight"][s]
        T = x.shape[0]
        p = torch.arange(T)
        for i in range(2):
            pre = f"model.layers.{i}"
            h = rms(x, hf[f"{pre}.input_layernorm.weight"])
            q = (TF.linear(h, hf[f"{pre}.self_attn.q_proj.weight"],
                           hf[f"{pre}.self_attn.q_proj.bias"])).view(T, 4, 16)
            k = (TF.linear(h, hf[f"{pre}.self_attn.k_proj.weight"],
                           hf[f"{pre}.self_attn.k_proj.bias"])).view(T, 2, 16)
            v = (TF.linear(h, hf[f"{pre}.self_attn.v_proj.weight"],
                           hf[f"{pre}.self_attn.v_proj.bias"])).view(T, 2, 16)
            q = _rope_ref(q, table, p)
            k = _rope_ref(k, table, p)
            kr = k.repeat_interleave(2, dim=1)
            vr = v.repeat_interleave(2, dim=1)
            att = torch.einsum("thd,shd->hts", q.float(), kr.float()) / 4.0
            mask = torch.triu(torch.ones(T, T, dtype=torch.bool), 1)
            att = att.masked_fill(mask, float("-inf")).softmax(-1)
            o = torch.einsum("hts,shd->thd", att, vr.float()).reshape(T, -1)
            x = x + TF.linear(o.to(x.dtype), hf[f"{pre}.self_attn.o_proj.weight"])
            h = rms(x, hf[f"{pre}.post_attention_layernorm.weight"])
            gate = TF.linear(h, hf[f"{pre}.mlp.gate_proj.weight"])
            up = TF.linear(h, hf[f"{pre}.mlp.up_proj.weight"])
            x = x + TF.linear(TF.silu(gate) * up, hf[f"{pre}.mlp.down_proj.weight"])
        x = rms(x, hf["model.norm.weight"])
        outs.append(TF.linear(x, hf["model.embed_tokens.weight"]))
    want = torch.cat(outs)
    assert torch.allclose(ours, want, atol=2e-4), float((ours - want).abs().max())


def test_export_import_roundtrip(tmp_path):
    cfg = ModelConfig(vocab_size=128, hidden_size=32, num_layers=2, num_heads=2,
                      num_kv_heads=1, head_dim=16, intermediate_size=64,
                      rope_theta=1e4, max_position=256, dtype="float32",
                      tie_word_embeddings=False)
    torch.manual_seed(1)
    m = CausalLM(cfg)
    out_dir = os.path.join(tmp_path, "export")
    hf_import.save_hf_checkpoint(m, out_dir)
    m2 = hf_import.load_pretrained(out_dir)
    for k, v in m.state_dict().items():
        assert torch.equal(v, m2.state_dict()[k]), k


def test_import_rejects_missing_tensor(tmp_path):
    from safetensors.torch import save_file
    cfg = tiny_hf_cfg(tmp_path)
    sd = make_hf_state(cfg)
    del sd["model.layers.1.mlp.down_proj.weight"]
    save_file(sd, os.path.join(tmp_path, "model.safetensors"))
    with pytest.raises(KeyError, match="down_proj"):
        hf_import.load_pretrained(str(tmp_path))


# --------------------------------------------------------------------------
# tokenizer seam
# --------------------------------------------------------------------------

@pytest.fixture(scope="module")
def tiny_tok(tmp_path_factory):
    from nanorlhf_amd.data.tokenizer import load_tokenizer, make_tiny_tokenizer
    d = tmp_path_factory.mktemp("tok")
    make_tiny_tokenizer(str(d))
    return load_tokenizer(str(d))


def test_tokenizer_pad_added(tiny_tok):
    assert tiny_tok.pad_token == "[PAD]"
    assert tiny_tok.pad_token_id is not None
    assert tiny_tok.chat_template is not None


def test_prepare_hh_prompts(tiny_tok):
    from nanorlhf_amd.data.tokenizer import prepare_hh_prompts
    recs = [{"chosen": "Human: What is your name? Assistant: I am Qwen."},
            {"chosen": "Human: hello there Assistant: hi"}]
    prompts = prepare_hh_prompts(recs, tiny_tok)
    assert len(prompts) == 2
    text = tiny_tok.decode(prompts[0])
    assert "What is your name?" in text and "<|im_start|>" in text
    assert text.rstrip().endswith("<|im_start|>assistant") or "assistant" in text


def test_prepare_math_prompts(tiny_tok):
    from nanorlhf_amd.data.tokenizer import prepare_math_prompts
    recs = [{"question": "What is 2+2?", "answer": "4"}]
    prompts, answers = prepare_math_prompts(recs, tiny_tok)
    assert len(prompts) == 1 and len(answers) == 1
    (text, gold), = answers.items()
    assert gold == "4" and "What is 2+2?" in text


def test_string_reward_contract(tiny_tok):
    from nanorlhf_amd.rewards import StringReward
    seen = {}

    def fn(texts):
        seen["texts"] = texts
        return torch.tensor([float(len(t)) for t in texts])

    sr = StringReward(fn, tiny_tok, mode="strings")
    rows = [tiny_tok("hello there")["input_ids"], tiny_tok("hi")["input_ids"]]
    out = sr(rows)
    assert out.shape == (2,) and out.dtype == torch.float32
    assert "hello there" in seen["texts"][0]


def test_string_reward_r1_mode(tiny_tok):
    from nanorlhf_amd.rewards import StringReward
    got = {}

    def fn(texts, responses_ids, tokenizer):
        got["responses"] = responses_ids
        got["tok"] = tokenizer
        return [1.0] * len(texts)

    sr = StringReward(fn, tiny_tok, mode="r1")
    rows = [tiny_tok("What is 2+2?")["input_ids"]]
    out = sr(rows, responses=[[5, 6]])
    assert float(out[0]) == 1.0 and got["responses"] == [[5, 6]]
    assert got["tok"] is tiny_tok


def test_trainer_with_string_reward(tmp_path, tiny_tok):
    """End-to-end: tiny model + StringReward through one training update —
    the real-model seam minus the (undownloadable) real weights."""
    from nanorlhf_amd.algos import grpo
    from nanorlhf_amd.models.qwen2 import CausalLM
    from nanorlhf_amd.rewards import StringReward

    vocab = max(1024, len(tiny_tok) + 1)
    policy = CausalLM.from_preset("tiny", vocab_size=vocab)
    ref = CausalLM.from_preset("tiny", vocab_size=vocab)
    cfg = grpo.GRPOConfig(
        output_dir=str(tmp_path / "out"), total_episodes=8, sample_n=2,
        per_device_train_batch_size=2, gradient_accumulation_steps=1,
        num_mini_batches=1, response_length=8, temperature=1.0,
        stop_token_id=None, pad_token_id=tiny_tok.pad_token_id,
        use_lora=False, save_steps=0, dtype="float32",
        score_token_budget=512, train_token_budget=0, kv_pool_tokens=4096,
        gradient_checkpointing=False)

    def reward(texts):
        return torch.tensor([1.0 if "e" in t else 0.0 for t in texts])

    prompts = [tiny_tok(s)["input_ids"] for s in
               ["hello there", "What is 2+2?", "You are Qwen", "hi"]]
    tr = grpo.make_trainer(cfg, policy, ref, StringReward(reward, tiny_tok),
                           prompts, device=torch.device("cpu"))
    tr.train(num_updates=1)
    assert tr.global_step == 1


def test_r1_real_data_seam_end_to_end(tmp_path, tiny_tok):
    """The r1 real-data path (examples/grpo_r1.py::run_real): tiny HF
    checkpoint + tokenizer files + (question, answer) jsonl through one
    sparse-GRPO update with the boxed-answer rule reward."""
    import importlib.util
    import json as _json

    from nanorlhf_amd.models.config import ModelConfig
    from nanorlhf_amd.models.hf_import import save_hf_checkpoint
    from nanorlhf_amd.models.qwen2 import CausalLM as _CLM

    # tiny HF-layout checkpoint with the tokenizer files beside it
    cfg = ModelConfig(vocab_size=max(2048, len(tiny_tok) + 8), hidden_size=64,
                      num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
                      intermediate_size=128, rope_theta=1e4, max_position=512,
                      dtype="float32", tie_word_embeddings=True)
    torch.manual_seed(0)
    save_hf_checkpoint(_CLM(cfg), str(tmp_path / "ckpt"))
    tiny_tok.save_pretrained(str(tmp_path / "ckpt"))

    data = tmp_path / "math.jsonl"
    with open(data, "w") as f:
        for q, a in [("What is 2+2?", "4"), ("What is 3*3?", "9"),
                     ("What is 10-4?", "6"), ("What is 1+1?", "2")]:
            f.write(_json.dumps({"question": q, "answer": a}) + "\n")

    spec = importlib.util.spec_from_file_location(
        "grpo_r1_example",
        os.path.join(os.path.dirname(__file__), "..", "examples", "grpo_r1.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    from nanorlhf_amd.algos.grpo import GRPOConfig
    run_cfg = GRPOConfig(
        output_dir=str(tmp_path / "out"), model_preset="tiny", dtype="float32",
        use_lora=True, lora_r=4, lora_alpha=8, per_device_train_batch_size=2,
        gradient_accumulation_steps=1, num_mini_batches=1, total_episodes=4,
        sample_n=2, response_length=8, temperature=1.0, sparse_filter=True,
        score_token_budget=2048, train_token_budget=0, kv_pool_tokens=4096,
        gradient_checkpointing=False, save_steps=0, kl_coef=0.0,
        missing_eos_penalty=None)
    tr = mod.run_real(str(tmp_path / "ckpt"), str(data), cfg=run_cfg,
                      num_updates=1)
    assert tr.global_step == 1
    # checkpoint written by trainer.save()
    assert any(p.name.startswith("checkpoint-")
               for p in (tmp_path / "out").iterdir())


def test_export_merged_hf_roundtrip(tmp_path):
    """Trained-model deploy path: LoRA-wrapped model → merged HF checkpoint
    → reload → logits equal the live merged forward."""
    from nanorlhf_amd.models.lora import LoraConfig, apply_lora
    torch.manual_seed(2)
    cfg = ModelConfig(vocab_size=256, hidden_size=64, num_layers=2, num_heads=4,
                      num_kv_heads=2, head_dim=16, intermediate_size=128,
                      rope_theta=1e4, max_position=256, dtype="float32",
                      tie_word_embeddings=True)
    m = CausalLM(cfg)
    apply_lora(m, LoraConfig(r=4, alpha=8))
    # give the adapters non-zero effect
    with torch.no_grad():
        for n, p in m.named_parameters():
            if "lora_B" in n:
                p.normal_(0, 0.05)
    seqs = [torch.randint(0, 256, (9,))]
    ids, cu, mx, pos = pack_sequences(seqs)
    ctx = CausalLM.train_ctx(cu, mx, pos)
    with torch.no_grad():
        want = m.logits(m(ids, ctx))
    out_dir = str(tmp_path / "deploy")
    hf_import.export_merged_hf(m, out_dir)
    # exported model is plain (no adapters) and reproduces the merged logits
    m2 = hf_import.load_pretrained(out_dir).eval()
    assert not any("lora" in k for k in m2.state_dict())
    with torch.no_grad():
        got = m2.logits(m2(ids, ctx))
    assert torch.allclose(got, want, atol=1e-4), float((got - want).abs().max())
    # the ORIGINAL model is untouched (still adapter-wrapped)
    assert any("lora_A" in k for k, _ in m.named_parameters())
