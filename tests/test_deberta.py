"""DeBERTa-v3 reward model parity vs the transformers implementation
(VERDICT missing #4: real deberta-v3-large weights must be hostable).

transformers is used ONLY as a local numerics oracle — random weights are
created with its DebertaV2ForSequenceClassification, exported, imported
into our model through the HF key map, and outputs compared."""
import pytest
import torch

transformers = pytest.importorskip("transformers")

from nanorlhf_amd.models.deberta import (DebertaConfig, DebertaV3Reward,
                                         build_relative_position,
                                         make_log_bucket_position)


def make_pair(position_biased_input=False, buckets=8, seed=0, vocab=128):
    from transformers import DebertaV2Config, DebertaV2ForSequenceClassification
    torch.manual_seed(seed)
    hf_cfg = DebertaV2Config(
        vocab_size=vocab, hidden_size=32, num_hidden_layers=2,
        num_attention_heads=2, intermediate_size=64,
        max_position_embeddings=64, position_buckets=buckets,
        relative_attention=True, pos_att_type=["p2c", "c2p"],
        norm_rel_ebd="layer_norm", share_att_key=True, num_labels=1,
        max_relative_positions=-1, position_biased_input=position_biased_input,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
        pooler_dropout=0.0, pooler_hidden_act="gelu")
    hf = DebertaV2ForSequenceClassification(hf_cfg).eval()
    cfg = DebertaConfig(vocab_size=vocab, hidden_size=32, num_layers=2,
                        num_heads=2, intermediate_size=64,
                        max_position_embeddings=64, position_buckets=buckets,
                        position_biased_input=position_biased_input,
                        num_labels=1)
    ours = DebertaV3Reward(cfg).eval()
    ours.load_hf_state({k: v.detach().clone() for k, v in hf.state_dict().items()})
    return hf, ours


@pytest.mark.parametrize("pbi", [False, True])
def test_deberta_matches_transformers(pbi):
    hf, ours = make_pair(position_biased_input=pbi)
    torch.manual_seed(1)
    ids = torch.randint(0, 128, (3, 20))
    mask = torch.ones(3, 20, dtype=torch.long)
    mask[0, 12:] = 0
    mask[2, 5:] = 0
    with torch.no_grad():
        want = hf(input_ids=ids, attention_mask=mask).logits
        got = ours(ids, mask)
    assert torch.allclose(got, want, atol=1e-4), float((got - want).abs().max())


def test_deberta_long_seq_buckets():
    """Sequence longer than the bucket span exercises the log buckets."""
    hf, ours = make_pair(buckets=8, seed=3)
    torch.manual_seed(2)
    ids = torch.randint(0, 128, (2, 60))
    mask = torch.ones(2, 60, dtype=torch.long)
    with torch.no_grad():
        want = hf(input_ids=ids, attention_mask=mask).logits
        got = ours(ids, mask)
    assert torch.allclose(got, want, atol=1e-4), float((got - want).abs().max())


def test_log_bucket_position_matches_hf():
    from transformers.models.deberta_v2.modeling_deberta_v2 import \
        make_log_bucket_position as hf_mlbp
    rel = (torch.arange(40)[:, None] - torch.arange(40)[None, :])
    ours = make_log_bucket_position(rel, 8, 64)
    want = hf_mlbp(rel, 8, 64).long()
    assert torch.equal(ours, want)


def test_build_relative_position_shape():
    r = build_relative_position(16, 16, 8, 64, "cpu")
    assert r.shape == (16, 16) and int(r[0, 0]) == 0


def test_deberta_reward_closure():
    """String-contract DebertaReward over a tiny tokenizer."""
    from nanorlhf_amd.data.tokenizer import load_tokenizer, make_tiny_tokenizer
    from nanorlhf_amd.models.deberta import DebertaReward
    import tempfile
    _, ours = make_pair(seed=5, vocab=2048)
    with tempfile.TemporaryDirectory() as d:
        make_tiny_tokenizer(d)
        tok = load_tokenizer(d)
        rm = DebertaReward(ours, tok, "cpu", batch_size=2)
        scores = rm(["hello there", "What is 2+2?", "You are Qwen"])
    assert scores.shape == (3,) and torch.isfinite(scores).all()


def test_deberta_hf_roundtrip_via_safetensors(tmp_path):
    """Full file-level path: save an HF-layout checkpoint, import with
    from_pretrained, outputs equal."""
    import json
    import os

    from safetensors.torch import save_file
    hf, ours = make_pair(seed=7)
    cfg_json = {
        "model_type": "deberta-v2", "vocab_size": 128, "hidden_size": 32,
        "num_hidden_layers": 2, "num_attention_heads": 2,
        "intermediate_size": 64, "max_position_embeddings": 64,
        "position_buckets": 8, "max_relative_positions": -1,
        "layer_norm_eps": 1e-7, "position_biased_input": False,
        "id2label": {"0": "LABEL_0"}, "pooler_hidden_act": "gelu",
    }
    with open(os.path.join(tmp_path, "config.json"), "w") as f:
        json.dump(cfg_json, f)
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
              os.path.join(tmp_path, "model.safetensors"))
    loaded = DebertaV3Reward.from_pretrained(str(tmp_path)).eval()
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        a = loaded(ids)
        b = ours(ids)
    assert torch.allclose(a, b, atol=1e-6)
