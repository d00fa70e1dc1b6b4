"""CPU model tests: shapes, packing, GQA, scalar heads, LoRA."""
import torch

from nanorlhf_amd.models import (AttnContext, CausalLM, LoraConfig, ScalarHeadModel,
                                 apply_lora, get_config, pack_sequences)
from nanorlhf_amd.models.lora import LoRALinear, merge_for_rollout, unmerge
from nanorlhf_amd.ops import build_rope_cache, rope_apply


def _fwd(model, seqs):
    ids, cu, mx, pos = pack_sequences(seqs)
    return model(ids, CausalLM.train_ctx(cu, mx, pos))


def test_packed_forward_matches_per_sequence():
    torch.manual_seed(0)
    m = CausalLM.from_preset("tiny").eval()
    seqs = [torch.randint(2, 1000, (n,)) for n in (9, 5, 13)]
    h_all = _fwd(m, seqs)
    off = 0
    for s in seqs:
        h_one = _fwd(m, [s])
        assert torch.allclose(h_all[off:off + len(s)], h_one, atol=1e-4)
        off += len(s)


def test_causal_dependence():
    """Changing a later token must not change earlier hidden states."""
    torch.manual_seed(0)
    m = CausalLM.from_preset("tiny").eval()
    s = torch.randint(2, 1000, (10,))
    h1 = _fwd(m, [s])
    s2 = s.clone()
    s2[-1] = (s2[-1] + 1) % 1000
    h2 = _fwd(m, [s2])
    assert torch.allclose(h1[:-1], h2[:-1], atol=1e-5)
    assert not torch.allclose(h1[-1], h2[-1], atol=1e-5)


def test_rope_orthogonal_and_inverse():
    table = build_rope_cache(32, 64, theta=1e4)
    x = torch.randn(5, 2, 32)
    pos = torch.arange(5)
    y = rope_apply(x, table, pos)
    # norms preserved (rotation)
    assert torch.allclose(x.norm(dim=-1), y.norm(dim=-1), atol=1e-4)
    # position 0 is identity
    assert torch.allclose(y[0], x[0], atol=1e-6)


def test_scalar_head_sequence_scores():
    torch.manual_seed(0)
    rm = ScalarHeadModel.from_preset("rm-tiny").eval()
    seqs = [torch.randint(2, 1000, (n,)) for n in (6, 11)]
    ids, cu, mx, _ = pack_sequences(seqs)
    s = rm.sequence_scores(ids, cu, mx)
    assert s.shape == (2,)
    v = rm.token_values(ids, cu, mx)
    assert v.shape == (ids.numel(),)


def test_bidirectional_rm_sees_future():
    torch.manual_seed(0)
    rm = ScalarHeadModel.from_preset("rm-tiny").eval()
    assert rm.cfg.bidirectional
    s = torch.randint(2, 1000, (8,))
    ids, cu, mx, _ = pack_sequences([s])
    v1 = rm.token_values.__wrapped__ if hasattr(rm.token_values, "__wrapped__") else None
    # change LAST token; FIRST token's score should change under bidirectional attn
    from nanorlhf_amd.models.qwen2 import AttnContext, make_positions
    ctx = AttnContext(mode="train", positions=make_positions(cu), cu_seqlens=cu, max_seqlen=mx)
    out1 = rm(ids, ctx)
    s2 = s.clone(); s2[-1] = (s2[-1] + 1) % 1000
    ids2, cu2, mx2, _ = pack_sequences([s2])
    ctx2 = AttnContext(mode="train", positions=make_positions(cu2), cu_seqlens=cu2, max_seqlen=mx2)
    out2 = rm(ids2, ctx2)
    assert not torch.allclose(out1[0], out2[0], atol=1e-6)


def test_lora_apply_merge_unmerge():
    torch.manual_seed(0)
    m = CausalLM.from_preset("tiny")
    apply_lora(m, LoraConfig(r=4, alpha=8))
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert any("lora_A" in n for n in trainable)
    assert any("embed_tokens" in n for n in trainable)
    assert not any(n.endswith("base.weight") for n in trainable)
    s = torch.randint(2, 1000, (7,))
    m.eval()
    h0 = _fwd(m, [s])
    # B is zero-init → merged == unmerged at init
    merge_for_rollout(m)
    h1 = _fwd(m, [s])
    unmerge(m)
    assert torch.allclose(h0, h1, atol=1e-5)
    # after perturbing B, lora path changes output
    for mod in m.modules():
        if isinstance(mod, LoRALinear):
            torch.nn.init.normal_(mod.lora_B, std=0.1)
    h2 = _fwd(m, [s])
    assert not torch.allclose(h0, h2, atol=1e-4)
    # merged path equals unmerged compute path
    merge_for_rollout(m)
    h3 = _fwd(m, [s])
    assert torch.allclose(h2, h3, atol=1e-4)
    unmerge(m)


def test_tied_and_untied_lm_head():
    tied = CausalLM(get_config("tiny", tie_word_embeddings=True))
    untied = CausalLM(get_config("tiny", tie_word_embeddings=False))
    assert tied.lm_head is None
    assert tied.lm_head_weight.data_ptr() == tied.model.embed_tokens.weight.data_ptr()
    assert untied.lm_head is not None
