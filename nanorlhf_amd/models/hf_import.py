"""HF-checkpoint import: real Qwen2/Qwen2.5 weights onto the fused layout.

The reference trains real Qwen2.5-1.5B-Instruct loaded via transformers
(`/root/reference/GRPO/grpo.py:218-224`); this module is the MI355X-native
equivalent seam — it reads a local HF checkpoint directory (config.json +
*.safetensors [+ model.safetensors.index.json for shards]) and maps the
per-projection tensors onto this package's fused qkv_proj / gate_up_proj
layout (models/qwen2.py).  No transformers modeling code is involved; only
safetensors file reading.

RoPE convention note: both HF Qwen2 and ops/rope.py use the NeoX
rotate-half layout, so q/k projection rows import verbatim (no permute).
"""
from __future__ import annotations

import json
import os
from typing import Optional

import torch

from .config import ModelConfig
from .qwen2 import CausalLM


def config_from_hf(path: str, **overrides) -> ModelConfig:
    """Build a ModelConfig from an HF checkpoint dir's config.json.

    Supports the Qwen2 architecture family (what the reference runs:
    Qwen2.5-{0.5,1.5,7}B, Qwen2-1.5B for r1)."""
    with open(os.path.join(path, "config.json")) as f:
        hf = json.load(f)
    archs = hf.get("architectures") or []
    if archs and not any("Qwen2" in a for a in archs):
        raise ValueError(f"unsupported HF architecture {archs}; Qwen2-family only")
    num_heads = hf["num_attention_heads"]
    head_dim = hf.get("head_dim") or hf["hidden_size"] // num_heads
    d = dict(
        vocab_size=hf["vocab_size"],
        hidden_size=hf["hidden_size"],
        num_layers=hf["num_hidden_layers"],
        num_heads=num_heads,
        num_kv_heads=hf.get("num_key_value_heads", num_heads),
        head_dim=head_dim,
        intermediate_size=hf["intermediate_size"],
        rms_eps=hf.get("rms_norm_eps", 1e-6),
        rope_theta=hf.get("rope_theta", 1e6),
        max_position=hf.get("max_position_embeddings", 32768),
        tie_word_embeddings=hf.get("tie_word_embeddings", False),
        qkv_bias=True,  # Qwen2 family uses attention biases
        dtype=hf.get("torch_dtype", "bfloat16"),
    )
    d.update(overrides)
    return ModelConfig(**d)


def _iter_safetensor_files(path: str) -> list[str]:
    index = os.path.join(path, "model.safetensors.index.json")
    if os.path.exists(index):
        with open(index) as f:
            weight_map = json.load(f)["weight_map"]
        return sorted({os.path.join(path, v) for v in weight_map.values()})
    files = sorted(
        os.path.join(path, f) for f in os.listdir(path) if f.endswith(".safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors files under {path}")
    return files


def load_hf_tensors(path: str) -> dict[str, torch.Tensor]:
    """Read every tensor from the checkpoint's safetensors shard(s)."""
    from safetensors.torch import load_file
    out: dict[str, torch.Tensor] = {}
    for f in _iter_safetensor_files(path):
        out.update(load_file(f))
    return out


def fuse_qwen2_state(hf_state: dict[str, torch.Tensor], cfg: ModelConfig,
                     dtype: Optional[torch.dtype] = None) -> dict[str, torch.Tensor]:
    """HF Qwen2 per-projection state dict → fused-layout state dict.

    q/k/v rows concatenate along dim 0 into qkv_proj (q first — matching the
    slicing order in models/qwen2.py Attention.forward); gate/up likewise
    into gate_up_proj."""
    sd: dict[str, torch.Tensor] = {}

    def take(name: str) -> torch.Tensor:
        if name not in hf_state:
            raise KeyError(f"HF checkpoint missing tensor {name!r}")
        t = hf_state[name]
        return t.to(dtype) if dtype is not None else t

    sd["model.embed_tokens.weight"] = take("model.embed_tokens.weight")
    for i in range(cfg.num_layers):
        p = f"model.layers.{i}"
        sd[f"{p}.input_layernorm.weight"] = take(f"{p}.input_layernorm.weight")
        sd[f"{p}.post_attention_layernorm.weight"] = take(
            f"{p}.post_attention_layernorm.weight")
        q_w = take(f"{p}.self_attn.q_proj.weight")
        k_w = take(f"{p}.self_attn.k_proj.weight")
        v_w = take(f"{p}.self_attn.v_proj.weight")
        sd[f"{p}.self_attn.qkv_proj.weight"] = torch.cat([q_w, k_w, v_w], dim=0)
        if cfg.qkv_bias:
            q_b = take(f"{p}.self_attn.q_proj.bias")
            k_b = take(f"{p}.self_attn.k_proj.bias")
            v_b = take(f"{p}.self_attn.v_proj.bias")
            sd[f"{p}.self_attn.qkv_proj.bias"] = torch.cat([q_b, k_b, v_b], dim=0)
        sd[f"{p}.self_attn.o_proj.weight"] = take(f"{p}.self_attn.o_proj.weight")
        gate = take(f"{p}.mlp.gate_proj.weight")
        up = take(f"{p}.mlp.up_proj.weight")
        sd[f"{p}.mlp.gate_up_proj.weight"] = torch.cat([gate, up], dim=0)
        sd[f"{p}.mlp.down_proj.weight"] = take(f"{p}.mlp.down_proj.weight")
    sd["model.norm.weight"] = take("model.norm.weight")
    if not cfg.tie_word_embeddings:
        sd["lm_head.weight"] = take("lm_head.weight")
    return sd


def unfuse_qwen2_state(model: CausalLM) -> dict[str, torch.Tensor]:
    """Inverse of fuse_qwen2_state: export this model back to HF Qwen2
    per-projection names (for interop / round-trip tests).  LoRA adapters,
    if applied, are NOT merged here — call models.lora.merge_lora first."""
    cfg = model.cfg
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()
          if "lora_" not in k}
    out: dict[str, torch.Tensor] = {}
    out["model.embed_tokens.weight"] = sd["model.embed_tokens.weight"]
    q_rows = cfg.num_heads * cfg.head_dim
    kv_rows = cfg.num_kv_heads * cfg.head_dim
    for i in range(cfg.num_layers):
        p = f"model.layers.{i}"
        out[f"{p}.input_layernorm.weight"] = sd[f"{p}.input_layernorm.weight"]
        out[f"{p}.post_attention_layernorm.weight"] = sd[f"{p}.post_attention_layernorm.weight"]
        # fused qkv may live under .base.weight when LoRA wraps the linear
        def fused(name):
            for cand in (f"{p}.{name}.weight", f"{p}.{name}.base.weight"):
                if cand in sd:
                    return sd[cand], sd.get(cand.replace(".weight", ".bias"))
            raise KeyError(f"{p}.{name}")
        qkv_w, qkv_b = fused("self_attn.qkv_proj")
        out[f"{p}.self_attn.q_proj.weight"] = qkv_w[:q_rows]
        out[f"{p}.self_attn.k_proj.weight"] = qkv_w[q_rows: q_rows + kv_rows]
        out[f"{p}.self_attn.v_proj.weight"] = qkv_w[q_rows + kv_rows:]
        if qkv_b is not None:
            out[f"{p}.self_attn.q_proj.bias"] = qkv_b[:q_rows]
            out[f"{p}.self_attn.k_proj.bias"] = qkv_b[q_rows: q_rows + kv_rows]
            out[f"{p}.self_attn.v_proj.bias"] = qkv_b[q_rows + kv_rows:]
        o_w, _ = fused("self_attn.o_proj")
        out[f"{p}.self_attn.o_proj.weight"] = o_w
        gu_w, _ = fused("mlp.gate_up_proj")
        out[f"{p}.mlp.gate_proj.weight"] = gu_w[: cfg.intermediate_size]
        out[f"{p}.mlp.up_proj.weight"] = gu_w[cfg.intermediate_size:]
        dn_w, _ = fused("mlp.down_proj")
        out[f"{p}.mlp.down_proj.weight"] = dn_w
    out["model.norm.weight"] = sd["model.norm.weight"]
    if not cfg.tie_word_embeddings:
        out["lm_head.weight"] = sd["lm_head.weight"]
    return out


def load_pretrained(path: str, dtype: Optional[str] = None,
                    **config_overrides) -> CausalLM:
    """Build a CausalLM from a local HF Qwen2 checkpoint directory.

    The MI355X-native replacement for the reference's
    `AutoModelForCausalLM.from_pretrained(..., attn_implementation=
    "flash_attention_2")` (GRPO/grpo.py:218-224)."""
    cfg = config_from_hf(path, **({"dtype": dtype} if dtype else {}),
                         **config_overrides)
    model = CausalLM(cfg)
    td = getattr(torch, cfg.dtype)
    hf_state = load_hf_tensors(path)
    fused = fuse_qwen2_state(hf_state, cfg, dtype=td)
    model = model.to(td)
    missing, unexpected = model.load_state_dict(fused, strict=False)
    # tied embeddings: lm_head is shared with embed_tokens (property), so the
    # only legitimately-missing keys are none at all
    if missing:
        raise RuntimeError(f"checkpoint did not cover model keys: {missing[:8]}")
    if unexpected:
        raise RuntimeError(f"unmapped checkpoint keys: {unexpected[:8]}")
    return model


def save_hf_checkpoint(model: CausalLM, path: str) -> None:
    """Write this model back out as an HF-layout Qwen2 checkpoint
    (config.json + model.safetensors) — interop with the wider ecosystem
    and the round-trip test harness."""
    from safetensors.torch import save_file
    os.makedirs(path, exist_ok=True)
    cfg = model.cfg
    hf_cfg = {
        "architectures": ["Qwen2ForCausalLM"],
        "model_type": "qwen2",
        "vocab_size": cfg.vocab_size,
        "hidden_size": cfg.hidden_size,
        "num_hidden_layers": cfg.num_layers,
        "num_attention_heads": cfg.num_heads,
        "num_key_value_heads": cfg.num_kv_heads,
        "head_dim": cfg.head_dim,
        "intermediate_size": cfg.intermediate_size,
        "rms_norm_eps": cfg.rms_eps,
        "rope_theta": cfg.rope_theta,
        "max_position_embeddings": cfg.max_position,
        "tie_word_embeddings": cfg.tie_word_embeddings,
        "torch_dtype": cfg.dtype,
    }
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(hf_cfg, f, indent=2)
    state = {k: v.contiguous() for k, v in unfuse_qwen2_state(model).items()}
    save_file(state, os.path.join(path, "model.safetensors"))


def export_merged_hf(model: CausalLM, path: str) -> None:
    """Export a (possibly LoRA-wrapped) trained policy as a plain HF Qwen2
    checkpoint with adapters merged into the base weights — the artifact a
    user deploys or hands to any HF-ecosystem tool.  (The reference writes
    merged checkpoints to disk every update to feed vLLM,
    grpo_trainer.py:131-140; here merging is only ever done on export.)"""
    import copy

    from .lora import LoRALinear

    model = copy.deepcopy(model).cpu()
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, LoRALinear):
                with torch.no_grad():
                    merged = (child.base.weight.float()
                              + (child.lora_B.float() @ child.lora_A.float())
                              * child.scaling).to(child.base.weight.dtype)
                    child.base.weight.copy_(merged)
                setattr(module, child_name, child.base)
    save_hf_checkpoint(model, path)
