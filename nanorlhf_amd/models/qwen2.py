"""MI355X-native Qwen2-architecture causal LM.

Replaces the reference's transformers Qwen2 + flash_attention_2 stack
(GRPO/grpo.py:218-224).  Design choices (MI355X-first, not a port):

  * packed varlen layout everywhere — no pad tokens reach compute; the
    collator packs, attention gets cu_seqlens (the reference instead builds
    pad masks in a local forward(), grpo_trainer.py:90-120);
  * hot ops route through nanorlhf_amd.ops (hand-written HIP on GPU);
  * plain projections stay torch.nn.functional.linear → rocBLAS/hipBLASLt;
  * the SAME module graph serves training, prefill and paged decode — the
    in-process sampler passes an AttnContext instead of booting a separate
    engine (deletes the reference's disk round trip, grpo_trainer.py:122-166).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .config import ModelConfig, get_config


@dataclass
class AttnContext:
    """Per-forward attention routing info."""
    mode: str                      # "train" | "prefill" | "decode"
    positions: torch.Tensor        # [T] int64 position ids
    cu_seqlens: Optional[torch.Tensor] = None   # [B+1] int32 (train/prefill)
    max_seqlen: int = 0
    # paged-KV fields (prefill writes, decode reads+writes)
    kv_caches: Optional[list] = None            # per layer (k_cache, v_cache)
    slots: Optional[torch.Tensor] = None        # [T] int64 flat slot ids
    block_tables: Optional[torch.Tensor] = None  # [B, P] int32 (decode)
    seq_lens: Optional[torch.Tensor] = None      # [B] int32 (decode)
    causal: bool = True


def dtype_of(cfg: ModelConfig):
    return getattr(torch, cfg.dtype)


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.layer_idx = layer_idx
        self.num_heads = cfg.num_heads
        self.num_kv_heads = cfg.num_kv_heads
        self.head_dim = cfg.head_dim
        H = cfg.hidden_size
        self.q_size = cfg.num_heads * cfg.head_dim
        self.kv_size = cfg.num_kv_heads * cfg.head_dim
        # fused qkv projection: one GEMM instead of three (MI355X-first —
        # bigger N per launch; the reference's transformers Qwen2 keeps
        # separate q/k/v linears)
        self.qkv_proj = nn.Linear(H, self.q_size + 2 * self.kv_size, bias=cfg.qkv_bias)
        self.o_proj = nn.Linear(cfg.num_heads * cfg.head_dim, H, bias=False)
        self.scale = cfg.head_dim ** -0.5

    def forward(self, x: torch.Tensor, rope_table: torch.Tensor, ctx: AttnContext):
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        q = qkv[:, : self.q_size].reshape(T, self.num_heads, self.head_dim)
        k = qkv[:, self.q_size: self.q_size + self.kv_size].reshape(
            T, self.num_kv_heads, self.head_dim)
        v = qkv[:, self.q_size + self.kv_size:].reshape(T, self.num_kv_heads, self.head_dim)
        q = ops.rope_apply(q, rope_table, ctx.positions)
        k = ops.rope_apply(k, rope_table, ctx.positions)

        if ctx.mode in ("prefill", "decode") and ctx.kv_caches is not None:
            k_cache, v_cache = ctx.kv_caches[self.layer_idx]
            ops.kv_append(k, v, ctx.slots, k_cache, v_cache)

        if ctx.mode == "decode":
            o = ops.paged_attn_decode(q, *ctx.kv_caches[self.layer_idx],
                                      ctx.block_tables, ctx.seq_lens, self.scale)
        else:
            o = ops.flash_attn_varlen(q, k, v, ctx.cu_seqlens, ctx.max_seqlen,
                                      self.scale, causal=ctx.causal)
        return self.o_proj(o.reshape(T, -1))


class MLP(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(ops.swiglu(self.gate_up_proj(x)))


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class Block(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.self_attn = Attention(cfg, layer_idx)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.mlp = MLP(cfg)

    def forward(self, delta, res, rope_table, ctx):
        """Carries the residual stream as (delta, res) with hidden =
        delta + res: the pending add fuses into the NEXT norm's kernel
        (ops.add_rms_norm), so no standalone residual-add kernels run."""
        eps = self.input_layernorm.eps
        a_in, h = ops.add_rms_norm(delta, res, self.input_layernorm.weight, eps)
        attn_delta = self.self_attn(a_in, rope_table, ctx)
        m_in, h2 = ops.add_rms_norm(attn_delta, h,
                                    self.post_attention_layernorm.weight, eps)
        return self.mlp(m_in), h2


class Transformer(nn.Module):
    """Shared backbone (causal or bidirectional by AttnContext.causal)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(Block(cfg, i) for i in range(cfg.num_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        table = ops.build_rope_cache(cfg.head_dim, cfg.max_position, cfg.rope_theta)
        self.register_buffer("rope_table", table, persistent=False)

    def forward(self, input_ids: torch.Tensor, ctx: AttnContext,
                checkpoint: bool = False) -> torch.Tensor:
        if self.rope_table.dtype != torch.float32:
            # keep the rope table fp32 even after model.to(bf16)
            self.rope_table = self.rope_table.float()
        delta = self.embed_tokens(input_ids)
        res = None
        for layer in self.layers:
            if checkpoint and torch.is_grad_enabled():
                delta, res = torch.utils.checkpoint.checkpoint(
                    layer, delta, res, self.rope_table, ctx, use_reentrant=False)
            else:
                delta, res = layer(delta, res, self.rope_table, ctx)
        y, _ = ops.add_rms_norm(delta, res, self.norm.weight, self.norm.eps)
        return y


class CausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.model = Transformer(cfg)
        if cfg.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.gradient_checkpointing = False
        self.apply(self._init_weights)

    def _init_weights(self, m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    @property
    def lm_head_weight(self) -> torch.Tensor:
        if self.lm_head is not None:
            return self.lm_head.weight
        return self.model.embed_tokens.weight

    def forward(self, input_ids: torch.Tensor, ctx: AttnContext) -> torch.Tensor:
        """Returns final hidden states [T, H] (logits are computed by the
        caller via ops.token_logprob_entropy or .logits())."""
        return self.model(input_ids, ctx, checkpoint=self.gradient_checkpointing)

    def logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return F.linear(hidden, self.lm_head_weight)

    @staticmethod
    def train_ctx(cu_seqlens: torch.Tensor, max_seqlen: int, positions: torch.Tensor,
                  causal: bool = True) -> AttnContext:
        return AttnContext(mode="train", positions=positions, cu_seqlens=cu_seqlens,
                           max_seqlen=max_seqlen, causal=causal)

    @classmethod
    def from_preset(cls, name: str, **overrides) -> "CausalLM":
        return cls(get_config(name, **overrides))


def make_positions(cu_seqlens: torch.Tensor) -> torch.Tensor:
    """position ids 0..len-1 per packed sequence."""
    device = cu_seqlens.device
    lens = (cu_seqlens[1:] - cu_seqlens[:-1]).long()
    return torch.cat([torch.arange(int(n), device=device) for n in lens]) if len(lens) else \
        torch.zeros(0, dtype=torch.long, device=device)


def pack_sequences(seqs: list[torch.Tensor], device="cpu"):
    """Pack a list of 1-D id tensors → (input_ids [T], cu_seqlens [B+1] int32,
    max_seqlen, positions [T])."""
    lens = [int(s.numel()) for s in seqs]
    cu = torch.zeros(len(seqs) + 1, dtype=torch.int32, device=device)
    cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=device), 0)
    ids = torch.cat([s.to(device) for s in seqs]) if seqs else torch.zeros(0, dtype=torch.long)
    pos = make_positions(cu)
    return ids.long(), cu, (max(lens) if lens else 0), pos
