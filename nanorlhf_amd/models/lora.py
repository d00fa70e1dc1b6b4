"""LoRA adapters — replaces the reference's peft dependency
(LoraConfig r=64, alpha=16, target q/k/v/o/gate_up/down + fully-trained
embed_tokens/lm_head via modules_to_save, GRPO/grpo.py:226-243).

MI355X-first detail: instead of peft's merge_and_unload-to-disk before every
rollout (grpo_trainer.py:131-140), `merge_for_rollout()` materializes
W + (alpha/r)·B·A into a cached HBM buffer that the in-process sampler reads
directly (288 GB leaves room for a merged copy), and `unmerge()` drops it.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


def quantize_fp8_weight(w_float: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-tensor e4m3 quantization: returns (w8 [N,K], scale fp32 scalar)."""
    amax = w_float.abs().amax().clamp(min=1e-12)
    scale = (amax / 448.0).float()
    w8 = (w_float / scale).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return w8, scale


def _scaled_linear(x: torch.Tensor, w8: torch.Tensor, scale_w: torch.Tensor,
                   bias: torch.Tensor | None) -> torch.Tensor:
    """y = x·w8ᵀ·scale_x·scale_w (+bias) on the hipBLASLt fp8 path.
    x is dynamically quantized per tensor."""
    amax = x.abs().amax().clamp(min=1e-12).float()
    scale_x = amax / 448.0
    x8 = (x.float() / scale_x).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    y = torch._scaled_mm(x8, w8.t(), scale_a=scale_x, scale_b=scale_w,
                         out_dtype=torch.bfloat16)
    if bias is not None:
        y = y + bias
    return y


@dataclass
class LoraConfig:
    r: int = 64
    alpha: int = 16
    dropout: float = 0.0
    target_modules: tuple = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")
    # fully-trained modules (reference modules_to_save: embed_tokens, lm_head)
    modules_to_save: tuple = ("embed_tokens", "lm_head")


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, r: int, alpha: int, dropout: float = 0.0):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        dtype = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(torch.zeros(r, base.in_features, dtype=dtype, device=dev))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r, dtype=dtype, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=5 ** 0.5)
        self.scaling = alpha / r
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
        self._merged: torch.Tensor | None = None
        self._merged_fp8: tuple | None = None  # (w8, scale) rollout-only
        self._weight_t: torch.Tensor | None = None  # Wᵀ cache for fused bwd

    @property
    def in_features(self):
        return self.base.in_features

    @property
    def out_features(self):
        return self.base.out_features

    @property
    def bias(self):
        return self.base.bias

    @property
    def weight(self):
        return self.base.weight

    def forward(self, x):
        if self._merged_fp8 is not None and not torch.is_grad_enabled():
            w8, sw = self._merged_fp8
            return _scaled_linear(x, w8, sw, self.base.bias)
        if self._merged is not None and not torch.is_grad_enabled():
            return F.linear(x, self._merged, self.base.bias)
        lx = self.dropout(x)
        from ..ops.lora import fused_lora_linear, fused_path_ok
        if (lx is x and fused_path_ok(self.base.weight, self.lora_A.shape[0])
                and x.dtype == torch.bfloat16):
            # single-launch fused HIP GEMM (csrc/lora.hip); Wᵀ cached once —
            # the base weight is frozen while LoRA trains
            if self._weight_t is None or self._weight_t.device != x.device:
                self._weight_t = self.base.weight.detach().t().contiguous()
            return fused_lora_linear(x, self.base.weight, self.base.bias,
                                     self.lora_A, self.lora_B, self.scaling,
                                     self._weight_t)
        y = self.base(x)
        return y + F.linear(F.linear(lx, self.lora_A), self.lora_B) * self.scaling

    def merge_for_rollout(self, quant: str | None = None):
        with torch.no_grad():
            merged = (self.base.weight.float()
                      + (self.lora_B.float() @ self.lora_A.float()) * self.scaling)
            if quant == "fp8_e4m3" and self.base.weight.is_cuda:
                # rollout-only OCP e4m3 weights: per-tensor scale, GEMMs go
                # through hipBLASLt's fp8 path (torch._scaled_mm) at ~2x the
                # bf16 MFMA rate.  Training weights untouched.
                self._merged_fp8 = quantize_fp8_weight(merged)
                self._merged = None
            else:
                self._merged = merged.to(self.base.weight.dtype)
                self._merged_fp8 = None

    def unmerge(self):
        self._merged = None
        self._merged_fp8 = None


def apply_lora(model: nn.Module, cfg: LoraConfig) -> nn.Module:
    """Wrap target linears with LoRA; freeze everything except adapters and
    modules_to_save.  Returns the (mutated) model."""
    for p in model.parameters():
        p.requires_grad_(False)
    for name, module in model.named_modules():
        for child_name, child in list(module.named_children()):
            if child_name in cfg.target_modules and isinstance(child, nn.Linear):
                setattr(module, child_name, LoRALinear(child, cfg.r, cfg.alpha, cfg.dropout))
    for name, module in model.named_modules():
        leaf = name.rsplit(".", 1)[-1]
        if leaf in cfg.modules_to_save:
            for p in module.parameters(recurse=True):
                p.requires_grad_(True)
    return model


def merge_for_rollout(model: nn.Module, quant: str | None = None):
    for m in model.modules():
        if isinstance(m, LoRALinear):
            m.merge_for_rollout(quant)


def unmerge(model: nn.Module):
    for m in model.modules():
        if isinstance(m, LoRALinear):
            m.unmerge()


def lora_state_dict(model: nn.Module, cfg: LoraConfig) -> dict:
    """Adapter + modules_to_save tensors only (what checkpoints serialize —
    mirrors peft save_pretrained + modules_to_save, grpo_trainer.py:321-327)."""
    out = {}
    for name, p in model.named_parameters():
        if "lora_A" in name or "lora_B" in name or p.requires_grad:
            out[name] = p.detach().cpu()
    return out
