"""Paged KV-cache primitives for the in-process sampler.

Replaces vLLM's paged attention.  The pool is two
tensors [num_pages, page_size, Hkv, D] (K and V) in bf16, sized against
288 GB HBM3E; sequences own page lists (block tables).

  * kv_append: scatter freshly-projected K/V rows into their page slots.
  * paged_attn_decode: one query token per sequence; HIP kernel does
    online-softmax attention over the sequence's pages, one workgroup per
    (sequence, kv-head) with the q-heads of that group held in registers
    (GQA), vectorized bf16 loads of KV lines, fp32 accumulation.
"""
from __future__ import annotations

import torch

from . import ext


def kv_append(k: torch.Tensor, v: torch.Tensor, slots: torch.Tensor,
              k_cache: torch.Tensor, v_cache: torch.Tensor) -> None:
    """k/v: [T, Hkv, D] bf16; slots: [T] int64 flat slot index
    (page * page_size + offset).  K cache is [P, ps, Hkv, D]; V cache is
    d-major per page, [P, Hkv, D, ps]."""
    if k.is_cuda:
        ext().kv_append(k.contiguous(), v.contiguous(), slots, k_cache, v_cache)
        return
    ps = k_cache.shape[1]
    kc = k_cache.view(-1, *k_cache.shape[2:])
    if k_cache.dtype == torch.float8_e4m3fn:
        # mirror the kernel's fragment-major fp8 K layout
        kc[slots] = k[..., _fp8_k_perm(k.shape[-1])].to(k_cache.dtype)
    else:
        kc[slots] = k
    pages = torch.div(slots, ps, rounding_mode="floor")
    offs = slots % ps
    # v_cache[page, h, d, off] = v[t, h, d]
    v_cache[pages, :, :, offs] = v


def _fp8_k_perm(D: int) -> torch.Tensor:
    """Fragment-major permutation for the fp8 K cache: stored index
    hi*(D/4) + c*8 + j holds logical element c*32 + hi*8 + j."""
    fwd = torch.empty(D, dtype=torch.long)
    for d in range(D):
        c, rem = divmod(d, 32)
        hi, j = divmod(rem, 8)
        fwd[hi * (D // 4) + c * 8 + j] = c * 32 + hi * 8 + j
    return fwd


def depermute_fp8_k(kc_float: torch.Tensor) -> torch.Tensor:
    """Undo the fragment-major byte permutation of the fp8 K cache
    (csrc/kvcache.hip kv_append)."""
    D = kc_float.shape[-1]
    idx = torch.empty(D, dtype=torch.long)
    for d in range(D):
        c, rem = divmod(d, 32)
        hi, j = divmod(rem, 8)
        idx[c * 32 + hi * 8 + j] = hi * (D // 4) + c * 8 + j
    return kc_float[..., idx.to(kc_float.device)]


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                      block_tables: torch.Tensor, seq_lens: torch.Tensor,
                      scale: float) -> torch.Tensor:
    """q: [B, Hq, D] bf16 (one new token per sequence); block_tables [B, P] int32;
    seq_lens [B] int32 (length INCLUDING the new token, whose K/V are already
    appended).  Returns [B, Hq, D] bf16."""
    if q.is_cuda:
        return ext().paged_attn_decode(q.contiguous(), k_cache, v_cache,
                                       block_tables.contiguous(), seq_lens.contiguous(),
                                       float(scale))
    # CPU reference.  On an fp8 cache this mirrors the native-fp8 kernel's
    # arithmetic (csrc/kvcache.hip): Q and the softmax P are ALSO e4m3 so
    # the whole QK^T/PV runs on mfma_..._fp8_fp8 — quantize them here too.
    fp8 = k_cache.dtype == torch.float8_e4m3fn
    B, Hq, D = q.shape
    page = k_cache.shape[1]
    Hkv = k_cache.shape[2]
    rep = Hq // Hkv
    out = torch.empty_like(q, dtype=torch.float32)
    kc = k_cache.view(-1, Hkv, D).float()
    if fp8:
        kc = depermute_fp8_k(kc)
    # V d-major [P, Hkv, D, ps] -> flat [P*ps, Hkv, D]
    vc = v_cache.permute(0, 3, 1, 2).reshape(-1, Hkv, D).float()

    def q8(x):
        return x.to(torch.float8_e4m3fn).float() if fp8 else x

    for b in range(B):
        L = int(seq_lens[b])
        pages = block_tables[b, : (L + page - 1) // page].long()
        slots = (pages.unsqueeze(1) * page + torch.arange(page)).reshape(-1)[:L]
        kk = kc[slots].repeat_interleave(rep, dim=1)  # [L, Hq, D]
        vv = vc[slots].repeat_interleave(rep, dim=1)
        att = torch.einsum("hd,lhd->hl", q8(q[b].float()), kk) * scale
        att = q8(torch.softmax(att, dim=-1))
        out[b] = torch.einsum("hl,lhd->hd", att, vv)
    return out.to(q.dtype)
