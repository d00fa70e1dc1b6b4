"""Paged KV-cache pool + page allocator.

The vLLM-engine replacement's memory core.
One (K, V) tensor pair per layer, [num_pages, page_size, Hkv, D] bf16,
sized against 288 GB HBM3E.  Sequences own page lists; pages are recycled
when a sequence finishes (continuous batching admits queued sequences into
the freed space)."""
from __future__ import annotations

import torch

from ..models.config import ModelConfig


class PagedKVCache:
    def __init__(self, cfg: ModelConfig, num_pages: int, page_size: int = 16,
                 device="cpu", dtype=torch.bfloat16):
        self.page_size = page_size
        self.num_pages = num_pages
        self.device = device
        kshape = (num_pages, page_size, cfg.num_kv_heads, cfg.head_dim)
        # V is stored d-major within each page ([page][Hkv][D][page_size]) so
        # the decode kernel's PV MFMA B-fragment is a contiguous 16-B load
        vshape = (num_pages, cfg.num_kv_heads, cfg.head_dim, page_size)
        self.layers = [
            (torch.zeros(kshape, dtype=dtype, device=device),
             torch.zeros(vshape, dtype=dtype, device=device))
            for _ in range(cfg.num_layers)
        ]
        self._free = list(range(num_pages - 1, -1, -1))

    @property
    def free_pages(self) -> int:
        return len(self._free)

    def alloc(self, n: int) -> list[int]:
        if n > len(self._free):
            raise RuntimeError(f"KV pool exhausted: want {n}, have {len(self._free)} pages")
        return [self._free.pop() for _ in range(n)]

    def free(self, pages: list[int]):
        self._free.extend(pages)

    @staticmethod
    def bytes_per_token(cfg: ModelConfig, dtype=torch.bfloat16) -> int:
        esize = torch.tensor([], dtype=dtype).element_size()
        return 2 * cfg.num_layers * cfg.num_kv_heads * cfg.head_dim * esize

    @classmethod
    def for_budget(cls, cfg: ModelConfig, max_tokens_total: int, page_size: int = 16,
                   device="cpu", dtype=torch.bfloat16) -> "PagedKVCache":
        pages = (max_tokens_total + page_size - 1) // page_size
        return cls(cfg, pages, page_size, device, dtype)


class SeqState:
    """Book-keeping for one in-flight sequence."""

    __slots__ = ("uid", "tokens", "prompt_len", "pages", "finished", "out_index",
                 "logprobs")

    def __init__(self, uid: int, prompt: list[int], out_index: int):
        self.uid = uid
        self.tokens = list(prompt)
        self.prompt_len = len(prompt)
        self.pages: list[int] = []
        self.finished = False
        self.out_index = out_index
        self.logprobs: list[float] = []  # per generated (response) token

    def __len__(self):
        return len(self.tokens)

    @property
    def response(self) -> list[int]:
        return self.tokens[self.prompt_len:]

    def slot_of(self, pos: int, page_size: int) -> int:
        return self.pages[pos // page_size] * page_size + pos % page_size

    def ensure_capacity(self, pool: PagedKVCache, upto: int):
        need = (upto + pool.page_size - 1) // pool.page_size
        if need > len(self.pages):
            self.pages.extend(pool.alloc(need - len(self.pages)))
