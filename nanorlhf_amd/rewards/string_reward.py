"""String-contract reward adapter.

The reference's reward plug-in contract is
`reward_func(pmt_and_responses: list[str], eos_token) -> FloatTensor[B]`
(GRPO/grpo.py:162-198) and, in r1 mode,
`reward_func(pmt_and_responses, responses_ids, tokenizer)`
(examples/r1-v0/grpo_r1.py:250).  The trainer engine works in token ids
(no strings on the hot path); this adapter decodes at the reward boundary
so user reward functions written against the reference's contract drop in
unchanged.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch


class StringReward:
    """Wraps a `fn(list[str], ...) -> FloatTensor` into the trainer's
    `reward_fn(list[list[int]]) -> FloatTensor` contract.

    mode:
      "strings"     -> fn(texts)                           (simplest)
      "eos"         -> fn(texts, eos_token)                (grpo.py:162)
      "r1"          -> fn(texts, responses_ids, tokenizer) (grpo_r1.py:250)
    strip_pad: remove the [PAD] token text before calling fn, as the
    reference does when decoding queries (grpo_trainer.py:484-485).
    """

    def __init__(self, fn: Callable, tokenizer, mode: str = "strings",
                 strip_pad: bool = True):
        if mode not in ("strings", "eos", "r1"):
            raise ValueError(f"unknown StringReward mode {mode!r}")
        self.fn = fn
        self.tokenizer = tokenizer
        self.mode = mode
        self.strip_pad = strip_pad

    def _decode(self, rows: list[list[int]]) -> list[str]:
        texts = self.tokenizer.batch_decode(
            [torch.tensor(r, dtype=torch.long) for r in rows],
            skip_special_tokens=False)
        if self.strip_pad and self.tokenizer.pad_token:
            texts = [t.replace(self.tokenizer.pad_token, "") for t in texts]
        return texts

    def __call__(self, rows: list[list[int]],
                 responses: Optional[list[list[int]]] = None) -> torch.Tensor:
        texts = self._decode(rows)
        if self.mode == "eos":
            out = self.fn(texts, self.tokenizer.eos_token)
        elif self.mode == "r1":
            out = self.fn(texts, responses, self.tokenizer)
        else:
            out = self.fn(texts)
        if not torch.is_tensor(out):
            out = torch.tensor(out, dtype=torch.float32)
        return out.float().cpu()
