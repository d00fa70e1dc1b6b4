"""Synthetic datasets (there is no network for hh-rlhf / MetaMathQA).

`hh_shaped_prompts` mimics the shape of the reference's prepared hh-rlhf
prompt set (first human turn + Qwen chat template, GRPO/grpo.py:249-270):
token-id sequences with the same length distribution (~tens to a few
hundred tokens), random content, fixed seed for reproducibility."""
from __future__ import annotations

import torch


def hh_shaped_prompts(num: int, vocab_size: int, seed: int = 0,
                      min_len: int = 24, max_len: int = 256) -> list[list[int]]:
    g = torch.Generator().manual_seed(seed)
    lens = torch.randint(min_len, max_len + 1, (num,), generator=g)
    out = []
    for i in range(num):
        # avoid ids 0/1 (pad/eos conventions in the synthetic vocab)
        ids = torch.randint(2, vocab_size, (int(lens[i]),), generator=g)
        out.append(ids.tolist())
    return out


def math_shaped_prompts(num: int, vocab_size: int, seed: int = 0) -> list[list[int]]:
    """MetaMathQA-shaped (longer prompts, r1 config)."""
    return hh_shaped_prompts(num, vocab_size, seed, min_len=64, max_len=512)


class PromptDataset(torch.utils.data.Dataset):
    def __init__(self, prompts: list[list[int]]):
        self.prompts = prompts

    def __len__(self):
        return len(self.prompts)

    def __getitem__(self, i):
        return self.prompts[i]
