"""Model-based reward: score sequences with a ScalarHeadModel reward model.

Reference role: the DeBERTa reward closure (GRPO/grpo.py:162-198) — batched
forward at reward_batch_size with the RM shuttled CPU↔GPU around the pass.
Here the shuttle is an OffloadEngine policy (288 GB usually keeps it
resident) and batching is token-budgeted varlen packing instead of a fixed
row count."""
from __future__ import annotations

import torch

from ..models.value_head import ScalarHeadModel
from ..utils.offload import OffloadEngine


class ModelReward:
    def __init__(self, model: ScalarHeadModel, device, token_budget: int = 65536,
                 offload: OffloadEngine | None = None, max_len: int = 512):
        """max_len mirrors the reference RM's tokenizer truncation —
        deberta-v3-large caps at 512 positions, so its rewards are computed
        on the first 512 tokens of question+response (GRPO/grpo.py:180-192)."""
        self.model = model
        self.device = torch.device(device)
        self.token_budget = token_budget
        self.offload = offload
        self.max_len = max_len

    @torch.no_grad()
    def __call__(self, sequences: list[list[int]]) -> torch.Tensor:
        """sequences: token ids of prompt+response per sample → scores [B] fp32.
        (The reference scores detokenized strings re-tokenized by the RM's own
        tokenizer; with a shared synthetic vocab we score ids directly — the
        string path lives in the entry scripts when a tokenizer is given.)"""
        if self.offload is not None:
            self.offload.model_to_device(self.model)
            self.offload.join_compute()
        self.model.eval()
        if self.max_len:
            sequences = [s[: self.max_len] for s in sequences]
        scores = torch.empty(len(sequences), dtype=torch.float32)
        i = 0
        while i < len(sequences):
            chunk = []
            total = 0
            while i < len(sequences) and (not chunk or total + len(sequences[i]) <= self.token_budget):
                total += len(sequences[i])
                chunk.append(i)
                i += 1
            lens = [len(sequences[j]) for j in chunk]
            V = self.model.cfg.vocab_size
            # the RM has its own vocab (the reference re-tokenizes detokenized
            # strings with the RM tokenizer, grpo.py:180-187); for the raw-id
            # path fold ids into the RM vocab deterministically
            ids = torch.tensor([t % V for j in chunk for t in sequences[j]],
                               dtype=torch.long, device=self.device)
            cu = torch.zeros(len(chunk) + 1, dtype=torch.int32, device=self.device)
            cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=self.device), 0)
            s = self.model.sequence_scores(ids, cu, max(lens))
            scores[torch.tensor(chunk)] = s.float().cpu()
        if self.offload is not None and self.offload.should_offload():
            self.offload.model_to_host(self.model)
        return scores
