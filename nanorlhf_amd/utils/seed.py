from __future__ import annotations

import random

import numpy as np
import torch


def set_seed(seed: int):
    """Global seeding; per-rank decorrelation uses the reference's offset
    seed + rank*100003 (GRPO/grpo_trainer.py:244)."""
    random.seed(seed)
    np.random.seed(seed % (2**32 - 1))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def rank_seed(seed: int, rank: int) -> int:
    return seed + rank * 100003
