"""Rule-based rewards.

MathRuleReward: boxed-answer correctness against a gold-answer map
(reference r1 reward_func, examples/r1-v0/grpo_r1.py:250-273: +1 correct,
0 otherwise, each equivalence check under a subprocess timeout).

constant_reward: the BASELINE config-#1 plumbing reward (always 1.0)."""
from __future__ import annotations

import torch

from .mathcheck import answers_equal, extract_boxed


def constant_reward(texts_or_ids, value: float = 1.0) -> torch.Tensor:
    return torch.full((len(texts_or_ids),), value, dtype=torch.float32)


class MathRuleReward:
    def __init__(self, gold_answers: dict[str, str] | list[str],
                 correct: float = 1.0, incorrect: float = 0.0,
                 timeout_s: float = 0.5, require_boxed: bool = True):
        """gold_answers: mapping prompt-key → gold answer string, or a list
        aligned with the rollout order (the reference hashes the train set's
        questions → answers, grpo_r1.py:237-240)."""
        self.gold = gold_answers
        self.correct = correct
        self.incorrect = incorrect
        self.timeout_s = timeout_s
        # require_boxed=False falls back to the LAST number in the text
        # (the reference's answer_extraction.py also has non-boxed extractors)
        self.require_boxed = require_boxed

    def _gold_for(self, i: int, key: str | None) -> str | None:
        if isinstance(self.gold, dict):
            return self.gold.get(key)
        if i < len(self.gold):
            return self.gold[i]
        return None

    def __call__(self, response_texts: list[str], keys: list[str] | None = None) -> torch.Tensor:
        scores = torch.full((len(response_texts),), self.incorrect, dtype=torch.float32)
        for i, text in enumerate(response_texts):
            gold = self._gold_for(i, keys[i] if keys else None)
            if gold is None:
                continue
            pred = extract_boxed(text)
            if pred is None and not self.require_boxed:
                import re
                nums = re.findall(r"-?\d+(?:\.\d+)?", text)
                pred = nums[-1] if nums else None
            if pred is None:
                continue
            if answers_equal(pred, gold, self.timeout_s):
                scores[i] = self.correct
        return scores
