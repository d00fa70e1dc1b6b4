"""Minimal training-callback protocol (the reference inherits transformers'
CallbackHandler/EarlyStoppingCallback machinery, GRPO/grpo_trainer.py:266-279;
note its early stopping is configured with patience 10^6 — effectively
disabled, grpo.py:89,281)."""
from __future__ import annotations


class TrainerCallback:
    def on_update_end(self, trainer, metrics: dict) -> bool:
        """Return True to request a training stop."""
        return False

    def on_save(self, trainer, ckpt_dir: str):
        pass


class EarlyStoppingCallback(TrainerCallback):
    def __init__(self, metric: str = "eval_objective/rlhf_reward_old",
                 patience: int = 10**6, greater_is_better: bool = True,
                 min_delta: float = 0.0):
        self.metric = metric
        self.patience = patience
        self.greater = greater_is_better
        self.min_delta = min_delta
        self.best = None
        self.bad = 0

    def on_update_end(self, trainer, metrics: dict) -> bool:
        if self.metric not in metrics:
            return False
        v = float(metrics[self.metric])
        improved = (self.best is None
                    or (v > self.best + self.min_delta if self.greater
                        else v < self.best - self.min_delta))
        if improved:
            self.best = v
            self.bad = 0
        else:
            self.bad += 1
        return self.bad >= self.patience
