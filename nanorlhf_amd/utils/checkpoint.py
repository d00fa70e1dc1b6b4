"""trl/transformers-compatible checkpoint layout (a BASELINE.json
requirement; reference behavior at GRPO/grpo_trainer.py:321-404).

Layout per save:
  output_dir/checkpoint-{global_step}/
    adapter/           policy adapter + modules_to_save tensors (safetensors)
    model_config.json  architecture config (ours; random-init reproducible)
    optimizer.pt       optimizer state
    scheduler.pt       LR scheduler state
    rng_state.pth      torch/np/python RNG states
    trainer_state.json step, episode, best-metric bookkeeping
    value_model/       (PPO only) critic weights — ppo_trainer.py:413-416

Rotation keeps `save_total_limit` newest checkpoints; the best checkpoint is
protected, with the reference's one-step-behind rule for `*_old` metrics
(grpo_trainer.py:352-382): an `eval_*_old` metric logged at step N describes
step N-1's weights, so best_checkpoint tracking lags one save."""
from __future__ import annotations

import json
import os
import random
import shutil

import numpy as np
import torch

PREFIX_CHECKPOINT_DIR = "checkpoint"


def _rng_state() -> dict:
    st = {
        "python": random.getstate(),
        "numpy": np.random.get_state(),
        "torch": torch.get_rng_state(),
    }
    if torch.cuda.is_available():
        st["cuda"] = torch.cuda.get_rng_state_all()
    return st


def load_rng_state(path: str):
    st = torch.load(os.path.join(path, "rng_state.pth"), weights_only=False)
    random.setstate(st["python"])
    np.random.set_state(st["numpy"])
    torch.set_rng_state(st["torch"])
    if torch.cuda.is_available() and "cuda" in st:
        torch.cuda.set_rng_state_all(st["cuda"])


class CheckpointManager:
    def __init__(self, output_dir: str, save_total_limit: int = 8,
                 metric_for_best: str | None = None, greater_is_better: bool = True):
        self.output_dir = output_dir
        self.save_total_limit = save_total_limit
        self.metric_for_best = metric_for_best
        self.greater_is_better = greater_is_better
        self.best_metric: float | None = None
        self.best_checkpoint: str | None = None
        self._prev_checkpoint: str | None = None  # one-step-behind for *_old

    def save(self, global_step: int, episode: int, policy_state: dict,
             model_config: dict, optimizer=None, scheduler=None,
             value_state: dict | None = None, metrics: dict | None = None) -> str:
        ckpt = os.path.join(self.output_dir, f"{PREFIX_CHECKPOINT_DIR}-{global_step}")
        os.makedirs(ckpt, exist_ok=True)
        # policy adapter / weights
        adapter_dir = os.path.join(ckpt, "adapter")
        os.makedirs(adapter_dir, exist_ok=True)
        try:
            import safetensors.torch as st
            st.save_file({k: v.contiguous() for k, v in policy_state.items()},
                         os.path.join(adapter_dir, "adapter_model.safetensors"))
        except ImportError:
            torch.save(policy_state, os.path.join(adapter_dir, "adapter_model.bin"))
        with open(os.path.join(ckpt, "model_config.json"), "w") as f:
            json.dump(model_config, f, indent=2)
        if optimizer is not None:
            torch.save(optimizer.state_dict(), os.path.join(ckpt, "optimizer.pt"))
        if scheduler is not None:
            torch.save(scheduler.state_dict(), os.path.join(ckpt, "scheduler.pt"))
        if value_state is not None:
            vdir = os.path.join(ckpt, "value_model")
            os.makedirs(vdir, exist_ok=True)
            torch.save(value_state, os.path.join(vdir, "pytorch_model.bin"))
        torch.save(_rng_state(), os.path.join(ckpt, "rng_state.pth"))

        # best-metric bookkeeping: metrics named *_old describe the PREVIOUS
        # checkpoint's weights (reference quirk, grpo_trainer.py:374-382)
        if metrics and self.metric_for_best and self.metric_for_best in metrics:
            val = float(metrics[self.metric_for_best])
            target = (self._prev_checkpoint
                      if self.metric_for_best.endswith("_old") else ckpt)
            if target is not None:
                improved = self.best_metric is None or (
                    val > self.best_metric if self.greater_is_better else val < self.best_metric)
                if improved:
                    self.best_metric = val
                    self.best_checkpoint = target

        state = {
            "global_step": global_step,
            "episode": episode,
            "best_metric": self.best_metric,
            "best_model_checkpoint": self.best_checkpoint,
        }
        with open(os.path.join(ckpt, "trainer_state.json"), "w") as f:
            json.dump(state, f, indent=2)
        self._prev_checkpoint = ckpt
        self._rotate()
        return ckpt

    def _rotate(self):
        if not self.save_total_limit:
            return
        ckpts = sorted(
            (d for d in os.listdir(self.output_dir)
             if d.startswith(PREFIX_CHECKPOINT_DIR + "-")),
            key=lambda d: int(d.split("-")[-1]),
        )
        keep = set(ckpts[-self.save_total_limit:])
        if self.best_checkpoint:
            keep.add(os.path.basename(self.best_checkpoint))
        for d in ckpts:
            if d not in keep:
                shutil.rmtree(os.path.join(self.output_dir, d), ignore_errors=True)

    @staticmethod
    def load_policy_state(ckpt: str) -> dict:
        p = os.path.join(ckpt, "adapter", "adapter_model.safetensors")
        if os.path.exists(p):
            import safetensors.torch as st
            return st.load_file(p)
        return torch.load(os.path.join(ckpt, "adapter", "adapter_model.bin"),
                          weights_only=True)

    @staticmethod
    def latest(output_dir: str) -> str | None:
        if not os.path.isdir(output_dir):
            return None
        ckpts = [d for d in os.listdir(output_dir) if d.startswith(PREFIX_CHECKPOINT_DIR + "-")]
        if not ckpts:
            return None
        return os.path.join(output_dir, max(ckpts, key=lambda d: int(d.split("-")[-1])))
