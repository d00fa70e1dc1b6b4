"""Metrics logging: console + JSONL file + optional wandb.

Keeps the reference's stable metric names (objective/kl_old,
objective/entropy_old, eval_objective/rlhf_reward_old, policy/approxkl_avg_new,
policy/clipfrac_avg_new, loss/policy_avg_new, policy/entropy_avg_new,
val/ratio_new, val/ratio_var_new, val/num_eos_tokens_old, lr, episode; PPO
adds loss/value_avg_new + val/clipfrac_avg_new; r1 adds eval_accuracy_new,
initial_accuracy, eval_response_length) so dashboards transfer."""
from __future__ import annotations

import json
import os
import sys
import time


class MetricsLogger:
    def __init__(self, output_dir: str | None = None, report_to: str = "none",
                 project: str | None = None, run_name: str | None = None,
                 rank: int = 0):
        self.rank = rank
        self.jsonl = None
        self.wandb = None
        if rank != 0:
            return
        if output_dir:
            os.makedirs(output_dir, exist_ok=True)
            self.jsonl = open(os.path.join(output_dir, "metrics.jsonl"), "a")
        if report_to == "wandb":
            try:
                import wandb
                self.wandb = wandb
                wandb.init(project=project or os.environ.get("WANDB_PROJECT", "nanorlhf-amd"),
                           name=run_name)
            except Exception as e:  # noqa: BLE001
                print(f"[logging] wandb unavailable ({e}); falling back to jsonl", file=sys.stderr)

    def log(self, metrics: dict, step: int):
        if self.rank != 0:
            return
        rec = {"step": step, "time": time.time(), **{k: _scalar(v) for k, v in metrics.items()}}
        if self.jsonl:
            self.jsonl.write(json.dumps(rec) + "\n")
            self.jsonl.flush()
        if self.wandb:
            self.wandb.log(metrics, step=step)
        shown = {k: (f"{v:.4g}" if isinstance(v, float) else v)
                 for k, v in rec.items() if k not in ("time",)}
        print(f"[metrics] {shown}")

    def log_samples(self, rows: list[dict], step: int, max_rows: int = 5):
        """Sample-completions table (reference rich table + wandb Table,
        grpo_trainer.py:712-724)."""
        if self.rank != 0:
            return
        for r in rows[:max_rows]:
            q = str(r.get("query", ""))[:120].replace("\n", " ")
            a = str(r.get("response", ""))[:160].replace("\n", " ")
            print(f"[sample step={step}] score={r.get('score')}: {q!r} -> {a!r}")
        if self.wandb:
            import wandb
            table = wandb.Table(columns=list(rows[0].keys()) if rows else ["query"])
            for r in rows[:max_rows]:
                table.add_data(*[str(v) for v in r.values()])
            self.wandb.log({"completions": table}, step=step)

    def close(self):
        if self.jsonl:
            self.jsonl.close()
        if self.wandb:
            self.wandb.finish()


def _scalar(v):
    try:
        import torch
        if isinstance(v, torch.Tensor):
            return float(v.detach().float().mean().item())
    except ImportError:
        pass
    if isinstance(v, float):
        return v
    if isinstance(v, (int, str, bool)) or v is None:
        return v
    return float(v)
