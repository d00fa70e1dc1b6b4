"""Length-bucketed dynamic batching (sparse-GRPO / r1 mode).

Re-implements the reference's `_create_batches` packer
(examples/r1-v0/grpo_r1_trainer.py:410-435): sort indices by length, pack
greedily while max_len_in_bucket * count <= token_budget.  Used for both
the scoring pass (rollout budget) and the update pass (train budget) — it is
a throughput feature (pad-free GEMMs), kept even though 288 GB HBM removes
the memory pressure that motivated it on A100-40G."""
from __future__ import annotations


def create_batches(lengths: list[int], token_budget: int) -> list[list[int]]:
    """Returns buckets of ORIGINAL indices; within a bucket lengths are close
    (sorted packing), and max(len)*count <= token_budget (each bucket holds
    at least one sequence)."""
    order = sorted(range(len(lengths)), key=lambda i: lengths[i])
    buckets: list[list[int]] = []
    cur: list[int] = []
    cur_max = 0
    for idx in order:
        ln = lengths[idx]
        new_max = max(cur_max, ln)
        if cur and new_max * (len(cur) + 1) > token_budget:
            buckets.append(cur)
            cur = [idx]
            cur_max = ln
        else:
            cur.append(idx)
            cur_max = new_max
    if cur:
        buckets.append(cur)
    return buckets
