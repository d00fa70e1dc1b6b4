"""Gradient data-parallelism: bucketed RCCL all-reduce overlapped with
backward.

MI355X-native replacement for accelerate's DDP wrap (grpo_trainer.py:314).
xGMI is 7 point-to-point links (~153 GB/s each) per GPU, so ring collectives
are per-link bound: buckets are sized LARGE (default 64 MiB) so RCCL's
multi-channel rings keep all links busy, and reduction runs on a side
stream concurrent with the remaining backward (SURVEY.md §2.3 xGMI note).
Gradient volume in the LoRA configs is small (adapters + embed/lm_head),
so overlap makes DP cost near-zero.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


class GradReducer:
    def __init__(self, params, bucket_bytes: int = 64 << 20):
        self.params = [p for p in params if p.requires_grad]
        self.bucket_bytes = bucket_bytes
        self.enabled = dist.is_initialized() and dist.get_world_size() > 1
        self._no_sync = False
        self._handles: list = []
        self._hooked = False
        if self.enabled:
            self._register_hooks()

    # --------------------------------------------------------------- buckets
    def _register_hooks(self):
        # reverse order of registration ~ backward completion order; bucket
        # greedily by byte budget
        self._buckets: list[list[torch.nn.Parameter]] = []
        cur, cur_bytes = [], 0
        for p in reversed(self.params):
            sz = p.numel() * p.element_size()
            if cur and cur_bytes + sz > self.bucket_bytes:
                self._buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            self._buckets.append(cur)
        self._pending: dict[int, set] = {}
        self._param_bucket: dict[int, int] = {}
        for bi, bucket in enumerate(self._buckets):
            for p in bucket:
                self._param_bucket[id(p)] = bi
        for p in self.params:
            p.register_post_accumulate_grad_hook(self._hook)
        self._reset_pending()

    def _reset_pending(self):
        self._pending = {bi: {id(p) for p in b} for bi, b in enumerate(self._buckets)}

    def _hook(self, p: torch.nn.Parameter):
        if self._no_sync or not self.enabled:
            return
        bi = self._param_bucket[id(p)]
        pend = self._pending[bi]
        pend.discard(id(p))
        if not pend:
            self._launch_bucket(bi)

    def _launch_bucket(self, bi: int):
        bucket = [p for p in self._buckets[bi] if p.grad is not None]
        if not bucket:
            return
        ws = dist.get_world_size()
        flat = torch._utils._flatten_dense_tensors([p.grad for p in bucket])
        flat.div_(ws)
        h = dist.all_reduce(flat, async_op=True)
        self._handles.append((h, flat, bucket))

    # ------------------------------------------------------------------- api
    def no_sync(self):
        reducer = self

        class _Ctx:
            def __enter__(self):
                reducer._no_sync = True

            def __exit__(self, *a):
                reducer._no_sync = False
                return False

        return _Ctx()

    def finalize(self):
        """Wait for in-flight reductions and scatter results back; call after
        backward of the LAST micro-batch (accumulation boundary — reference
        accelerator.accumulate semantics, grpo_trainer.py:642,690)."""
        if not self.enabled:
            return
        # launch any bucket whose params all have grads but whose hook order
        # didn't complete (e.g. grads precomputed under no_sync)
        for bi, pend in self._pending.items():
            if pend and all(p.grad is not None for p in self._buckets[bi]):
                self._launch_bucket(bi)
        for h, flat, bucket in self._handles:
            h.wait()
            for p, g in zip(bucket, torch._utils._unflatten_dense_tensors(flat, [p.grad for p in bucket])):
                p.grad.copy_(g)
        self._handles.clear()
        self._reset_pending()
