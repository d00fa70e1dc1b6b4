"""Async host-offload engine.

Replaces the reference's synchronous blocking `.to('cpu')/.to('cuda')`
shuttles (policy grpo_trainer.py:124,164; ref model :525,622; reward model
grpo.py:164,195; optimizer state :475,625 via state_to_device :168-172).

MI355X-first: offload is a POLICY, not a reflex — with 288 GB HBM3E the
1.5B-class policy+ref+reward all stay resident and `maybe_offload` is a
no-op; for 7B+ configs (or user-forced), transfers use pinned host buffers
+ hipMemcpyAsync on a dedicated side stream so the copy overlaps compute
(`torch.cuda.Stream` → HIP stream on ROCm)."""
from __future__ import annotations

import torch


class OffloadEngine:
    def __init__(self, device: torch.device, enabled: bool | None = None,
                 reserve_gb: float = 24.0):
        self.device = device
        self.reserve_bytes = int(reserve_gb * 2**30)
        self.enabled = enabled  # None → auto by memory pressure
        self.stream = torch.cuda.Stream(device) if (torch.cuda.is_available()
                                                    and device.type == "cuda") else None
        self._pinned: dict[int, torch.Tensor] = {}

    # ------------------------------------------------------------------ policy
    def should_offload(self, extra_bytes_needed: int = 0) -> bool:
        if self.enabled is not None:
            return self.enabled
        if self.device.type != "cuda":
            return False
        free, _total = torch.cuda.mem_get_info(self.device)
        return free < self.reserve_bytes + extra_bytes_needed

    # ------------------------------------------------------------------ moves
    def _pinned_like(self, t: torch.Tensor) -> torch.Tensor:
        key = id(t)
        buf = self._pinned.get(key)
        if buf is None or buf.shape != t.shape or buf.dtype != t.dtype:
            buf = torch.empty_like(t, device="cpu", pin_memory=True)
            self._pinned[key] = buf
        return buf

    def model_to_host(self, model: torch.nn.Module, non_blocking: bool = True):
        if self.stream is None:
            model.to("cpu")
            return
        with torch.cuda.stream(self.stream):
            for p in list(model.parameters()) + list(model.buffers()):
                if p.device.type != "cuda":
                    continue
                buf = self._pinned_like(p.data)
                buf.copy_(p.data, non_blocking=non_blocking)
                p.data = buf

    def model_to_device(self, model: torch.nn.Module, non_blocking: bool = True):
        if self.stream is None:
            model.to(self.device)
            return
        with torch.cuda.stream(self.stream):
            for p in list(model.parameters()) + list(model.buffers()):
                if p.device.type == "cuda":
                    continue
                p.data = p.data.to(self.device, non_blocking=non_blocking)

    def optimizer_state_to(self, optimizer: torch.optim.Optimizer, device):
        """state_to_device equivalent (grpo_trainer.py:168-172), async."""
        if self.stream is not None:
            ctx = torch.cuda.stream(self.stream)
        else:
            ctx = _null_ctx()
        with ctx:
            for group in optimizer.param_groups:
                for p in group["params"]:
                    st = optimizer.state.get(p)
                    if not st:
                        continue
                    for k, v in st.items():
                        if isinstance(v, torch.Tensor):
                            if str(device) == "cpu" and v.device.type == "cuda":
                                buf = self._pinned_like(v)
                                buf.copy_(v, non_blocking=True)
                                st[k] = buf
                            elif str(device) != "cpu" and v.device.type == "cpu":
                                st[k] = v.to(device, non_blocking=True)

    def synchronize(self):
        if self.stream is not None:
            self.stream.synchronize()


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
