"""Async host-offload engine.

Replaces the reference's synchronous blocking `.to('cpu')/.to('cuda')`
shuttles (policy grpo_trainer.py:124,164; ref model :525,622; reward model
grpo.py:164,195; optimizer state :475,625 via state_to_device :168-172).

MI355X-first: offload is a POLICY, not a reflex — with 288 GB HBM3E the
1.5B-class policy+ref+reward all stay resident and `should_offload` is a
no-op; for 7B+ configs (or user-forced), transfers use pinned host buffers
+ hipMemcpyAsync on a dedicated side stream so the copy overlaps compute
(`torch.cuda.Stream` → HIP stream on ROCm).

Stream-ordering contract (the round-1 race fix):
  * device→host: the side stream first WAITS on the compute stream (pending
    kernels that write the tensors must land before the copy reads them),
    and each source GPU tensor is `record_stream`ed on the side stream so
    the caching allocator cannot recycle its block before the copy drains.
  * host→device: copies run on the side stream and record a ready event;
    `join_compute()` makes the compute stream wait on that event (no host
    sync), `synchronize()` host-blocks (needed before CPU reads of the
    pinned buffers).
"""
from __future__ import annotations

import itertools

import torch


class OffloadEngine:
    def __init__(self, device: torch.device, enabled: bool | None = None,
                 reserve_gb: float = 24.0):
        self.device = device
        self.reserve_bytes = int(reserve_gb * 2**30)
        self.enabled = enabled  # None → auto by memory pressure
        self.stream = torch.cuda.Stream(device) if (torch.cuda.is_available()
                                                    and device.type == "cuda") else None
        self._pinned: dict[tuple, torch.Tensor] = {}
        self._ready_event: torch.cuda.Event | None = None

    # ------------------------------------------------------------------ policy
    def should_offload(self, extra_bytes_needed: int = 0) -> bool:
        if self.enabled is not None:
            return self.enabled
        if self.device.type != "cuda":
            return False
        free, _total = torch.cuda.mem_get_info(self.device)
        return free < self.reserve_bytes + extra_bytes_needed

    # ------------------------------------------------------------------ moves
    def _pinned_buf(self, key: tuple, t: torch.Tensor) -> torch.Tensor:
        """Stable-keyed pinned host buffer (keys survive device round trips —
        keying by id(tensor) would leak one buffer per shuttle)."""
        buf = self._pinned.get(key)
        if buf is None or buf.shape != t.shape or buf.dtype != t.dtype:
            buf = torch.empty_like(t, device="cpu", pin_memory=True)
            self._pinned[key] = buf
        return buf

    @staticmethod
    def _named_tensors(model: torch.nn.Module):
        return itertools.chain(model.named_parameters(), model.named_buffers())

    def model_to_host(self, model: torch.nn.Module, non_blocking: bool = True):
        if self.stream is None:
            model.to("cpu")
            return
        # copies must run after any compute that produced/uses these params
        self.stream.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(self.stream):
            for name, p in self._named_tensors(model):
                if p.device.type != "cuda":
                    continue
                buf = self._pinned_buf((id(model), name), p.data)
                buf.copy_(p.data, non_blocking=non_blocking)
                # block allocator reuse of the GPU tensor until the copy drains
                p.data.record_stream(self.stream)
                p.data = buf

    def model_to_device(self, model: torch.nn.Module, non_blocking: bool = True):
        if self.stream is None:
            model.to(self.device)
            return
        with torch.cuda.stream(self.stream):
            for name, p in self._named_tensors(model):
                if p.device.type == "cuda":
                    continue
                p.data = p.data.to(self.device, non_blocking=non_blocking)
        self._record_ready()

    def optimizer_state_to(self, optimizer: torch.optim.Optimizer, device):
        """state_to_device equivalent (grpo_trainer.py:168-172), async."""
        to_host = str(device) == "cpu"
        if self.stream is not None:
            if to_host:
                self.stream.wait_stream(torch.cuda.current_stream(self.device))
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
                        if not isinstance(v, torch.Tensor):
                            continue
                        if to_host and v.device.type == "cuda":
                            buf = self._pinned_buf((id(p), k), v)
                            buf.copy_(v, non_blocking=True)
                            if self.stream is not None:
                                v.record_stream(self.stream)
                            st[k] = buf
                        elif not to_host and v.device.type == "cpu":
                            st[k] = v.to(device, non_blocking=True)
        if not to_host:
            self._record_ready()

    # ------------------------------------------------------------- sync points
    def _record_ready(self):
        if self.stream is not None:
            self._ready_event = torch.cuda.Event()
            self._ready_event.record(self.stream)

    def join_compute(self):
        """Compute-stream-side barrier: subsequent kernels on the current
        stream see the transferred tensors; the host does not block."""
        if self.stream is None:
            return
        if self._ready_event is not None:
            torch.cuda.current_stream(self.device).wait_event(self._ready_event)
        else:
            torch.cuda.current_stream(self.device).wait_stream(self.stream)

    def synchronize(self):
        """Host-side barrier (required before CPU code reads pinned buffers)."""
        if self.stream is not None:
            self.stream.synchronize()


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
