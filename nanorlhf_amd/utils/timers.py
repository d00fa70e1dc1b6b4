"""Phase timers + HBM high-water tracking — a real observability subsystem
(the reference only had a per-update `s/episode` print and commented-out
memory hooks, SURVEY.md §5 "Tracing / profiling")."""
from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager

import torch


class PhaseTimers:
    """Wall-clock per named phase (rollout / reward / score / update / ...).
    CUDA-synchronizing at boundaries when on GPU so the numbers line up with
    rocprof kernel time."""

    def __init__(self, sync_cuda: bool = True):
        self.totals: dict[str, float] = defaultdict(float)
        self.counts: dict[str, int] = defaultdict(int)
        self.sync_cuda = sync_cuda

    @contextmanager
    def phase(self, name: str):
        if self.sync_cuda and torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.sync_cuda and torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            self.totals[name] += dt
            self.counts[name] += 1

    def snapshot_and_reset(self) -> dict[str, float]:
        out = {f"time/{k}": v for k, v in self.totals.items()}
        if torch.cuda.is_available():
            out["mem/hbm_peak_gb"] = torch.cuda.max_memory_allocated() / 2**30
            torch.cuda.reset_peak_memory_stats()
        self.totals.clear()
        self.counts.clear()
        return out
