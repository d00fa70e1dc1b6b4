"""Distributed runtime: one process per GPU, RCCL over xGMI.

Replaces the reference's implicit accelerate layer.
Backend "nccl" IS RCCL on ROCm; CPU tests use gloo.  Collectives used:
broadcast (run timestamp sync, grpo_trainer.py:241-242), all_reduce/
all_gather for metrics (:729-740), and the gradient all-reduce of ddp.py.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: int = 1800) -> tuple[int, int, int]:
    """Returns (rank, local_rank, world_size); degenerates to (0,0,1) when not
    launched under torchrun."""
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 0, 1
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    world = int(os.environ["WORLD_SIZE"])
    if world == 1 and not dist.is_initialized():
        return 0, 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=timeout_s))
    return rank, local_rank, world


def is_main() -> bool:
    return (not dist.is_initialized()) or dist.get_rank() == 0


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def barrier():
    if dist.is_initialized():
        dist.barrier()


def broadcast_scalar(value: float, src: int = 0, device="cpu") -> float:
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.broadcast(t, src)
    return float(t.item())


def gather_mean(value: torch.Tensor | float, device="cpu") -> float:
    """Cross-rank mean of a scalar metric (the reference's
    accelerator.gather(x).mean(), grpo_trainer.py:729-740)."""
    if isinstance(value, torch.Tensor):
        value = float(value.detach().float().mean().item())
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return float(t.item()) / dist.get_world_size()


def gather_sum(value: float, device="cpu") -> float:
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return float(t.item())


def metric_device() -> str:
    """Collectives under nccl/RCCL must use GPU tensors; gloo wants CPU."""
    if dist.is_initialized() and dist.get_backend() == "nccl":
        return "cuda"
    return "cpu"
