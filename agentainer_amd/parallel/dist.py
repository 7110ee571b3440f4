"""Distributed runtime over RCCL/xGMI (torch.distributed backend "nccl").

The reference has no data-plane communication beyond HTTP-over-bridge
(SURVEY.md §5 "Distributed communication backend"); the MI355X engine's
tensor traffic runs on RCCL collectives over the 8-GPU xGMI hive:

  * TP (Llama-3-70B, 8-way): column-parallel qkv/gate_up, row-parallel
    o/down with all-reduce per sub-layer (models/llama.py).
  * EP (Mixtral): expert all-to-all (models/mixtral.py).

Control-plane design ("SPMD workers, replicated scheduler state"): rank 0
owns the scheduler; step plans (sequence ids + token ids only) are
broadcast; every rank executes the same plan against its own shard and
its own KV pool. Page allocation is deterministic, so per-rank page
tables evolve identically with no per-step tensor metadata exchange.
CPU CI runs the identical code over gloo (world_size 2).
"""

from __future__ import annotations

import datetime
import os
import pickle
from typing import Any, List, Optional

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def init_distributed(backend: Optional[str] = None, timeout_s: float = 300.0):
    """Initialize torch.distributed from torchrun env (no-op for world 1).
    backend: nccl (== RCCL on ROCm) when GPUs are visible, else gloo."""
    if env_world() <= 1 or dist.is_initialized():
        return dist.is_initialized()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_s))
    return True


def is_tp() -> bool:
    return dist.is_initialized() and dist.get_world_size() > 1


def tp_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def tp_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def tp_group():
    return dist.group.WORLD if dist.is_initialized() else None


def broadcast_obj(obj: Any = None, src: int = 0) -> Any:
    """Broadcast a picklable object from src to all ranks (control plane —
    step plans, lifecycle commands; tensor traffic uses collectives)."""
    if not is_tp():
        return obj
    holder: List[Any] = [obj]
    dist.broadcast_object_list(holder, src=src)
    return holder[0]


def barrier():
    if is_tp():
        dist.barrier()


def all_reduce_(t: torch.Tensor, group=None):
    if is_tp():
        dist.all_reduce(t, group=group or tp_group())
    return t
