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
    backend: nccl (== RCCL on ROCm) when GPUs are visible, else gloo.

    When the default backend is nccl, a SECOND gloo group is created for
    the control plane (step-plan broadcasts): control traffic must be one
    cheap CPU-tensor broadcast, never a pickled object shipped through
    GPU memory on the same RCCL channel the compute collectives use."""
    global _plan_group
    if env_world() <= 1 or dist.is_initialized():
        return dist.is_initialized()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_s))
    if backend != "gloo":
        _plan_group = dist.new_group(backend="gloo",
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


# ---------------- plan channel (control-plane broadcasts) ----------------
#
# Per-decode-step latency is the TP headline cost: the
# round-1 design pickled every step plan through broadcast_object_list,
# which on an nccl default group means host->GPU->xGMI->host per step.
# The plan channel instead broadcasts ONE fixed-capacity int64 CPU tensor
# over gloo (loopback, tens of microseconds): hot commands (decode rows,
# barriers) are pure int vectors; everything else falls back to a header
# + pickled-bytes pair on the same channel.

_plan_group = None
# op + args + up to 4096 async rows at 6 ints each (gloo broadcast is
# latency-bound — 128 B and 192 KB cost the same, tools/plan_latency.py)
_PLAN_CAP = 64 + 6 * 4096
_plan_buf: Optional[torch.Tensor] = None
_OP_PICKLE = 0  # engine op codes start at 1 (llm.OP_*)


def _plan_pg():
    return _plan_group  # None => default (gloo) group


def _buf() -> torch.Tensor:
    global _plan_buf
    if _plan_buf is None:
        _plan_buf = torch.zeros(_PLAN_CAP, dtype=torch.int64)
    return _plan_buf


def send_ints(vec: List[int], src: int = 0) -> None:
    """Rank src: broadcast a small int vector (vec[0] must be an engine op
    code >= 1). ONE gloo CPU-tensor broadcast — the decode hot path."""
    if not is_tp():
        return
    buf = _buf()
    n = len(vec)
    assert 1 + n <= _PLAN_CAP and vec[0] >= 1
    buf[0] = n
    buf[1:1 + n] = torch.tensor(vec, dtype=torch.int64)
    dist.broadcast(buf, src=src, group=_plan_pg())


def send_obj(obj: Any, src: int = 0) -> None:
    """Rank src: broadcast an arbitrary picklable command (bind/unbind/
    prefill plans/...): header then bytes, both over the gloo channel."""
    if not is_tp():
        return
    payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    buf = _buf()
    buf[0] = 2
    buf[1] = _OP_PICKLE
    buf[2] = len(payload)
    dist.broadcast(buf, src=src, group=_plan_pg())
    bt = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
    dist.broadcast(bt, src=src, group=_plan_pg())


def recv_cmd(src: int = 0) -> Any:
    """Worker ranks: receive the next command. Returns a List[int] for int
    vector commands (first element = engine op code) or the unpickled
    object for everything else."""
    buf = _buf()
    dist.broadcast(buf, src=src, group=_plan_pg())
    n = int(buf[0])
    vec = buf[1:1 + n].tolist()
    if vec and vec[0] == _OP_PICKLE:
        nbytes = int(vec[1])
        bt = torch.empty(nbytes, dtype=torch.uint8)
        dist.broadcast(bt, src=src, group=_plan_pg())
        return pickle.loads(bt.numpy().tobytes())
    return vec


def broadcast_obj(obj: Any = None, src: int = 0) -> Any:
    """Compatibility wrapper over the plan channel: rank src sends `obj`,
    other ranks receive it. (Old callers did broadcast_obj(None) on the
    worker side; that is recv_cmd now, kept working here.)"""
    if not is_tp():
        return obj
    if dist.get_rank() == src:
        send_obj(obj, src=src)
        return obj
    return recv_cmd(src=src)


def barrier():
    if is_tp():
        dist.barrier()


def all_reduce_(t: torch.Tensor, group=None):
    if is_tp():
        dist.all_reduce(t, group=group or tp_group())
    return t
