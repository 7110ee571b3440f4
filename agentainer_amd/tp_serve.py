"""Tensor-parallel server entry point.

Launch under torchrun (one rank per GPU over RCCL/xGMI):

    AGENTAINER_ENGINE_TP_DEGREE=8 \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 -m agentainer_amd.tp_serve

Rank 0 hosts the REST server + scheduler; ranks > 0 run SPMD workers
executing the broadcast step plans (engine.run_worker)."""

from __future__ import annotations

import os

import torch

from . import parallel as par
from .config import load_config
from .engine.llm import LLMEngine
from .service import Runtime
from .store import Store


def main():
    par.init_distributed()
    rank = par.tp_rank()
    cfg = load_config()
    world = par.tp_size()
    cfg.data["engine"]["tp_degree"] = world
    device = "cuda" if torch.cuda.is_available() else "cpu"
    root = cfg.state_root if rank == 0 else os.path.join(
        cfg.state_root, f"worker-{rank}")
    os.makedirs(root, exist_ok=True)
    store = Store(os.path.join(root, "state"),
                  sync=cfg.get("store", "sync", "interval"))
    engine = LLMEngine(store, cfg, device=device, state_root=root)
    if rank == 0:
        from .api.server import run_server

        rt = Runtime(cfg, engine=engine, store=store, state_root=root)
        run_server(rt)
    else:
        engine.run_worker()


if __name__ == "__main__":
    main()
