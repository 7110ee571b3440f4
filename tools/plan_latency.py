#!/usr/bin/env python3
"""Measure the TP control-plane cost: one decode-step plan broadcast on
the gloo plan channel (parallel.dist.send_ints/recv_cmd), vs the pickled
broadcast_object_list it replaced.

This is the per-token host overhead rank 0 adds to every TP decode step
(the round-1 path pickled the plan per step). Runs on CPU
(world 2 over loopback) — the gloo channel is the SAME one used under
RCCL on MI355X (init_distributed creates it alongside the nccl group),
so loopback latency here is the real per-step cost there.

    python tools/plan_latency.py          # spawns both ranks itself
"""
import os
import socket
import sys
import time

import torch
import torch.multiprocessing as mp


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def worker(rank, world, port, iters, batch):
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist

    from agentainer_amd import parallel as par

    par.init_distributed(backend="gloo")
    vec = [1, 0, batch] + list(range(batch)) + [7] * batch  # decode plan
    plan_obj = ("decode", "llama3-70b", [(f"agent-{i}", 7) for i in range(batch)])

    # ---- int-vector plan channel (the round-2 hot path) ----
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        if rank == 0:
            par.send_ints(vec)
        else:
            par.recv_cmd()
    dist.barrier()
    t_ints = (time.perf_counter() - t0) / iters

    # ---- pickled broadcast_object_list (the round-1 path) ----
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        holder = [plan_obj if rank == 0 else None]
        dist.broadcast_object_list(holder, src=0)
    dist.barrier()
    t_obj = (time.perf_counter() - t0) / iters

    if rank == 0:
        print(f"batch={batch:4d}  plan-channel {t_ints*1e6:7.1f} us/step   "
              f"pickled broadcast_object_list {t_obj*1e6:7.1f} us/step   "
              f"({t_obj/t_ints:.1f}x)")
    dist.barrier()
    dist.destroy_process_group()


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    for batch in (16, 64, 256):
        port = _free_port()
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=worker, args=(r, 2, port, iters, batch))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0


if __name__ == "__main__":
    main()
