#!/usr/bin/env python3
"""RCCL small-message all-reduce latency probe (decode-
time TP all-reduce on the 7-link xGMI hive is latency-bound; ring is
wrong-shaped for tiny tensors). Run on an 8-GPU node:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 tools/allreduce_probe.py

Measures stock RCCL all-reduce for decode-shaped bf16 tensors
([B, 8192], B = 1..64 — the Llama-3-70B hidden size) plus the prefill
sizes, so the one-shot-vs-ring decision is made from
data the moment multi-GPU hardware is available. Single-GPU boxes and
CPU/gloo worlds run too (for plumbing checks), with the backend noted.
"""
import os
import time

import torch
import torch.distributed as dist


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    has_gpu = torch.cuda.is_available()
    if world < 2:
        print("world_size < 2: nothing to reduce (launch under torchrun)")
        return
    backend = "nccl" if has_gpu else "gloo"
    if has_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend)
    dev = "cuda" if has_gpu else "cpu"
    dtype = torch.bfloat16 if has_gpu else torch.float32
    shapes = ([(b, 8192) for b in (1, 4, 16, 64)]          # 70B decode
              + [(b, 4096) for b in (16, 64)]              # 8B decode
              + [(2048, 8192), (8192, 8192)])              # prefill-scale
    if rank == 0:
        print(f"# backend={backend} world={world} dtype={dtype}")
        print(f"{'shape':>14s} {'bytes':>12s} {'us/op':>9s} {'alg GB/s':>9s}")
    for shape in shapes:
        t = torch.randn(*shape, dtype=dtype, device=dev)
        for _ in range(20):  # warmup
            dist.all_reduce(t)
        if has_gpu:
            torch.cuda.synchronize()
        dist.barrier()
        iters = 200 if t.numel() * t.element_size() < (1 << 24) else 20
        t0 = time.perf_counter()
        for _ in range(iters):
            dist.all_reduce(t)
        if has_gpu:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        nbytes = t.numel() * t.element_size()
        # ring algorithmic bandwidth: 2*(n-1)/n * bytes / time
        algbw = 2 * (world - 1) / world * nbytes / dt / 1e9
        if rank == 0:
            print(f"{str(tuple(shape)):>14s} {nbytes:12d} {dt*1e6:9.1f} "
                  f"{algbw:9.2f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
