#!/usr/bin/env python3
"""bench.py — concurrent-agent chat throughput on MI355X.

Measures BASELINE.json's headline metric: concurrent-agent chat req/sec
(+ p50 E2E latency) for N stateful agents sharing one model under the
continuous-batching engine, synthetic prompts, random-init weights, greedy
bf16 decode, WAL + history writes included in the timed region (the full
config-3 serving path).

A STEP is one engine iteration: admit+prefill (token-budgeted) then one
batched decode step for every running sequence. The harness keeps every
agent closed-loop (one outstanding chat request each; a completed request
is immediately replaced), so the decode batch stays at the agent count.

Scaling mode is WEAK data parallelism: each rank (GPU) runs its own engine
replica serving its own `--agents` agents (the reference's replica fan-out
semantics); ranks synchronize only at the timing barriers.
Whole-job value = sum of per-rank completed requests / max elapsed.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Tensor-parallel mode (BASELINE config 4, Llama-3-70B TP=8 over xGMI):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --tp 8 --steps K --warmup W
Rank 0 schedules ONE engine sharded across all ranks (broadcast step
plans, RCCL all-reduce per sub-layer); value = rank 0's completed
requests / barrier-bounded elapsed.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import tempfile
import time

import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import OP_BARRIER, GenRequest, LLMEngine
from agentainer_amd.registry import Manager
from agentainer_amd.store import Store
from agentainer_amd.wal import RequestManager


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--warmup", type=int, default=60)
    p.add_argument("--agents", type=int, default=64)
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree (= world size); default "
                        "model becomes llama3-70b (config 4)")
    p.add_argument("--model", default=None,
                   help="default: llama3-8b on GPU, tiny-llama on CPU "
                        "(llama3-70b / tiny-llama-tp when --tp > 1)")
    p.add_argument("--prompt-len", type=int, default=256)
    p.add_argument("--gen-len", type=int, default=64)
    p.add_argument("--max-batch-tokens", type=int, default=16384)
    p.add_argument("--expert-fp8", action="store_true",
                   help="fp8 MFMA expert decode GEMMs (Mixtral, config 5)")
    p.add_argument("--kv-fp8", action="store_true",
                   help="OCP e4m3 KV cache (half the KV bytes, 2x agents)")
    p.add_argument("--expert-fp4", action="store_true",
                   help="MXFP4 block-scaled expert GEMMs (quarter bytes)")
    p.add_argument("--dense-quant", default="", choices=["", "fp8", "mxfp4"],
                   help="opt-in quantized dense decode projections (turbo; "
                        "NOT the headline dtype)")
    return p.parse_args()


class ClosedLoopClient:
    """One agent's closed loop: WAL append -> submit -> on completion WAL
    ack + history write -> next request."""

    def __init__(self, agent, inst, store, wal, prompt_len, gen_len, seed):
        self.agent = agent
        self.inst = inst
        self.store = store
        self.wal = wal
        self.prompt_len = prompt_len
        self.gen_len = gen_len
        self.rng = torch.Generator().manual_seed(seed)
        self.active = None
        self.active_wal = None
        self.completed = []  # (t_submit, t_done)

    def _prompt_tokens(self):
        v = self.inst.cfg.vocab_size
        return torch.randint(3, v, (self.prompt_len,), generator=self.rng).tolist()

    def submit(self):
        toks = self._prompt_tokens()
        wal_req = self.wal.store_request(self.agent.id, "POST", "/chat",
                                         body={"tokens": self.prompt_len})
        req = GenRequest(agent_id=self.agent.id, prompt_tokens=toks,
                         max_new=self.gen_len, temperature=0.0, top_p=1.0, seed=0)
        b = self.inst.binding(self.agent.id)
        with self.inst._lock:
            b.queue.put(req)
            self.inst._pump_agent(b)
        self.active = req
        self.active_wal = wal_req

    def poll(self, record: bool):
        if self.active is not None and self.active.done.is_set():
            req = self.active
            self.wal.store_response(self.agent.id, self.active_wal.id,
                                    {"tokens": len(req.generated)})
            hist = f"agent:{self.agent.id}:conversations"
            self.store.rpush(hist, {"user": "bench", "tokens": len(req.generated),
                                    "ts": req.fin_t})
            if self.store.llen(hist) > 50:
                self.store.ltrim(hist, -50, -1)
            if record:
                self.completed.append((req.enq_t, req.fin_t))
            # KV hygiene: schedule a conversation reset so seq length stays
            # bounded. pending_reset is applied AT ADMISSION inside the
            # engine (plan-mirrored), so TP workers reset in lockstep and
            # no pages are freed under an in-flight speculative step.
            b = self.inst.binding(self.agent.id)
            if b is not None:
                b.pending_reset = True
            self.submit()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    has_gpu = torch.cuda.is_available()
    device = "cuda" if has_gpu else "cpu"
    if has_gpu:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group(backend="nccl" if has_gpu else "gloo")

    tp = max(args.tp, 1)
    if tp > 1 and world != tp:
        raise SystemExit(f"--tp {tp} requires a torchrun world of {tp} "
                         f"(got WORLD_SIZE={world})")
    if tp > 1:
        model_name = args.model or ("llama3-70b" if has_gpu else "tiny-llama-tp")
    else:
        model_name = args.model or ("llama3-8b" if has_gpu else "tiny-llama")
    if not has_gpu and args.model is None:
        # CPU fallback sizes so the no-GPU smoke run finishes in seconds
        args.agents = min(args.agents, 4)
        args.prompt_len = min(args.prompt_len, 32)
        args.gen_len = min(args.gen_len, 8)
        args.steps = min(args.steps, 40)
        args.warmup = min(args.warmup, 8)
    # any driver-chosen --steps must complete requests INSIDE the timed
    # region (a gen_len longer than the window yields a vacuous 0 req/s):
    # cap generation length at half the timed steps
    args.gen_len = max(1, min(args.gen_len, max(args.steps // 2, 1)))

    tmp = tempfile.mkdtemp(prefix=f"bench-rank{rank}-")
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = tmp
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["max_batch_tokens"] = args.max_batch_tokens
    cfg.data["engine"]["max_decode_batch"] = max(args.agents, 1)
    cfg.data["engine"]["expert_fp8"] = bool(args.expert_fp8)
    cfg.data["engine"]["expert_fp4"] = bool(args.expert_fp4)
    cfg.data["engine"]["dense_quant"] = args.dense_quant
    if args.kv_fp8:
        cfg.data["engine"]["kv_dtype"] = "fp8"
    if tp > 1:
        cfg.data["engine"]["tp_degree"] = tp
    store = Store(os.path.join(tmp, "state"), sync="interval")
    engine = LLMEngine(store, cfg, device=device, state_root=tmp)
    if tp > 1 and rank != 0:
        # SPMD worker: execute rank 0's broadcast step plans (incl. the
        # "barrier" timing fences) until shutdown
        engine.run_worker()
        dist.destroy_process_group()
        return
    manager = Manager(store, engine, cfg)
    wal = RequestManager(store)

    def fence():
        """Synchronize + barrier on every rank (TP: via the plan channel)."""
        if tp > 1:
            import agentainer_amd.parallel as par
            par.send_ints([OP_BARRIER])
        if has_gpu:
            torch.cuda.synchronize()
        if dist:
            dist.barrier()

    t_load0 = time.time()
    agents = []
    for i in range(args.agents):
        a = manager.deploy(name=f"bench-{i}", model=model_name,
                           sampling={"max_tokens": args.gen_len})
        manager.start(a.id)
        agents.append(a)
    inst = engine._instances[model_name]
    if rank == 0:
        print(f"# model={model_name} load+attach {time.time()-t_load0:.1f}s "
              f"pages={inst.kvm.n_pages}", file=sys.stderr)

    clients = [ClosedLoopClient(a, inst, store, wal, args.prompt_len,
                                args.gen_len, seed=1000 * rank + i)
               for i, a in enumerate(agents)]
    for c in clients:
        c.submit()

    def run_steps(n, record):
        for _ in range(n):
            inst.step()
            for c in clients:
                c.poll(record)

    # warmup
    run_steps(args.warmup, record=False)
    for c in clients:
        c.completed.clear()

    # timed region
    fence()
    t0 = time.time()
    run_steps(args.steps, record=True)
    fence()
    t1 = time.time()

    elapsed = t1 - t0
    n_done = sum(len(c.completed) for c in clients)
    e2es = sorted(d - s for c in clients for (s, d) in c.completed)
    decode_tokens = inst.decode_tokens
    prefill_tokens = inst.prefill_tokens

    if dist and tp == 1:
        # weak DP: whole-job aggregate = sum of completions / MAX elapsed,
        # with the e2e populations of EVERY rank merged so p50/p99 are
        # whole-job too (not rank-0-only)
        dd = device if has_gpu else "cpu"
        stats = torch.tensor([elapsed, float(n_done)],
                             dtype=torch.float64, device=dd)
        gathered = [torch.zeros_like(stats) for _ in range(world)]
        dist.all_gather(gathered, stats)
        elapsed = max(float(g[0]) for g in gathered)  # MAX over ranks
        n_done = sum(float(g[1]) for g in gathered)
        n = torch.tensor([len(e2es)], dtype=torch.int64, device=dd)
        ns = [torch.zeros_like(n) for _ in range(world)]
        dist.all_gather(ns, n)
        width = max(int(x.item()) for x in ns)
        if width > 0:
            buf = torch.full((width,), float("nan"), dtype=torch.float64,
                             device=dd)
            if e2es:
                buf[: len(e2es)] = torch.tensor(e2es, dtype=torch.float64,
                                                device=dd)
            bufs = [torch.zeros_like(buf) for _ in range(world)]
            dist.all_gather(bufs, buf)
            e2es = sorted(v for b in bufs for v in b.tolist() if v == v)

    if rank == 0:
        p50 = statistics.median(e2es) if e2es else None
        p99 = (e2es[min(len(e2es) - 1, int(0.99 * len(e2es)))] if e2es else None)
        out = {
            "metric": "concurrent_agent_chat_req_per_s",
            "value": round(n_done / elapsed, 3),
            "unit": "req/s",
            "n_gpus": world if has_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "strong" if tp > 1 else "weak",
            "vs_baseline": None,  # reference publishes no number
            "dtype": (f"bf16+{args.dense_quant}w" if args.dense_quant else
                      "bf16+fp4exp" if args.expert_fp4 else
                      "bf16+fp8exp" if args.expert_fp8 else
                      "bf16+fp8kv" if args.kv_fp8 else "bf16"),
            "data": "synthetic",
            "config": {
                "model": model_name,
                "agents_per_gpu": args.agents if tp == 1 else None,
                "agents": args.agents * (max(world, 1) if tp == 1 else 1),
                "global_batch": args.agents * (max(world, 1) if tp == 1 else 1),
                "prompt_len": args.prompt_len,
                "gen_len": args.gen_len,
                "seq_len": args.prompt_len + args.gen_len,
                "parallelism": (f"tp{world}" if tp > 1 else f"dp{max(world,1)}"),
                "p50_e2e_s": round(p50, 4) if p50 is not None else None,
                "p99_e2e_s": round(p99, 4) if p99 is not None else None,
                "decode_tokens_rank0": decode_tokens,
                "prefill_tokens_rank0": prefill_tokens,
            },
        }
        print(json.dumps(out))
    if tp > 1:
        engine.shutdown()  # broadcasts shutdown so workers exit run_worker
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
