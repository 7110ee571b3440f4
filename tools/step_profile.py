"""Profile one decode step of llama3-8b: top kernels by GPU time.

Usage (GPU box):  python tools/step_profile.py [--agents 64] [--graph 0|1]
"""

import argparse
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, ".")
from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import GenRequest, LLMEngine
from agentainer_amd.registry import Manager
from agentainer_amd.store import Store

p = argparse.ArgumentParser()
p.add_argument("--agents", type=int, default=64)
p.add_argument("--graph", type=int, default=1)
p.add_argument("--model", default="llama3-8b")
p.add_argument("--ctx", type=int, default=144)
args = p.parse_args()

tmp = tempfile.mkdtemp()
cfg = load_config(path="/nonexistent.yaml", env={})
cfg.data["store"]["path"] = tmp
cfg.data["engine"]["sync_mode"] = True
cfg.data["engine"]["graph_capture"] = bool(args.graph)
cfg.data["engine"]["max_decode_batch"] = args.agents
store = Store(os.path.join(tmp, "state"), sync="never")
engine = LLMEngine(store, cfg, device="cuda", state_root=tmp)
manager = Manager(store, engine, cfg)

agents = []
for i in range(args.agents):
    a = manager.deploy(name=f"p{i}", model=args.model,
                       sampling={"max_tokens": 100000})
    manager.start(a.id)
    agents.append(a)
inst = engine._instances[args.model]

# prefill everyone to ctx tokens, then keep decoding
for a in agents:
    req = GenRequest(agent_id=a.id, prompt_tokens=list(range(3, 3 + args.ctx)),
                     max_new=100000, temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
for _ in range(8):  # prefill + a few decode steps as warmup
    inst.step()
torch.cuda.synchronize()

# timed steady-state decode
t0 = time.time()
N = 30
for _ in range(N):
    inst.step()
torch.cuda.synchronize()
t1 = time.time()
print(f"steady decode: {(t1 - t0) / N * 1000:.3f} ms/step "
      f"(batch {args.agents}, graph={bool(args.graph)})")

from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    for _ in range(10):
        inst.step()
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=18))
