"""Prefix-sharing cold-start benchmark: N agents sharing one long system
prompt — total wall + prefill tokens for every agent's FIRST chat, with
COW sharing on vs off.

python tools/prefix_bench.py [n_agents] [sys_tokens]
"""
import sys
import time

import torch

sys.path.insert(0, ".")
from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import GenRequest, LLMEngine
from agentainer_amd.registry import Manager
from agentainer_amd.store import Store

N_AGENTS = int(sys.argv[1]) if len(sys.argv) > 1 else 64
SYS_TOK = int(sys.argv[2]) if len(sys.argv) > 2 else 1024

engine = None
for sharing in (True, False):
    import gc
    import tempfile
    if engine is not None:
        engine.shutdown()
        del engine  # release the previous 200+ GB pool BEFORE reallocating
        gc.collect()
        torch.cuda.empty_cache()
    tmp = tempfile.mkdtemp(prefix=f"pfx-{sharing}-")
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = tmp
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 48.0
    cfg.data["engine"]["prefix_sharing"] = sharing
    store = Store(tmp + "/state", sync="never")
    torch.manual_seed(0)
    engine = LLMEngine(store, cfg, device="cuda", state_root=tmp)
    manager = Manager(store, engine, cfg)
    sysprompt = ("rule %d: always be terse. " * (SYS_TOK // 25))[:SYS_TOK - 12]
    agents = []
    for i in range(N_AGENTS):
        a = manager.deploy(name=f"p{i}", model="llama3-8b",
                           system_prompt=sysprompt,
                           sampling={"max_tokens": 4})
        manager.start(a.id)
        agents.append(a)
    inst = engine._instances["llama3-8b"]
    full = inst.tokenizer.encode(f"[system] {sysprompt}\n")
    n_full = (len(full) // inst.kvm.page_size) * inst.kvm.page_size
    pk = full[:n_full]  # identical prompt in BOTH modes
    reqs = []
    torch.cuda.synchronize()
    t0 = time.time()
    for a in agents:
        prompt = list(pk) + inst.tokenizer.encode(f"[user] hi from {a.id}\n")
        rq = GenRequest(agent_id=a.id, prompt_tokens=prompt, max_new=4,
                        temperature=0.0, top_p=1.0, seed=0)
        b = inst.binding(a.id)
        with inst._lock:
            b.queue.put(rq)
            inst._pump_agent(b)
        reqs.append(rq)
    steps = 0
    while not all(r.done.is_set() for r in reqs) and steps < 5000:
        inst.step()
        steps += 1
    torch.cuda.synchronize()
    wall = time.time() - t0
    assert all(r.done.is_set() and not r.error for r in reqs), \
        [r.error for r in reqs if r.error][:2]
    print(f"sharing={sharing}: {N_AGENTS} agents x {len(pk)}-tok prefix | "
          f"wall {wall:.2f}s  prefill_tokens {inst.prefill_tokens}  "
          f"used_pages {inst.kvm.used_pages}")
