"""Phase-level timing of the decode step (sync boundaries added)."""
import os, sys, tempfile, time
import torch
sys.path.insert(0, ".")
from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import GenRequest, LLMEngine
from agentainer_amd.registry import Manager
from agentainer_amd.store import Store

tmp = tempfile.mkdtemp()
cfg = load_config(path="/nonexistent.yaml", env={})
cfg.data["store"]["path"] = tmp
cfg.data["engine"]["sync_mode"] = True
store = Store(os.path.join(tmp, "state"), sync="never")
engine = LLMEngine(store, cfg, device="cuda", state_root=tmp)
manager = Manager(store, engine, cfg)
agents = []
for i in range(64):
    a = manager.deploy(name=f"p{i}", model="llama3-8b",
                       sampling={"max_tokens": 100000})
    manager.start(a.id)
    agents.append(a)
inst = engine._instances["llama3-8b"]
for a in agents:
    req = GenRequest(agent_id=a.id, prompt_tokens=list(range(3, 147)),
                     max_new=100000, temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
for _ in range(8):
    inst.step()
torch.cuda.synchronize()

# instrument _decode_gpu phases by monkeypatching pieces
import agentainer_amd.engine.llm as L
T = {"prep": 0.0, "replay": 0.0, "sample": 0.0, "book": 0.0}
orig_sample = inst._sample
def timed_step():
    with inst._lock:
        reqs = [r for r in inst.running if not r.done.is_set()][:inst.max_decode_batch]
    kvm = inst.kvm
    t0 = time.time()
    B = len(reqs)
    bucket = min(inst._bucket(B), max(inst.max_decode_batch, 1))
    row_ids = []
    for r in reqs:
        b = inst._bindings[r.agent_id]
        kvm.ensure_decode_page(b.seq_id)
        row_ids.append(kvm.slot(b.seq_id))
    entry = inst._get_graph(bucket)
    entry["rows_pin"][:B] = torch.tensor(row_ids, dtype=torch.long)
    entry["rows_pin"][B:] = inst._pad_slot
    entry["ids_pin"][:B] = torch.tensor([r.generated[-1] for r in reqs], dtype=torch.long)
    entry["rows"].copy_(entry["rows_pin"], non_blocking=True)
    entry["ids"].copy_(entry["ids_pin"], non_blocking=True)
    torch.cuda.synchronize(); t1 = time.time()
    entry["graph"].replay()
    torch.cuda.synchronize(); t2 = time.time()
    logits = entry["logits"][:B]
    toks = orig_sample(logits, reqs)
    torch.cuda.synchronize(); t3 = time.time()
    with inst._lock:
        for r in reqs:
            b = inst._bindings.get(r.agent_id)
            if b is not None:
                kvm.advance_host(b.seq_id)
        for r, t in zip(reqs, toks):
            r.generated.append(int(t))
            inst._finish_or_run(r, int(t))
    t4 = time.time()
    T["prep"] += t1-t0; T["replay"] += t2-t1; T["sample"] += t3-t2; T["book"] += t4-t3

N = 30
t0 = time.time()
for _ in range(N):
    timed_step()
wall = time.time() - t0
print(f"instrumented wall {wall/N*1000:.3f} ms/step")
for k, v in T.items():
    print(f"  {k:8s} {v/N*1000:8.3f} ms")
