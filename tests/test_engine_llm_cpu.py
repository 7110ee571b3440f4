"""LLM engine on CPU (tiny-llama, reference ops): continuous batching,
multi-tenancy, KV checkpoint/restore determinism, WAL integration."""

import threading

import pytest
import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import LLMEngine
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store


@pytest.fixture()
def llm_runtime(tmp_path):
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path / "root")
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    s = Store(str(tmp_path / "root" / "state"), sync="interval")
    eng = LLMEngine(s, cfg, device="cpu", state_root=str(tmp_path / "root"))
    rt = Runtime(cfg, engine=eng, store=s, state_root=str(tmp_path / "root"))
    yield rt
    rt.shutdown()


def _mk_agent(rt, name="llm1", **kw):
    kw.setdefault("sampling", {"max_tokens": 8})
    a = rt.agents.deploy(name=name, model="tiny-llama", **kw)
    rt.agents.start(a.id)
    return a


def test_chat_roundtrip_and_determinism(llm_runtime):
    rt = llm_runtime
    a = _mk_agent(rt)
    st, p1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "hello"})
    assert st == 200
    assert p1["tokens"] == 8
    assert p1["model"] == "tiny-llama"
    # same prompt from a fresh agent => identical greedy output
    b = _mk_agent(rt, name="llm2")
    st, p2 = rt.agent_request(b.id, "POST", "/chat", body={"message": "hello"})
    assert st == 200
    assert p2["response"] == p1["response"]
    # history stored
    hist = rt.store.lrange(f"agent:{a.id}:conversations")
    assert len(hist) == 1 and hist[0]["assistant"] == p1["response"]


def test_multi_turn_kv_grows(llm_runtime):
    rt = llm_runtime
    a = _mk_agent(rt)
    inst = rt.engine._instances["tiny-llama"]
    rt.agent_request(a.id, "POST", "/chat", body={"message": "turn one"})
    len1 = inst.kvm.seq_len(a.id)
    rt.agent_request(a.id, "POST", "/chat", body={"message": "turn two"})
    len2 = inst.kvm.seq_len(a.id)
    assert len2 > len1 > 0


def test_stop_resume_restores_kv_exactly(llm_runtime):
    """stop = KV offload; resume = KV upload. The continuation after a
    stop/resume must equal the uninterrupted continuation (config 4's
    KV-restore correctness)."""
    rt = llm_runtime
    a = _mk_agent(rt, name="ckpt-a")
    b = _mk_agent(rt, name="ckpt-b")  # control: never stopped
    r1a = rt.agent_request(a.id, "POST", "/chat", body={"message": "alpha"})[1]
    r1b = rt.agent_request(b.id, "POST", "/chat", body={"message": "alpha"})[1]
    assert r1a["response"] == r1b["response"]
    # stop a (offloads KV to host), then resume (restores)
    rt.agents.stop(a.id)
    agent = rt.agents.get(a.id)
    assert agent.kv_offloaded is True
    inst = rt.engine._instances["tiny-llama"]
    assert not inst.kvm.has_seq(a.id)
    rt.agents.resume(a.id)
    assert inst.kvm.seq_len(a.id) > 0
    r2a = rt.agent_request(a.id, "POST", "/chat", body={"message": "beta"})[1]
    r2b = rt.agent_request(b.id, "POST", "/chat", body={"message": "beta"})[1]
    assert r2a["response"] == r2b["response"]


def test_kv_checkpoint_survives_engine_restart(tmp_path):
    """Disk-persisted pinned-host checkpoint: a NEW engine (server restart)
    restores the conversation KV."""
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    root = str(tmp_path / "root")
    s = Store(root + "/state", sync="always")
    eng = LLMEngine(s, cfg, device="cpu", state_root=root)
    rt = Runtime(cfg, engine=eng, store=s, state_root=root)
    a = rt.agents.deploy(name="persist", model="tiny-llama",
                         sampling={"max_tokens": 8})
    rt.agents.start(a.id)
    r1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "alpha"})[1]
    rt.agents.stop(a.id)  # offload + disk persist
    rt.shutdown()
    # new process: fresh store handle + fresh engine
    s2 = Store(root + "/state", sync="always")
    eng2 = LLMEngine(s2, cfg, device="cpu", state_root=root)
    rt2 = Runtime(cfg, engine=eng2, store=s2, state_root=root)
    rt2.agents.resume(a.id)
    inst = eng2._instances["tiny-llama"]
    assert inst.kvm.seq_len(a.id) > 0  # restored from disk checkpoint
    # control continuation
    b = rt2.agents.deploy(name="ctl", model="tiny-llama",
                          sampling={"max_tokens": 8})
    rt2.agents.start(b.id)
    rt2.agent_request(b.id, "POST", "/chat", body={"message": "alpha"})
    r2a = rt2.agent_request(a.id, "POST", "/chat", body={"message": "beta"})[1]
    r2b = rt2.agent_request(b.id, "POST", "/chat", body={"message": "beta"})[1]
    assert r2a["response"] == r2b["response"]
    rt2.shutdown()


def test_concurrent_agents_batched(llm_runtime):
    """8 agents chat concurrently; scheduler batches them; all succeed and
    each agent's response equals the single-agent greedy output."""
    rt = llm_runtime
    agents = [_mk_agent(rt, name=f"c{i}") for i in range(8)]
    results = {}
    def do(agent):
        st, p = rt.agent_request(agent.id, "POST", "/chat",
                                 body={"message": "same prompt"})
        results[agent.id] = (st, p)
    threads = [threading.Thread(target=do, args=(a,)) for a in agents]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert len(results) == 8
    outs = set()
    for st, p in results.values():
        assert st == 200
        outs.add(p["response"])
    assert len(outs) == 1  # identical greedy output for identical prompts


def test_crash_replay_llm(llm_runtime):
    """Kill-replay against the real engine: queued requests replay after
    resume and produce the deterministic greedy response."""
    rt = llm_runtime
    a = _mk_agent(rt, name="crash")
    ref = _mk_agent(rt, name="ref")
    want = rt.agent_request(ref.id, "POST", "/chat", body={"message": "rep"})[1]
    rt.agents.stop(a.id)  # simulate downtime
    st, payload = rt.agent_request(a.id, "POST", "/chat", body={"message": "rep"})
    assert st == 202
    rid = payload["data"]["request_id"]
    rt.agents.resume(a.id)
    n = rt.replay.tick()
    assert n == 1
    done = rt.requests.get(a.id, rid)
    assert done.status == "completed"
    assert done.response["response"] == want["response"]


def test_paused_agent_queues(llm_runtime):
    rt = llm_runtime
    a = _mk_agent(rt, name="pausey")
    rt.agents.pause(a.id)
    st, _ = rt.agent_request(a.id, "POST", "/chat", body={"message": "x"})
    assert st == 202  # paused => queued, not dispatched
    rt.agents.resume(a.id)
    assert rt.replay.tick() == 1


def test_engine_stats_shape(llm_runtime):
    rt = llm_runtime
    a = _mk_agent(rt, name="stats")
    rt.agent_request(a.id, "POST", "/chat", body={"message": "m"})
    st = rt.engine.stats()
    assert a.id in st["agents"]
    assert st["agents"][a.id]["requests"] == 1
    assert st["agents"][a.id]["kv_pages"] > 0
    assert "tiny-llama" in st["models"]
    assert st["models"]["tiny-llama"]["decode_tokens"] > 0


def test_chat_stream_matches_blocking(llm_runtime):
    """Token-streamed chat produces the same text as the blocking path
    (greedy decode, fresh agents, same prompt)."""
    rt = llm_runtime
    a = _mk_agent(rt, name="st-a")
    b = _mk_agent(rt, name="st-b")
    blocking = rt.engine.chat(a.id, "stream parity")
    events = list(rt.engine.chat_stream(b.id, "stream parity"))
    final = events[-1]
    assert final.get("done") is True
    assert final["response"] == blocking["response"]
    assert final["tokens"] == blocking["tokens"]
    toks = [e for e in events[:-1] if e.get("token") is not None]
    assert len(toks) == blocking["tokens"]
    assert "".join(e["text"] for e in events[:-1]) == blocking["response"]
    # history recorded for the streamed turn too
    hist = rt.store.lrange(f"agent:{b.id}:conversations")
    assert len(hist) == 1 and hist[0]["assistant"] == final["response"]


def test_runtime_stream_wal_ack(llm_runtime):
    rt = llm_runtime
    a = _mk_agent(rt, name="st-wal")
    status, gen = rt.agent_request_stream(a.id, {"message": "ack me"})
    assert status == 200
    events = list(gen)
    assert events[-1].get("done") is True
    done = rt.requests.by_queue(a.id, "completed")
    assert len(done) == 1
    assert done[0].response["response"] == events[-1]["response"]


def test_chunked_prefill_matches_unchunked(tmp_path):
    """A prompt longer than max_batch_tokens prefills across several
    engine steps (decode interleaves); the generated tokens must equal a
    single-shot prefill of the same prompt with the same weights."""
    from agentainer_amd.config import load_config
    from agentainer_amd.store import Store

    outs = {}
    for tag, mbt in (("big", 8192), ("small", 48)):
        cfg = load_config(path="/nonexistent.yaml", env={})
        root = str(tmp_path / tag)
        cfg.data["store"]["path"] = root
        cfg.data["engine"]["kv_pool_gb"] = 0.01
        cfg.data["engine"]["max_batch_tokens"] = mbt
        s = Store(root + "/state", sync="interval")
        torch.manual_seed(11)
        eng = LLMEngine(s, cfg, device="cpu", state_root=root)
        rt = Runtime(cfg, engine=eng, store=s, state_root=root)
        try:
            a = rt.agents.deploy(name="long", model="tiny-llama",
                                 sampling={"max_tokens": 5})
            rt.agents.start(a.id)
            long_msg = "alpha beta gamma " * 12  # ~200 byte tokens
            st, p = rt.agent_request(a.id, "POST", "/chat",
                                     body={"message": long_msg})
            assert st == 200, p
            assert p["tokens"] == 5
            outs[tag] = p["response"]
            inst = rt.engine._instances["tiny-llama"]
            # the whole prompt landed in KV exactly once
            assert inst.kvm.seq_len(a.id) >= len(long_msg)
            if tag == "small":
                assert inst.prefill_tokens > mbt  # really ran multiple slices
                assert not inst._chunking
        finally:
            rt.shutdown()
    assert outs["big"] == outs["small"]


def test_chunked_prefill_streams_and_truncates(tmp_path):
    """Chunked prefill composes with SSE streaming (first token after the
    last slice) and with context-truncation resets."""
    from agentainer_amd.config import load_config
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "r")
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    cfg.data["engine"]["max_batch_tokens"] = 40
    s = Store(root + "/state", sync="interval")
    eng = LLMEngine(s, cfg, device="cpu", state_root=root)
    rt = Runtime(cfg, engine=eng, store=s, state_root=root)
    try:
        a = rt.agents.deploy(name="lng", model="tiny-llama",
                             sampling={"max_tokens": 4})
        rt.agents.start(a.id)
        msg = "delta echo foxtrot " * 8  # ~150 tokens >> 40 budget
        events = list(rt.engine.chat_stream(a.id, msg))
        assert events[-1]["done"] is True and events[-1]["tokens"] == 4
        toks = [e for e in events[:-1] if e.get("token") is not None]
        assert len(toks) == 4
        inst = rt.engine._instances["tiny-llama"]
        assert not inst._chunking
        # keep chatting until the pool forces a truncation reset, chunked
        for i in range(30):
            st, p = rt.agent_request(a.id, "POST", "/chat",
                                     body={"message": msg})
            if st != 200:
                break
            assert p["tokens"] == 4
        assert st == 200  # kept serving through resets
    finally:
        rt.shutdown()


def test_no_page_or_slot_leaks_across_lifecycle(llm_runtime):
    """Pages and sequence slots return to the pool across repeated
    chat -> stop -> resume -> remove cycles (leak regression: refcounted
    prefix pages + checkpoint restore paths both allocate)."""
    rt = llm_runtime
    seed = rt.agents.deploy(name="leak-seed", model="tiny-llama",
                            system_prompt="always reply in fewer than ten words",
                            sampling={"max_tokens": 4})
    rt.agents.start(seed.id)
    rt.agent_request(seed.id, "POST", "/chat", body={"message": "warm"})
    inst = rt.engine._instances["tiny-llama"]
    free0 = inst.kvm.free_pages
    slots0 = len(inst.kvm._free_slots)
    for i in range(3):
        a = rt.agents.deploy(name=f"leak-{i}", model="tiny-llama",
                             system_prompt="always reply in fewer than ten words",
                             sampling={"max_tokens": 4})
        rt.agents.start(a.id)
        rt.agent_request(a.id, "POST", "/chat", body={"message": f"x{i}"})
        rt.agents.stop(a.id)     # offload to checkpoint
        rt.agents.resume(a.id)   # restore (fresh pages)
        rt.agent_request(a.id, "POST", "/chat", body={"message": f"y{i}"})
        rt.agents.remove(a.id, request_manager=rt.requests)
    assert inst.kvm.free_pages == free0, (free0, inst.kvm.free_pages)
    assert len(inst.kvm._free_slots) == slots0
    assert not any(s.startswith("agent-leak") for s in inst.kvm._seqs)


def test_remove_purges_disk_checkpoint(llm_runtime, tmp_path):
    rt = llm_runtime
    import glob as _glob
    a = _mk_agent(rt, name="purge-me")
    rt.agent_request(a.id, "POST", "/chat", body={"message": "hi"})
    rt.agents.stop(a.id)  # creates a disk checkpoint
    ckpts = _glob.glob(f"{rt.engine.state_root}/kv_ckpt/{a.id}*.pt")
    assert ckpts, "stop should persist a checkpoint"
    rt.agents.remove(a.id, request_manager=rt.requests)
    assert not _glob.glob(f"{rt.engine.state_root}/kv_ckpt/{a.id}*.pt")


def test_multi_model_cohosting(llm_runtime):
    """Two model FAMILIES share one engine/GPU: agents on tiny-llama and
    tiny-mixtral serve interleaved, each instance with its own KV pool
    and scheduler; lifecycle ops on one family don't disturb the other."""
    rt = llm_runtime
    a = _mk_agent(rt, name="llama-side")
    m = rt.agents.deploy(name="moe-side", model="tiny-mixtral",
                         sampling={"max_tokens": 4})
    rt.agents.start(m.id)
    r1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "one"})
    r2 = rt.agent_request(m.id, "POST", "/chat", body={"message": "one"})
    assert r1[0] == 200 and r2[0] == 200
    assert set(rt.engine._instances) == {"tiny-llama", "tiny-mixtral"}
    # stop the llama agent; the mixtral agent keeps serving
    rt.agents.stop(a.id)
    r3 = rt.agent_request(m.id, "POST", "/chat", body={"message": "two"})
    assert r3[0] == 200
    # engine stats report both models
    st = rt.engine.stats()
    assert {"tiny-llama", "tiny-mixtral"} <= set(st["models"])


def test_decode_batch_cap_is_fair(tmp_path):
    """running > max_decode_batch must round-robin, not starve the tail
    (found as a 34.6s p99 at 512 agents against the 256-row cap)."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import GenRequest, LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path / "root")
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["max_decode_batch"] = 2   # oversubscribe: 6 agents
    s = Store(str(tmp_path / "root" / "state"), sync="never")
    eng = LLMEngine(s, cfg, device="cpu", state_root=str(tmp_path / "root"))
    man = Manager(s, eng, cfg)
    reqs = []
    for i in range(6):
        a = man.deploy(name=f"fair{i}", model="tiny-llama",
                       sampling={"max_tokens": 5})
        man.start(a.id)
        inst = eng._instances["tiny-llama"]
        r = GenRequest(agent_id=a.id, prompt_tokens=list(range(3, 19)),
                       max_new=5, temperature=0.0, top_p=1.0, seed=0)
        b = inst.binding(a.id)
        with inst._lock:
            b.queue.put(r)
            inst._pump_agent(b)
        reqs.append(r)
    # each of the 6 rows needs 4 decode steps at 2 rows/step => ~12 decode
    # steps + prefills; WITH starvation the last rows would need the first
    # ones to fully finish first, which still converges — so assert the
    # stronger property: progress interleaves (no row finishes 5 tokens
    # before every row has at least 1)
    inst = eng._instances["tiny-llama"]
    for _ in range(40):
        inst.step()
        lens = [len(r.generated) for r in reqs]
        if max(lens) >= 5:
            assert min(lens) >= 1, f"tail starved: {lens}"
        if all(r.done.is_set() for r in reqs):
            break
    if inst.async_decode:
        inst.drain_async()
    assert all(r.done.is_set() and not r.error for r in reqs)
    eng.shutdown()
    s.close()
