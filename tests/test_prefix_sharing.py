"""Copy-on-write prefix sharing: agents with a common system prompt share
the whole-page head of their KV sequence (refcounted pages, adopted at
first prefill). Correctness bar: shared and unshared runs produce the SAME
greedy tokens, and page accounting balances under every lifecycle op.

No reference analog (the reference gives every container its own
filesystem); this is the MI355X-native win from putting every agent's
conversation state in one HBM page pool.
"""

import pytest
import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.kvcache import KVCacheManager
from agentainer_amd.engine.llm import LLMEngine
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store

SYSPROMPT = ("You are a terse assistant for the flight-booking team. "
             "Answer in one sentence and never guess prices.")


# ---------------------------------------------------------------- kvm unit

def _kvm(n_pages=64):
    return KVCacheManager(n_layers=2, n_kv=2, head_dim=16, page_size=16,
                          n_pages=n_pages, device="cpu")


def test_adopt_shares_pages_and_refcounts():
    kvm = _kvm()
    kvm.create_seq("pfx")
    kvm.append_slots("pfx", 32)  # 2 full pages
    free0 = kvm.free_pages
    kvm.create_seq("a")
    kvm.adopt_prefix("a", "pfx", 32)
    assert kvm.free_pages == free0  # no new pages allocated
    assert kvm._seqs["a"].pages == kvm._seqs["pfx"].pages
    assert kvm.seq_len("a") == 32
    # appends go to a NEW page, never the shared ones
    slots = kvm.append_slots("a", 1)
    assert kvm._seqs["a"].pages[-1] not in kvm._seqs["pfx"].pages
    assert slots[0] // 16 == kvm._seqs["a"].pages[-1]


def test_release_order_either_way():
    for order in (("a", "b", "pfx"), ("pfx", "a", "b")):
        kvm = _kvm()
        kvm.create_seq("pfx")
        kvm.append_slots("pfx", 16)
        base = kvm.free_pages
        for sid in ("a", "b"):
            kvm.create_seq(sid)
            kvm.adopt_prefix(sid, "pfx", 16)
            kvm.append_slots(sid, 3)
        assert kvm.free_pages == base - 2  # one private page each
        for sid in order:
            kvm.free_seq(sid)
        assert kvm.free_pages == base + 1  # everything back, page freed once
        assert not kvm._refs


def test_reset_seq_releases_shared_pages():
    kvm = _kvm()
    kvm.create_seq("pfx")
    kvm.append_slots("pfx", 16)
    kvm.create_seq("a")
    kvm.adopt_prefix("a", "pfx", 16)
    free0 = kvm.free_pages
    kvm.reset_seq("a")  # context truncation on an adopter
    assert kvm.free_pages == free0  # shared page NOT freed (pfx owns it)
    assert kvm._seqs["pfx"].pages and not kvm._refs
    kvm.free_seq("pfx")
    assert kvm.free_pages == free0 + 1


def test_adopt_requires_empty_and_whole_pages():
    kvm = _kvm()
    kvm.create_seq("pfx")
    kvm.append_slots("pfx", 20)
    kvm.create_seq("a")
    with pytest.raises(ValueError):
        kvm.adopt_prefix("a", "pfx", 20)  # not a page multiple
    with pytest.raises(ValueError):
        kvm.adopt_prefix("a", "pfx", 32)  # longer than source
    kvm.adopt_prefix("a", "pfx", 16)
    with pytest.raises(ValueError):
        kvm.adopt_prefix("a", "pfx", 16)  # adopter not empty


def test_can_append_after_reset_ignores_shared_pages():
    kvm = _kvm(n_pages=4)  # pages 1..3 usable (page 0 = scratch)
    kvm.create_seq("pfx")
    kvm.append_slots("pfx", 32)  # 2 pages
    kvm.create_seq("a")
    kvm.adopt_prefix("a", "pfx", 32)
    kvm.append_slots("a", 1)     # last page
    assert kvm.free_pages == 0
    # resetting "a" frees only its private page — 2 shared stay with pfx
    assert kvm.can_append_after_reset("a", 16)
    assert not kvm.can_append_after_reset("a", 32)


# ------------------------------------------------------------- engine level

def _runtime(tmp_path, tag, prefix_sharing=True):
    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / tag)
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    cfg.data["engine"]["prefix_sharing"] = prefix_sharing
    s = Store(root + "/state", sync="interval")
    eng = LLMEngine(s, cfg, device="cpu", state_root=root)
    return Runtime(cfg, engine=eng, store=s, state_root=root)


def _chat(rt, aid, msg):
    st, p = rt.agent_request(aid, "POST", "/chat", body={"message": msg})
    assert st == 200, p
    return p


def _deploy(rt, name):
    a = rt.agents.deploy(name=name, model="tiny-llama",
                         system_prompt=SYSPROMPT,
                         sampling={"max_tokens": 8})
    rt.agents.start(a.id)
    return a


def test_engine_prefix_sharing_same_tokens_less_memory(tmp_path):
    shared = _runtime(tmp_path, "shared", prefix_sharing=True)
    plain = _runtime(tmp_path, "plain", prefix_sharing=False)
    try:
        outs, agents = {}, {}
        for rt, key in ((shared, "s"), (plain, "p")):
            # identical random-init weights: the instance is created at
            # first attach, so seed right before deploying
            torch.manual_seed(0)
            a1 = _deploy(rt, "agent1")
            a2 = _deploy(rt, "agent2")
            agents[key] = (a1, a2)
            outs[key] = [_chat(rt, a1.id, "where is my bag?")["response"],
                         _chat(rt, a2.id, "can I change seats?")["response"],
                         _chat(rt, a1.id, "thanks!")["response"]]
        a1, a2 = agents["s"]
        # greedy outputs identical with and without sharing
        assert outs["s"] == outs["p"]
        si = shared.engine._instances["tiny-llama"]
        pi = plain.engine._instances["tiny-llama"]
        assert len(si._prefixes) == 1
        assert si.kvm._refs  # pages actually shared
        n_pfx_pages = len(SYSPROMPT.encode()) // si.kvm.page_size
        assert n_pfx_pages >= 2
        # sharing saves at least (n_agents - 1) * (n_pfx_pages - 1) pages
        # net of the prefix sequence's own copy
        assert si.kvm.used_pages < pi.kvm.used_pages
        # detach an adopter: shared pages survive for the other agent
        shared.agents.stop(a1.id)
        assert si.kvm._refs
        p = _chat(shared, a2.id, "one more thing")
        assert p["tokens"] == 8  # still serving off the shared prefix
    finally:
        shared.shutdown()
        plain.shutdown()


def test_prefix_not_adopted_for_short_prompt(tmp_path):
    rt = _runtime(tmp_path, "short", prefix_sharing=True)
    try:
        a = rt.agents.deploy(name="s1", model="tiny-llama",
                             system_prompt="hi", sampling={"max_tokens": 4})
        rt.agents.start(a.id)
        _chat(rt, a.id, "hello")
        inst = rt.engine._instances["tiny-llama"]
        assert not inst._prefixes  # "[system] hi\n" < one 16-token page
        assert not inst.kvm._refs
    finally:
        rt.shutdown()


def test_stats_expose_sharing(tmp_path):
    rt = _runtime(tmp_path, "stats", prefix_sharing=True)
    try:
        for n in ("s1", "s2"):
            a = rt.agents.deploy(name=n, model="tiny-llama",
                                 system_prompt=SYSPROMPT,
                                 sampling={"max_tokens": 3})
            rt.agents.start(a.id)
            rt.agent_request(a.id, "POST", "/chat", body={"message": "x"})
        st = rt.engine.stats()["models"]["tiny-llama"]
        assert st["shared_prefixes"] == 1
        assert st["kv_pages_shared"] >= 2
        assert st["kv_dtype"] == "bfloat16"
    finally:
        rt.shutdown()
