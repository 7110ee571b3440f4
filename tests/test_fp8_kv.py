"""fp8 (OCP e4m3) KV cache: half the KV bytes per token — 2x the agents
per GPU. Scale-free storage (vLLM-style fp8_e4m3 semantics); opt in with
engine.kv_dtype: fp8. These CPU tests pin the reference-path plumbing and
the checkpoint bit-stability; kernel numerics live in test_gpu_ops.py.
"""

import pytest
import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.kvcache import KVCacheManager
from agentainer_amd.engine.llm import LLMEngine, ModelInstance
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store


def _runtime(tmp_path, tag, kv_dtype="fp8"):
    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / tag)
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    cfg.data["engine"]["kv_dtype"] = kv_dtype
    s = Store(root + "/state", sync="interval")
    torch.manual_seed(3)
    eng = LLMEngine(s, cfg, device="cpu", state_root=root)
    return Runtime(cfg, engine=eng, store=s, state_root=root)


def _agent(rt, name="f1"):
    a = rt.agents.deploy(name=name, model="tiny-llama",
                         sampling={"max_tokens": 6})
    rt.agents.start(a.id)
    return a


def test_fp8_pool_doubles_pages():
    cfg = load_config(path="/nonexistent.yaml", env={})
    ecfg = dict(cfg.get("engine"))
    ecfg["kv_pool_gb"] = 0.01
    from agentainer_amd.models.llama import LLAMA_CONFIGS
    lc = LLAMA_CONFIGS["tiny-llama"]
    n_bf16 = ModelInstance._pool_pages(lc, 16, "cpu", dict(ecfg, kv_dtype="bf16"))
    n_fp8 = ModelInstance._pool_pages(lc, 16, "cpu", dict(ecfg, kv_dtype="fp8"))
    assert n_fp8 == 2 * n_bf16


def test_fp8_chat_and_multiturn(tmp_path):
    rt = _runtime(tmp_path, "fp8")
    try:
        a = _agent(rt)
        inst = rt.engine._instances["tiny-llama"]
        assert inst.kvm.dtype == torch.float8_e4m3fn
        assert inst.kvm.k_caches[0].dtype == torch.float8_e4m3fn
        st, p1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "one"})
        assert st == 200 and p1["tokens"] == 6
        len1 = inst.kvm.seq_len(a.id)
        st, p2 = rt.agent_request(a.id, "POST", "/chat", body={"message": "two"})
        assert st == 200 and inst.kvm.seq_len(a.id) > len1
        # deterministic: same prompt on a fresh agent = same fp8 output
        b = _agent(rt, name="f2")
        st, q1 = rt.agent_request(b.id, "POST", "/chat", body={"message": "one"})
        assert q1["response"] == p1["response"]
    finally:
        rt.shutdown()


def test_fp8_checkpoint_roundtrip_bit_exact(tmp_path):
    """stop/resume of an fp8-KV agent: the continuation equals the
    uninterrupted control (checkpoint preserves the e4m3 bytes)."""
    rt = _runtime(tmp_path, "ckpt")
    try:
        a = _agent(rt, name="ck-a")
        b = _agent(rt, name="ck-b")
        r1a = rt.agent_request(a.id, "POST", "/chat", body={"message": "go"})[1]
        r1b = rt.agent_request(b.id, "POST", "/chat", body={"message": "go"})[1]
        assert r1a["response"] == r1b["response"]
        rt.agents.stop(a.id)
        rt.agents.resume(a.id)
        r2a = rt.agent_request(a.id, "POST", "/chat", body={"message": "on"})[1]
        r2b = rt.agent_request(b.id, "POST", "/chat", body={"message": "on"})[1]
        assert r2a["response"] == r2b["response"]
    finally:
        rt.shutdown()


def test_fp8_reference_attention_consistency():
    """Reference decode attention over an fp8 cache == fp32 attention over
    the dequantized cache (the CPU reference is the oracle the HIP kernel
    is tested against on the GPU)."""
    from agentainer_amd.ops import reference as R

    torch.manual_seed(0)
    n_kv, D, PS, P, B, n_q = 2, 16, 4, 9, 2, 4
    ctx = 10
    kc = torch.randn(P, n_kv, D // 8, PS, 8).to(torch.float8_e4m3fn)
    vc = torch.randn(P, n_kv, PS, D).to(torch.float8_e4m3fn)
    pt = torch.arange(1, 1 + B * 3, dtype=torch.int32).reshape(B, 3)
    sl = torch.full((B,), ctx, dtype=torch.int32)
    q = torch.randn(B, n_q, D, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    R.paged_decode_attention(out, q, kc, vc, pt, sl, 0.25)
    # oracle: same math on explicitly dequantized fp32 copies
    out32 = torch.empty_like(q)
    R.paged_decode_attention(out32, q, kc.float().to(torch.float8_e4m3fn),
                             vc.float().to(torch.float8_e4m3fn), pt, sl, 0.25)
    assert torch.equal(out, out32)
    assert out.abs().sum() > 0


def test_fp8_prefix_sharing_composes(tmp_path):
    """COW prefix sharing works with fp8 pages (shared bytes are shared
    regardless of dtype)."""
    rt = _runtime(tmp_path, "pfx")
    try:
        sp = "You are terse and always answer within one short sentence."
        a = rt.agents.deploy(name="p1", model="tiny-llama", system_prompt=sp,
                             sampling={"max_tokens": 4})
        rt.agents.start(a.id)
        b = rt.agents.deploy(name="p2", model="tiny-llama", system_prompt=sp,
                             sampling={"max_tokens": 4})
        rt.agents.start(b.id)
        assert rt.agent_request(a.id, "POST", "/chat", body={"message": "x"})[0] == 200
        assert rt.agent_request(b.id, "POST", "/chat", body={"message": "y"})[0] == 200
        inst = rt.engine._instances["tiny-llama"]
        assert inst.kvm._refs  # pages shared
    finally:
        rt.shutdown()
