"""Mixtral MoE on CPU: router math, engine chat, determinism."""

import pytest
import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import LLMEngine
from agentainer_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralMoE
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store


def test_moe_routing_matches_manual():
    """Dense-routed MoE == explicit per-token top-k expert mixture."""
    torch.manual_seed(0)
    cfg = MIXTRAL_CONFIGS["tiny-mixtral"]
    moe = MixtralMoE(cfg)
    for p in moe.parameters():
        p.data.normal_(0, 0.05)
    h = torch.randn(5, cfg.hidden_size, dtype=torch.bfloat16)
    out = moe(h)

    # manual: route each token through its top-k experts explicitly
    logits = torch.nn.functional.linear(h, moe.router).float()
    probs = torch.softmax(logits, -1)
    topv, topi = probs.topk(cfg.top_k, -1)
    topv = topv / topv.sum(-1, keepdim=True)
    want = torch.zeros(5, cfg.hidden_size)
    for t in range(5):
        for j in range(cfg.top_k):
            e = int(topi[t, j])
            gu = torch.nn.functional.linear(h[t:t + 1], moe.gate_up[e])
            gate, up = gu[:, :moe.inter].float(), gu[:, moe.inter:].float()
            act = (torch.nn.functional.silu(gate) * up).to(torch.bfloat16)
            eo = torch.nn.functional.linear(act, moe.down[e]).float()
            want[t] += float(topv[t, j]) * eo[0]
    assert torch.allclose(out.float(), want, atol=0.05, rtol=0.05)


@pytest.fixture()
def mixtral_rt(tmp_path):
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path / "root")
    cfg.data["engine"]["kv_pool_gb"] = 0.02
    s = Store(str(tmp_path / "root" / "state"), sync="interval")
    eng = LLMEngine(s, cfg, device="cpu", state_root=str(tmp_path / "root"))
    rt = Runtime(cfg, engine=eng, store=s, state_root=str(tmp_path / "root"))
    yield rt
    rt.shutdown()


def test_mixtral_chat_roundtrip(mixtral_rt):
    rt = mixtral_rt
    a = rt.agents.deploy(name="mx", model="tiny-mixtral",
                         sampling={"max_tokens": 6})
    rt.agents.start(a.id)
    st, p1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "hello"})
    assert st == 200 and p1["model"] == "tiny-mixtral" and p1["tokens"] == 6
    b = rt.agents.deploy(name="mx2", model="tiny-mixtral",
                         sampling={"max_tokens": 6})
    rt.agents.start(b.id)
    st, p2 = rt.agent_request(b.id, "POST", "/chat", body={"message": "hello"})
    assert p2["response"] == p1["response"]  # greedy determinism


def test_mixtral_stop_resume_kv(mixtral_rt):
    rt = mixtral_rt
    a = rt.agents.deploy(name="ck", model="tiny-mixtral",
                         sampling={"max_tokens": 6})
    ctl = rt.agents.deploy(name="ctl", model="tiny-mixtral",
                           sampling={"max_tokens": 6})
    rt.agents.start(a.id)
    rt.agents.start(ctl.id)
    r1 = rt.agent_request(a.id, "POST", "/chat", body={"message": "one"})[1]
    rt.agent_request(ctl.id, "POST", "/chat", body={"message": "one"})
    rt.agents.stop(a.id)
    rt.agents.resume(a.id)
    r2a = rt.agent_request(a.id, "POST", "/chat", body={"message": "two"})[1]
    r2b = rt.agent_request(ctl.id, "POST", "/chat", body={"message": "two"})[1]
    assert r2a["response"] == r2b["response"]


def test_sparse_prefill_experts_match_dense():
    """Prefill-scale batches route sparsely (gather per expert, 2/8 of
    the FLOPs); the output must match the dense-routed reference."""
    import torch

    from agentainer_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralMoE

    torch.manual_seed(4)
    cfg = MIXTRAL_CONFIGS["tiny-mixtral"]
    moe = MixtralMoE(cfg)
    g = torch.Generator().manual_seed(4)
    for p in moe.parameters():
        p.data.normal_(0, 0.05, generator=g)
    h = torch.randn(100, cfg.hidden_size, dtype=torch.bfloat16)
    sparse = moe(h.clone())                    # T=100 >= SPARSE_MIN_TOKENS
    try:
        moe.SPARSE_MIN_TOKENS = 10**9          # force the dense path
        dense = moe(h.clone())
    finally:
        del moe.SPARSE_MIN_TOKENS              # restore class attribute
    diff = (sparse.float() - dense.float()).abs().max().item()
    assert diff < 0.05, diff
    # decode-scale batches stay dense (hipGraph-capturable)
    small = moe(h[:8].clone())
    assert small.shape == (8, cfg.hidden_size)
