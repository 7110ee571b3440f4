"""GPU end-to-end engine tests (tiny-llama on MI355X): generation runs on
the HIP kernels, determinism, KV checkpoint restore on device."""

import tempfile

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from agentainer_amd import ops
from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import GenRequest, LLMEngine
from agentainer_amd.registry import Manager
from agentainer_amd.store import Store


@pytest.fixture()
def gpu_rt(tmp_path):
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path)
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 2.0
    store = Store(str(tmp_path / "state"), sync="never")
    engine = LLMEngine(store, cfg, device="cuda", state_root=str(tmp_path))
    manager = Manager(store, engine, cfg)
    yield engine, manager, store
    engine.shutdown()


def _gen(engine, manager, agent, prompt, max_new=8):
    inst = engine._instances[agent.model]
    req = GenRequest(agent_id=agent.id, prompt_tokens=prompt, max_new=max_new,
                     temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(agent.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
    for _ in range(max_new + 8):
        inst.step()
        if req.done.is_set():
            break
    torch.cuda.synchronize()
    assert req.done.is_set() and not req.error, req.error
    return req.generated


def test_hip_extension_required(gpu_rt):
    assert ops.hip_available()


def test_generation_deterministic(gpu_rt):
    engine, manager, _ = gpu_rt
    a = manager.deploy(name="g1", model="tiny-llama")
    manager.start(a.id)
    b = manager.deploy(name="g2", model="tiny-llama")
    manager.start(b.id)
    prompt = list(range(3, 35))
    out1 = _gen(engine, manager, a, prompt)
    out2 = _gen(engine, manager, b, prompt)
    assert out1 == out2
    assert len(out1) == 8


def test_model_forward_matches_cpu_reference(gpu_rt):
    """Tiny model, same seed: GPU HIP-kernel forward logits vs CPU
    reference-op forward logits (the whole-stack numerics gate)."""
    engine, manager, store = gpu_rt
    a = manager.deploy(name="x", model="tiny-llama")
    manager.start(a.id)
    inst = engine._instances["tiny-llama"]
    prompt = list(range(3, 35))
    gpu_tokens = _gen(engine, manager, a, prompt, max_new=4)

    # CPU twin with identical weights
    from agentainer_amd.engine.llm import LLMEngine as CpuEngine
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    with tempfile.TemporaryDirectory() as td:
        cstore = Store(td + "/state", sync="never")
        ceng = CpuEngine(cstore, cfg, device="cpu", state_root=td)
        cman = Manager(cstore, ceng, cfg)
        ca = cman.deploy(name="x", model="tiny-llama")
        cman.start(ca.id)
        cinst = ceng._instances["tiny-llama"]
        # copy GPU weights onto the CPU twin (same init seed but device
        # RNGs differ — exact copy removes that variable)
        sd = {k: v.cpu() for k, v in inst.model.state_dict().items()}
        cinst.model.load_state_dict(sd)
        cpu_tokens = _gen(ceng, cman, ca, prompt, max_new=4)
    assert gpu_tokens == cpu_tokens, (
        f"GPU kernels diverge from CPU reference: {gpu_tokens} vs {cpu_tokens}")


def test_kv_checkpoint_roundtrip_gpu(gpu_rt):
    engine, manager, _ = gpu_rt
    a = manager.deploy(name="ck", model="tiny-llama")
    manager.start(a.id)
    ctl = manager.deploy(name="ctl", model="tiny-llama")
    manager.start(ctl.id)
    p1 = list(range(3, 35))
    assert _gen(engine, manager, a, p1) == _gen(engine, manager, ctl, p1)
    manager.stop(a.id)   # offload to pinned host
    manager.resume(a.id)  # upload back
    p2 = list(range(40, 72))
    out_a = _gen(engine, manager, a, p2)
    out_ctl = _gen(engine, manager, ctl, p2)
    assert out_a == out_ctl  # restored KV produces identical continuation


def test_mixtral_generation_gpu(gpu_rt):
    """tiny-mixtral end-to-end on the HIP kernels + MoE routing."""
    engine, manager, _ = gpu_rt
    a = manager.deploy(name="mx", model="tiny-mixtral")
    manager.start(a.id)
    b = manager.deploy(name="mx2", model="tiny-mixtral")
    manager.start(b.id)
    prompt = list(range(3, 35))
    out1 = _gen(engine, manager, a, prompt)
    out2 = _gen(engine, manager, b, prompt)
    assert out1 == out2 and len(out1) == 8


def test_mixtral_matches_cpu_reference(gpu_rt):
    engine, manager, store = gpu_rt
    a = manager.deploy(name="mxr", model="tiny-mixtral")
    manager.start(a.id)
    inst = engine._instances["tiny-mixtral"]
    prompt = list(range(3, 35))
    gpu_tokens = _gen(engine, manager, a, prompt, max_new=4)
    from agentainer_amd.engine.llm import LLMEngine as CpuEngine
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    with tempfile.TemporaryDirectory() as td:
        cstore = Store(td + "/state", sync="never")
        ceng = CpuEngine(cstore, cfg, device="cpu", state_root=td)
        cman = Manager(cstore, ceng, cfg)
        ca = cman.deploy(name="mxr", model="tiny-mixtral")
        cman.start(ca.id)
        cinst = ceng._instances["tiny-mixtral"]
        sd = {k: v.cpu() for k, v in inst.model.state_dict().items()}
        cinst.model.load_state_dict(sd)
        cpu_tokens = _gen(ceng, cman, ca, prompt, max_new=4)
    assert gpu_tokens == cpu_tokens, (gpu_tokens, cpu_tokens)
