"""Watchdog + fault-injection: a wedged engine step (stuck HIP stream
analog) must never deadlock the recovery path (SURVEY.md §5 failure
detection; reference triple-detection semantics monitor.go:207-251).

The injected fault is a step that blocks forever while holding the step
mutex — exactly what a hung kernel does to the engine thread. The
watchdog verdict (instance.stuck()) must flip health to down, and the
health monitor's restart chain must COMPLETE (stop force-detaches
instead of blocking behind the wedged step; start fails loudly -> agent
status 'failed') rather than hanging the single monitor thread.
"""

import threading
import time

import pytest

from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import EngineDead, LLMEngine
from agentainer_amd.health.monitor import HealthMonitor
from agentainer_amd.registry import FAILED, RUNNING, STOPPED
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store


@pytest.fixture()
def rt(tmp_path):
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path / "root")
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    s = Store(str(tmp_path / "root" / "state"), sync="interval")
    eng = LLMEngine(s, cfg, device="cpu", state_root=str(tmp_path / "root"))
    r = Runtime(cfg, engine=eng, store=s, state_root=str(tmp_path / "root"))
    yield r
    r.shutdown()


def _wedge(inst, release: threading.Event):
    """Make the engine thread block forever inside a step (mutex held)."""
    inst.watchdog_timeout_s = 0.3

    def stuck_admit():
        release.wait(timeout=60.0)
        return []

    inst._admit = stuck_admit
    inst._wake.set()  # kick the loop into the wedged step


def test_stuck_step_trips_watchdog_and_health(rt):
    a = rt.agents.deploy(name="w1", model="tiny-llama", auto_restart=True,
                         sampling={"max_tokens": 4})
    rt.agents.start(a.id)
    inst = rt.engine._instances["tiny-llama"]
    release = threading.Event()
    try:
        _wedge(inst, release)
        deadline = time.time() + 5.0
        while time.time() < deadline and not inst.stuck():
            time.sleep(0.05)
        assert inst.stuck()
        assert rt.engine.health_probe(a.id) is False
    finally:
        release.set()


def test_stuck_engine_restart_does_not_deadlock(rt):
    """HealthMonitor.check_one -> restart -> stop ->
    detach must not block forever on the step mutex a wedged step holds."""
    a = rt.agents.deploy(name="w2", model="tiny-llama", auto_restart=True,
                         sampling={"max_tokens": 4})
    rt.agents.start(a.id)
    inst = rt.engine._instances["tiny-llama"]
    release = threading.Event()
    try:
        _wedge(inst, release)
        while not inst.stuck():
            time.sleep(0.05)
        mon = HealthMonitor(rt.store, rt.agents, retries=1)
        mon.start_monitoring(a.id)
        done = threading.Event()
        result = {}

        def run_check():
            result["status"] = mon.check_one(a.id)
            done.set()

        t = threading.Thread(target=run_check, daemon=True)
        t.start()
        # the whole unhealthy -> restart -> force-detach chain completes
        # well inside the old-deadlock horizon
        assert done.wait(timeout=20.0), "health check deadlocked on stuck engine"
        assert result["status"]["healthy"] is False
        # restart ran: stop force-detached (no hang); start re-binds, which
        # raises EngineDead against the wedged instance -> status 'failed'
        agent = rt.agents.get(a.id)
        assert agent.status in (FAILED, STOPPED, RUNNING)
        assert agent.status != RUNNING or not inst.stuck()
    finally:
        release.set()


def test_bind_on_stuck_engine_raises(rt):
    a = rt.agents.deploy(name="w3", model="tiny-llama",
                         sampling={"max_tokens": 4})
    rt.agents.start(a.id)
    inst = rt.engine._instances["tiny-llama"]
    release = threading.Event()
    try:
        _wedge(inst, release)
        while not inst.stuck():
            time.sleep(0.05)
        with pytest.raises(EngineDead):
            inst.bind(a, seq_id="new-seq", ckpt=None)
        # export_kv gives up instead of hanging
        assert rt.engine.export_kv(a.id) is None
    finally:
        release.set()


def test_clear_resets_engine_kv(rt):
    """/clear must reset the KV sequence, not just the
    history list — otherwise the next chat attends to the cleared
    conversation and doubles the system prompt."""
    a = rt.agents.deploy(name="c1", model="tiny-llama",
                         system_prompt="sys prompt here",
                         sampling={"max_tokens": 4})
    rt.agents.start(a.id)
    inst = rt.engine._instances["tiny-llama"]
    st, _ = rt.agent_request(a.id, "POST", "/chat", body={"message": "one"})
    assert st == 200
    st, _ = rt.agent_request(a.id, "POST", "/chat", body={"message": "two"})
    assert st == 200
    len_two_turns = inst.kvm.seq_len(a.id)
    st, p = rt.agent_request(a.id, "POST", "/clear", body={})
    assert st == 200 and p["status"] == "cleared"
    st, _ = rt.agent_request(a.id, "POST", "/chat", body={"message": "one"})
    assert st == 200
    len_after_clear = inst.kvm.seq_len(a.id)
    # post-clear KV = exactly one turn's worth (system prompt re-prefilled
    # onto an EMPTY sequence), strictly less than the two-turn context
    assert len_after_clear < len_two_turns
    # and a fresh agent's first turn matches it exactly (same prompt shape)
    b = rt.agents.deploy(name="c2", model="tiny-llama",
                         system_prompt="sys prompt here",
                         sampling={"max_tokens": 4})
    rt.agents.start(b.id)
    rt.agent_request(b.id, "POST", "/chat", body={"message": "one"})
    assert inst.kvm.seq_len(b.id) == len_after_clear


def test_admission_reservation_fails_cleanly(rt):
    """Admission reserves prompt+max_new KV room up front: an oversized
    request fails with an explicit error instead of starving mid-prefill
    (chunked-prefill OutOfPages)."""
    inst_probe = rt.agents.deploy(name="r0", model="tiny-llama",
                                  sampling={"max_tokens": 4})
    rt.agents.start(inst_probe.id)
    inst = rt.engine._instances["tiny-llama"]
    n_free = inst.kvm.free_pages
    ps = inst.kvm.page_size
    # fill most of the pool with one agent's long conversation
    hog = rt.agents.deploy(name="hog", model="tiny-llama",
                           sampling={"max_tokens": 4})
    rt.agents.start(hog.id)
    filler = "x" * int(n_free * ps * 0.8)
    st, p = rt.agent_request(hog.id, "POST", "/chat",
                             body={"message": filler[: inst.max_batch_tokens * 3]})
    # either served (chunked) or failed explicitly -- never hung
    assert st in (200, 500, 503), p
    # now a second big request: must resolve quickly with a clean verdict
    t0 = time.time()
    st2, p2 = rt.agent_request(inst_probe.id, "POST", "/chat",
                               body={"message": "y" * (n_free * ps * 2)})
    assert time.time() - t0 < 30.0
    if st2 != 200:
        blob = str(p2)
        assert ("KV pool exhausted" in blob or "prompt too long" in blob
                or "cap" in blob), p2
