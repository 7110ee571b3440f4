"""Reconciler convergence, health monitor auto-restart, metrics history."""

import time

from agentainer_amd.registry import FAILED, PAUSED, RUNNING, STOPPED


def _started(rt, name="r1", **kw):
    a = rt.agents.deploy(name=name, model="echo", **kw)
    rt.agents.start(a.id)
    return a


# ---------- reconciler ----------

def test_reconciler_adopts_engine_truth(runtime):
    a = _started(runtime)
    # engine-side pause without going through the manager
    runtime.engine.pause(a.id)
    runtime.reconciler.sync_agent(a.id)
    assert runtime.agents.get(a.id).status == PAUSED
    runtime.engine.unpause(a.id)
    runtime.reconciler.sync_agent(a.id)
    assert runtime.agents.get(a.id).status == RUNNING


def test_reconciler_marks_vanished_stopped(runtime):
    a = _started(runtime)
    runtime.engine.detach(a.id, offload_kv=False)  # attachment vanishes
    runtime.reconciler.sync_agent(a.id)
    assert runtime.agents.get(a.id).status == STOPPED


def test_reconciler_auto_restarts(runtime):
    a = _started(runtime, auto_restart=True)
    runtime.engine.detach(a.id, offload_kv=False)
    runtime.reconciler.sync_agent(a.id)
    assert runtime.agents.get(a.id).status == RUNNING  # restart-policy analog
    assert runtime.engine.is_attached(a.id)


def test_reconciler_detaches_orphans(runtime):
    a = _started(runtime)
    # delete the registry record behind the manager's back
    runtime.store.delete(f"agent:{a.id}")
    runtime.store.srem("agents:list", a.id)
    runtime.reconciler.sync_all()
    assert not runtime.engine.is_attached(a.id)


# ---------- health ----------

def test_health_probe_and_status(runtime):
    a = _started(runtime)
    runtime.health.start_monitoring(a.id, {"interval": 0.01, "retries": 3})
    st = runtime.health.check_one(a.id)
    assert st["healthy"] is True
    stored = runtime.health.get_status(a.id)
    assert stored["healthy"] is True
    assert a.id in runtime.health.get_all_statuses()


def test_health_auto_restart_after_retries(runtime):
    a = _started(runtime, auto_restart=True)
    runtime.health.start_monitoring(a.id, {"interval": 0.0, "retries": 3})
    runtime.engine.crash()
    # first 2 failures: no restart yet
    runtime.health.check_one(a.id)
    runtime.health.check_one(a.id)
    assert runtime.health.get_status(a.id)["consecutive_failures"] == 2
    runtime.engine.recover()  # allow the restart's attach to succeed
    runtime.health.check_one(a.id)  # 3rd failure => restart
    assert runtime.engine.is_attached(a.id)
    assert runtime.agents.get(a.id).status == RUNNING


def test_health_event_driven_registration(runtime):
    """Status events (pattern pub/sub) register/deregister monitors — the
    fixed version of the reference's broken Subscribe."""
    runtime.health.start()
    try:
        a = _started(runtime, name="ev")
        assert a.id in runtime.health.monitored_ids()
        runtime.agents.stop(a.id)
        assert a.id not in runtime.health.monitored_ids()
    finally:
        runtime.health.stop()


# ---------- metrics ----------

def test_metrics_sample_and_history(runtime):
    a = _started(runtime)
    for i in range(3):
        runtime.agent_request(a.id, "POST", "/chat", body={"message": f"m{i}"})
    now = time.time()
    s1 = runtime.metrics.sample_agent(
        a.id, runtime.engine.stats()["agents"][a.id], now=now)
    assert s1["requests_total"] == 3
    assert s1["e2e_p50_s"] >= 0
    cur = runtime.metrics.get_metrics(a.id)
    assert cur["requests_total"] == 3
    runtime.agent_request(a.id, "POST", "/chat", body={"message": "m3"})
    s2 = runtime.metrics.sample_agent(
        a.id, runtime.engine.stats()["agents"][a.id], now=now + 10)
    assert s2["req_per_s"] > 0
    hist = runtime.metrics.get_metrics_history(a.id, duration_s=3600, now=now + 11)
    assert len(hist) == 2


def test_metrics_history_trim(runtime):
    a = _started(runtime)
    old = time.time() - 25 * 3600
    runtime.metrics.sample_agent(a.id, {"tokens": 1, "requests": 1}, now=old)
    runtime.metrics.sample_agent(a.id, {"tokens": 2, "requests": 2}, now=time.time())
    hist = runtime.metrics.get_metrics_history(a.id, duration_s=48 * 3600)
    assert len(hist) == 1  # 24h retention trim


# ---------- engine watchdog ----------

def test_watchdog_marks_stuck_engine_unhealthy(tmp_path):
    """A step stuck past watchdog_timeout_s fails the health probe, and
    the health monitor's retry counter picks it up (stuck-kernel path)."""
    import time as _time

    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    cfg.data["engine"]["watchdog_timeout_s"] = 0.05
    store = Store(str(tmp_path / "state"), sync="never")
    eng = LLMEngine(store, cfg, device="cpu", state_root=str(tmp_path))
    man = Manager(store, eng, cfg)
    a = man.deploy(name="w", model="tiny-llama")
    man.start(a.id)
    assert eng.health_probe(a.id) is True
    inst = eng._instances["tiny-llama"]
    inst._step_started = _time.time() - 1.0  # simulate a wedged step
    assert inst.stuck() is True
    assert eng.health_probe(a.id) is False
    assert eng.stats()["models"]["tiny-llama"]["stuck"] is True
    inst._step_started = None
    assert eng.health_probe(a.id) is True
    store.close()


# ---------------- device metrics (xGMI / HBM) -----------------------------

def test_device_metrics_sampler_stub(tmp_path):
    """xGMI link counters flow into metrics:current:device with per-
    interval rates (SURVEY.md §5 metrics row: xGMI link throughput)."""
    import json
    import stat

    from agentainer_amd.metrics import DeviceMetricsSampler

    state = tmp_path / "n"
    state.write_text("0")
    script = tmp_path / "smi"
    script.write_text(f"""#!/usr/bin/env python3
import json
i = int(open({str(state)!r}).read() or 0)
open({str(state)!r}, "w").write(str(i + 1))
print(json.dumps({{"gpu_data": [{{"gpu": 0, "xgmi": {{
    "link_0_read_kb": 1000 * (i + 1), "link_0_write_kb": 500 * (i + 1)}}}}]}}))
""")
    script.chmod(script.stat().st_mode | stat.S_IEXEC)
    s = DeviceMetricsSampler(cmd=[str(script)])
    first = s.sample(now=100.0)
    assert "xgmi_counters" in first and "xgmi_per_s" not in first
    second = s.sample(now=110.0)
    rates = second["xgmi_per_s"]
    read_key = [k for k in rates if "read" in k][0]
    assert abs(rates[read_key] - 100.0) < 1e-6  # 1000 KB over 10 s


def test_device_metrics_sampler_disables_after_failures(tmp_path):
    from agentainer_amd.metrics import DeviceMetricsSampler

    s = DeviceMetricsSampler(cmd=[str(tmp_path / "missing")])
    for _ in range(3):
        assert s.sample() == {}
    assert not s.enabled
    assert s.sample() == {}  # permanently off, no subprocess attempts


def test_collector_publishes_device_sample(tmp_path):
    """Engine HBM fields land in metrics:current:device even without a
    working amd-smi (CPU path: engine stats only, no device sampler)."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.metrics import MetricsCollector
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    store = Store(str(tmp_path / "state"), sync="never")
    eng = EchoEngine(store)
    # fake a GPU-shaped stats() payload
    orig = eng.stats

    def stats():
        out = orig()
        out["hbm_total_bytes"] = 288 << 30
        out["hbm_free_bytes"] = 100 << 30
        return out

    eng.stats = stats
    man = Manager(store, eng, cfg)
    col = MetricsCollector(store, man)
    col.sample_all(now=100.0)
    dev = store.get("metrics:current:device")
    assert dev["hbm_total_bytes"] == 288 << 30
    store.close()
