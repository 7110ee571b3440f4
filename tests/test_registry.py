"""Registry FSM + lifecycle semantics (SURVEY.md §7.1 semantic mappings)."""

import pytest

from agentainer_amd.engine.base import ModelNotFound
from agentainer_amd.registry import (
    CREATED, FAILED, PAUSED, RUNNING, STOPPED, AgentNotFound,
)


def _deploy(rt, name="a1", **kw):
    return rt.agents.deploy(name=name, model="echo", **kw)


def test_deploy_touches_no_engine(runtime):
    a = _deploy(runtime)
    assert a.status == CREATED
    assert not runtime.engine.is_attached(a.id)  # no container at deploy (§3.2)
    got = runtime.agents.get(a.id)
    assert got.name == "a1" and got.model == "echo"


def test_deploy_unknown_model_rejected(runtime):
    with pytest.raises(ModelNotFound):
        runtime.agents.deploy(name="x", model="not-a-model")


def test_start_stop_restart(runtime):
    a = _deploy(runtime)
    runtime.agents.start(a.id)
    assert runtime.agents.get(a.id).status == RUNNING
    assert runtime.engine.is_attached(a.id)
    runtime.agents.stop(a.id)
    g = runtime.agents.get(a.id)
    assert g.status == STOPPED
    assert g.kv_offloaded is True  # stop = drain + offload KV
    assert not runtime.engine.is_attached(a.id)
    runtime.agents.restart(a.id)
    assert runtime.agents.get(a.id).status == RUNNING


def test_pause_resume_keeps_attachment(runtime):
    a = _deploy(runtime)
    runtime.agents.start(a.id)
    runtime.agents.pause(a.id)
    assert runtime.agents.get(a.id).status == PAUSED
    assert runtime.engine.is_attached(a.id)  # KV resident
    assert runtime.engine.engine_status(a.id) == "paused"
    runtime.agents.resume(a.id)
    assert runtime.agents.get(a.id).status == RUNNING


def test_pause_requires_running(runtime):
    a = _deploy(runtime)
    with pytest.raises(Exception):
        runtime.agents.pause(a.id)


def test_resume_is_universal_rehydrator(runtime):
    # resume works from created/stopped/failed (agent.go:255-311)
    a = _deploy(runtime)
    runtime.agents.resume(a.id)
    assert runtime.agents.get(a.id).status == RUNNING
    runtime.agents.stop(a.id)
    runtime.agents.resume(a.id)
    assert runtime.agents.get(a.id).status == RUNNING


def test_remove_purges(runtime):
    a = _deploy(runtime)
    runtime.agents.start(a.id)
    runtime.requests.store_request(a.id, "POST", "/chat", body={"message": "hi"})
    runtime.agents.remove(a.id, request_manager=runtime.requests)
    with pytest.raises(AgentNotFound):
        runtime.agents.get(a.id)
    assert not runtime.engine.is_attached(a.id)
    assert runtime.requests.pending(a.id) == []
    assert runtime.store.keys(f"agent:{a.id}:requests:*") == []


def test_list_sorted_by_creation(runtime):
    ids = [_deploy(runtime, name=f"n{i}").id for i in range(3)]
    assert [a.id for a in runtime.agents.list()] == ids


def test_status_events_published(runtime):
    events = []
    runtime.store.subscribe("agent:status:*", lambda ch, m: events.append(m))
    a = _deploy(runtime)
    runtime.agents.start(a.id)
    runtime.agents.stop(a.id)
    assert events == [CREATED, RUNNING, STOPPED]


def test_registry_survives_restart(tmp_path):
    """Agent records persist across a store reopen (crash durability)."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    path = str(tmp_path / "state")
    s = Store(path, sync="always")
    m = Manager(s, EchoEngine(s))
    a = m.deploy(name="persist", model="echo", auto_restart=True)
    m.start(a.id)
    # crash: no close
    s2 = Store(path)
    m2 = Manager(s2, EchoEngine(s2))
    g = m2.get(a.id)
    assert g.name == "persist"
    assert g.status == RUNNING  # desired state preserved; reconciler fixes runtime state
    assert g.auto_restart is True
    s2.close()


def test_double_start_and_double_stop_are_clean(runtime):
    """Idempotency-adjacent: re-starting a running agent and re-stopping a
    stopped one either no-op or raise cleanly — never corrupt state."""
    a = runtime.agents.deploy(name="dbl", model="echo")
    runtime.agents.start(a.id)
    try:
        runtime.agents.start(a.id)  # second start
    except Exception:
        pass
    assert runtime.agents.get(a.id).status == "running"
    st, p = runtime.agent_request(a.id, "POST", "/chat", body={"message": "ok"})
    assert st == 200
    runtime.agents.stop(a.id)
    try:
        runtime.agents.stop(a.id)
    except Exception:
        pass
    assert runtime.agents.get(a.id).status == "stopped"
    runtime.agents.resume(a.id)
    st, _ = runtime.agent_request(a.id, "POST", "/chat", body={"message": "ok"})
    assert st == 200
