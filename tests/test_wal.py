"""Request WAL: state transitions, retry/dead-letter, TTL, replay, crash."""

from agentainer_amd.registry import RUNNING
from agentainer_amd.wal import COMPLETED, FAILED, PENDING, RequestManager, ReplayWorker


def _deploy_started(rt, name="w1"):
    a = rt.agents.deploy(name=name, model="echo")
    rt.agents.start(a.id)
    return a


def test_store_and_complete(runtime):
    a = _deploy_started(runtime)
    req = runtime.requests.store_request(a.id, "POST", "/chat", body={"message": "hi"})
    assert req.status == PENDING
    assert [r.id for r in runtime.requests.pending(a.id)] == [req.id]
    assert runtime.requests.agents_with_pending() == [a.id]
    done = runtime.requests.store_response(a.id, req.id, {"response": "ok"})
    assert done.status == COMPLETED
    assert done.response == {"response": "ok"}
    assert runtime.requests.pending(a.id) == []
    assert [r.id for r in runtime.requests.by_queue(a.id, "completed")] == [req.id]
    assert runtime.requests.agents_with_pending() == []


def test_ttl_set(runtime):
    a = _deploy_started(runtime)
    req = runtime.requests.store_request(a.id, "POST", "/chat", body={})
    ttl = runtime.store.ttl(f"agent:{a.id}:requests:{req.id}")
    assert ttl is not None and 23 * 3600 < ttl <= 24 * 3600  # requests.go:106


def test_retry_then_dead_letter(runtime):
    a = _deploy_started(runtime)
    req = runtime.requests.store_request(a.id, "POST", "/chat", body={})
    for i in range(1, 3):
        r = runtime.requests.mark_failed(a.id, req.id, f"err{i}")
        assert r.status == PENDING and r.retry_count == i
    r = runtime.requests.mark_failed(a.id, req.id, "err3")
    assert r.status == FAILED and r.retry_count == 3  # max_retries=3 dead-letter
    assert runtime.requests.pending(a.id) == []
    assert [x.id for x in runtime.requests.by_queue(a.id, "failed")] == [req.id]


def test_queue_202_when_not_running(runtime):
    a = runtime.agents.deploy(name="q", model="echo")  # created, not started
    status, payload = runtime.agent_request(a.id, "POST", "/chat",
                                            body={"message": "hello"})
    assert status == 202
    rid = payload["data"]["request_id"]
    assert rid is not None
    assert [r.id for r in runtime.requests.pending(a.id)] == [rid]


def test_replay_after_resume(runtime):
    """The signature crash->queue->resume->replay flow."""
    a = _deploy_started(runtime)
    # crash the engine mid-life
    runtime.engine.crash()
    runtime.reconciler.sync_agent(a.id)
    assert runtime.agents.get(a.id).status != RUNNING
    # requests arriving while down are queued with 202
    st, payload = runtime.agent_request(a.id, "POST", "/chat", body={"message": "m1"})
    assert st == 202
    st, _ = runtime.agent_request(a.id, "POST", "/chat", body={"message": "m2"})
    assert st == 202
    assert len(runtime.requests.pending(a.id)) == 2
    # recover + resume
    runtime.engine.recover()
    runtime.agents.resume(a.id)
    n = runtime.replay.tick()
    assert n == 2
    assert runtime.requests.pending(a.id) == []
    completed = runtime.requests.by_queue(a.id, "completed")
    assert len(completed) == 2
    assert all(c.response and "echo" in c.response["response"] for c in completed)
    # single response store per request (no duplicate completed entries)
    assert len(runtime.store.lrange(f"agent:{a.id}:requests:completed")) == 2


def test_replay_skips_non_running(runtime):
    a = runtime.agents.deploy(name="s", model="echo")
    runtime.agent_request(a.id, "POST", "/chat", body={"message": "x"})
    assert runtime.replay.tick() == 0  # agent not running => untouched
    assert len(runtime.requests.pending(a.id)) == 1


def test_replay_marks_failed_on_app_error(runtime):
    a = _deploy_started(runtime)
    runtime.engine.fail_on = "boom"
    runtime.agents.pause(a.id)
    st, _ = runtime.agent_request(a.id, "POST", "/chat", body={"message": "boom"})
    assert st == 202  # paused => queued
    runtime.agents.resume(a.id)
    for _ in range(3):
        runtime.replay.tick()
    reqs = runtime.requests.by_queue(a.id, "failed")
    assert len(reqs) == 1 and reqs[0].retry_count == 3


def test_wal_survives_crash(tmp_path):
    """Pending WAL entries persist across process death (fsync discipline)."""
    from agentainer_amd.store import Store

    path = str(tmp_path / "state")
    s = Store(path, sync="always")
    rm = RequestManager(s)
    req = rm.store_request("agent-x", "POST", "/chat", body={"message": "survive"})
    # crash: no close
    s2 = Store(path)
    rm2 = RequestManager(s2)
    pend = rm2.pending("agent-x")
    assert [r.id for r in pend] == [req.id]
    assert pend[0].body == {"message": "survive"}
    assert rm2.agents_with_pending() == ["agent-x"]
    s2.close()


def test_hot_path_completes_request(runtime):
    a = _deploy_started(runtime)
    st, payload = runtime.agent_request(a.id, "POST", "/chat", body={"message": "hi"})
    assert st == 200
    assert "echo" in payload["response"]
    done = runtime.requests.by_queue(a.id, "completed")
    assert len(done) == 1 and done[0].response == payload


def test_replay_skips_inflight(runtime):
    """A request mid-dispatch (slow generation) must NOT be re-dispatched
    by the replay worker; after a crash the in-flight set is empty so it
    replays (at-least-once preserved)."""
    a = _deploy_started(runtime, name="slow")
    req = runtime.requests.store_request(a.id, "POST", "/chat",
                                         body={"message": "slow one"})
    runtime._inflight.add(req.id)   # simulate live dispatch in progress
    assert runtime.replay.tick() == 0
    assert len(runtime.requests.pending(a.id)) == 1
    runtime._inflight.discard(req.id)  # dispatch died (crash analog)
    assert runtime.replay.tick() == 1
    assert runtime.requests.pending(a.id) == []


def test_client_supplied_id_cannot_clobber(runtime):
    """A duplicate X-Agentainer-Request-ID must not overwrite an existing
    WAL record — the second request gets a fresh id."""
    a = _deploy_started(runtime, name="idem")
    r1 = runtime.requests.store_request(a.id, "POST", "/chat",
                                        body={"message": "one"},
                                        req_id="fixed-id")
    assert r1.id == "fixed-id"
    runtime.requests.store_response(a.id, r1.id, {"response": "done"})
    r2 = runtime.requests.store_request(a.id, "POST", "/chat",
                                        body={"message": "two"},
                                        req_id="fixed-id")
    assert r2.id != "fixed-id"
    kept = runtime.requests.get(a.id, "fixed-id")
    assert kept.status == COMPLETED and kept.response == {"response": "done"}
