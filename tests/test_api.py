"""REST API surface + auth + proxy semantics, via FastAPI TestClient."""

import pytest
from fastapi.testclient import TestClient

from agentainer_amd.api import create_app
from agentainer_amd.config import DEFAULT_TOKEN


@pytest.fixture()
def client(runtime):
    app = create_app(runtime)
    with TestClient(app) as c:
        c.runtime = runtime
        yield c


AUTH = {"Authorization": f"Bearer {DEFAULT_TOKEN}"}


def _deploy(client, name="api1", **kw):
    r = client.post("/agents", json={"name": name, "model": "echo", **kw},
                    headers=AUTH)
    assert r.status_code == 200, r.text
    return r.json()["data"]


def test_health_public(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["success"]


def test_auth_required(client):
    assert client.get("/agents").status_code == 401
    assert client.post("/agents", json={}).status_code == 401
    # query-param token accepted (server.go:449-478)
    assert client.get(f"/agents?token={DEFAULT_TOKEN}").status_code == 200


def test_deploy_validation(client):
    r = client.post("/agents", json={"name": "", "model": "echo"}, headers=AUTH)
    assert r.status_code == 422
    r = client.post("/agents", json={"name": "x" * 65, "model": "echo"}, headers=AUTH)
    assert r.status_code == 422
    r = client.post("/agents", json={"name": "ok", "model": "m" * 257}, headers=AUTH)
    assert r.status_code == 422
    r = client.post("/agents", json={"name": "ok", "model": "echo",
                                     "env": {f"K{i}": "v" for i in range(51)}},
                    headers=AUTH)
    assert r.status_code == 422
    r = client.post("/agents", json={"name": "ok", "model": "no-such-model"},
                    headers=AUTH)
    assert r.status_code == 400


def test_lifecycle_roundtrip(client):
    a = _deploy(client)
    aid = a["id"]
    assert a["status"] == "created"
    assert client.post(f"/agents/{aid}/start", headers=AUTH).status_code == 200
    got = client.get(f"/agents/{aid}", headers=AUTH).json()["data"]
    assert got["status"] == "running"
    assert client.post(f"/agents/{aid}/pause", headers=AUTH).status_code == 200
    assert client.post(f"/agents/{aid}/resume", headers=AUTH).status_code == 200
    assert client.post(f"/agents/{aid}/stop", headers=AUTH).status_code == 200
    assert client.post(f"/agents/{aid}/restart", headers=AUTH).status_code == 200
    assert client.delete(f"/agents/{aid}", headers=AUTH).status_code == 200
    assert client.get(f"/agents/{aid}", headers=AUTH).status_code == 404


def test_lifecycle_conflict(client):
    a = _deploy(client)
    r = client.post(f"/agents/{a['id']}/pause", headers=AUTH)  # not running
    assert r.status_code == 409


def test_proxy_chat_unauthenticated(client):
    a = _deploy(client)
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    r = client.post(f"/agent/{a['id']}/chat", json={"message": "hi"})
    assert r.status_code == 200
    assert "echo" in r.json()["response"]
    # history endpoint
    h = client.get(f"/agent/{a['id']}/history")
    assert len(h.json()["history"]) == 1
    # clear
    assert client.post(f"/agent/{a['id']}/clear", json={}).status_code == 200
    assert client.get(f"/agent/{a['id']}/history").json()["history"] == []


def test_proxy_queues_when_stopped(client):
    a = _deploy(client)
    r = client.post(f"/agent/{a['id']}/chat", json={"message": "queued"})
    assert r.status_code == 202
    rid = r.json()["data"]["request_id"]
    # request visible via management API
    reqs = client.get(f"/agents/{a['id']}/requests", headers=AUTH).json()["data"]
    assert [x["id"] for x in reqs["pending"]] == [rid]
    one = client.get(f"/agents/{a['id']}/requests/{rid}", headers=AUTH).json()["data"]
    assert one["status"] == "pending"
    # start + manual replay endpoint
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    rep = client.post(f"/agents/{a['id']}/requests/{rid}/replay", headers=AUTH)
    assert rep.status_code == 200
    one = client.get(f"/agents/{a['id']}/requests/{rid}", headers=AUTH).json()["data"]
    assert one["status"] == "completed"


def test_invoke_real_dispatch(client):
    """invoke is a real authenticated dispatch (ref stub fixed, §7.4)."""
    a = _deploy(client)
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    r = client.post(f"/agents/{a['id']}/invoke",
                    json={"body": {"message": "ping"}}, headers=AUTH)
    assert r.status_code == 200
    assert "ping" in r.json()["data"]["response"]


def test_per_agent_token(client):
    a = _deploy(client, name="tok", token="secret-tok")
    hdr = {"Authorization": "Bearer secret-tok"}
    assert client.get(f"/agents/{a['id']}", headers=hdr).status_code == 200
    # wrong token still rejected
    bad = {"Authorization": "Bearer nope"}
    assert client.get(f"/agents/{a['id']}", headers=bad).status_code == 401


def test_health_and_metrics_endpoints(client):
    a = _deploy(client)
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    client.runtime.health.start_monitoring(a["id"])
    h = client.get(f"/agents/{a['id']}/health", headers=AUTH)
    assert h.status_code == 200 and h.json()["data"]["healthy"] is True
    allh = client.get("/health/agents", headers=AUTH)
    assert a["id"] in allh.json()["data"]
    client.post(f"/agent/{a['id']}/chat", json={"message": "m"})
    client.runtime.metrics.sample_all()
    m = client.get(f"/agents/{a['id']}/metrics", headers=AUTH).json()["data"]
    assert m["requests_total"] == 1
    hist = client.get(f"/agents/{a['id']}/metrics/history", headers=AUTH).json()["data"]
    assert len(hist) == 1


def test_audit_and_logs_endpoints(client):
    a = _deploy(client)
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    audit = client.get("/audit", headers=AUTH).json()["data"]
    actions = [e["action"] for e in audit]
    assert "deploy" in actions and "start" in actions
    r = client.get("/audit", params={"action": "deploy"}, headers=AUTH)
    assert all(e["action"] == "deploy" for e in r.json()["data"])


def test_backup_endpoints(client):
    _deploy(client, name="b1")
    r = client.post("/backups", json={"name": "apisnap"}, headers=AUTH)
    assert r.status_code == 200
    bid = r.json()["data"]["id"]
    assert any(b["id"] == bid for b in client.get("/backups", headers=AUTH).json()["data"])
    rr = client.post(f"/backups/{bid}/restore", headers=AUTH)
    assert rr.status_code == 200 and rr.json()["data"][0]["name"] == "b1-restored"
    assert client.delete(f"/backups/{bid}", headers=AUTH).status_code == 200


def test_chat_sse_streaming(client):
    """SSE streaming on the chat proxy: token events then a final done
    event matching the blocking /chat contract; WAL-acked on completion."""
    import json as _json

    a = _deploy(client, name="sse1")
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    with client.stream("POST", f"/agent/{a['id']}/chat",
                       json={"message": "stream me", "stream": True}) as r:
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/event-stream")
        events = []
        for line in r.iter_lines():
            if line.startswith("data: "):
                events.append(_json.loads(line[len("data: "):]))
    assert len(events) >= 2
    final = events[-1]
    assert final.get("done") is True
    assert "stream me" in final["response"]
    # chunk texts concatenate to the full response
    text = "".join(e["text"] for e in events[:-1])
    assert final["response"] in text or text.strip() == final["response"]
    # the request was WAL-acked as completed
    rq = client.get(f"/agents/{a['id']}/requests", headers=AUTH).json()["data"]
    assert len(rq["completed"]) == 1
    assert rq["completed"][0]["response"]["response"] == final["response"]
    assert rq["pending"] == []


def test_chat_sse_not_running_queues(client):
    a = _deploy(client, name="sse2")  # never started
    r = client.post(f"/agent/{a['id']}/chat?stream=1",
                    json={"message": "queued"})
    assert r.status_code == 202
    assert r.json()["data"]["status"] == "pending"


def test_chat_sse_client_disconnect_leaves_pending(client):
    """A client abandoning an SSE stream mid-generation must not lose the
    request: the generator's cleanup leaves the WAL entry pending (no
    ack), and the replay worker completes it later (at-least-once)."""
    rt = client.runtime
    a = _deploy(client, name="sse-drop")
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    status, gen = rt.agent_request_stream(a["id"], {"message": "dropped"})
    assert status == 200
    first = next(gen)  # start consuming, then abandon
    assert "text" in first or "token" in first
    gen.close()  # GeneratorExit inside the stream
    # in-flight marker released; entry not acked
    assert not rt._inflight
    pend = rt.requests.pending(a["id"])
    assert len(pend) == 1
    # replay completes it
    assert rt.replay.tick() == 1
    assert rt.requests.pending(a["id"]) == []
    done = rt.requests.by_queue(a["id"], "completed")
    assert len(done) == 1 and "dropped" in done[0].response["response"]



def test_prometheus_metrics_endpoint(client):
    a = _deploy(client, name="prom")
    client.post(f"/agents/{a['id']}/start", headers=AUTH)
    client.post(f"/agent/{a['id']}/chat", json={"message": "hi"})
    r = client.get(f"/metrics/prometheus?token={DEFAULT_TOKEN}")
    assert r.status_code == 200
    body = r.text
    assert "agentainer_agents_attached" in body
    assert body.count("# TYPE agentainer_agents_attached") == 1
    # unauthenticated scrape rejected
    assert client.get("/metrics/prometheus").status_code == 401


def test_threadpool_sized_for_concurrent_generations(tmp_path):
    """anyio's default 40-token limiter silently capped the server at ~40
    concurrent generations (each blocking chat parks a worker thread);
    the startup hook must raise it to server.threadpool_size."""
    import anyio

    from agentainer_amd.api.server import create_app
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.service import Runtime
    from agentainer_amd.store import Store
    from starlette.testclient import TestClient

    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "root")
    cfg.data["store"]["path"] = root
    s = Store(root + "/state", sync="never")
    rt = Runtime(cfg, engine=EchoEngine(s), store=s, state_root=root)
    app = create_app(rt)

    @app.get("/__limiter")  # test-only: the limiter is per-event-loop
    async def _lim():
        return {"tokens": anyio.to_thread
                .current_default_thread_limiter().total_tokens}

    with TestClient(app) as client:  # runs startup events
        tokens = client.get("/__limiter").json()["tokens"]
        assert tokens >= 1024, tokens
    rt.shutdown()
