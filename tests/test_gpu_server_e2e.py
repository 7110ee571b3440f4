"""Full-stack E2E on MI355X: real server process (uvicorn + threaded
engine + llama3-8b), concurrent HTTP chat load, lifecycle + KV restore
through the REST API."""

import concurrent.futures
import sys
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

sys.path.insert(0, ".")
from test_crash_integration import Server, _free_port  # noqa: E402


@pytest.mark.timeout(900)
def test_server_llama8b_concurrent_chat(tmp_path):
    root = str(tmp_path / "root")
    srv = Server(root, _free_port(), device="cuda", kv_pool_gb=16.0)
    try:
        srv.start(timeout=300)  # model load + pool allocation
        agents = []
        for i in range(8):
            st, resp = srv.call("POST", "/agents", {
                "name": f"prod-{i}", "model": "llama3-8b",
                "auto_restart": True, "sampling": {"max_tokens": 32}})
            assert st == 200, resp
            agents.append(resp["data"]["id"])
        for aid in agents:
            assert srv.call("POST", f"/agents/{aid}/start", timeout=300)[0] == 200

        def one_chat(args):
            aid, i = args
            st, out = srv.call("POST", f"/agent/{aid}/chat",
                               body={"message": f"msg-{i}"}, auth=False,
                               timeout=300)
            assert st == 200, out
            assert out["tokens"] == 32
            return out["e2e_s"]

        t0 = time.time()
        jobs = [(aid, i) for i in range(5) for aid in agents]  # 40 chats
        with concurrent.futures.ThreadPoolExecutor(max_workers=16) as ex:
            lat = list(ex.map(one_chat, jobs))
        wall = time.time() - t0
        print(f"# 40 concurrent chats in {wall:.2f}s "
              f"({len(jobs)/wall:.1f} req/s), mean e2e {sum(lat)/len(lat):.3f}s",
              file=sys.stderr)

        # multi-turn KV continuity + stop/resume via REST
        aid = agents[0]
        h1 = srv.call("GET", f"/agent/{aid}/history", auth=False)[1]["history"]
        assert len(h1) == 5, (h1, srv.log_tail())  # no replay duplicates
        assert srv.call("POST", f"/agents/{aid}/stop")[0] == 200
        assert srv.call("POST", f"/agents/{aid}/resume")[0] == 200
        st, out = srv.call("POST", f"/agent/{aid}/chat",
                           body={"message": "after resume"}, auth=False,
                           timeout=300)
        assert st == 200 and out["tokens"] == 32, (st, out, srv.log_tail())

        # engine metrics live
        st, m = srv.call("GET", "/metrics/engine")
        assert st == 200
        assert m["data"]["models"]["llama3-8b"]["decode_tokens"] > 0
        assert m["data"].get("hbm_total_bytes", 0) > 0

        # SSE streaming through the full GPU stack (graph-captured decode
        # feeding per-token events): 32 token frames + final done frame
        import json as _json

        import httpx
        events = []
        with httpx.stream("POST", f"{srv.base}/agent/{aid}/chat",
                          json={"message": "stream on gpu", "stream": True},
                          timeout=300) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                if line.startswith("data: "):
                    events.append(_json.loads(line[len("data: "):]))
        assert events[-1].get("done") is True, events[-1:]
        assert events[-1]["tokens"] == 32
        assert sum(1 for e in events[:-1] if e.get("token") is not None) == 32
    finally:
        srv.terminate()


@pytest.mark.timeout(600)
def test_kill_and_replay_gpu(tmp_path):
    """BASELINE config 4's crash contract on DEVICE: SIGKILL the server
    while requests are queued against a GPU-resident model; restart on
    the same state root; auto-restart + WAL replay regenerate every
    pending request through the HIP decode path."""
    root = str(tmp_path / "root")
    port = _free_port()
    srv = Server(root, port, device="cuda", kv_pool_gb=2.0)
    try:
        srv.start(timeout=300)
        st, resp = srv.call("POST", "/agents", {
            "name": "gpu-crash", "model": "tiny-llama", "auto_restart": True,
            "sampling": {"max_tokens": 8}})
        assert st == 200, resp
        aid = resp["data"]["id"]
        assert srv.call("POST", f"/agents/{aid}/start", timeout=300)[0] == 200
        st, warm = srv.call("POST", f"/agent/{aid}/chat",
                            body={"message": "warm"}, auth=False, timeout=300)
        assert st == 200 and warm["tokens"] == 8
        assert srv.call("POST", f"/agents/{aid}/stop")[0] == 200
        rids = []
        for i in range(3):
            st, q = srv.call("POST", f"/agent/{aid}/chat",
                             body={"message": f"q-{i}"}, auth=False)
            assert st == 202
            rids.append(q["data"]["request_id"])
        srv.kill9()
    finally:
        srv.terminate()

    srv2 = Server(root, port, device="cuda", kv_pool_gb=2.0)
    try:
        srv2.start(timeout=300)
        deadline = time.time() + 240
        done = {}
        while time.time() < deadline and len(done) < len(rids):
            for rid in rids:
                if rid in done:
                    continue
                st, r = srv2.call("GET", f"/agents/{aid}/requests/{rid}")
                if st == 200 and r["data"]["status"] == "completed":
                    done[rid] = r["data"]["response"]
            time.sleep(0.5)
        assert len(done) == len(rids), f"{len(done)}/{len(rids)} replayed"
        for rid in rids:
            assert done[rid]["tokens"] == 8
    finally:
        srv2.terminate()
