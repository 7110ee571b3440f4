"""Kill-and-replay integration tests against a REAL server process.

Re-creates the reference's absent-but-named integration suites
(Makefile:41-57: test-persistence, test-crash, test-network) and the
manual acceptance procedure of RESILIENT_AGENTS.md:399-440:
deploy -> fire requests -> SIGKILL the server mid-queue -> restart ->
verify every pending request replays to completion.

Uses the echo engine (CPU) and tiny-llama; the same WAL/replay path runs
under the MI355X engine.
"""

import json
import os
import signal
import socket
import subprocess
import sys
import time

import httpx
import pytest

TOKEN = "agentainer-default-token"
AUTH = {"Authorization": f"Bearer {TOKEN}"}


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class Server:
    def __init__(self, root: str, port: int, device: str = "echo",
                 kv_pool_gb: float = 0.02):
        self.root = root
        self.port = port
        self.device = device
        self.kv_pool_gb = kv_pool_gb  # 0 = engine auto-size (GPU runs)
        self.proc = None
        self.base = f"http://127.0.0.1:{port}"

    def start(self, timeout=60.0):
        env = dict(os.environ)
        env.update({
            "AGENTAINER_STORE_PATH": self.root,
            "AGENTAINER_SERVER_PORT": str(self.port),
            "AGENTAINER_ENGINE_KV_POOL_GB": str(self.kv_pool_gb),
            "AGENTAINER_FEATURES_REPLAY_INTERVAL_S": "0.2",
            "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        })
        self.log_path = os.path.join(self.root, f"server-{self.port}.log")
        os.makedirs(self.root, exist_ok=True)
        self._log_f = open(self.log_path, "ab")
        self.proc = subprocess.Popen(
            [sys.executable, "-m", "agentainer_amd.cli", "server",
             "--engine-device", self.device],
            env=env, stdout=self._log_f, stderr=subprocess.STDOUT,
            start_new_session=True)
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                r = httpx.get(self.base + "/health", timeout=2.0)
                if r.status_code == 200:
                    return
            except httpx.HTTPError:
                pass
            if self.proc.poll() is not None:
                raise RuntimeError(f"server died rc={self.proc.returncode}")
            time.sleep(0.2)
        raise TimeoutError("server did not come up")

    def log_tail(self, n=4000) -> str:
        try:
            with open(self.log_path, "rb") as f:
                return f.read()[-n:].decode("utf-8", "replace")
        except OSError:
            return "(no log)"

    def kill9(self):
        """SIGKILL — the docker-kill analog. No flush, no goodbye."""
        os.killpg(os.getpgid(self.proc.pid), signal.SIGKILL)
        self.proc.wait(timeout=10)

    def terminate(self):
        if self.proc and self.proc.poll() is None:
            os.killpg(os.getpgid(self.proc.pid), signal.SIGTERM)
            try:
                self.proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                self.kill9()

    def call(self, method, path, body=None, params=None, auth=True, timeout=30.0):
        headers = AUTH if auth else {}
        r = httpx.request(method, self.base + path, json=body, params=params,
                          headers=headers, timeout=timeout)
        return r.status_code, (r.json() if r.content else {})


@pytest.mark.timeout(300)
def test_kill_and_replay_echo(tmp_path):
    root = str(tmp_path / "root")
    port = _free_port()
    srv = Server(root, port)
    try:
        srv.start()
        # deploy + start, verify live chat
        st, resp = srv.call("POST", "/agents", {
            "name": "crashy", "model": "echo", "auto_restart": True})
        assert st == 200, resp
        aid = resp["data"]["id"]
        assert srv.call("POST", f"/agents/{aid}/start")[0] == 200
        st, chat = srv.call("POST", f"/agent/{aid}/chat",
                            body={"message": "live"}, auth=False)
        assert st == 200 and "live" in chat["response"]
        # stop the agent; queue requests (202 + persisted pending)
        assert srv.call("POST", f"/agents/{aid}/stop")[0] == 200
        rids = []
        for i in range(5):
            st, q = srv.call("POST", f"/agent/{aid}/chat",
                             body={"message": f"queued-{i}"}, auth=False)
            assert st == 202
            rids.append(q["data"]["request_id"])
        # SIGKILL mid-queue
        srv.kill9()
    finally:
        srv.terminate()

    # restart on the same state root: recovery must auto-restart the agent
    # (auto_restart=True) and replay every pending request
    srv2 = Server(root, port)
    try:
        srv2.start()
        deadline = time.time() + 60
        done = {}
        while time.time() < deadline and len(done) < len(rids):
            for rid in rids:
                if rid in done:
                    continue
                st, r = srv2.call("GET", f"/agents/{aid}/requests/{rid}")
                if st == 200 and r["data"]["status"] == "completed":
                    done[rid] = r["data"]["response"]
            time.sleep(0.3)
        assert len(done) == len(rids), f"only {len(done)}/{len(rids)} replayed"
        for i, rid in enumerate(rids):
            assert f"queued-{i}" in done[rid]["response"]
        # conversation history survived the crash too
        st, hist = srv2.call("GET", f"/agent/{aid}/history", auth=False)
        assert st == 200 and len(hist["history"]) >= len(rids)
    finally:
        srv2.terminate()


@pytest.mark.timeout(300)
def test_kill_and_replay_llm(tmp_path):
    """Same contract against the real LLM engine (tiny-llama): in-flight
    KV dies with the process; replay regenerates deterministically."""
    root = str(tmp_path / "root")
    port = _free_port()
    srv = Server(root, port, device="cpu")
    try:
        srv.start(timeout=120)
        st, resp = srv.call("POST", "/agents", {
            "name": "llm-crash", "model": "tiny-llama", "auto_restart": True,
            "sampling": {"max_tokens": 6}})
        assert st == 200, resp
        aid = resp["data"]["id"]
        assert srv.call("POST", f"/agents/{aid}/start")[0] == 200
        st, chat = srv.call("POST", f"/agent/{aid}/chat",
                            body={"message": "warm"}, auth=False, timeout=120)
        assert st == 200 and chat["tokens"] == 6
        assert srv.call("POST", f"/agents/{aid}/stop")[0] == 200
        st, q = srv.call("POST", f"/agent/{aid}/chat",
                         body={"message": "survive-me"}, auth=False)
        assert st == 202
        rid = q["data"]["request_id"]
        srv.kill9()
    finally:
        srv.terminate()

    srv2 = Server(root, port, device="cpu")
    try:
        srv2.start(timeout=120)
        deadline = time.time() + 90
        status = None
        while time.time() < deadline:
            st, r = srv2.call("GET", f"/agents/{aid}/requests/{rid}")
            if st == 200:
                status = r["data"]["status"]
                if status == "completed":
                    assert r["data"]["response"]["tokens"] == 6
                    break
            time.sleep(0.5)
        assert status == "completed", f"request stuck in {status}"
    finally:
        srv2.terminate()


@pytest.mark.timeout(120)
def test_network_isolation(tmp_path):
    """Management API requires auth; the agent proxy does not (the
    reference's network-isolation contract, server.go:68-107)."""
    root = str(tmp_path / "root")
    port = _free_port()
    srv = Server(root, port)
    try:
        srv.start()
        assert srv.call("GET", "/agents", auth=False)[0] == 401
        assert srv.call("POST", "/agents", {"name": "x", "model": "echo"},
                        auth=False)[0] == 401
        st, resp = srv.call("POST", "/agents", {"name": "iso", "model": "echo"})
        aid = resp["data"]["id"]
        srv.call("POST", f"/agents/{aid}/start")
        # proxy path works unauthenticated
        st, _ = srv.call("POST", f"/agent/{aid}/chat",
                         body={"message": "open"}, auth=False)
        assert st == 200
        # lifecycle of someone else's agent is rejected without the token
        assert srv.call("POST", f"/agents/{aid}/stop", auth=False)[0] == 401
    finally:
        srv.terminate()


@pytest.mark.timeout(300)
def test_kill_mid_stream(tmp_path):
    """BASELINE config 4's crash contract, literally: SIGKILL the server
    while a generation is IN FLIGHT; after restart the unacked request is
    still pending and replays to completion (greedy => the regenerated
    response is the one the caller would have gotten)."""
    import threading

    root = str(tmp_path / "root")
    port = _free_port()
    srv = Server(root, port, device="cpu")
    killed = {}
    try:
        srv.start(timeout=120)
        st, resp = srv.call("POST", "/agents", {
            "name": "midstream", "model": "tiny-llama", "auto_restart": True,
            "sampling": {"max_tokens": 300}})  # ~seconds of CPU decode
        aid = resp["data"]["id"]
        assert srv.call("POST", f"/agents/{aid}/start")[0] == 200

        def fire():
            try:
                srv.call("POST", f"/agent/{aid}/chat",
                         body={"message": "cut me off"}, auth=False,
                         timeout=120)
            except Exception as exc:  # connection dies with the server
                killed["client_error"] = type(exc).__name__

        th = threading.Thread(target=fire)
        th.start()
        # wait until the request is admitted (visible as pending in the WAL)
        deadline = time.time() + 30
        rid = None
        while time.time() < deadline and rid is None:
            st, r = srv.call("GET", f"/agents/{aid}/requests")
            pend = r.get("data", {}).get("pending", [])
            if pend:
                rid = pend[0]["id"]
            time.sleep(0.05)
        assert rid, 'request never became pending'
        time.sleep(0.3)  # let decoding actually start
        srv.kill9()      # mid-stream
        th.join(timeout=30)
    finally:
        srv.terminate()

    srv2 = Server(root, port, device="cpu")
    try:
        srv2.start(timeout=120)
        deadline = time.time() + 120
        rec = None
        while time.time() < deadline:
            st, r = srv2.call("GET", f"/agents/{aid}/requests/{rid}")
            if st == 200 and r["data"]["status"] == "completed":
                rec = r["data"]
                break
            time.sleep(0.5)
        assert rec is not None, "mid-stream request did not replay"
        assert rec["response"]["tokens"] == 300  # full regeneration
    finally:
        srv2.terminate()


@pytest.mark.timeout(120)
def test_logs_follow_stream(tmp_path):
    """`GET /logs/stream` (the `agentainer logs -f` tail): lines published
    after the subscription arrive as SSE frames over a REAL server
    (TestClient's ASGI transport buffers, so this needs uvicorn)."""
    import threading

    root = str(tmp_path / "root")
    srv = Server(root, _free_port())
    try:
        srv.start()
        st, resp = srv.call("POST", "/agents", {"name": "tailed", "model": "echo"})
        aid = resp["data"]["id"]

        def traffic():
            time.sleep(0.4)
            srv.call("POST", f"/agents/{aid}/start")
            srv.call("POST", f"/agent/{aid}/chat", body={"message": "make logs"},
                     auth=False)

        th = threading.Thread(target=traffic)
        th.start()
        got = []
        deadline = time.time() + 30
        with httpx.stream("GET", f"{srv.base}/logs/stream",
                          headers=AUTH, timeout=30) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                if time.time() > deadline:
                    break
                if line.startswith("data: "):
                    got.append(json.loads(line[len("data: "):]))
                if any(g.get("action") == "start" or
                       "start" in str(g.get("message", "")) for g in got):
                    break
        th.join(timeout=10)
        assert got, "no log lines streamed"
    finally:
        srv.terminate()
