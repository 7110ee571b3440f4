#!/usr/bin/env python3
"""Full-stack HTTP serving load: sustained closed-loop chat req/s against
a REAL server process (uvicorn + auth/proxy + WAL + engine + history).

This measures the metric the reference anchors at the proxy tier
("thousands of requests/second", NETWORK_ARCHITECTURE.md:448) on the
WHOLE durable LLM path: N agents, C closed-loop HTTP clients, a timed
window, whole-run req/s + p50/p99 E2E (client-observed).

    python tools/server_load.py [--model llama3-8b] [--agents 64]
        [--window 45] [--gen 32] [--device cuda]

Spawns its own server on a free port; prints one JSON line.
"""
import argparse
import concurrent.futures
import json
import os
import signal
import socket
import statistics
import subprocess
import sys
import tempfile
import threading
import time

import httpx

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TOKEN = "agentainer-default-token"
AUTH = {"Authorization": f"Bearer {TOKEN}"}


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=None)
    p.add_argument("--agents", type=int, default=64)
    p.add_argument("--window", type=float, default=45.0)
    p.add_argument("--warmup", type=float, default=10.0)
    p.add_argument("--gen", type=int, default=32)
    p.add_argument("--prompt-words", type=int, default=48)
    p.add_argument("--clear-every", type=int, default=1,
                   help="clear each conversation every N turns (0 = never; "
                        "without clearing, context grows unboundedly and "
                        "the workload hardens over the window)")
    p.add_argument("--device", default=None, help="cuda|cpu|echo")
    args = p.parse_args()
    import torch

    has_gpu = torch.cuda.is_available()
    device = args.device or ("cuda" if has_gpu else "echo")
    model = args.model or ("llama3-8b" if device == "cuda" else "echo")
    if device != "cuda":
        args.agents = min(args.agents, 8)
        args.window = min(args.window, 10.0)
        args.warmup = min(args.warmup, 2.0)

    # SERVER_LOAD_LOGDIR: put the state root (incl. server.log) somewhere
    # inspectable — e.g. gpurun_out/ so a crashed server's log survives
    root = (os.environ.get("SERVER_LOAD_LOGDIR")
            or tempfile.mkdtemp(prefix="srvload-"))
    os.makedirs(root, exist_ok=True)
    port = _free_port()
    env = dict(os.environ)
    env.update({"AGENTAINER_STORE_PATH": root,
                "AGENTAINER_SERVER_PORT": str(port),
                "PYTHONPATH": ROOT})
    log = open(os.path.join(root, "server.log"), "ab")
    srv = subprocess.Popen(
        [sys.executable, "-m", "agentainer_amd.cli", "server",
         "--engine-device", device],
        env=env, cwd=ROOT, stdout=log, stderr=subprocess.STDOUT,
        start_new_session=True)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.time() + 180
        while time.time() < deadline:
            try:
                if httpx.get(base + "/health", timeout=2.0).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.3)
        else:
            raise TimeoutError("server did not come up")

        aids = []
        with httpx.Client(timeout=120.0) as c:
            for i in range(args.agents):
                r = c.post(base + "/agents", headers=AUTH, json={
                    "name": f"load-{i}", "model": model,
                    "sampling": {"max_tokens": args.gen}})
                aid = r.json()["data"]["id"]
                c.post(base + f"/agents/{aid}/start", headers=AUTH)
                aids.append(aid)

        stop = threading.Event()
        rec_on = threading.Event()
        lat = []
        ttfts = []
        lat_lock = threading.Lock()
        counts = [0] * args.agents
        prompt = " ".join(f"w{i}" for i in range(args.prompt_words))

        def client(i):
            aid = aids[i]
            turns = 0
            with httpx.Client(timeout=120.0) as c:
                while not stop.is_set():
                    t0 = time.time()
                    r = c.post(base + f"/agent/{aid}/chat",
                               json={"message": prompt})
                    dt = time.time() - t0
                    if r.status_code != 200:
                        continue
                    turns += 1
                    if args.clear_every and turns % args.clear_every == 0:
                        c.post(base + f"/agent/{aid}/clear", json={})
                    if rec_on.is_set():
                        tf = None
                        try:
                            tf = r.json().get("ttft_s")
                        except ValueError:
                            pass
                        with lat_lock:
                            lat.append(dt)
                            counts[i] += 1
                            if tf is not None:
                                ttfts.append(float(tf))

        def engine_stats():
            try:
                r = httpx.get(base + "/metrics/engine", headers=AUTH,
                              timeout=10.0)
                m = (r.json()["data"].get("models") or {})
                return next(iter(m.values()), {})
            except Exception:
                return {}

        with concurrent.futures.ThreadPoolExecutor(args.agents) as ex:
            futs = [ex.submit(client, i) for i in range(args.agents)]
            time.sleep(args.warmup)
            st0 = engine_stats()
            rec_on.set()
            t0 = time.time()
            time.sleep(args.window)
            elapsed = time.time() - t0
            rec_on.clear()
            st1 = engine_stats()
            stop.set()
            for f in futs:
                f.result(timeout=150)
        dsteps = (st1.get("steps", 0) or 0) - (st0.get("steps", 0) or 0)
        dtok = (st1.get("decode_tokens", 0) or 0) - (st0.get("decode_tokens", 0) or 0)
        lat.sort()
        n = len(lat)
        out = {
            "tool": "server_load",
            "model": model, "device": device, "agents": args.agents,
            "gen_len": args.gen, "window_s": round(elapsed, 2),
            "http_req_per_s": round(n / elapsed, 2),
            "p50_e2e_s": round(statistics.median(lat), 4) if lat else None,
            "p99_e2e_s": (round(lat[min(n - 1, int(0.99 * n))], 4)
                          if lat else None),
            "completed": n,
            "ttft_p50_s": (round(statistics.median(ttfts), 4)
                           if ttfts else None),
            "ttft_p99_s": (round(sorted(ttfts)[min(len(ttfts) - 1,
                                                   int(0.99 * len(ttfts)))], 4)
                           if ttfts else None),
            # engine-side view of the same window: starved batch shows as
            # low decode occupancy; a starved engine thread as low steps/s
            "engine_steps_per_s": round(dsteps / elapsed, 1),
            "engine_decode_tok_per_s": round(dtok / elapsed, 1),
            "decode_occupancy": (round(dtok / dsteps / args.agents, 3)
                                 if dsteps else None),
            "engine_mode": {k: st1.get(k) for k in
                            ("use_graph", "async_decode", "graph_buckets",
                             "step_ms_ema", "phase_ms")},
        }
        print(json.dumps(out))
    finally:
        try:
            os.killpg(os.getpgid(srv.pid), signal.SIGTERM)
            srv.wait(timeout=20)
        except Exception:
            try:
                os.killpg(os.getpgid(srv.pid), signal.SIGKILL)
            except Exception:
                pass


if __name__ == "__main__":
    main()
