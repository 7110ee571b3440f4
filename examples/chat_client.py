#!/usr/bin/env python3
"""End-to-end walkthrough against a running server (the analog of the
reference's examples/gpt-agent quickstart). Start a server first:

    python -m agentainer_amd.cli server            # MI355X
    python -m agentainer_amd.cli server --engine-device echo   # CPU demo
"""
import os
import sys
import time

import httpx

BASE = os.environ.get("AGENTAINER_URL", "http://127.0.0.1:8081")
TOKEN = os.environ.get("AGENTAINER_TOKEN", "agentainer-default-token")
AUTH = {"Authorization": f"Bearer {TOKEN}"}
MODEL = sys.argv[1] if len(sys.argv) > 1 else "echo"


def call(method, path, auth=True, **kw):
    r = httpx.request(method, BASE + path, headers=AUTH if auth else {},
                      timeout=120, **kw)
    r.raise_for_status()
    return r.json()


agent = call("POST", "/agents", json={
    "name": "demo", "model": MODEL, "auto_restart": True,
    "system_prompt": "You are a demo agent.",
    "sampling": {"max_tokens": 32}})["data"]
aid = agent["id"]
print("deployed", aid)
call("POST", f"/agents/{aid}/start")

for msg in ["hello there", "tell me more"]:
    out = call("POST", f"/agent/{aid}/chat", auth=False,
               json={"message": msg})
    print(f"user: {msg}\nagent: {out['response']}")

print("history:", len(call("GET", f"/agent/{aid}/history",
                           auth=False)["history"]), "turns")

# stop (KV -> pinned host checkpoint); queued request while down
call("POST", f"/agents/{aid}/stop")
r = httpx.post(f"{BASE}/agent/{aid}/chat", json={"message": "queued while down"},
               timeout=30)
assert r.status_code == 202
rid = r.json()["data"]["request_id"]
print("queued:", rid)

# resume (KV restored) -> the replay worker completes the queued request
call("POST", f"/agents/{aid}/resume")
for _ in range(40):
    rec = call("GET", f"/agents/{aid}/requests/{rid}")["data"]
    if rec["status"] == "completed":
        print("replayed:", rec["response"]["response"])
        break
    time.sleep(0.5)

print("metrics:", call("GET", f"/agents/{aid}/metrics")["data"])

# streaming (SSE): tokens as they decode
print("streamed: ", end="", flush=True)
with httpx.stream("POST", f"{BASE}/agent/{aid}/chat",
                  json={"message": "stream this", "stream": True},
                  timeout=120) as r:
    import json as _json
    for line in r.iter_lines():
        if line.startswith("data: "):
            ev = _json.loads(line[len("data: "):])
            if ev.get("done"):
                print(f"  [{ev.get('tokens', '?')} tokens]")
            elif "text" in ev:
                print(ev["text"], end="", flush=True)
