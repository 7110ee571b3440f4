#!/usr/bin/env python3
"""Second example-agent flavor (the reference ships two distinct agents,
examples/gpt-agent and examples/gemini-agent): an OUT-OF-PROCESS consumer
agent.

In the reference, agent containers share state through Agentainer's
Redis (examples/gpt-agent/app.py:39-67): one agent writes conversation
history, another process can read it. This runtime embeds the store, so
the out-of-process consumer story runs over the REST surface instead:
the summarizer process

  1. deploys its own `summarizer` agent (a second model binding),
  2. watches a source agent's conversation via GET /agent/{id}/history
     (the Redis-list read analog), and
  3. periodically chats a digest request into its own conversation,
     clearing its context first (/clear = history wipe + KV reset).

Start a server, then:
    python examples/summarizer_agent.py            # echo engine demo
    python examples/summarizer_agent.py llama3-8b  # on MI355X
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


# --- the source conversation (any already-running agent works too) ------
src = call("POST", "/agents", json={
    "name": "worker", "model": MODEL,
    "system_prompt": "You are a terse task assistant.",
    "sampling": {"max_tokens": 24}})["data"]
call("POST", f"/agents/{src['id']}/start")
for msg in ["plan the rollout", "what are the risks?", "draft the email"]:
    call("POST", f"/agent/{src['id']}/chat", auth=False,
         json={"message": msg})

# --- the summarizer agent (this process's own binding) ------------------
summ = call("POST", "/agents", json={
    "name": "summarizer", "model": MODEL, "auto_restart": True,
    "system_prompt": "Summarize conversations in one short sentence.",
    "sampling": {"max_tokens": 48, "temperature": 0.0}})["data"]
call("POST", f"/agents/{summ['id']}/start")

seen = 0
for round_no in range(3):
    # Redis-consumer analog: read the worker's history list over REST
    hist = call("GET", f"/agent/{src['id']}/history", auth=False)["history"]
    if len(hist) > seen:
        new = hist[seen:]
        seen = len(hist)
        digest_input = " | ".join(
            f"{t.get('user', '')} -> {t.get('assistant', '')}" for t in new)
        # fresh context per digest: /clear wipes history AND resets KV
        call("POST", f"/agent/{summ['id']}/clear", auth=False, json={})
        out = call("POST", f"/agent/{summ['id']}/chat", auth=False,
                   json={"message": f"Summarize: {digest_input}"})
        print(f"[digest {round_no}] {out['response']}")
    # next worker turn while the summarizer sleeps
    call("POST", f"/agent/{src['id']}/chat", auth=False,
         json={"message": f"follow-up {round_no}"})
    time.sleep(0.2)

print("worker metrics:", call("GET", f"/agent/{src['id']}/metrics",
                              auth=False)["metrics"])
print("summarizer history:",
      len(call("GET", f"/agent/{summ['id']}/history", auth=False)["history"]),
      "turn(s)")
