"""Echo engine — the mocked LLM of BASELINE.json config 1.

Plays the role of examples/gpt-agent/app.py (the reference's canonical
Flask agent): /chat-style responses, conversation history kept in the
runtime store under `agent:{id}:conversations` trimmed to 50 entries with
the last 3 used as context (reference examples/gpt-agent/app.py:39-67,
89-92), and a metrics hash `agent:{id}:metrics` (app.py:66).

No GPU, no model weights: the response echoes the message plus context
size. Used by the CPU test suite as the fake engine backend for FSM, WAL,
retry/dead-letter and reconciler convergence tests.
"""

from __future__ import annotations

import threading
import time
from typing import Any, Dict, List

from ..store import Store
from .base import ModelNotFound

ECHO_MODELS = ("echo", "echo-stub", "gpt-echo")


class EchoEngine:
    def __init__(self, store: Store, fail_on: str = ""):
        self.store = store
        self._attached: Dict[str, Dict[str, Any]] = {}
        self._lock = threading.RLock()
        self._crashed = False
        self.fail_on = fail_on  # test hook: message substring that raises

    # ---------- EngineBackend ----------

    def validate_model(self, model: str) -> None:
        if model not in ECHO_MODELS:
            raise ModelNotFound(f"unknown echo model {model!r} (expected one of {ECHO_MODELS})")

    def attach(self, agent) -> None:
        with self._lock:
            self._attached[agent.id] = {
                "agent": agent, "paused": False, "since": time.time(),
                "requests": 0, "tokens": 0,
            }

    def detach(self, agent_id: str, offload_kv: bool = True) -> bool:
        with self._lock:
            self._attached.pop(agent_id, None)
            return bool(offload_kv)

    def pause(self, agent_id: str) -> None:
        with self._lock:
            if agent_id in self._attached:
                self._attached[agent_id]["paused"] = True

    def unpause(self, agent_id: str) -> None:
        with self._lock:
            if agent_id in self._attached:
                self._attached[agent_id]["paused"] = False

    def is_attached(self, agent_id: str) -> bool:
        with self._lock:
            return agent_id in self._attached

    def attached_ids(self) -> List[str]:
        with self._lock:
            return list(self._attached.keys())

    def engine_status(self, agent_id: str) -> str:
        with self._lock:
            st = self._attached.get(agent_id)
            if st is None:
                return "missing"
            return "paused" if st["paused"] else "running"

    def health_probe(self, agent_id: str) -> bool:
        return self.engine_status(agent_id) == "running" and not self._crashed

    # ---------- chat ----------

    def chat(self, agent_id: str, message: str, **kwargs: Any) -> Dict[str, Any]:
        from ..wal import EngineUnavailable

        with self._lock:
            st = self._attached.get(agent_id)
        if self._crashed or st is None:
            raise EngineUnavailable(f"agent {agent_id} is not attached")
        if st["paused"]:
            raise EngineUnavailable(f"agent {agent_id} is paused")
        if self.fail_on and self.fail_on in message:
            raise RuntimeError("injected failure")

        hist_key = f"agent:{agent_id}:conversations"
        context = self.store.lrange(hist_key, -3, -1)  # last-3 context, app.py:89-92
        reply = f"echo({len(context)}): {message}"
        self.store.rpush(hist_key, {"user": message, "assistant": reply, "ts": time.time()})
        n = self.store.llen(hist_key)
        if n > 50:  # trim to 50, app.py:62-63
            self.store.ltrim(hist_key, n - 50, -1)
        with self._lock:
            st["requests"] += 1
            st["tokens"] += len(reply.split())
        self.store.hset(f"agent:{agent_id}:metrics", "total_requests", st["requests"])
        return {"response": reply, "model": "echo", "context_turns": len(context)}

    def chat_stream(self, agent_id: str, message: str, **kwargs: Any):
        """SSE analog: word-chunked stream of the echo reply, then the
        final payload event (same contract as LLMEngine.chat_stream)."""
        payload = self.chat(agent_id, message, **kwargs)
        for word in payload["response"].split(" "):
            yield {"token": None, "text": word + " "}
        yield {"done": True, **payload}

    # ---------- stats / fault injection ----------

    def stats(self) -> Dict[str, Any]:
        with self._lock:
            return {
                "engine": "echo",
                "agents": {
                    aid: {"requests": st["requests"], "tokens": st["tokens"],
                          "paused": st["paused"]}
                    for aid, st in self._attached.items()
                },
            }

    def crash(self) -> None:
        """Test hook: simulate engine death (docker kill analog)."""
        with self._lock:
            self._crashed = True
            self._attached.clear()

    def recover(self) -> None:
        with self._lock:
            self._crashed = False
