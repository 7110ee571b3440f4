"""Replay worker — re-delivers pending WAL requests to recovered agents.

Rebuilds the reference's `internal/requests/replay_worker.go`:

  * 5-second tick (replay_worker.go:37).
  * Scans agents with pending requests (via the index set, not a KEYS scan).
  * Per agent: skip unless the registry says RUNNING (replay_worker.go:166-189).
  * Replays each pending request FIFO through the same dispatch path the
    proxy uses, flagged `replay=True` so it is not re-stored
    (replay_worker.go:119-163; header X-Agentainer-Replay).
  * Skips requests already at max retries (replay_worker.go:101-105).

Delivery contract is at-least-once (RESILIENT_AGENTS.md:305-324): a request
whose dispatch died mid-flight stays pending and will be replayed again;
with greedy decoding a replayed chat regenerates deterministically
(SURVEY.md §7.3 "mid-stream crash replay", option (a)).

Instead of HTTP round-trips through the proxy (the reference replays via
http://localhost:8081/agent/{id}…), dispatch here is a direct in-process
call into the scheduler — the whole proxy stack collapsed per SURVEY.md §3.4.
"""

from __future__ import annotations

import threading
import traceback
from typing import Callable, Optional

from ..registry import RUNNING, Manager
from .requests import RequestManager

# dispatch(agent_id, request, replay) -> response payload; raises on failure
DispatchFn = Callable[..., object]


class EngineUnavailable(Exception):
    """Dispatch failed because the engine/agent is down — leave the request
    pending (the crash-capture path, reference server.go:597-605)."""


class ReplayWorker:
    def __init__(self, requests: RequestManager, agents: Manager,
                 dispatch: DispatchFn, interval_s: float = 5.0,
                 inflight: Optional[set] = None):
        self.requests = requests
        self.agents = agents
        self.dispatch = dispatch
        self.interval_s = interval_s
        # ids currently being dispatched by the live proxy path: a request
        # is only "orphaned" (replayable) if no dispatch holds it. In-memory
        # on purpose — a crash empties it, so everything pending at death
        # replays (at-least-once), while live slow generations are not
        # double-dispatched.
        self.inflight = inflight if inflight is not None else set()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        if self._thread is not None:
            return
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, name="replay-worker", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=self.interval_s + 1.0)
            self._thread = None

    def _run(self) -> None:
        while not self._stop.wait(self.interval_s):
            try:
                self.tick()
            except Exception:
                traceback.print_exc()

    def tick(self) -> int:
        """One replay sweep; returns number of requests replayed. Public so
        tests (and crash-recovery on boot) can drive it synchronously."""
        replayed = 0
        for agent_id in self.requests.agents_with_pending():
            agent = self.agents.try_get(agent_id)
            if agent is None or agent.status != RUNNING:
                continue
            for req in self.requests.pending(agent_id):
                if req.retry_count >= req.max_retries:
                    continue
                if req.id in self.inflight:
                    continue  # live dispatch in progress — not orphaned
                try:
                    resp = self.dispatch(agent_id, req, replay=True)
                except EngineUnavailable:
                    break  # agent went down again; keep everything pending
                except Exception as exc:  # noqa: BLE001
                    self.requests.mark_failed(agent_id, req.id, str(exc))
                    continue
                self.requests.store_response(agent_id, req.id, resp)
                replayed += 1
        return replayed
