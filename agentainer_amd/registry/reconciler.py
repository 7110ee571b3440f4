"""Reconciler — converges registry desired-state with engine ground truth.

Rebuilds both reference reconcilers with one component (they were duplicated
only to avoid a Go import cycle, SURVEY.md §2 quick-sync row):

  * `internal/sync/state_sync.go`: periodic full reconcile (10s,
    main.go:325) + event-driven single-agent sync, publishing status
    changes on `agent:status:{id}` (state_sync.go:312-317).
  * `pkg/agentsync/quick_sync.go`: synchronous SyncAgent after every
    lifecycle op and SyncAll before every List.

Mapping (analog of the Docker-state map, state_sync.go:216-229): the
engine's view {running, paused, missing} is the source of truth for
runtime status; the registry holds desired state + metadata. An agent the
registry believes RUNNING/PAUSED whose engine attachment is gone is marked
STOPPED (container-vanished case, state_sync.go:126-213) — and restarted
here if auto_restart is set (replacing Docker's restart-policy `always`,
agent.go:481-495).
"""

from __future__ import annotations

import threading
import traceback
from typing import List, Optional

from . import agent as reg
from .agent import Manager


class Reconciler:
    def __init__(self, manager: Manager, interval_s: float = 10.0,
                 auto_restart: bool = True):
        self.manager = manager
        self.interval_s = interval_s
        self.auto_restart_enabled = auto_restart
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ---------- lifecycle ----------

    def start(self) -> None:
        self.sync_all()  # initial sync (state_sync.go:49-54)
        if self._thread is None:
            self._stop.clear()
            self._thread = threading.Thread(target=self._run, name="reconciler", daemon=True)
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=self.interval_s + 1.0)
            self._thread = None

    def _run(self) -> None:
        while not self._stop.wait(self.interval_s):
            try:
                self.sync_all()
            except Exception:
                traceback.print_exc()

    # ---------- reconcile ----------

    def sync_agent(self, agent_id: str) -> Optional[str]:
        """Converge one agent; returns the (possibly new) status."""
        m = self.manager
        agent = m.try_get(agent_id)
        if agent is None:
            # registry entry gone but engine still holds it: detach (orphan
            # container case, state_sync.go:126-213)
            if m.engine.is_attached(agent_id):
                m.engine.detach(agent_id, offload_kv=False)
            return None
        eng = m.engine.engine_status(agent.id)
        new_status = agent.status
        if eng == "running":
            new_status = reg.RUNNING
        elif eng == "paused":
            new_status = reg.PAUSED
        elif eng == "missing":
            if agent.status in (reg.RUNNING, reg.PAUSED):
                new_status = reg.STOPPED  # attachment vanished
        if new_status != agent.status:
            m._set_status(agent, new_status)
            if (new_status == reg.STOPPED and agent.auto_restart
                    and self.auto_restart_enabled):
                try:
                    m.start(agent.id)
                    new_status = reg.RUNNING
                except Exception:
                    traceback.print_exc()
        return new_status

    def sync_all(self) -> List[str]:
        synced = []
        ids = {a.id for a in self.manager.list()}
        ids.update(self.manager.engine.attached_ids())
        for aid in sorted(ids):
            self.sync_agent(aid)
            synced.append(aid)
        return synced
