"""Agent registry + lifecycle FSM — the core of the control plane.

Rebuilds the reference's `internal/agent/agent.go` Manager with an
in-process inference engine in place of the Docker daemon:

  * Agent record (reference agent.go:43-59) — model-shard fields replace
    container-image fields.
  * FSM created/running/stopped/paused/failed (agent.go:21-29).
  * Deploy registers only — no engine memory is touched, mirroring
    "Deploy creates no container" (agent.go:104-142).
  * Start = attach model shard + allocate KV + open admission
    (analog of container create+start, agent.go:144-181).
  * Stop = drain + offload KV to pinned host + free pages (agent.go:183-215).
  * Pause = close admission, KV stays resident (Docker pause, agent.go:224-246).
  * Resume = the universal rehydrator for paused/stopped/failed/created,
    re-attaching and restoring KV if needed (agent.go:255-311).
  * Remove = purge registry entry + WAL queues (agent.go:313-370).
  * Status changes publish on `agent:status:{id}` (state_sync.go:312-317).

IDs are UUIDs (deliberate fix of the reference's UnixNano IDs,
SURVEY.md §7.4 / agent.go:594-596).
"""

from __future__ import annotations

import threading
import time
import uuid
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List, Optional

from ..config.config import Config
from ..store import Store

# FSM states (reference agent.go:21-29)
CREATED = "created"
RUNNING = "running"
STOPPED = "stopped"
PAUSED = "paused"
FAILED = "failed"
STATUSES = (CREATED, RUNNING, STOPPED, PAUSED, FAILED)


class AgentError(Exception):
    pass


class AgentNotFound(AgentError):
    pass


@dataclass
class Agent:
    """The persisted agent record (analog of reference agent.go:43-59)."""

    id: str
    name: str
    model: str                       # model family id or weights path (was: Image)
    status: str = CREATED
    dtype: str = "bf16"
    tp_degree: int = 1
    kv_budget: int = 0               # bytes of HBM KV budget; 0 = engine default
    max_context: int = 8192
    env: Dict[str, str] = field(default_factory=dict)
    auto_restart: bool = False
    token: Optional[str] = None      # per-agent token — honored (ref stores but ignores it)
    health_check: Optional[Dict[str, Any]] = None
    system_prompt: str = ""
    sampling: Dict[str, Any] = field(default_factory=dict)
    created_at: float = 0.0
    updated_at: float = 0.0
    kv_offloaded: bool = False       # True when stop() parked the KV in host memory

    def to_dict(self) -> Dict[str, Any]:
        return asdict(self)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Agent":
        known = {f for f in Agent.__dataclass_fields__}
        return Agent(**{k: v for k, v in d.items() if k in known})


class Manager:
    """Agent lifecycle manager over a Store and an engine backend.

    The engine backend must implement the `EngineBackend` protocol
    (agentainer_amd.engine.base): validate_model, attach, detach, pause,
    unpause, is_attached, attached_ids.
    """

    def __init__(self, store: Store, engine, config: Optional[Config] = None):
        self.store = store
        self.engine = engine
        self.config = config
        self._lock = threading.RLock()

    # ---------- persistence helpers ----------

    def _save(self, agent: Agent) -> None:
        agent.updated_at = time.time()
        self.store.set(f"agent:{agent.id}", agent.to_dict())
        self.store.sadd("agents:list", agent.id)
        self.store.flush()

    def _set_status(self, agent: Agent, status: str) -> None:
        if agent.status != status:
            agent.status = status
            self._save(agent)
            self.store.publish(f"agent:status:{agent.id}", status)
        else:
            self._save(agent)

    # ---------- queries ----------

    def get(self, agent_id: str) -> Agent:
        d = self.store.get(f"agent:{agent_id}")
        if d is None:
            raise AgentNotFound(f"agent {agent_id} not found")
        return Agent.from_dict(d)

    def try_get(self, agent_id: str) -> Optional[Agent]:
        d = self.store.get(f"agent:{agent_id}")
        return Agent.from_dict(d) if d is not None else None

    def list(self) -> List[Agent]:
        out = []
        for aid in self.store.smembers("agents:list"):
            a = self.try_get(aid)
            if a is not None:
                out.append(a)
        out.sort(key=lambda a: a.created_at)
        return out

    # ---------- lifecycle ----------

    def deploy(
        self,
        name: str,
        model: str,
        *,
        dtype: str = "bf16",
        tp_degree: int = 1,
        kv_budget: int = 0,
        max_context: int = 8192,
        env: Optional[Dict[str, str]] = None,
        auto_restart: bool = False,
        token: Optional[str] = None,
        health_check: Optional[Dict[str, Any]] = None,
        system_prompt: str = "",
        sampling: Optional[Dict[str, Any]] = None,
    ) -> Agent:
        """Register an agent. Validates the model exists (analog of the
        image-must-exist check, agent.go:106-112) but allocates nothing."""
        with self._lock:
            self.engine.validate_model(model)
            agent = Agent(
                id=f"agent-{uuid.uuid4().hex[:12]}",
                name=name, model=model, dtype=dtype, tp_degree=tp_degree,
                kv_budget=kv_budget, max_context=max_context,
                env=dict(env or {}), auto_restart=auto_restart, token=token,
                health_check=health_check, system_prompt=system_prompt,
                sampling=dict(sampling or {}),
                created_at=time.time(),
            )
            self._save(agent)
            self.store.publish(f"agent:status:{agent.id}", CREATED)
            return agent

    def start(self, agent_id: str) -> Agent:
        with self._lock:
            agent = self.get(agent_id)
            if agent.status == RUNNING and self.engine.is_attached(agent.id):
                return agent
            try:
                self.engine.attach(agent)
                agent.kv_offloaded = False
                self._set_status(agent, RUNNING)
            except Exception:
                self._set_status(agent, FAILED)
                raise
            return agent

    def stop(self, agent_id: str) -> Agent:
        with self._lock:
            agent = self.get(agent_id)
            if self.engine.is_attached(agent.id):
                # drain + offload KV to pinned host (replaces "container keeps
                # its filesystem" durability, SURVEY.md §2.3 KV-cache manager row)
                offloaded = self.engine.detach(agent.id, offload_kv=True)
                agent.kv_offloaded = bool(offloaded)
            self._set_status(agent, STOPPED)
            return agent

    def restart(self, agent_id: str) -> Agent:
        # reference Restart = Stop + Start (agent.go:217-222)
        self.stop(agent_id)
        return self.start(agent_id)

    def pause(self, agent_id: str) -> Agent:
        with self._lock:
            agent = self.get(agent_id)
            if agent.status != RUNNING:
                raise AgentError(f"agent {agent_id} is {agent.status}, not running")
            self.engine.pause(agent.id)  # admission closed, KV resident
            self._set_status(agent, PAUSED)
            return agent

    def resume(self, agent_id: str) -> Agent:
        """Universal rehydrator (reference agent.go:255-311): paused agents
        unpause; stopped/failed/created agents (re)attach with KV restore."""
        with self._lock:
            agent = self.get(agent_id)
            if agent.status == PAUSED and self.engine.is_attached(agent.id):
                self.engine.unpause(agent.id)
                self._set_status(agent, RUNNING)
                return agent
            return self.start(agent_id)

    def remove(self, agent_id: str, request_manager=None) -> None:
        """Purge agent + its WAL queues + health/metrics keys (agent.go:313-370)."""
        with self._lock:
            agent = self.get(agent_id)
            if self.engine.is_attached(agent.id):
                self.engine.detach(agent.id, offload_kv=False)
            if hasattr(self.engine, "purge_agent"):
                self.engine.purge_agent(agent.id)  # drop KV checkpoints
            if request_manager is not None:
                request_manager.purge_agent(agent.id)
            self.store.delete(f"agent:{agent.id}")
            self.store.srem("agents:list", agent.id)
            self.store.delete(f"health:{agent.id}")
            self.store.delete(f"metrics:current:{agent.id}")
            self.store.delete(f"metrics:history:{agent.id}")
            self.store.delete(f"agent:{agent.id}:conversations")
            self.store.flush()
            self.store.publish(f"agent:status:{agent.id}", "removed")

    def get_logs(self, agent_id: str, limit: int = 200) -> List[Dict[str, Any]]:
        """Per-agent engine log lines (analog of docker logs, agent.go:411-429)."""
        self.get(agent_id)  # existence check
        entries = self.store.lrange(f"agent:{agent_id}:log", -limit, -1)
        return entries
