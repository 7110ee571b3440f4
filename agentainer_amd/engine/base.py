"""EngineBackend protocol — what the control plane needs from a data plane.

The reference's data plane is the Docker daemon (pkg/docker/client.go:10-28);
ours is an in-process inference engine. This interface is the seam: the
registry Manager, reconciler, health monitor and metrics collector all talk
to an EngineBackend, so the control plane runs identically over the echo
stub (BASELINE.json config 1, CPU) and the LLM engine (configs 2-5, MI355X).
"""

from __future__ import annotations

from typing import Any, Dict, List, Protocol, runtime_checkable


class ModelNotFound(Exception):
    """Analog of deploy's image-must-exist failure (reference agent.go:106-112)."""


@runtime_checkable
class EngineBackend(Protocol):
    def validate_model(self, model: str) -> None:
        """Raise ModelNotFound if `model` is not a known family or weights path."""
        ...

    def attach(self, agent) -> None:
        """Bind the agent: load/refcount the model shard, allocate its KV
        binding, restore offloaded KV if any, open admission."""
        ...

    def detach(self, agent_id: str, offload_kv: bool = True) -> bool:
        """Drain the agent's in-flight work and release its KV pages.
        offload_kv=True parks the KV image in host memory first; returns
        True iff a KV image was offloaded."""
        ...

    def pause(self, agent_id: str) -> None:
        """Close admission; KV stays resident in HBM."""
        ...

    def unpause(self, agent_id: str) -> None:
        ...

    def is_attached(self, agent_id: str) -> bool:
        ...

    def attached_ids(self) -> List[str]:
        """Engine-side ground truth for the reconciler (replaces listing
        Docker containers by label, state_sync.go:86-95)."""
        ...

    def engine_status(self, agent_id: str) -> str:
        """'running' | 'paused' | 'missing' — the engine's own view."""
        ...

    def chat(self, agent_id: str, message: str, **kwargs: Any) -> Dict[str, Any]:
        """Synchronous chat completion for the agent (admission -> batch ->
        decode -> response). Raises EngineUnavailable-compatible errors when
        the agent/engine is down."""
        ...

    def stats(self) -> Dict[str, Any]:
        """Engine counters: per-agent tokens, batch occupancy, KV usage."""
        ...

    def health_probe(self, agent_id: str) -> bool:
        """Liveness probe for the health monitor (replaces the HTTP GET
        /health probe through the proxy, monitor.go:225-234)."""
        ...
