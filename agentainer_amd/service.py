"""Runtime — server bootstrap wiring + the request hot path.

Analog of the reference's `runServer` (cmd/agentainer/main.go:284-356):
composes store, engine, registry manager, reconciler (10s), request WAL +
replay worker (5s), health monitor, metrics collector, logger/audit and
backup manager.

The hot path: where the reference did
  proxy -> Redis write-ahead -> HTTP to container -> Redis ack  (>=5 RTs)
this runtime does
  WAL append (fsync) -> engine dispatch (in-process) -> WAL ack.

`agent_request` implements the full proxy contract of server.go:493-615:
  * write-ahead store of every non-replay request (server.go:504-522),
  * 202 {request_id, status: pending} when the agent is not running
    (server.go:525-541),
  * dispatch; on success store the response (server.go:588-594),
  * on engine-down leave the request PENDING for replay (server.go:597-605),
  * on other errors mark_failed -> retry/dead-letter (server.go:606-613).

The per-agent endpoint surface mirrors the canonical example agent
(examples/gpt-agent/app.py:32-179): `/` `/health` `/chat` `/history`
`/clear` `/metrics`.
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional, Tuple

from .backup import BackupManager
from .config import Config, load_config
from .engine.base import ModelNotFound
from .engine.echo import EchoEngine
from .health import HealthMonitor
from .logs import Logger, set_global_logger
from .metrics import MetricsCollector
from .registry import Manager, RUNNING
from .registry.reconciler import Reconciler
from .store import Store
from .wal import EngineUnavailable, Request, RequestManager, ReplayWorker


def make_engine(config: Config, store: Store):
    """Engine factory: echo stub (CPU) or the LLM engine (MI355X)."""
    device = config.get("engine", "device", "auto")
    if device == "echo":
        return EchoEngine(store)
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if device == "cpu" or (device == "auto" and not has_gpu):
        # CPU fallback runs the echo engine AND tiny CPU LLMs for tests
        from .engine.llm import LLMEngine
        return LLMEngine(store, config, device="cpu")
    from .engine.llm import LLMEngine
    return LLMEngine(store, config, device="cuda")


class Runtime:
    def __init__(self, config: Optional[Config] = None, engine=None,
                 store: Optional[Store] = None, state_root: Optional[str] = None):
        self.config = config or load_config()
        root = state_root or self.config.state_root
        os.makedirs(root, exist_ok=True)
        self.store = store or Store(os.path.join(root, "state"),
                                    sync=self.config.get("store", "sync", "interval"))
        self.logger = Logger(self.store, os.path.join(root, "logs"),
                             level=self.config.get("logging", "level", "info"))
        set_global_logger(self.logger)
        self.engine = engine if engine is not None else make_engine(self.config, self.store)
        self.agents = Manager(self.store, self.engine, self.config)
        feats = self.config.get("features")
        self.requests = RequestManager(
            self.store,
            ttl_s=float(feats.get("request_ttl_s", 24 * 3600)),
            max_retries=int(feats.get("max_retries", 3)),
        )
        self.persistence_enabled = bool(feats.get("request_persistence", True))
        self._inflight: set = set()
        self.replay = ReplayWorker(self.requests, self.agents, self._dispatch_replay,
                                   interval_s=float(feats.get("replay_interval_s", 5.0)),
                                   inflight=self._inflight)
        self.reconciler = Reconciler(self.agents, interval_s=10.0)
        # device-level fault probing (amd-smi: ECC counts, device
        # presence) on GPU engines — SURVEY.md §5 failure detection
        gpu_fault = None
        if (getattr(engine, "device", "cpu").startswith("cuda")
                and bool(self.config.get("health", "gpu_fault_probe", True))):
            from .health import GpuFaultDetector

            gpu_fault = GpuFaultDetector(
                interval_s=float(self.config.get("health",
                                                 "gpu_fault_interval_s", 30.0)))
        self.health = HealthMonitor(
            self.store, self.agents,
            interval_s=float(self.config.get("health", "interval_s", 30.0)),
            timeout_s=float(self.config.get("health", "timeout_s", 5.0)),
            retries=int(self.config.get("health", "retries", 3)),
            gpu_fault=gpu_fault,
        )
        dev_sampler = None
        if getattr(engine, "device", "cpu").startswith("cuda"):
            from .metrics import DeviceMetricsSampler

            dev_sampler = DeviceMetricsSampler()
        self.metrics = MetricsCollector(
            self.store, self.agents,
            sample_interval_s=float(self.config.get("metrics", "sample_interval_s", 10.0)),
            device_sampler=dev_sampler,
        )
        self.backups = BackupManager(self.store, self.agents, os.path.join(root, "backups"))
        self._started = False

    # ---------- lifecycle ----------

    def start_workers(self) -> None:
        """Start background loops (reconciler, replay, health, metrics)."""
        if self._started:
            return
        self._started = True
        self.reconciler.start()
        if self.persistence_enabled:
            self.replay.start()
        self.health.start()
        self.metrics.start()
        self.logger.info("runtime started", component="service")

    def shutdown(self) -> None:
        if not self._started:
            self.store.close()
            return
        self._started = False
        self.metrics.stop()
        self.health.stop()
        self.replay.stop()
        self.reconciler.stop()
        shutdown_engine = getattr(self.engine, "shutdown", None)
        if shutdown_engine:
            shutdown_engine()
        self.logger.info("runtime stopped", component="service")
        self.store.close()

    # ---------- the hot path ----------

    def agent_request(self, agent_id: str, method: str, path: str,
                      body: Any = None, headers: Optional[Dict[str, str]] = None,
                      replay: bool = False, req_id: Optional[str] = None,
                      ) -> Tuple[int, Dict[str, Any]]:
        """Proxy-equivalent entry point. Returns (http_status, payload)."""
        agent = self.agents.try_get(agent_id)
        if agent is None:
            return 404, {"success": False, "message": f"agent {agent_id} not found"}

        req: Optional[Request] = None
        if self.persistence_enabled and not replay:
            req = self.requests.store_request(agent_id, method, path,
                                              headers=headers, body=body, req_id=req_id)
        elif replay and req_id:
            req = self.requests.get(agent_id, req_id)

        if agent.status != RUNNING:
            # queue semantics (server.go:525-541)
            return 202, {
                "success": True,
                "message": "agent not running; request queued",
                "data": {"request_id": req.id if req else None, "status": "pending"},
            }

        t0 = time.time()
        if req is not None:
            self._inflight.add(req.id)
        try:
            payload = self._serve(agent, method, path, body,
                                  trace_id=req.id if req else "")
        except EngineUnavailable as exc:
            # crash-capture: leave pending (server.go:597-605)
            return 503, {"success": False,
                         "message": f"agent unavailable: {exc}",
                         "data": {"request_id": req.id if req else None,
                                  "status": "pending"}}
        except Exception as exc:  # noqa: BLE001
            if req is not None:
                self.requests.mark_failed(agent_id, req.id, str(exc))
            return 500, {"success": False, "message": str(exc),
                         "data": {"request_id": req.id if req else None}}
        finally:
            if req is not None:
                self._inflight.discard(req.id)
        e2e = time.time() - t0
        if req is not None:
            self.requests.store_response(agent_id, req.id, payload)
        self.metrics.observe_request(agent_id, e2e,
                                     tokens=payload.get("tokens", 0)
                                     if isinstance(payload, dict) else 0)
        return 200, payload

    def agent_request_stream(self, agent_id: str, body: Any = None,
                             headers: Optional[Dict[str, str]] = None):
        """Streaming variant of the /chat hot path (SSE). Returns
        (status, payload) for non-streamable outcomes (404/202), else
        (200, generator) — the generator yields event dicts and performs
        the WAL ack + metrics write when the generation completes. A
        stream cut by an engine crash leaves the WAL entry PENDING, so
        the replay worker regenerates it (same at-least-once contract as
        the blocking path)."""
        agent = self.agents.try_get(agent_id)
        if agent is None:
            return 404, {"success": False, "message": f"agent {agent_id} not found"}
        req: Optional[Request] = None
        if self.persistence_enabled:
            req = self.requests.store_request(agent_id, "POST", "/chat",
                                              headers=headers, body=body)
        if agent.status != RUNNING:
            return 202, {
                "success": True,
                "message": "agent not running; request queued",
                "data": {"request_id": req.id if req else None, "status": "pending"},
            }
        message = body.get("message", "") if isinstance(body, dict) else str(body)
        kwargs = dict(body.get("sampling", {})) if isinstance(body, dict) else {}
        if req is not None:
            kwargs["trace_id"] = req.id

        def run():
            t0 = time.time()
            if req is not None:
                self._inflight.add(req.id)
            final = None
            try:
                for ev in self.engine.chat_stream(agent.id, message, **kwargs):
                    if ev.get("done"):
                        final = ev
                    yield ev
            except EngineUnavailable as exc:
                yield {"error": f"agent unavailable: {exc}",
                       "request_id": req.id if req else None,
                       "status": "pending"}
                return
            except Exception as exc:  # noqa: BLE001
                if req is not None:
                    self.requests.mark_failed(agent_id, req.id, str(exc))
                yield {"error": str(exc),
                       "request_id": req.id if req else None}
                return
            finally:
                if req is not None:
                    self._inflight.discard(req.id)
            if final is not None:
                payload = {k: v for k, v in final.items() if k != "done"}
                if req is not None:
                    self.requests.store_response(agent_id, req.id, payload)
                self.metrics.observe_request(
                    agent_id, time.time() - t0,
                    tokens=payload.get("tokens", 0) or 0)

        return 200, run()

    def _dispatch_replay(self, agent_id: str, req: Request, replay: bool = True) -> Any:
        """ReplayWorker dispatch: direct engine path, response stored by the
        worker (single store — quirk fix vs replay_worker.go:158)."""
        agent = self.agents.get(agent_id)
        if agent.status != RUNNING:
            raise EngineUnavailable(f"agent {agent_id} is {agent.status}")
        return self._serve(agent, req.method, req.path, req.body,
                           trace_id=req.id)

    # ---------- per-agent endpoint surface (gpt-agent app.py contract) ----------

    def _serve(self, agent, method: str, path: str, body: Any,
               trace_id: str = "") -> Dict[str, Any]:
        path = "/" + path.strip("/")
        body = body or {}
        if path == "/chat":
            message = body.get("message", "") if isinstance(body, dict) else str(body)
            kwargs = dict(body.get("sampling", {})) if isinstance(body, dict) else {}
            if trace_id:
                kwargs["trace_id"] = trace_id
            out = self.engine.chat(agent.id, message, **kwargs)
            return out
        if path == "/health":
            ok = self.engine.health_probe(agent.id)
            if not ok:
                raise EngineUnavailable(f"agent {agent.id} unhealthy")
            return {"status": "healthy", "agent_id": agent.id}
        if path == "/history":
            return {"history": self.store.lrange(f"agent:{agent.id}:conversations")}
        if path == "/clear":
            self.store.delete(f"agent:{agent.id}:conversations")
            # also reset the engine-side KV sequence: a cleared history
            # with resident KV would leave the next turn attending to the
            # old conversation (and double the system prompt)
            reset = getattr(self.engine, "reset_conversation", None)
            if reset is not None:
                reset(agent.id)
            return {"status": "cleared"}
        if path == "/metrics":
            return {"metrics": self.store.hgetall(f"agent:{agent.id}:metrics")}
        if path == "/":
            return {"agent_id": agent.id, "name": agent.name, "model": agent.model,
                    "status": agent.status}
        raise ValueError(f"unknown agent endpoint {path!r}")

    # ---------- crash recovery ----------

    def recover(self, block: bool = False) -> int:
        """Boot-time recovery: reconcile, restart auto-restart agents and
        kick off replay of their pending WAL (the resume -> replay flow,
        SURVEY.md §3.5). Replay runs in the background by default so a
        server with long pending generations still starts serving
        immediately (the replay worker owns the retry cadence)."""
        import threading

        self.reconciler.sync_all()
        for agent in self.agents.list():
            if agent.auto_restart and agent.status != RUNNING:
                try:
                    self.agents.start(agent.id)
                except Exception:
                    self.logger.error(f"auto-restart of {agent.id} failed",
                                      component="service", agent_id=agent.id)
        if block:
            return self.replay.tick()
        if self.persistence_enabled:
            threading.Thread(target=self.replay.tick, name="boot-replay",
                             daemon=True).start()
        return 0
