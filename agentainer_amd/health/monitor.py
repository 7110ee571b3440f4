"""Health monitor — per-agent liveness probes with auto-restart.

Rebuilds `internal/health/monitor.go`:

  * Defaults: 30s interval / 5s timeout / 3 retries (monitor.go:117-129).
  * Probe = engine liveness check (replaces the HTTP GET through the proxy,
    monitor.go:225-234 — the engine is in-process, no HTTP needed).
  * Status persisted at `health:{id}` with a 24h TTL (monitor.go:267-270).
  * failure_count >= retries and auto_restart => Manager.restart + reset
    (monitor.go:273-297).
  * Registration is event-driven via store pub/sub on `agent:status:*` —
    with a pattern subscription that actually works, fixing the reference's
    Subscribe-vs-PSubscribe bug (monitor.go:301) — plus
    bootstrap of already-running agents at server start (monitor.go:70-84).

One monitor thread polls all registered agents (an in-process probe is
microseconds; the reference's goroutine-per-agent was shaped by HTTP).
"""

from __future__ import annotations

import json
import subprocess
import threading
import time
import traceback
from typing import Any, Dict, List, Optional

from ..registry import RUNNING, Manager
from ..store import Store

DEFAULT_INTERVAL_S = 30.0
DEFAULT_TIMEOUT_S = 5.0
DEFAULT_RETRIES = 3
STATUS_TTL_S = 24 * 3600.0


def _sum_keys(obj: Any, needle: str) -> int:
    """Recursively sum every integer field whose key contains `needle`
    (amd-smi's JSON nesting varies across versions; count totals don't)."""
    total = 0
    if isinstance(obj, dict):
        for k, v in obj.items():
            if needle in k.lower() and isinstance(v, (int, float)):
                total += int(v)
            else:
                total += _sum_keys(v, needle)
    elif isinstance(obj, list):
        for v in obj:
            total += _sum_keys(v, needle)
    return total


class GpuFaultDetector:
    """Device-level GPU fault detection ("HBM ECC /
    'GPU fell off the bus' detection via amd-smi"). The reference detects
    container death three ways (events/reconcile/probes,
    state_sync.go:253-309, monitor.go:207-251); the MI355X analog of the
    hardware dying under an agent is the accelerator itself faulting —
    which an IN-PROCESS probe cannot see (a hung HIP stream blocks the
    caller), so this probes via an external `amd-smi` subprocess under a
    hard timeout.

    Faults reported:
      * gpu_missing      — amd-smi sees no GPU / exits nonzero / times out
      * ecc_uncorrectable — the uncorrectable-ECC total INCREASED since
                            the baseline sample
    Correctable ECC increases are recorded in the status but are not a
    fault (HBM scrubbing handles them)."""

    def __init__(self, interval_s: float = 30.0, timeout_s: float = 30.0,
                 cmd: Optional[List[str]] = None):
        # timeout covers amd-smi's cold start (first invocation on a fresh
        # node measures >10s while it enumerates devices)
        self.interval_s = interval_s
        self.timeout_s = timeout_s
        self.cmd = cmd or ["amd-smi", "metric", "--ecc", "--json"]
        self.fault: Optional[str] = None
        self.status: Dict[str, Any] = {}
        self._baseline_ue: Optional[int] = None
        self._baseline_ce: int = 0
        self._next_at = 0.0

    def _sample(self) -> Dict[str, Any]:
        try:
            out = subprocess.run(self.cmd, capture_output=True, text=True,
                                 timeout=self.timeout_s)
        except (subprocess.TimeoutExpired, OSError) as e:
            return {"ok": False, "error": type(e).__name__}
        if out.returncode != 0:
            return {"ok": False, "error": f"rc={out.returncode}",
                    "stderr": out.stderr[-500:]}
        try:
            blob = json.loads(out.stdout)
        except ValueError:
            return {"ok": False, "error": "unparseable amd-smi output"}
        if isinstance(blob, dict) and isinstance(blob.get("gpu_data"), list):
            gpus = blob["gpu_data"]  # amd-smi 26.x: {"gpu_data": [...]}
        elif isinstance(blob, list):
            gpus = blob
        else:
            gpus = [blob]
        if not gpus:
            return {"ok": False, "error": "no GPUs reported"}
        return {"ok": True,
                "n_gpus": len(gpus),
                "uncorrectable": _sum_keys(blob, "uncorrect"),
                "correctable": _sum_keys(blob, "correctable")
                - _sum_keys(blob, "uncorrect")}

    def check(self, now: Optional[float] = None) -> Optional[str]:
        """Rate-limited probe; returns the current fault verdict."""
        now = now if now is not None else time.time()
        if now < self._next_at:
            return self.fault
        self._next_at = now + self.interval_s
        s = self._sample()
        s["last_check"] = now
        self.status = s
        if not s["ok"]:
            self.fault = "gpu_missing"
            return self.fault
        ue = s["uncorrectable"]
        if self._baseline_ue is None:
            self._baseline_ue = ue
            self._baseline_ce = s["correctable"]
        if ue > self._baseline_ue:
            self.fault = "ecc_uncorrectable"
        else:
            self.fault = None
        return self.fault


class HealthMonitor:
    def __init__(self, store: Store, manager: Manager,
                 interval_s: float = DEFAULT_INTERVAL_S,
                 timeout_s: float = DEFAULT_TIMEOUT_S,
                 retries: int = DEFAULT_RETRIES,
                 gpu_fault: Optional[GpuFaultDetector] = None):
        self.store = store
        self.manager = manager
        self.interval_s = interval_s
        self.timeout_s = timeout_s
        self.retries = retries
        self.gpu_fault = gpu_fault
        self._watch: Dict[str, Dict[str, Any]] = {}
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._unsub = None

    # ---------- lifecycle ----------

    def start(self) -> None:
        # bootstrap: monitor everything already running (monitor.go:70-84)
        for agent in self.manager.list():
            if agent.status == RUNNING:
                self.start_monitoring(agent.id, agent.health_check)
        self._unsub = self.store.subscribe("agent:status:*", self._on_status_event)
        if self._thread is None:
            self._stop.clear()
            self._thread = threading.Thread(target=self._run, name="health-monitor", daemon=True)
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._unsub:
            self._unsub()
            self._unsub = None
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None

    def _on_status_event(self, channel: str, status: str) -> None:
        agent_id = channel.rsplit(":", 1)[-1]
        if status == RUNNING:
            agent = self.manager.try_get(agent_id)
            self.start_monitoring(agent_id, agent.health_check if agent else None)
        elif status in ("stopped", "removed", "failed"):
            self.stop_monitoring(agent_id)

    # ---------- registration ----------

    def start_monitoring(self, agent_id: str, check_config: Optional[Dict[str, Any]] = None) -> None:
        cfg = check_config or {}
        with self._lock:
            self._watch[agent_id] = {
                "interval": float(cfg.get("interval", self.interval_s)),
                "retries": int(cfg.get("retries", self.retries)),
                "failures": 0,
                "next_at": 0.0,
            }

    def stop_monitoring(self, agent_id: str) -> None:
        with self._lock:
            self._watch.pop(agent_id, None)

    def monitored_ids(self):
        with self._lock:
            return list(self._watch)

    # ---------- probing ----------

    def _run(self) -> None:
        while not self._stop.wait(1.0):
            try:
                self.check_due()
            except Exception:
                traceback.print_exc()

    def check_due(self, now: Optional[float] = None) -> None:
        now = now if now is not None else time.time()
        if self.gpu_fault is not None:
            self.gpu_fault.check(now)
            self.store.set("health:gpu",
                           {"fault": self.gpu_fault.fault,
                            **self.gpu_fault.status}, ttl=STATUS_TTL_S)
        with self._lock:
            due = [aid for aid, w in self._watch.items() if w["next_at"] <= now]
        for aid in due:
            self.check_one(aid, now=now)

    def check_one(self, agent_id: str, now: Optional[float] = None) -> Dict[str, Any]:
        now = now if now is not None else time.time()
        with self._lock:
            w = self._watch.get(agent_id)
        if w is None:
            return {}
        try:
            healthy = bool(self.manager.engine.health_probe(agent_id))
        except Exception:
            healthy = False
        gpu_fault = self.gpu_fault.fault if self.gpu_fault is not None else None
        if gpu_fault is not None:
            healthy = False  # every agent on a faulted device is down
        with self._lock:
            w["next_at"] = now + w["interval"]
            if healthy:
                w["failures"] = 0
            else:
                w["failures"] += 1
            failures = w["failures"]
            threshold = w["retries"]
        status = {
            "agent_id": agent_id,
            "healthy": healthy,
            "consecutive_failures": failures,
            "last_check": now,
        }
        if gpu_fault is not None:
            status["gpu_fault"] = gpu_fault
        self.store.set(f"health:{agent_id}", status, ttl=STATUS_TTL_S)
        if not healthy and failures >= threshold:
            agent = self.manager.try_get(agent_id)
            if agent is not None and agent.auto_restart:
                try:
                    self.manager.restart(agent_id)  # monitor.go:273-297
                    with self._lock:
                        if agent_id in self._watch:
                            self._watch[agent_id]["failures"] = 0
                except Exception:
                    traceback.print_exc()
        return status

    # ---------- queries ----------

    def get_status(self, agent_id: str) -> Optional[Dict[str, Any]]:
        return self.store.get(f"health:{agent_id}")

    def get_all_statuses(self) -> Dict[str, Dict[str, Any]]:
        out = {}
        for key in self.store.keys("health:*"):
            st = self.store.get(key)
            if st:
                out[key.split(":", 1)[1]] = st
        return out
