"""Metrics collector — engine counters with current + 24h history.

Rebuilds `pkg/metrics/collector.go` with engine counters replacing Docker
container stats (SURVEY.md §5 "Metrics"): tokens/s, requests, TTFT, p50/p99
E2E latency, batch occupancy, KV/HBM usage per agent — same storage and
query API shape as the reference:

  * 10s sampling (collector.go:205).
  * `metrics:current:{id}` with 1h TTL + 24h sorted-set history
    `metrics:history:{id}` keyed by unix ts, trimmed by score
    (collector.go:300-322).
  * get_metrics / get_metrics_history (collector.go:157-200).

Auto-start actually works here (pattern pub/sub + a live registry), unlike
the reference where storage stubs left the collector dormant
(SURVEY.md §2 metrics row, §7.4).
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

CURRENT_TTL_S = 3600.0
HISTORY_RETENTION_S = 24 * 3600.0
SAMPLE_INTERVAL_S = 10.0


def _numeric_leaves(obj: Any, needle: str, path: str = "") -> Dict[str, float]:
    """Flatten numeric JSON leaves whose key path contains `needle`."""
    out: Dict[str, float] = {}
    if isinstance(obj, dict):
        for k, v in obj.items():
            p = f"{path}.{k}" if path else str(k)
            if isinstance(v, (int, float)) and needle in p.lower():
                out[p] = float(v)
            else:
                out.update(_numeric_leaves(v, needle, p))
    elif isinstance(obj, list):
        for i, v in enumerate(obj):
            out.update(_numeric_leaves(v, needle, f"{path}[{i}]"))
    return out


class DeviceMetricsSampler:
    """Device-level counters via amd-smi (SURVEY.md §5 metrics row: the
    engine-counter story needs HBM + xGMI link throughput, which only
    the platform tool exposes). Best-effort: xGMI counters exist only on
    multi-GPU hives; repeated failures disable the sampler so a missing
    tool never stalls the 10s metrics loop."""

    def __init__(self, timeout_s: float = 30.0,
                 cmd: Optional[List[str]] = None):
        self.cmd = cmd or ["amd-smi", "metric", "--json"]
        self.timeout_s = timeout_s
        self._fails = 0
        self._prev: Dict[str, float] = {}
        self._prev_ts: Optional[float] = None

    @property
    def enabled(self) -> bool:
        return self._fails < 3

    def sample(self, now: Optional[float] = None) -> Dict[str, Any]:
        if not self.enabled:
            return {}
        now = now if now is not None else time.time()
        try:
            out = subprocess.run(self.cmd, capture_output=True, text=True,
                                 timeout=self.timeout_s)
            blob = json.loads(out.stdout)
        except Exception:
            self._fails += 1
            return {}
        self._fails = 0
        xgmi = _numeric_leaves(blob, "xgmi")
        result: Dict[str, Any] = {}
        if xgmi:
            result["xgmi_counters"] = xgmi
            if self._prev_ts is not None:
                dt = max(1e-9, now - self._prev_ts)
                rates = {k: (v - self._prev.get(k, v)) / dt
                         for k, v in xgmi.items()
                         if v >= self._prev.get(k, v)}
                # only counter-like fields produce meaningful rates, but
                # publishing all deltas keeps this schema-agnostic
                result["xgmi_per_s"] = rates
            self._prev = xgmi
            self._prev_ts = now
        return result


class LatencyWindow:
    """Fixed-size reservoir of recent request latencies for p50/p99."""

    def __init__(self, cap: int = 2048):
        self.cap = cap
        self._vals: List[float] = []
        self._lock = threading.Lock()

    def add(self, v: float) -> None:
        with self._lock:
            self._vals.append(v)
            if len(self._vals) > self.cap:
                self._vals = self._vals[-self.cap:]

    def percentiles(self, ps=(50, 99)) -> Dict[str, float]:
        with self._lock:
            vals = sorted(self._vals)
        if not vals:
            return {f"p{p}": 0.0 for p in ps}
        out = {}
        for p in ps:
            i = min(len(vals) - 1, max(0, int(round(p / 100.0 * (len(vals) - 1)))))
            out[f"p{p}"] = vals[i]
        return out


class MetricsCollector:
    def __init__(self, store: Store, manager: Manager,
                 sample_interval_s: float = SAMPLE_INTERVAL_S,
                 history_retention_s: float = HISTORY_RETENTION_S,
                 device_sampler: Optional[DeviceMetricsSampler] = None):
        self.store = store
        self.manager = manager
        self.sample_interval_s = sample_interval_s
        self.history_retention_s = history_retention_s
        self.device_sampler = device_sampler
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._latency: Dict[str, LatencyWindow] = {}
        self._prev: Dict[str, Dict[str, Any]] = {}
        self._lock = threading.Lock()

    # ---------- request-path hooks (called by the dispatcher) ----------

    def observe_request(self, agent_id: str, e2e_s: float, ttft_s: Optional[float] = None,
                        tokens: int = 0) -> None:
        with self._lock:
            win = self._latency.setdefault(agent_id, LatencyWindow())
        win.add(e2e_s)
        self.store.hset(f"agent:{agent_id}:metrics", "last_e2e_s", e2e_s)
        if ttft_s is not None:
            self.store.hset(f"agent:{agent_id}:metrics", "last_ttft_s", ttft_s)

    # ---------- lifecycle ----------

    def start(self) -> None:
        if self._thread is None:
            self._stop.clear()
            self._thread = threading.Thread(target=self._run, name="metrics-collector", daemon=True)
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None

    def _run(self) -> None:
        while not self._stop.wait(self.sample_interval_s):
            try:
                self.sample_all()
            except Exception:
                traceback.print_exc()

    # ---------- sampling ----------

    def sample_all(self, now: Optional[float] = None) -> None:
        now = now if now is not None else time.time()
        try:
            engine_stats = self.manager.engine.stats()
        except Exception:
            engine_stats = {}
        per_agent = engine_stats.get("agents", {})
        for agent in self.manager.list():
            if agent.status != RUNNING:
                continue
            self.sample_agent(agent.id, per_agent.get(agent.id, {}), now=now)
        # device-level sample: HBM from the engine, xGMI from amd-smi
        dev: Dict[str, Any] = {"ts": now}
        for k in ("hbm_total_bytes", "hbm_free_bytes", "hbm_torch_allocated"):
            if k in engine_stats:
                dev[k] = engine_stats[k]
        if self.device_sampler is not None:
            dev.update(self.device_sampler.sample(now))
        if len(dev) > 1:
            self.store.set("metrics:current:device", dev, ttl=CURRENT_TTL_S)
            self.store.zadd("metrics:history:device", now, json.dumps(dev))
            self.store.zremrangebyscore("metrics:history:device", 0,
                                        now - self.history_retention_s)

    def sample_agent(self, agent_id: str, eng: Dict[str, Any],
                     now: Optional[float] = None) -> Dict[str, Any]:
        now = now if now is not None else time.time()
        prev = self._prev.get(agent_id, {})
        dt = max(1e-9, now - prev.get("ts", now - self.sample_interval_s))
        tokens = float(eng.get("tokens", 0))
        requests = float(eng.get("requests", 0))
        with self._lock:
            win = self._latency.get(agent_id)
        pct = win.percentiles() if win else {"p50": 0.0, "p99": 0.0}
        sample = {
            "ts": now,
            "tokens_total": tokens,
            "requests_total": requests,
            "tokens_per_s": max(0.0, (tokens - float(prev.get("tokens", tokens))) / dt),
            "req_per_s": max(0.0, (requests - float(prev.get("requests", requests))) / dt),
            "e2e_p50_s": pct["p50"],
            "e2e_p99_s": pct["p99"],
            "kv_bytes": eng.get("kv_bytes", 0),
            "kv_pages": eng.get("kv_pages", 0),
            "batch_occupancy": eng.get("batch_occupancy", 0.0),
        }
        self._prev[agent_id] = {"ts": now, "tokens": tokens, "requests": requests}
        self.store.set(f"metrics:current:{agent_id}", sample, ttl=CURRENT_TTL_S)
        self.store.zadd(f"metrics:history:{agent_id}", now, json.dumps(sample))
        self.store.zremrangebyscore(f"metrics:history:{agent_id}", 0,
                                    now - self.history_retention_s)
        return sample

    # ---------- queries (collector.go:157-200 shape) ----------

    def get_metrics(self, agent_id: str) -> Optional[Dict[str, Any]]:
        return self.store.get(f"metrics:current:{agent_id}")

    def get_metrics_history(self, agent_id: str, duration_s: float = HISTORY_RETENTION_S,
                            now: Optional[float] = None) -> List[Dict[str, Any]]:
        now = now if now is not None else time.time()
        duration_s = min(duration_s, HISTORY_RETENTION_S)  # 24h cap, server.go:793-796
        rows = self.store.zrangebyscore(f"metrics:history:{agent_id}", now - duration_s, now)
        return [json.loads(m) for m, _ in rows]
