"""Structured + audit logging — triple sink with rotation and retention.

Rebuilds `internal/logging/logger.go`:

  * Structured LogEntry / AuditEntry (user, action, resource, result, IP,
    UA) — logger.go:29-53.
  * Triple sink: JSONL files under <state>/logs/ + store sorted sets
    `logs:entries` / `audit:entries` (7d retention, logger.go:346-348) +
    console (logger.go:122-147).
  * Size-based rotation at 100 MB with 7d file cleanup (logger.go:375-452).
  * Filtered queries get_logs/get_audit_logs (logger.go:200-290).
  * Tail streaming via store pub/sub channel `logs:stream` — and unlike the
    reference, entries ARE published to it (the reference's TailLogs
    subscribed to a channel nothing wrote, SURVEY.md §2 logging row).
  * Global-logger pattern (logger.go:495-535).
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Any, Dict, List, Optional

from ..store import Store

LEVELS = ("debug", "info", "warn", "error", "fatal")
RETENTION_S = 7 * 24 * 3600.0
ROTATE_BYTES = 100 * 1024 * 1024


class Logger:
    def __init__(self, store: Optional[Store], log_dir: str,
                 level: str = "info", console: bool = False,
                 retention_s: float = RETENTION_S, rotate_bytes: int = ROTATE_BYTES):
        self.store = store
        self.log_dir = log_dir
        self.level = level
        self.console = console
        self.retention_s = retention_s
        self.rotate_bytes = rotate_bytes
        self._lock = threading.RLock()
        os.makedirs(log_dir, exist_ok=True)
        self._files: Dict[str, Any] = {}
        self._last_rotate_check = 0.0

    # ---------- files ----------

    def _path(self, name: str) -> str:
        return os.path.join(self.log_dir, name + ".log")

    def _file(self, name: str):
        f = self._files.get(name)
        if f is None:
            f = open(self._path(name), "a", encoding="utf-8")
            self._files[name] = f
        return f

    def _maybe_rotate(self, name: str) -> None:
        # hourly check in the reference (logger.go:375-452); here per-write
        # with a 60s throttle.
        now = time.time()
        if now - self._last_rotate_check < 60:
            return
        self._last_rotate_check = now
        path = self._path(name)
        try:
            if os.path.getsize(path) >= self.rotate_bytes:
                f = self._files.pop(name, None)
                if f:
                    f.close()
                os.replace(path, f"{path}.{int(now)}")
        except OSError:
            pass
        # 7d file cleanup
        try:
            for fn in os.listdir(self.log_dir):
                full = os.path.join(self.log_dir, fn)
                if "." in fn and fn.rsplit(".", 1)[-1].isdigit():
                    if now - os.path.getmtime(full) > self.retention_s:
                        os.unlink(full)
        except OSError:
            pass

    def _write(self, name: str, entry: Dict[str, Any], zkey: str) -> None:
        line = json.dumps(entry, separators=(",", ":"), default=str)
        with self._lock:
            f = self._file(name)
            f.write(line + "\n")
            f.flush()
            self._maybe_rotate(name)
        if self.store is not None:
            now = entry.get("ts", time.time())
            self.store.zadd(zkey, now, line)
            self.store.zremrangebyscore(zkey, 0, now - self.retention_s)
            self.store.publish("logs:stream", line)
        if self.console:
            print(line)

    # ---------- structured log ----------

    def log(self, level: str, message: str, component: str = "", agent_id: str = "",
            **fields: Any) -> Dict[str, Any]:
        if LEVELS.index(level) < LEVELS.index(self.level):
            return {}
        entry = {"ts": time.time(), "level": level, "component": component,
                 "agent_id": agent_id, "message": message}
        if fields:
            entry["fields"] = fields
        self._write("agentainer", entry, "logs:entries")
        if agent_id and self.store is not None:
            self.store.rpush(f"agent:{agent_id}:log", entry)
            n = self.store.llen(f"agent:{agent_id}:log")
            if n > 1000:
                self.store.ltrim(f"agent:{agent_id}:log", n - 1000, -1)
        return entry

    def debug(self, msg: str, **kw): return self.log("debug", msg, **kw)
    def info(self, msg: str, **kw): return self.log("info", msg, **kw)
    def warn(self, msg: str, **kw): return self.log("warn", msg, **kw)
    def error(self, msg: str, **kw): return self.log("error", msg, **kw)

    # ---------- audit ----------

    def audit(self, user: str, action: str, resource: str, result: str,
              ip: str = "", user_agent: str = "", details: Any = None) -> Dict[str, Any]:
        entry = {"ts": time.time(), "user": user, "action": action,
                 "resource": resource, "result": result, "ip": ip,
                 "user_agent": user_agent}
        if details is not None:
            entry["details"] = details
        self._write("audit", entry, "audit:entries")
        return entry

    # ---------- queries ----------

    def _query(self, zkey: str, since_s: float, filters: Dict[str, Any],
               limit: int) -> List[Dict[str, Any]]:
        if self.store is None:
            return []
        now = time.time()
        rows = self.store.zrangebyscore(zkey, now - since_s, now + 1)
        out = []
        for line, _ in rows:
            try:
                e = json.loads(line)
            except json.JSONDecodeError:
                continue
            if all(e.get(k) == v for k, v in filters.items() if v):
                out.append(e)
        return out[-limit:]

    def get_logs(self, since_s: float = RETENTION_S, level: str = "",
                 component: str = "", agent_id: str = "",
                 limit: int = 500) -> List[Dict[str, Any]]:
        return self._query("logs:entries", since_s,
                           {"level": level, "component": component, "agent_id": agent_id},
                           limit)

    def get_audit_logs(self, since_s: float = RETENTION_S, user: str = "",
                       action: str = "", resource: str = "",
                       limit: int = 500) -> List[Dict[str, Any]]:
        return self._query("audit:entries", since_s,
                           {"user": user, "action": action, "resource": resource}, limit)

    def tail(self, callback) -> Any:
        """Subscribe callback(line) to live log stream; returns unsubscribe fn."""
        return self.store.subscribe("logs:stream", lambda ch, msg: callback(msg))

    def close(self) -> None:
        with self._lock:
            for f in self._files.values():
                f.close()
            self._files.clear()


_global: Optional[Logger] = None
_global_lock = threading.Lock()


def set_global_logger(lg: Logger) -> None:
    global _global
    with _global_lock:
        _global = lg


def get_logger() -> Logger:
    global _global
    with _global_lock:
        if _global is None:
            _global = Logger(None, os.path.expanduser("~/.agentainer_amd/logs"))
        return _global
