"""Embedded persistent KV store — the runtime's durable state plane.

The reference uses an external Redis 7 for *everything*: agent registry,
request WAL, health store, metrics TSDB, log/audit store and pub/sub bus
(reference internal/storage/storage.go:21-76). An external
store process is the wrong shape for an in-process MI355X inference engine —
and this image has no Redis — so the rebuild embeds a crash-safe store:

  * in-memory structures (strings, lists, sorted sets, hashes, sets) with
    per-key TTL, mirroring the Redis subset the reference uses;
  * durability via an append-only file (AOF) of JSON ops + snapshot
    compaction, so a SIGKILL'd server recovers registry + WAL state on
    restart (the crash-replay contract of SURVEY.md §3.5);
  * in-process pub/sub with glob *pattern* subscriptions — deliberately
    fixing the reference's broken plain-Subscribe-on-glob bug
    (reference internal/health/monitor.go:301).

Thread-safe. `sync` policy: "always" fsyncs every append (WAL-critical),
"interval" relies on OS flush + explicit flush(), "never" keeps pure memory
(unit tests).
"""

from __future__ import annotations

import fnmatch
import json
import os
import threading
import time
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional, Tuple

_TOMBSTONE = object()


@dataclass
class _Entry:
    kind: str  # str|list|zset|hash|set
    value: Any
    expires_at: Optional[float] = None  # unix seconds


class Store:
    """Durable dict-of-structures with TTL, AOF persistence and pub/sub."""

    SNAPSHOT_OPS = 50_000  # compact AOF after this many appended ops

    def __init__(self, path: Optional[str] = None, sync: str = "interval"):
        self._lock = threading.RLock()
        self._data: Dict[str, _Entry] = {}
        self._subs: List[Tuple[str, Callable[[str, str], None]]] = []
        self._path = path
        self._sync = sync if path else "never"
        self._aof = None
        self._aof_ops = 0
        if path:
            os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
            self._load()
            self._aof = open(self._aof_path(), "a", encoding="utf-8")

    # ---------- persistence ----------

    def _aof_path(self) -> str:
        return self._path + ".aof"

    def _snap_path(self) -> str:
        return self._path + ".snap"

    def _load(self) -> None:
        snap = self._snap_path()
        if os.path.exists(snap):
            try:
                with open(snap, "r", encoding="utf-8") as f:
                    raw = json.load(f)
                for k, (kind, val, exp) in raw.items():
                    if kind == "zset":
                        val = {m: float(s) for m, s in val.items()}
                    elif kind == "set":
                        val = set(val)
                    self._data[k] = _Entry(kind, val, exp)
            except (json.JSONDecodeError, OSError):
                self._data = {}
        aof = self._aof_path()
        if os.path.exists(aof):
            with open(aof, "r", encoding="utf-8") as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        op = json.loads(line)
                    except json.JSONDecodeError:
                        continue  # torn tail write from a crash — ignore
                    try:
                        self._apply(op, replay=True)
                    except Exception:
                        continue
        self._sweep_expired()

    def _append(self, op: List[Any]) -> None:
        if self._aof is None:
            return
        self._aof.write(json.dumps(op, separators=(",", ":")) + "\n")
        if self._sync == "always":
            self._aof.flush()
            os.fsync(self._aof.fileno())
        self._aof_ops += 1
        if self._aof_ops >= self.SNAPSHOT_OPS:
            self._compact_locked()

    def flush(self) -> None:
        """WAL commit point. Always pushes buffered ops to the OS (survives
        a SIGKILL of this process); fsyncs to media only under sync=always
        (power-loss durability — same split as Redis AOF everysec/always)."""
        with self._lock:
            if self._aof is not None:
                self._aof.flush()
                if self._sync == "always":
                    os.fsync(self._aof.fileno())

    def _compact_locked(self) -> None:
        if self._path is None:
            return
        self._sweep_expired()
        raw = {}
        for k, e in self._data.items():
            val = e.value
            if e.kind == "set":
                val = sorted(val)
            raw[k] = (e.kind, val, e.expires_at)
        tmp = self._snap_path() + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump(raw, f)
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, self._snap_path())
        if self._aof is not None:
            self._aof.close()
        with open(self._aof_path(), "w", encoding="utf-8") as f:
            f.flush()
            os.fsync(f.fileno())
        self._aof = open(self._aof_path(), "a", encoding="utf-8")
        self._aof_ops = 0

    def compact(self) -> None:
        with self._lock:
            self._compact_locked()

    def close(self) -> None:
        with self._lock:
            if self._aof is not None:
                self._aof.flush()
                os.fsync(self._aof.fileno())
                self._aof.close()
                self._aof = None

    # ---------- expiry ----------

    def _alive(self, key: str) -> Optional[_Entry]:
        e = self._data.get(key)
        if e is None:
            return None
        if e.expires_at is not None and e.expires_at <= time.time():
            del self._data[key]
            return None
        return e

    def _sweep_expired(self) -> None:
        now = time.time()
        dead = [k for k, e in self._data.items() if e.expires_at is not None and e.expires_at <= now]
        for k in dead:
            del self._data[k]

    # ---------- op application (shared by live calls and AOF replay) ----------

    def _apply(self, op: List[Any], replay: bool = False):
        name = op[0]
        return getattr(self, "_do_" + name)(*op[1:])

    def _mutate(self, op: List[Any]):
        with self._lock:
            out = self._apply(op)
            self._append(op)
            return out

    # ---------- strings ----------

    def _do_set(self, key: str, value: Any, expires_at: Optional[float] = None):
        self._data[key] = _Entry("str", value, expires_at)

    def set(self, key: str, value: Any, ttl: Optional[float] = None) -> None:
        exp = time.time() + ttl if ttl else None
        self._mutate(["set", key, value, exp])

    def get(self, key: str, default: Any = None) -> Any:
        with self._lock:
            e = self._alive(key)
            return e.value if e is not None and e.kind == "str" else default

    def _do_del(self, key: str):
        self._data.pop(key, None)

    def delete(self, key: str) -> None:
        self._mutate(["del", key])

    def exists(self, key: str) -> bool:
        with self._lock:
            return self._alive(key) is not None

    def keys(self, pattern: str = "*") -> List[str]:
        with self._lock:
            self._sweep_expired()
            return [k for k in self._data.keys() if fnmatch.fnmatchcase(k, pattern)]

    def _do_expire(self, key: str, expires_at: Optional[float]):
        e = self._data.get(key)
        if e is not None:
            e.expires_at = expires_at

    def expire(self, key: str, ttl: Optional[float]) -> None:
        exp = time.time() + ttl if ttl is not None else None
        self._mutate(["expire", key, exp])

    def ttl(self, key: str) -> Optional[float]:
        with self._lock:
            e = self._alive(key)
            if e is None or e.expires_at is None:
                return None
            return max(0.0, e.expires_at - time.time())

    def _do_incr(self, key: str, by: int):
        e = self._alive(key)
        cur = int(e.value) if e is not None else 0
        cur += by
        self._data[key] = _Entry("str", cur, e.expires_at if e else None)
        return cur

    def incr(self, key: str, by: int = 1) -> int:
        return self._mutate(["incr", key, by])

    # ---------- lists ----------

    def _list(self, key: str, create: bool = False) -> Optional[List[Any]]:
        e = self._alive(key)
        if e is None:
            if not create:
                return None
            e = _Entry("list", [])
            self._data[key] = e
        return e.value

    def _do_rpush(self, key: str, *vals):
        lst = self._list(key, create=True)
        lst.extend(vals)
        return len(lst)

    def rpush(self, key: str, *vals: Any) -> int:
        return self._mutate(["rpush", key, *vals])

    def _do_lpush(self, key: str, *vals):
        lst = self._list(key, create=True)
        for v in vals:
            lst.insert(0, v)
        return len(lst)

    def lpush(self, key: str, *vals: Any) -> int:
        return self._mutate(["lpush", key, *vals])

    def _drop_if_empty(self, key: str) -> None:
        """Redis semantics: a collection key vanishes when its last
        element is removed (an empty list/zset/hash/set never lingers in
        keys() scans)."""
        e = self._data.get(key)
        if e is not None and e.kind != "str" and not e.value:
            del self._data[key]

    def _do_lpop(self, key: str):
        lst = self._list(key)
        if not lst:
            return None
        v = lst.pop(0)
        self._drop_if_empty(key)
        return v

    def lpop(self, key: str) -> Any:
        return self._mutate(["lpop", key])

    def lrange(self, key: str, start: int = 0, stop: int = -1) -> List[Any]:
        with self._lock:
            lst = self._list(key)
            if lst is None:
                return []
            if stop == -1:
                return list(lst[start:])
            return list(lst[start : stop + 1])

    def _do_lrem(self, key: str, value: Any):
        lst = self._list(key)
        if lst is None:
            return 0
        n = lst.count(value)
        self._data[key].value = [v for v in lst if v != value]
        self._drop_if_empty(key)
        return n

    def lrem(self, key: str, value: Any) -> int:
        return self._mutate(["lrem", key, value])

    def _do_ltrim(self, key: str, start: int, stop: int):
        lst = self._list(key)
        if lst is None:
            return
        self._data[key].value = lst[start : (None if stop == -1 else stop + 1)]
        self._drop_if_empty(key)

    def ltrim(self, key: str, start: int, stop: int) -> None:
        self._mutate(["ltrim", key, start, stop])

    def llen(self, key: str) -> int:
        with self._lock:
            lst = self._list(key)
            return len(lst) if lst else 0

    # ---------- sorted sets (member -> score) ----------

    def _zset(self, key: str, create: bool = False) -> Optional[Dict[str, float]]:
        e = self._alive(key)
        if e is None:
            if not create:
                return None
            e = _Entry("zset", {})
            self._data[key] = e
        return e.value

    def _do_zadd(self, key: str, score: float, member: str):
        z = self._zset(key, create=True)
        z[member] = float(score)

    def zadd(self, key: str, score: float, member: str) -> None:
        self._mutate(["zadd", key, score, member])

    def zrangebyscore(self, key: str, lo: float, hi: float) -> List[Tuple[str, float]]:
        with self._lock:
            z = self._zset(key)
            if z is None:
                return []
            out = [(m, s) for m, s in z.items() if lo <= s <= hi]
            out.sort(key=lambda t: (t[1], t[0]))
            return out

    def _do_zremrangebyscore(self, key: str, lo: float, hi: float):
        z = self._zset(key)
        if z is None:
            return 0
        dead = [m for m, s in z.items() if lo <= s <= hi]
        for m in dead:
            del z[m]
        self._drop_if_empty(key)
        return len(dead)

    def zremrangebyscore(self, key: str, lo: float, hi: float) -> int:
        return self._mutate(["zremrangebyscore", key, lo, hi])

    def zcard(self, key: str) -> int:
        with self._lock:
            z = self._zset(key)
            return len(z) if z else 0

    # ---------- hashes ----------

    def _hash(self, key: str, create: bool = False) -> Optional[Dict[str, Any]]:
        e = self._alive(key)
        if e is None:
            if not create:
                return None
            e = _Entry("hash", {})
            self._data[key] = e
        return e.value

    def _do_hset(self, key: str, field_: str, value: Any):
        h = self._hash(key, create=True)
        h[field_] = value

    def hset(self, key: str, field_: str, value: Any) -> None:
        self._mutate(["hset", key, field_, value])

    def hget(self, key: str, field_: str, default: Any = None) -> Any:
        with self._lock:
            h = self._hash(key)
            return h.get(field_, default) if h else default

    def hgetall(self, key: str) -> Dict[str, Any]:
        with self._lock:
            h = self._hash(key)
            return dict(h) if h else {}

    # ---------- sets ----------

    def _set(self, key: str, create: bool = False) -> Optional[set]:
        e = self._alive(key)
        if e is None:
            if not create:
                return None
            e = _Entry("set", set())
            self._data[key] = e
        return e.value

    def _do_sadd(self, key: str, *members):
        s = self._set(key, create=True)
        for m in members:
            s.add(m)

    def sadd(self, key: str, *members: str) -> None:
        self._mutate(["sadd", key, *members])

    def _do_srem(self, key: str, *members):
        s = self._set(key)
        if s is None:
            return
        for m in members:
            s.discard(m)
        self._drop_if_empty(key)

    def srem(self, key: str, *members: str) -> None:
        self._mutate(["srem", key, *members])

    def smembers(self, key: str) -> List[str]:
        with self._lock:
            s = self._set(key)
            return sorted(s) if s else []

    # ---------- pub/sub (in-process; glob patterns WORK, unlike the reference) ----------

    def subscribe(self, pattern: str, callback: Callable[[str, str], None]) -> Callable[[], None]:
        """Subscribe callback(channel, message) to a glob pattern. Returns unsubscribe fn."""
        ent = (pattern, callback)
        with self._lock:
            self._subs.append(ent)

        def _unsub():
            with self._lock:
                if ent in self._subs:
                    self._subs.remove(ent)

        return _unsub

    def publish(self, channel: str, message: str) -> int:
        with self._lock:
            targets = [cb for pat, cb in self._subs if fnmatch.fnmatchcase(channel, pat)]
        for cb in targets:
            try:
                cb(channel, message)
            except Exception:
                pass
        return len(targets)
