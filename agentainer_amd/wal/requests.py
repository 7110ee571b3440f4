"""Request write-ahead log — at-least-once delivery with crash replay.

Rebuilds the reference's `internal/requests/requests.go` contract:

  * Request record: uuid id, agent id, method/path/headers/body, status
    pending/completed/failed, retry_count, max_retries=3 (requests.go:26-41,
    :95). The reference's vestigial "processing" status is dropped on
    purpose (SURVEY.md §2 Request-WAL row: never set).
  * store_request: persist the record with a 24h TTL and push its id onto
    the agent's pending queue — the write-ahead step of the proxy hot path
    (requests.go:63-117, server.go:504-522).
  * store_response: mark completed, remove from pending, push to completed
    (requests.go:119-194). Stored exactly once — the reference's
    double-store on replay (replay_worker.go:158 + server.go:588-594) is a
    quirk we fix.
  * mark_failed: retry_count++; stays pending below max_retries, else moves
    to the failed dead-letter queue (requests.go:227-275).

An index set `requests:agents_pending` replaces the reference's O(N)
blocking `KEYS agent:*:requests:pending` scan (replay_worker.go:60,
SURVEY.md §7.4). Durability: every mutation ends with store.flush() (fsync)
so a SIGKILL never loses an admitted request.
"""

from __future__ import annotations

import time
import uuid
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List, Optional

from ..store import Store

PENDING = "pending"
COMPLETED = "completed"
FAILED = "failed"

DEFAULT_TTL_S = 24 * 3600.0   # requests.go:106
DEFAULT_MAX_RETRIES = 3       # requests.go:95

_PENDING_AGENTS = "requests:agents_pending"


@dataclass
class Request:
    id: str
    agent_id: str
    method: str
    path: str
    headers: Dict[str, str] = field(default_factory=dict)
    body: Any = None
    status: str = PENDING
    retry_count: int = 0
    max_retries: int = DEFAULT_MAX_RETRIES
    created_at: float = 0.0
    processed_at: Optional[float] = None
    response: Any = None
    error: Optional[str] = None

    def to_dict(self) -> Dict[str, Any]:
        return asdict(self)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Request":
        known = set(Request.__dataclass_fields__)
        return Request(**{k: v for k, v in d.items() if k in known})


class RequestManager:
    def __init__(self, store: Store, ttl_s: float = DEFAULT_TTL_S,
                 max_retries: int = DEFAULT_MAX_RETRIES):
        self.store = store
        self.ttl_s = ttl_s
        self.max_retries = max_retries

    # ---------- keys ----------

    @staticmethod
    def _rec(agent_id: str, req_id: str) -> str:
        return f"agent:{agent_id}:requests:{req_id}"

    @staticmethod
    def _q(agent_id: str, which: str) -> str:
        return f"agent:{agent_id}:requests:{which}"

    # ---------- WAL ops ----------

    def store_request(self, agent_id: str, method: str, path: str,
                      headers: Optional[Dict[str, str]] = None, body: Any = None,
                      req_id: Optional[str] = None) -> Request:
        if req_id is not None and self.store.get(self._rec(agent_id, req_id)):
            # a client-supplied id (X-Agentainer-Request-ID) must never
            # clobber an existing WAL record — fall back to a fresh uuid
            req_id = None
        req = Request(
            id=req_id or str(uuid.uuid4()), agent_id=agent_id, method=method,
            path=path, headers=dict(headers or {}), body=body,
            max_retries=self.max_retries, created_at=time.time(),
        )
        self.store.set(self._rec(agent_id, req.id), req.to_dict(), ttl=self.ttl_s)
        self.store.rpush(self._q(agent_id, "pending"), req.id)
        self.store.sadd(_PENDING_AGENTS, agent_id)
        self.store.flush()  # the WAL commit point
        return req

    def get(self, agent_id: str, req_id: str) -> Optional[Request]:
        d = self.store.get(self._rec(agent_id, req_id))
        return Request.from_dict(d) if d is not None else None

    def _put(self, req: Request) -> None:
        ttl = self.store.ttl(self._rec(req.agent_id, req.id))
        self.store.set(self._rec(req.agent_id, req.id), req.to_dict(),
                       ttl=ttl if ttl is not None else self.ttl_s)

    def store_response(self, agent_id: str, req_id: str, response: Any) -> Optional[Request]:
        req = self.get(agent_id, req_id)
        if req is None:
            return None
        if req.status != PENDING:
            # idempotent under at-least-once delivery: a duplicate success
            # (client retry, replay race) must not re-append the completed
            # queue or resurrect a dead-lettered record (the reference
            # double-stored here — replay_worker.go:158)
            return req
        req.status = COMPLETED
        req.processed_at = time.time()
        req.response = response
        req.error = None
        self._put(req)
        self.store.lrem(self._q(agent_id, "pending"), req_id)
        self.store.rpush(self._q(agent_id, "completed"), req_id)
        self._maybe_clear_pending_index(agent_id)
        self.store.flush()
        return req

    def mark_failed(self, agent_id: str, req_id: str, error: str) -> Optional[Request]:
        req = self.get(agent_id, req_id)
        if req is None:
            return None
        if req.status != PENDING:
            return req  # completed/dead-lettered records never regress
        req.retry_count += 1
        req.error = error
        if req.retry_count >= req.max_retries:
            # dead-letter (requests.go:227-275)
            req.status = FAILED
            req.processed_at = time.time()
            self.store.lrem(self._q(agent_id, "pending"), req_id)
            self.store.rpush(self._q(agent_id, "failed"), req_id)
        self._put(req)
        self._maybe_clear_pending_index(agent_id)
        self.store.flush()
        return req

    def _maybe_clear_pending_index(self, agent_id: str) -> None:
        if self.store.llen(self._q(agent_id, "pending")) == 0:
            self.store.srem(_PENDING_AGENTS, agent_id)

    # ---------- queries ----------

    def pending(self, agent_id: str) -> List[Request]:
        out = []
        for rid in self.store.lrange(self._q(agent_id, "pending")):
            r = self.get(agent_id, rid)
            if r is not None:
                out.append(r)
        return out

    def by_queue(self, agent_id: str, which: str) -> List[Request]:
        out = []
        for rid in self.store.lrange(self._q(agent_id, which)):
            r = self.get(agent_id, rid)
            if r is not None:
                out.append(r)
        return out

    def all_requests(self, agent_id: str) -> Dict[str, List[Request]]:
        return {
            "pending": self.pending(agent_id),
            "completed": self.by_queue(agent_id, "completed"),
            "failed": self.by_queue(agent_id, "failed"),
        }

    def agents_with_pending(self) -> List[str]:
        """Index-set lookup (fixes the reference's KEYS scan)."""
        out = []
        for aid in self.store.smembers(_PENDING_AGENTS):
            if self.store.llen(self._q(aid, "pending")) > 0:
                out.append(aid)
            else:
                self.store.srem(_PENDING_AGENTS, aid)
        return out

    def purge_agent(self, agent_id: str) -> None:
        """Delete queues + request records for a removed agent (agent.go:337-365)."""
        for which in ("pending", "completed", "failed"):
            self.store.delete(self._q(agent_id, which))
        for key in self.store.keys(f"agent:{agent_id}:requests:*"):
            self.store.delete(key)
        self.store.srem(_PENDING_AGENTS, agent_id)
        self.store.flush()
