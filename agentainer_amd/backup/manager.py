"""Backup manager — snapshot/restore agent configs + conversation state.

Rebuilds `internal/backup/manager.go`:

  * Backup record {id=backup-{unix}, name, description, created_at,
    agents[], version} (manager.go:21-28).
  * create: snapshots every agent's config + its conversation history and
    per-agent store keys (the analog of tarring volume dirs,
    manager.go:61-130 — our "volumes" are the agent's durable store state
    and optional offloaded KV images).
  * restore: re-deploys each agent as `{name}-restored` and reinstates its
    conversation history (manager.go:132-191).
  * list/load/delete (manager.go:193-238) and export to a single tar.gz
    (manager.go:396-456).

Unified behind the manager (and exposed over REST) rather than the
reference's CLI-direct construction (SURVEY.md §1 note).
"""

from __future__ import annotations

import json
import os
import tarfile
import time
from typing import Any, Dict, List, Optional

from ..registry import Agent, Manager
from ..store import Store

BACKUP_VERSION = "1"


class BackupError(Exception):
    pass


class BackupManager:
    def __init__(self, store: Store, manager: Manager, backup_dir: str):
        self.store = store
        self.manager = manager
        self.backup_dir = backup_dir
        os.makedirs(backup_dir, exist_ok=True)

    def _path(self, backup_id: str) -> str:
        return os.path.join(self.backup_dir, backup_id + ".json")

    # ---------- create ----------

    def create(self, name: str, description: str = "",
               agent_ids: Optional[List[str]] = None) -> Dict[str, Any]:
        agents = self.manager.list()
        if agent_ids is not None:
            wanted = set(agent_ids)
            agents = [a for a in agents if a.id in wanted]
        backup = {
            "id": f"backup-{int(time.time())}-{os.urandom(3).hex()}",
            "name": name,
            "description": description,
            "created_at": time.time(),
            "version": BACKUP_VERSION,
            "agents": [],
        }
        for a in agents:
            backup["agents"].append({
                "config": a.to_dict(),
                "conversations": self.store.lrange(f"agent:{a.id}:conversations"),
                "metrics": self.store.hgetall(f"agent:{a.id}:metrics"),
            })
        with open(self._path(backup["id"]), "w", encoding="utf-8") as f:
            json.dump(backup, f)
            f.flush()
            os.fsync(f.fileno())
        return backup

    # ---------- restore ----------

    def restore(self, backup_id: str) -> List[Agent]:
        backup = self.load(backup_id)
        restored = []
        for entry in backup["agents"]:
            cfg = entry["config"]
            agent = self.manager.deploy(
                name=f"{cfg['name']}-restored",  # manager.go:132-191 naming
                model=cfg["model"],
                dtype=cfg.get("dtype", "bf16"),
                tp_degree=cfg.get("tp_degree", 1),
                kv_budget=cfg.get("kv_budget", 0),
                max_context=cfg.get("max_context", 8192),
                env=cfg.get("env") or {},
                auto_restart=cfg.get("auto_restart", False),
                token=cfg.get("token"),
                health_check=cfg.get("health_check"),
                system_prompt=cfg.get("system_prompt", ""),
                sampling=cfg.get("sampling") or {},
            )
            for conv in entry.get("conversations", []):
                self.store.rpush(f"agent:{agent.id}:conversations", conv)
            for k, v in (entry.get("metrics") or {}).items():
                self.store.hset(f"agent:{agent.id}:metrics", k, v)
            restored.append(agent)
        return restored

    # ---------- list / load / delete / export ----------

    def list(self) -> List[Dict[str, Any]]:
        out = []
        for fn in sorted(os.listdir(self.backup_dir)):
            if fn.endswith(".json"):
                try:
                    with open(os.path.join(self.backup_dir, fn), encoding="utf-8") as f:
                        b = json.load(f)
                    out.append({"id": b["id"], "name": b["name"],
                                "description": b.get("description", ""),
                                "created_at": b["created_at"],
                                "n_agents": len(b.get("agents", []))})
                except (json.JSONDecodeError, KeyError, OSError):
                    continue
        return out

    def load(self, backup_id: str) -> Dict[str, Any]:
        path = self._path(backup_id)
        if not os.path.exists(path):
            raise BackupError(f"backup {backup_id} not found")
        with open(path, encoding="utf-8") as f:
            return json.load(f)

    def delete(self, backup_id: str) -> None:
        path = self._path(backup_id)
        if not os.path.exists(path):
            raise BackupError(f"backup {backup_id} not found")
        os.unlink(path)

    def export(self, backup_id: str, out_path: str) -> str:
        """Bundle a backup into one tar.gz (manager.go:396-456)."""
        self.load(backup_id)  # existence check
        with tarfile.open(out_path, "w:gz") as tar:
            tar.add(self._path(backup_id), arcname=backup_id + ".json")
        return out_path

    def import_(self, tar_path: str) -> List[str]:
        """Import an exported bundle; returns backup ids."""
        ids = []
        with tarfile.open(tar_path, "r:gz") as tar:
            for member in tar.getmembers():
                if not member.name.endswith(".json") or "/" in member.name:
                    continue
                f = tar.extractfile(member)
                if f is None:
                    continue
                data = json.load(f)
                if "id" not in data or "agents" not in data:
                    continue
                with open(self._path(data["id"]), "w", encoding="utf-8") as out:
                    json.dump(data, out)
                ids.append(data["id"])
        return ids
