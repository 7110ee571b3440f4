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
        kv_dir = os.path.join(self.backup_dir, backup["id"] + ".kv")
        engine = getattr(self.manager, "engine", None)
        for a in agents:
            entry = {
                "config": a.to_dict(),
                "conversations": self.store.lrange(f"agent:{a.id}:conversations"),
                "metrics": self.store.hgetall(f"agent:{a.id}:metrics"),
            }
            # conversation KV snapshot (live agents offload copy-on-read):
            # restore becomes KV-exact instead of replay-regenerated
            if engine is not None and hasattr(engine, "export_kv"):
                try:
                    ckpt = engine.export_kv(a.id)
                except Exception:  # noqa: BLE001 — backup stays best-effort
                    ckpt = None
                if ckpt is not None:
                    import torch
                    os.makedirs(kv_dir, exist_ok=True)
                    torch.save({"length": ckpt.length, "n_pages": ckpt.n_pages,
                                "data": ckpt.data.cpu()},
                               os.path.join(kv_dir, a.id + ".pt"))
                    entry["kv"] = a.id + ".pt"
            backup["agents"].append(entry)
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
            kv_name = entry.get("kv")
            engine = getattr(self.manager, "engine", None)
            if kv_name and engine is not None and hasattr(engine, "import_kv"):
                p = os.path.join(self.backup_dir, backup_id + ".kv", kv_name)
                if os.path.exists(p):
                    import torch

                    from ..engine.kvcache import KVCheckpoint
                    d = torch.load(p, map_location="cpu", weights_only=True)
                    engine.import_kv(agent.id, KVCheckpoint(
                        length=d["length"], n_pages=d["n_pages"],
                        data=d["data"]))
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
        """Bundle a backup (+ any KV checkpoints) into one tar.gz
        (manager.go:396-456)."""
        self.load(backup_id)  # existence check
        with tarfile.open(out_path, "w:gz") as tar:
            tar.add(self._path(backup_id), arcname=backup_id + ".json")
            kv_dir = os.path.join(self.backup_dir, backup_id + ".kv")
            if os.path.isdir(kv_dir):
                for fn in sorted(os.listdir(kv_dir)):
                    tar.add(os.path.join(kv_dir, fn),
                            arcname=f"{backup_id}.kv/{fn}")
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
            # KV checkpoint payloads: "<backup_id>.kv/<agent_id>.pt"
            for member in tar.getmembers():
                parts = member.name.split("/")
                if (len(parts) != 2 or not parts[0].endswith(".kv")
                        or not parts[1].endswith(".pt")
                        or ".." in member.name):
                    continue
                f = tar.extractfile(member)
                if f is None:
                    continue
                d = os.path.join(self.backup_dir, parts[0])
                os.makedirs(d, exist_ok=True)
                with open(os.path.join(d, parts[1]), "wb") as out:
                    out.write(f.read())
        return ids
