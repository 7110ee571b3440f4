"""Deployment YAML schema — `kind: AgentDeployment`.

Mirrors the reference's k8s-flavored multi-agent deployment file
(reference internal/config/deployment.go:14-74): apiVersion/kind/metadata +
spec.agents[], per-agent replicas (expanded to `name-N`,
deployment.go:165-174), env, resources, autoRestart, token, healthCheck and
dependencies (validated to exist, deployment.go:151-155), with env-var
expansion in the file content (deployment.go:96-97).

Model-shard fields replace container-image fields: `model` (family id or
safetensors path), `dtype`, `tp_degree`, `kv_budget` (per-agent KV-cache
byte budget, parsed like the reference's ParseMemory: 512M/2G/512Mi/2Gi,
deployment.go:290-337) and `max_context`.
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

_MEM_RE = re.compile(r"^\s*(\d+(?:\.\d+)?)\s*([KMGT]i?)?B?\s*$", re.IGNORECASE)
_MULT = {
    None: 1,
    "k": 1000, "m": 1000**2, "g": 1000**3, "t": 1000**4,
    "ki": 1024, "mi": 1024**2, "gi": 1024**3, "ti": 1024**4,
}


def parse_memory(s: Any) -> int:
    """'512M'/'2G'/'512Mi'/'2Gi' (or int bytes) -> bytes. Reference deployment.go:290-337."""
    if isinstance(s, (int, float)):
        return int(s)
    m = _MEM_RE.match(str(s))
    if not m:
        raise ValueError(f"unparseable memory quantity: {s!r}")
    qty, unit = m.groups()
    mult = _MULT.get(unit.lower() if unit else None)
    if mult is None:
        raise ValueError(f"unknown memory unit in {s!r}")
    return int(float(qty) * mult)


@dataclass
class AgentSpec:
    """One agent entry of spec.agents[] after replica expansion."""

    name: str
    model: str
    dtype: str = "bf16"
    tp_degree: int = 1
    kv_budget: int = 0           # bytes; 0 = engine default share
    max_context: int = 8192
    env: Dict[str, str] = field(default_factory=dict)
    auto_restart: bool = False
    token: Optional[str] = None
    health_check: Optional[Dict[str, Any]] = None
    system_prompt: str = ""
    sampling: Dict[str, Any] = field(default_factory=dict)  # temperature/top_p/top_k/max_tokens
    dependencies: List[str] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "name": self.name, "model": self.model, "dtype": self.dtype,
            "tp_degree": self.tp_degree, "kv_budget": self.kv_budget,
            "max_context": self.max_context, "env": self.env,
            "auto_restart": self.auto_restart, "token": self.token,
            "health_check": self.health_check, "system_prompt": self.system_prompt,
            "sampling": self.sampling, "dependencies": self.dependencies,
        }


class DeploymentError(ValueError):
    pass


def load_deployment(path: str) -> List[AgentSpec]:
    """Parse + validate an AgentDeployment file; expand replicas to name-N."""
    with open(path, "r", encoding="utf-8") as f:
        text = f.read()
    text = os.path.expandvars(text)  # ${VAR} expansion, deployment.go:96-97
    doc = yaml.safe_load(text) or {}
    if doc.get("kind") != "AgentDeployment":
        raise DeploymentError(f"kind must be AgentDeployment, got {doc.get('kind')!r}")
    spec = doc.get("spec") or {}
    agents_raw = spec.get("agents") or []
    if not agents_raw:
        raise DeploymentError("spec.agents is empty")

    names = set()
    for a in agents_raw:
        if not a.get("name"):
            raise DeploymentError("agent missing name")
        names.add(a["name"])

    out: List[AgentSpec] = []
    for a in agents_raw:
        deps = list(a.get("dependencies") or [])
        for d in deps:
            if d not in names:
                raise DeploymentError(f"agent {a['name']}: unknown dependency {d!r}")
        resources = a.get("resources") or {}
        kv_budget = parse_memory(resources.get("kv_budget", 0)) if resources.get("kv_budget") else 0
        base = dict(
            model=a.get("model", ""),
            dtype=a.get("dtype", "bf16"),
            tp_degree=int(a.get("tp_degree", 1)),
            kv_budget=kv_budget,
            max_context=int(resources.get("max_context", a.get("max_context", 8192))),
            env=dict(a.get("env") or {}),
            auto_restart=bool(a.get("autoRestart", a.get("auto_restart", False))),
            token=a.get("token"),
            health_check=a.get("healthCheck", a.get("health_check")),
            system_prompt=a.get("systemPrompt", a.get("system_prompt", "")),
            sampling=dict(a.get("sampling") or {}),
            dependencies=deps,
        )
        if not base["model"]:
            raise DeploymentError(f"agent {a['name']}: model is required")
        replicas = int(a.get("replicas", 1))
        if replicas <= 1:
            out.append(AgentSpec(name=a["name"], **base))
        else:
            # replica fan-out, reference deployment.go:165-174
            for i in range(1, replicas + 1):
                out.append(AgentSpec(name=f"{a['name']}-{i}", **base))
    return out
