"""Server configuration: defaults < YAML file < AGENTAINER_* env overrides.

Same precedence contract as the reference's viper setup
(reference internal/config/config.go:49-107): defaults, then
`./config.yaml` / `~/.agentainer/config.yaml` / `/etc/agentainer/config.yaml`,
then environment variables `AGENTAINER_<SECTION>_<KEY>`.

The docker/redis sections of the reference become engine/store sections:
model paths, dtype, KV-page budget and tensor-parallel degree replace
image/socket fields (SURVEY.md §5 "Config / flag system").
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict, Optional

import yaml

DEFAULT_TOKEN = "agentainer-default-token"  # PoC shared token (reference config.go:~75)

DEFAULTS: Dict[str, Dict[str, Any]] = {
    "server": {
        "host": "127.0.0.1",
        "port": 8081,
        # worker-thread cap for blocking request handlers; each in-flight
        # generation parks one thread, so size for max concurrent agents
        "threadpool_size": 1024,
    },
    "store": {
        # state root: registry AOF, WAL, logs, backups
        "path": "~/.agentainer_amd",
        "sync": "interval",  # always|interval|never
    },
    "engine": {
        "device": "auto",          # auto|cuda|cpu
        "dtype": "bf16",
        "kv_page_size": 16,        # tokens per KV page
        "kv_pool_gb": 0.0,         # 0 => auto-size from free HBM
        "kv_dtype": "bf16",        # fp8 = e4m3 pages (2x agents per GPU)
        "max_batch_tokens": 8192,  # prefill token budget per step
        "max_decode_batch": 256,
        "max_seqs": 2048,          # sequence-slot budget (device mirrors)
        "graph_capture": True,     # hipGraph-capture the decode step
        "async_decode": True,      # one-step-lag speculative resolution
        "prefill_stream": True,    # prefill on a side HIP stream
        "prefix_sharing": True,    # COW system-prompt KV across agents
        "moe_ep_mode": "auto",     # auto|a2a|dense (Mixtral EP dispatch)
        "moe_a2a_min_tokens": 128,
        "dense_quant": "",         # fp8|mxfp4 decode projections (turbo)
        "expert_fp8": False,       # e4m3 Mixtral expert GEMMs
        "expert_fp4": False,       # MXFP4 Mixtral expert GEMMs
        "tp_degree": 1,
    },
    "security": {
        "api_token": DEFAULT_TOKEN,
    },
    "features": {
        # gates the entire WAL subsystem, like the reference's
        # features.request_persistence (config.go:45-47, main.go:334-341)
        "request_persistence": True,
        "replay_interval_s": 5.0,   # replay_worker.go:37
        "request_ttl_s": 24 * 3600,  # requests.go:106
        "max_retries": 3,            # requests.go:95
    },
    "health": {
        "interval_s": 30.0,  # monitor.go:117-129 defaults
        "timeout_s": 5.0,
        "retries": 3,
        # amd-smi device-level fault probe (ECC / device presence) on GPU
        "gpu_fault_probe": True,
        "gpu_fault_interval_s": 30.0,
    },
    "metrics": {
        "sample_interval_s": 10.0,  # collector.go:205
        "history_retention_s": 24 * 3600.0,  # collector.go:319-321
    },
    "logging": {
        "level": "info",
        "retention_s": 7 * 24 * 3600.0,  # logger.go:97
        "rotate_bytes": 100 * 1024 * 1024,  # logger.go:96
    },
}

_SEARCH_PATHS = [
    "./config.yaml",
    "~/.agentainer_amd/config.yaml",
    "/etc/agentainer_amd/config.yaml",
]


def _deep_merge(base: Dict[str, Any], over: Dict[str, Any]) -> Dict[str, Any]:
    out = copy.deepcopy(base)
    for k, v in (over or {}).items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def _coerce(val: str, like: Any) -> Any:
    if isinstance(like, bool):
        return val.lower() in ("1", "true", "yes", "on")
    if isinstance(like, int) and not isinstance(like, bool):
        return int(val)
    if isinstance(like, float):
        return float(val)
    return val


class Config:
    """Nested dict with attribute-ish access: cfg.get('engine','dtype')."""

    def __init__(self, data: Dict[str, Any]):
        self.data = data

    def get(self, section: str, key: Optional[str] = None, default: Any = None) -> Any:
        sec = self.data.get(section, {})
        if key is None:
            return sec
        return sec.get(key, default)

    def __getitem__(self, section: str) -> Dict[str, Any]:
        return self.data[section]

    @property
    def state_root(self) -> str:
        return os.path.expanduser(self.get("store", "path"))


def load_config(path: Optional[str] = None, env: Optional[Dict[str, str]] = None) -> Config:
    """defaults < yaml < env(AGENTAINER_SECTION_KEY)."""
    data = copy.deepcopy(DEFAULTS)
    candidates = [path] if path else [os.path.expanduser(p) for p in _SEARCH_PATHS]
    for p in candidates:
        if p and os.path.exists(p):
            with open(p, "r", encoding="utf-8") as f:
                loaded = yaml.safe_load(f) or {}
            data = _deep_merge(data, loaded)
            break
    env = env if env is not None else dict(os.environ)
    for name, val in env.items():
        if not name.startswith("AGENTAINER_"):
            continue
        parts = name[len("AGENTAINER_"):].lower().split("_", 1)
        if len(parts) != 2:
            continue
        section, key = parts
        if section in data and key in data[section]:
            data[section][key] = _coerce(val, data[section][key])
    return Config(data)
