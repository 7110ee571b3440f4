"""Config precedence + deployment YAML schema."""

import pytest

from agentainer_amd.config import (
    DeploymentError, load_config, load_deployment, parse_memory,
)


def test_defaults(tmp_path):
    cfg = load_config(path=str(tmp_path / "none.yaml"), env={})
    assert cfg.get("server", "port") == 8081
    assert cfg.get("features", "request_persistence") is True
    assert cfg.get("features", "max_retries") == 3
    assert cfg.get("health", "interval_s") == 30.0


def test_yaml_overrides(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("server:\n  port: 9000\nengine:\n  dtype: fp8\n")
    cfg = load_config(path=str(p), env={})
    assert cfg.get("server", "port") == 9000
    assert cfg.get("engine", "dtype") == "fp8"
    assert cfg.get("server", "host") == "127.0.0.1"  # default survives


def test_env_overrides_yaml(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("server:\n  port: 9000\n")
    cfg = load_config(path=str(p), env={
        "AGENTAINER_SERVER_PORT": "9100",
        "AGENTAINER_FEATURES_REQUEST_PERSISTENCE": "false",
    })
    assert cfg.get("server", "port") == 9100
    assert cfg.get("features", "request_persistence") is False


def test_parse_memory():
    assert parse_memory("512M") == 512 * 1000**2
    assert parse_memory("2G") == 2 * 1000**3
    assert parse_memory("512Mi") == 512 * 1024**2
    assert parse_memory("2Gi") == 2 * 1024**3
    assert parse_memory(1024) == 1024
    with pytest.raises(ValueError):
        parse_memory("12 parsecs")


DEPLOY_YAML = """
apiVersion: v1
kind: AgentDeployment
metadata:
  name: fleet
spec:
  agents:
    - name: worker
      model: echo
      replicas: 3
      resources:
        kv_budget: 1Gi
        max_context: 4096
      autoRestart: true
      env:
        ROLE: worker
    - name: boss
      model: echo
      dependencies: [worker]
"""


def test_deployment_replicas(tmp_path):
    p = tmp_path / "d.yaml"
    p.write_text(DEPLOY_YAML)
    specs = load_deployment(str(p))
    names = [s.name for s in specs]
    assert names == ["worker-1", "worker-2", "worker-3", "boss"]
    assert specs[0].kv_budget == 1024**3
    assert specs[0].max_context == 4096
    assert specs[0].auto_restart is True
    assert specs[0].env == {"ROLE": "worker"}
    assert specs[3].dependencies == ["worker"]


def test_deployment_validation(tmp_path):
    p = tmp_path / "bad.yaml"
    p.write_text("kind: Wrong\nspec: {agents: []}\n")
    with pytest.raises(DeploymentError):
        load_deployment(str(p))
    p.write_text(
        "kind: AgentDeployment\nspec:\n  agents:\n"
        "    - name: a\n      model: echo\n      dependencies: [ghost]\n")
    with pytest.raises(DeploymentError, match="ghost"):
        load_deployment(str(p))


def test_env_overrides_quant_and_kv_options(tmp_path):
    """Every engine option added this round is env-overridable (the env
    layer only applies to keys present in DEFAULTS — regression guard)."""
    cfg = load_config(path="/nonexistent.yaml", env={
        "AGENTAINER_ENGINE_KV_DTYPE": "fp8",
        "AGENTAINER_ENGINE_DENSE_QUANT": "mxfp4",
        "AGENTAINER_ENGINE_EXPERT_FP4": "true",
        "AGENTAINER_ENGINE_PREFIX_SHARING": "false",
        "AGENTAINER_ENGINE_MAX_SEQS": "4096",
    })
    e = cfg.get("engine")
    assert e["kv_dtype"] == "fp8"
    assert e["dense_quant"] == "mxfp4"
    assert e["expert_fp4"] is True
    assert e["prefix_sharing"] is False
    assert e["max_seqs"] == 4096
