from .config import Config, DEFAULT_TOKEN, DEFAULTS, load_config
from .deployment import AgentSpec, DeploymentError, load_deployment, parse_memory

__all__ = [
    "Config", "DEFAULT_TOKEN", "DEFAULTS", "load_config",
    "AgentSpec", "DeploymentError", "load_deployment", "parse_memory",
]
