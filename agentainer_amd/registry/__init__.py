from .agent import (
    CREATED, FAILED, PAUSED, RUNNING, STOPPED, STATUSES,
    Agent, AgentError, AgentNotFound, Manager,
)

__all__ = [
    "CREATED", "FAILED", "PAUSED", "RUNNING", "STOPPED", "STATUSES",
    "Agent", "AgentError", "AgentNotFound", "Manager",
]
