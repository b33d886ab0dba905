"""ID generation for agents, tasks, actions, commands and secrets."""

from __future__ import annotations

import secrets
import time

_ALPHABET = "0123456789abcdefghijklmnopqrstuvwxyz"


def _rand(n: int) -> str:
    return "".join(secrets.choice(_ALPHABET) for _ in range(n))


def agent_id(prefix: str = "agent") -> str:
    return f"{prefix}_{int(time.time() * 1000):x}{_rand(6)}"


def task_id() -> str:
    return f"task_{_rand(12)}"


def action_id() -> str:
    return f"act_{_rand(12)}"


def command_id() -> str:
    return f"cmd_{_rand(12)}"


def connection_id() -> str:
    return f"mcp_{_rand(12)}"


def request_id() -> str:
    return f"req_{_rand(12)}"
