"""Profiles: named bundles of model pool + capability groups + consensus
parameters, with the capability-group -> action mapping and the runtime
ActionGate.

Behavior-parity with the reference (reference:
lib/quoracle/profiles/capability_groups.ex:8-46, action_gate.ex:30-69,
resolver.ex; profile columns per migrations 20260105050308/20260208210722/
20260225180000).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

ALWAYS_ALLOWED = [
    "wait", "orient", "todo", "send_message", "fetch_web", "answer_engine",
    "generate_images", "learn_skills", "create_skill", "batch_sync", "batch_async",
]

GROUP_ACTIONS: Dict[str, List[str]] = {
    "hierarchy": ["spawn_child", "dismiss_child", "adjust_budget"],
    "local_execution": ["execute_shell", "call_mcp", "record_cost",
                        "search_secrets", "generate_secret"],
    "file_read": ["file_read"],
    "file_write": ["file_write", "search_secrets", "generate_secret"],
    "external_api": ["call_api", "record_cost", "search_secrets", "generate_secret"],
}

VALID_GROUPS = ["file_read", "file_write", "external_api", "hierarchy",
                "local_execution"]

GROUP_DESCRIPTIONS = {
    "file_read": "Read files from the filesystem",
    "file_write": "Write and edit files on the filesystem",
    "external_api": "Make HTTP requests to external APIs",
    "hierarchy": "Spawn and manage child agents",
    "local_execution": "Execute shell commands and MCP calls",
}


class InvalidGroupError(Exception):
    pass


class ActionNotAllowedError(Exception):
    pass


def allowed_actions(capability_groups: List[str]) -> List[str]:
    for group in capability_groups:
        if group not in GROUP_ACTIONS:
            raise InvalidGroupError(group)
    out = list(ALWAYS_ALLOWED)
    for group in capability_groups:
        for action in GROUP_ACTIONS[group]:
            if action not in out:
                out.append(action)
    return out


def action_allowed(action: str, capability_groups: Optional[List[str]]) -> bool:
    if capability_groups is None:
        return True
    try:
        return action in allowed_actions(capability_groups)
    except InvalidGroupError:
        return False


def check_action(action: str, capability_groups: Optional[List[str]]) -> None:
    """Runtime gate called by the action router before dispatch."""
    if not action_allowed(action, capability_groups):
        raise ActionNotAllowedError(action)


def filter_actions(actions: List[str], capability_groups: Optional[List[str]]) -> List[str]:
    """Capability-filtered action list for the system prompt's schema section."""
    if capability_groups is None:
        return list(actions)
    try:
        allowed = set(allowed_actions(capability_groups))
    except InvalidGroupError:
        allowed = set(ALWAYS_ALLOWED)
    return [a for a in actions if a in allowed]


@dataclass
class Profile:
    name: str
    description: str = ""
    model_pool: List[str] = field(default_factory=list)
    capability_groups: List[str] = field(default_factory=list)
    max_refinement_rounds: int = 4
    force_reflection: bool = False

    def to_dict(self) -> dict:
        return {
            "name": self.name, "description": self.description,
            "model_pool": self.model_pool,
            "capability_groups": self.capability_groups,
            "max_refinement_rounds": self.max_refinement_rounds,
            "force_reflection": self.force_reflection,
        }

    @classmethod
    def from_dict(cls, data: dict) -> "Profile":
        return cls(
            name=data["name"], description=data.get("description", ""),
            model_pool=list(data.get("model_pool") or []),
            capability_groups=list(data.get("capability_groups") or []),
            max_refinement_rounds=int(data.get("max_refinement_rounds", 4)),
            force_reflection=bool(data.get("force_reflection", False)),
        )


class ProfileNotFoundError(Exception):
    pass


class ProfileStore:
    """In-memory profile registry, optionally backed by the persistence store."""

    def __init__(self, store=None):
        self._profiles: Dict[str, Profile] = {}
        self._store = store
        if store is not None:
            for row in store.list_profiles():
                self._profiles[row["name"]] = Profile.from_dict(row)

    def put(self, profile: Profile) -> None:
        if profile.max_refinement_rounds < 0 or profile.max_refinement_rounds > 9:
            raise ValueError("max_refinement_rounds must be 0-9")
        for group in profile.capability_groups:
            if group not in VALID_GROUPS:
                raise InvalidGroupError(group)
        self._profiles[profile.name] = profile
        if self._store is not None:
            self._store.save_profile(profile.to_dict())

    def resolve(self, name: str) -> Profile:
        try:
            return self._profiles[name]
        except KeyError:
            raise ProfileNotFoundError(name) from None

    def exists(self, name: str) -> bool:
        return name in self._profiles

    def names(self) -> List[str]:
        return list(self._profiles)

    def catalog(self) -> List[Dict[str, str]]:
        """name + description pairs — shown to parents when delegating
        (reference: README 'profile names and descriptions are visible
        to parent agents')."""
        return [{"name": p.name, "description": p.description}
                for p in self._profiles.values()]
