"""Agent registry: unique-id actor table + tree queries.

Replaces the reference's Elixir Registry + RegistryQueries (reference:
lib/quoracle/agent/registry_queries.ex).  Injected per task-runtime; duplicate
registration raises, matching the reference's unique-key Registry contract.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


class DuplicateAgentError(Exception):
    pass


@dataclass
class AgentEntry:
    agent_id: str
    actor: Any                 # agent.core.AgentActor
    task_id: str
    parent_id: Optional[str] = None
    meta: Dict[str, Any] = field(default_factory=dict)


class Registry:
    def __init__(self) -> None:
        self._agents: Dict[str, AgentEntry] = {}

    def register(self, agent_id: str, actor: Any, task_id: str,
                 parent_id: Optional[str] = None,
                 meta: Optional[Dict[str, Any]] = None) -> AgentEntry:
        if agent_id in self._agents:
            raise DuplicateAgentError(agent_id)
        entry = AgentEntry(agent_id=agent_id, actor=actor, task_id=task_id,
                           parent_id=parent_id, meta=meta or {})
        self._agents[agent_id] = entry
        return entry

    def unregister(self, agent_id: str) -> None:
        self._agents.pop(agent_id, None)

    def lookup(self, agent_id: str) -> Optional[AgentEntry]:
        return self._agents.get(agent_id)

    def alive(self, agent_id: str) -> bool:
        return agent_id in self._agents

    def all_ids(self) -> List[str]:
        return list(self._agents)

    # -- tree queries ----------------------------------------------------------
    def children_of(self, agent_id: str) -> List[str]:
        return [e.agent_id for e in self._agents.values() if e.parent_id == agent_id]

    def parent_of(self, agent_id: str) -> Optional[str]:
        entry = self._agents.get(agent_id)
        return entry.parent_id if entry else None

    def siblings_of(self, agent_id: str) -> List[str]:
        entry = self._agents.get(agent_id)
        if entry is None or entry.parent_id is None:
            return []
        return [e.agent_id for e in self._agents.values()
                if e.parent_id == entry.parent_id and e.agent_id != agent_id]

    def agents_for_task(self, task_id: str) -> List[str]:
        return [e.agent_id for e in self._agents.values() if e.task_id == task_id]

    def descendants_of(self, agent_id: str) -> List[str]:
        """All transitive children, leaves last visited (BFS order)."""
        out: List[str] = []
        frontier = self.children_of(agent_id)
        while frontier:
            out.extend(frontier)
            nxt: List[str] = []
            for child in frontier:
                nxt.extend(self.children_of(child))
            frontier = nxt
        return out
