"""AgentState: everything an agent knows, serializable for checkpoint/resume.

The native counterpart of the reference's 60+-field Core.State struct
(reference: lib/quoracle/agent/core/state.ex:68-170).  Per-model conversation
histories are the logical checkpoint; on GPU the paged KV cache is *derived*
state, rebuilt by re-prefill after resume.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


def history_entry(type_: str, content: Any, **extra: Any) -> Dict[str, Any]:
    entry = {"type": type_, "content": content, "ts": time.time()}
    entry.update(extra)
    return entry


@dataclass
class PendingAction:
    action_id: str
    action: str
    params: Dict[str, Any]
    wait: Any  # False | True | int seconds
    started_at: float = field(default_factory=time.time)
    acked: bool = False


@dataclass
class AgentState:
    agent_id: str
    task_id: str
    parent_id: Optional[str] = None

    # config
    profile: Optional[str] = None
    model_pool: List[str] = field(default_factory=list)
    capability_groups: List[str] = field(default_factory=list)
    max_refinement_rounds: int = 4
    force_reflection: bool = False
    role: Optional[str] = None
    cognitive_style: Optional[str] = None
    output_style: Optional[str] = None
    delegation_strategy: Optional[str] = None
    constraints: List[str] = field(default_factory=list)   # accumulated downstream
    system_prompt_fields: Dict[str, Any] = field(default_factory=dict)
    grove: Optional[Dict[str, Any]] = None                  # parsed GROVE manifest
    grove_vars: Dict[str, str] = field(default_factory=dict)
    active_skills: List[Dict[str, Any]] = field(default_factory=list)
    sibling_context: List[Dict[str, Any]] = field(default_factory=list)

    # per-model conversation state (newest-first entries, like the reference)
    model_histories: Dict[str, List[Dict[str, Any]]] = field(default_factory=dict)
    context_lessons: Dict[str, List[Dict[str, Any]]] = field(default_factory=dict)
    model_states: Dict[str, Optional[Dict[str, Any]]] = field(default_factory=dict)

    # runtime
    status: str = "initializing"   # initializing|ready|busy|waiting|terminating
    todos: List[Dict[str, Any]] = field(default_factory=list)
    children: Dict[str, Dict[str, Any]] = field(default_factory=dict)
    pending_actions: Dict[str, PendingAction] = field(default_factory=dict)
    message_queue: List[Dict[str, Any]] = field(default_factory=list)
    wait_generation: int = 0        # guards stale wait timers
    consensus_in_flight: bool = False
    dismissing: set = field(default_factory=set)   # child ids being dismissed
    consensus_failures: int = 0
    cached_system_prompt: Optional[str] = None

    # budget: mode "root" (task-level limit), "allocated", or "na"
    budget_mode: str = "na"
    budget_allocated: Optional[float] = None
    budget_spent: float = 0.0
    budget_committed: float = 0.0   # escrowed to children

    def init_model_maps(self) -> None:
        for model in self.model_pool:
            self.model_histories.setdefault(model, [])
            self.context_lessons.setdefault(model, [])
            self.model_states.setdefault(model, None)

    def append_history(self, entry: Dict[str, Any],
                       models: Optional[List[str]] = None) -> None:
        """Prepend (newest-first) to each model's history."""
        for model in models if models is not None else self.model_pool:
            self.model_histories.setdefault(model, []).insert(0, entry)

    # -- checkpoint -------------------------------------------------------------
    def to_checkpoint(self) -> Dict[str, Any]:
        return {
            "agent_id": self.agent_id,
            "task_id": self.task_id,
            "parent_id": self.parent_id,
            "profile": self.profile,
            "model_pool": self.model_pool,
            "capability_groups": self.capability_groups,
            "max_refinement_rounds": self.max_refinement_rounds,
            "force_reflection": self.force_reflection,
            "role": self.role,
            "cognitive_style": self.cognitive_style,
            "output_style": self.output_style,
            "delegation_strategy": self.delegation_strategy,
            "constraints": self.constraints,
            "system_prompt_fields": self.system_prompt_fields,
            "grove": self.grove,
            "grove_vars": self.grove_vars,
            "active_skills": self.active_skills,
            "sibling_context": self.sibling_context,
            "model_histories": self.model_histories,
            "context_lessons": self.context_lessons,
            "model_states": self.model_states,
            "todos": self.todos,
            "children": self.children,
            "budget_mode": self.budget_mode,
            "budget_allocated": self.budget_allocated,
            "budget_spent": self.budget_spent,
            "budget_committed": self.budget_committed,
        }

    @classmethod
    def from_checkpoint(cls, data: Dict[str, Any]) -> "AgentState":
        state = cls(agent_id=data["agent_id"], task_id=data["task_id"],
                    parent_id=data.get("parent_id"))
        for key, value in data.items():
            # only public data fields: a corrupt/malicious checkpoint must
            # not reach __dict__/__class__ or private attributes
            if not key.startswith("_") and hasattr(state, key):
                setattr(state, key, value)
        state.init_model_maps()
        return state
