"""Supervisor: agent lifecycle — spawn, dismiss, tree termination, restarts.

The native rebuild of the reference's Agent.DynSup + Actions.Spawn +
TreeTerminator (reference: lib/quoracle/agent/dyn_sup.ex:29-132,
actions/spawn.ex:72-331, spawn/config_builder.ex, spawn/topology_resolver.ex,
agent/tree_terminator.ex, dismiss_child/cost_transaction.ex).

Spawn is async like the reference: the child_id and escrow commit happen
synchronously inside the action; agent startup runs in a background task that
casts child_spawned / spawn_failed back to the parent.  Dismissal terminates
the subtree leaves-first and settles costs (escrow released, child spend
absorbed into the parent's spent).
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, Dict, List, Optional

from ..budget import tracker as budget_mod
from ..governance import groves as groves_mod
from ..governance.profiles import ProfileNotFoundError
from ..registry import DuplicateAgentError
from ..utils import ids
from .core import AgentActor
from .state import AgentState

logger = logging.getLogger(__name__)

MAX_RESTARTS = 5
RESTART_WINDOW_S = 60.0


class SpawnError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


def resolve_spawn_contract(grove: Optional[Dict[str, Any]],
                           parent_skills: List[str],
                           child_skills: List[str]) -> Dict[str, Any]:
    """Find the grove-topology edge matching (parent skills -> child skills)
    and return its auto_inject map (reference: spawn_contract_resolver.ex).

    Edge shape: {"from": [skill...], "to": [skill...], "auto_inject":
    {"profile": ..., "skills": [...], "constraints": [...]}}.  An edge with no
    "from"/"to" matches anything.
    """
    if not grove:
        return {}
    topology = grove.get("topology") or {}
    for edge in topology.get("edges") or []:
        frm = edge.get("from")
        to = edge.get("to")
        if frm and not set(frm) & set(parent_skills or []):
            continue
        if to and not set(to) & set(child_skills or []):
            continue
        return edge.get("auto_inject") or {}
    return {}


class Supervisor:
    def __init__(self, runtime):
        self.runtime = runtime
        runtime.supervisor = self
        self._restarts: Dict[str, List[float]] = {}

    # -- low-level start/terminate ------------------------------------------------
    def start_agent(self, state: AgentState) -> AgentActor:
        actor = AgentActor(state, self.runtime)
        self.runtime.registry.register(state.agent_id, actor, state.task_id,
                                       parent_id=state.parent_id)
        self.runtime.store.save_agent(
            state.agent_id, state.task_id, state.parent_id,
            config={"profile": state.profile, "model_pool": state.model_pool,
                    "role": state.role},
            state=state.to_checkpoint(), status="running")
        actor.start()
        self.runtime.bus.agent_spawned(state.agent_id, state.parent_id,
                                       state.task_id)
        return actor

    async def terminate_agent(self, agent_id: str, reason: str = "normal") -> None:
        entry = self.runtime.registry.lookup(agent_id)
        if entry is None:
            return
        await entry.actor.stop(reason)

    async def terminate_tree(self, agent_id: str, reason: str = "dismissed") -> None:
        """Leaves-first recursive termination (reference: tree_terminator.ex)."""
        for child_id in self.runtime.registry.children_of(agent_id):
            await self.terminate_tree(child_id, reason)
        await self.terminate_agent(agent_id, reason)

    # -- spawn_child action ----------------------------------------------------------
    async def spawn_child_action(self, parent: AgentActor,
                                 params: Dict[str, Any]) -> Dict[str, Any]:
        state = parent.state

        # dismiss-vs-spawn race guard: refuse to spawn while the parent has
        # ANY dismissal in flight (reference: spawn.ex:73-107 returns
        # :parent_dismissing whenever the parent is dismissing children)
        if state.dismissing:
            return {"error": "parent_dismissing",
                    "dismissing": sorted(state.dismissing)}

        child_id = ids.agent_id()

        # Topology contract: grove may auto-inject profile/skills/constraints
        parent_skill_names = [s.get("name") for s in state.active_skills]
        child_skills = list(params.get("skills") or [])
        inject = resolve_spawn_contract(state.grove, parent_skill_names,
                                        child_skills)
        profile_name = params.get("profile") or inject.get("profile")
        if not profile_name:
            return {"error": "missing_profile"}
        try:
            profile = self.runtime.profiles.resolve(profile_name)
        except ProfileNotFoundError:
            return {"error": "unknown_profile", "profile": profile_name}

        # Budget escrow (sync, so over-spawning is impossible)
        budget_allocated: Optional[float] = None
        if params.get("budget") is not None:
            try:
                budget_allocated = budget_mod.parse_amount(params["budget"])
                view = budget_mod.BudgetView(state.budget_mode,
                                             state.budget_allocated,
                                             state.budget_spent,
                                             state.budget_committed)
                state.budget_committed = budget_mod.lock_allocation(
                    view, budget_allocated)
            except budget_mod.BudgetError as exc:
                return {"error": exc.reason}

        state.children[child_id] = {
            "status": "spawning",
            "task_description": params.get("task_description", ""),
            "budget": budget_allocated,
        }

        asyncio.ensure_future(self._spawn_background(
            parent, child_id, profile, params, inject, budget_allocated))
        return {"status": "spawning", "child_id": child_id,
                "budget_allocated": budget_allocated}

    async def _spawn_background(self, parent: AgentActor, child_id: str,
                                profile, params: Dict[str, Any],
                                inject: Dict[str, Any],
                                budget_allocated: Optional[float]) -> None:
        state = parent.state
        try:
            child_state = self._build_child_config(
                parent, child_id, profile, params, inject, budget_allocated)
            actor = None
            last_exc: Optional[Exception] = None
            for _attempt in range(self.runtime.config.spawn_retries):
                try:
                    actor = self.start_agent(child_state)
                    break
                except DuplicateAgentError:
                    child_state.agent_id = ids.agent_id()
                    continue
                except Exception as exc:  # noqa: BLE001
                    last_exc = exc
                    await asyncio.sleep(0.01)
            if actor is None:
                raise SpawnError(f"start_failed: {last_exc}")

            narrative = await self._ancestor_narrative(parent)
            initial = _initial_message(params, narrative)
            await actor.deliver({"type": "user_message", "content": initial})
            state.children[child_id]["status"] = "running"
            await parent.deliver({"type": "child_spawned", "child_id": child_id})
        except Exception as exc:  # noqa: BLE001
            logger.exception("spawn of %s failed", child_id)
            # release escrow and notify parent
            if budget_allocated is not None:
                state.budget_committed = budget_mod.release_allocation(
                    state.budget_committed, budget_allocated)
            state.children.pop(child_id, None)
            await parent.deliver({"type": "spawn_failed", "child_id": child_id,
                                  "reason": str(exc)})

    async def _ancestor_narrative(self, parent: AgentActor) -> str:
        """Lineage context for the child: the parent's recent decision
        trail, LLM-summarized through the summarization-role model when it
        is long (reference: spawn/config_builder.ex:125,230); truncation is
        the fallback when no summarizer is reachable."""
        state = parent.state
        history = max(state.model_histories.values(), key=len, default=[])
        decisions = [e for e in history if e.get("type") == "decision"][:6]
        own_task = ""
        for e in reversed(history):              # oldest entries last in scan
            if e.get("type") in ("prompt", "event"):
                own_task = str(e.get("content", ""))[:300]
                break
        if not decisions and not own_task:
            return ""
        lines = [f"Parent's task: {own_task}" if own_task else ""]
        for e in reversed(decisions):            # oldest first
            c = e.get("content")
            if isinstance(c, dict):
                lines.append(f"- {c.get('action')}: "
                             f"{str(c.get('reasoning', ''))[:200]}")
        raw = "\n".join(l for l in lines if l)
        if len(raw) <= 1200:
            return raw
        model = (self.runtime.config.model_roles.get("summarization")
                 or (state.model_pool[0] if state.model_pool else None))
        if model is None:
            return raw[:1200]
        try:
            from ..engine.api import GenerateRequest
            engine = self.runtime.engines.engine_for(model)
            result = await engine.generate(GenerateRequest(
                model_key=model,
                messages=[{"role": "user",
                           "content": "Summarize this agent's work so far "
                                      "in <=5 bullet points for a child "
                                      "agent inheriting a subtask:\n"
                                      + raw[:6000]}],
                temperature=0.3, max_tokens=512))
            if result.ok and result.text.strip():
                return result.text.strip()
        except Exception:  # noqa: BLE001 — narrative is best-effort
            pass
        return raw[:1200]

    def _build_child_config(self, parent: AgentActor, child_id: str, profile,
                            params: Dict[str, Any], inject: Dict[str, Any],
                            budget_allocated: Optional[float]) -> AgentState:
        """ConfigBuilder parity: inherit constraints/grove/context; merge
        topology-injected skills/constraints (reference: spawn/config_builder.ex)."""
        state = parent.state
        constraints = list(state.constraints)
        if params.get("downstream_constraints"):
            constraints.append(params["downstream_constraints"])
        for constraint in inject.get("constraints") or []:
            if constraint not in constraints:
                constraints.append(constraint)

        grove = state.grove
        grove_vars = dict(state.grove_vars)
        grove_vars.update(params.get("grove_vars") or {})
        if grove is not None and grove_vars:
            grove = groves_mod.substitute_grove_vars(grove, grove_vars)

        skill_names = list(params.get("skills") or [])
        for name in inject.get("skills") or []:
            if name not in skill_names:
                skill_names.append(name)
        active_skills = []
        loader = parent.skill_loader()
        for name in skill_names:
            try:
                skill = loader.load(name)
                active_skills.append({"name": skill["name"],
                                      "description": skill["description"],
                                      "content": skill["content"]})
            except Exception:
                logger.warning("skill %s not found at spawn", name)

        child = AgentState(
            agent_id=child_id,
            task_id=state.task_id,
            parent_id=state.agent_id,
            profile=profile.name,
            model_pool=list(profile.model_pool),
            capability_groups=list(profile.capability_groups),
            max_refinement_rounds=profile.max_refinement_rounds,
            force_reflection=profile.force_reflection,
            role=params.get("role"),
            cognitive_style=params.get("cognitive_style"),
            output_style=params.get("output_style"),
            delegation_strategy=params.get("delegation_strategy"),
            constraints=constraints,
            grove=grove,
            grove_vars=grove_vars,
            active_skills=active_skills,
            sibling_context=list(params.get("sibling_context") or []),
            budget_mode="allocated" if budget_allocated is not None else state.budget_mode,
            budget_allocated=budget_allocated
            if budget_allocated is not None else None,
        )
        child.init_model_maps()
        return child

    # -- dismiss_child action ---------------------------------------------------------
    async def dismiss_child_action(self, parent: AgentActor, child_id: str,
                                   reason: Optional[str]) -> Dict[str, Any]:
        state = parent.state
        if child_id not in state.children:
            return {"error": "not_a_direct_child", "child_id": child_id}
        state.dismissing.add(child_id)
        child_info = state.children.get(child_id, {})

        async def _background():
            try:
                # Cost absorption: subtree spend rolls into the parent, escrow
                # is released (reference: dismiss_child/cost_transaction.ex)
                subtree = [child_id] + self.runtime.registry.descendants_of(child_id)
                subtree_spent = self.runtime.store.total_cost(subtree)
                await self.terminate_tree(child_id, reason or "dismissed")
                allocated = child_info.get("budget")
                if allocated is not None:
                    state.budget_committed = budget_mod.release_allocation(
                        state.budget_committed, allocated)
                state.budget_spent += subtree_spent
                state.children.pop(child_id, None)
                await parent.deliver({"type": "child_dismissed",
                                      "child_id": child_id, "reason": reason})
            finally:
                state.dismissing.discard(child_id)

        asyncio.ensure_future(_background())
        return {"status": "dismissing", "child_id": child_id}

    # -- restart policy -----------------------------------------------------------------
    def schedule_restart(self, agent_id: str) -> None:
        """Crash recovery: restore the agent from its persisted checkpoint
        when the restart window allows (reference: DynamicSupervisor policy,
        dyn_sup.ex:54-60); beyond the window the agent stays failed."""
        if not self.record_crash(agent_id):
            self.runtime.store.update_agent_status(agent_id, "failed")
            self.runtime.bus.log(agent_id, "error",
                                 "restart limit reached; agent failed")
            return

        async def _restart():
            row = self.runtime.store.get_agent(agent_id)
            if row is None or row.get("state") is None:
                self.runtime.store.update_agent_status(agent_id, "failed")
                return
            try:
                state = AgentState.from_checkpoint(row["state"])
                self.start_agent(state)
                self.runtime.bus.log(agent_id, "warning",
                                     "agent restarted after crash")
            except Exception:  # noqa: BLE001
                logger.exception("restart of %s failed", agent_id)
                self.runtime.store.update_agent_status(agent_id, "failed")

        asyncio.ensure_future(_restart())

    def record_crash(self, agent_id: str) -> bool:
        """True if the agent may restart (max 5 restarts / 60 s, like the
        reference's DynamicSupervisor policy)."""
        now = time.monotonic()
        window = [t for t in self._restarts.get(agent_id, [])
                  if now - t < RESTART_WINDOW_S]
        window.append(now)
        self._restarts[agent_id] = window
        return len(window) <= MAX_RESTARTS


def _initial_message(params: Dict[str, Any], narrative: str = "") -> str:
    parts = [f"# Task\n{params.get('task_description', '')}"]
    if narrative:
        parts.append(f"# Lineage context\n{narrative}")
    if params.get("success_criteria"):
        parts.append(f"# Success criteria\n{params['success_criteria']}")
    if params.get("immediate_context"):
        parts.append(f"# Context\n{params['immediate_context']}")
    if params.get("approach_guidance"):
        parts.append(f"# Suggested approach\n{params['approach_guidance']}")
    siblings = params.get("sibling_context") or []
    if siblings:
        lines = ["# Sibling agents (their scopes are OFF-LIMITS to you)"]
        for sib in siblings:
            lines.append(f"- {sib.get('agent_id', '?')}: {sib.get('task', '')}")
        parts.append("\n".join(lines))
    return "\n\n".join(parts)
