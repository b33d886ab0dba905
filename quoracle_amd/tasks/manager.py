"""Task lifecycle: creation, pause, restore, delete, boot revival.

Behavior-parity with the reference (reference: lib/quoracle/tasks/
task_manager.ex:39-229,448-470, task_restorer.ex:30-120,
boot/agent_revival.ex:1-45): a task = profile validation -> task row -> root
agent spawn with assembled prompt fields.  Pause terminates the tree
leaves-first and marks rows "paused"; restore rebuilds agents from persisted
checkpoints in topological order with per-agent failure isolation (a failed
agent skips its whole subtree); boot revival restores every "running" task
and finalizes stuck "pausing" ones.  The GPU KV cache is derived state: the
restored histories re-prefill on the next consensus cycle.
"""

from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional

from ..agent.state import AgentState
from ..agent.supervisor import Supervisor
from ..governance.profiles import ProfileNotFoundError
from ..registry import DuplicateAgentError
from ..utils import ids

logger = logging.getLogger(__name__)


class TaskError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


class TaskManager:
    def __init__(self, runtime):
        self.runtime = runtime
        self.supervisor = runtime.supervisor or Supervisor(runtime)

    # -- create -----------------------------------------------------------------
    async def create_task(
        self,
        prompt: str,
        profile_name: str,
        *,
        budget_limit: Optional[float] = None,
        global_context: Optional[str] = None,
        initial_constraints: Optional[List[str]] = None,
        grove: Optional[Dict[str, Any]] = None,
        role: Optional[str] = None,
        cognitive_style: Optional[str] = None,
        output_style: Optional[str] = None,
        system_prompt: Optional[str] = None,
        task_id: Optional[str] = None,
        # Task Work fields (reference: README "Create a Task" +
        # fields/prompt_field_manager.ex): assembled into the first user
        # prompt exactly like child spawns
        success_criteria: Optional[str] = None,
        immediate_context: Optional[str] = None,
        approach_guidance: Optional[str] = None,
        skills: Optional[List[str]] = None,
        delegation_strategy: Optional[str] = None,
    ) -> Dict[str, Any]:
        try:
            profile = self.runtime.profiles.resolve(profile_name)
        except ProfileNotFoundError:
            raise TaskError("unknown_profile") from None

        task_id = task_id or ids.task_id()
        self.runtime.store.save_task({
            "task_id": task_id, "status": "running", "prompt": prompt,
            "profile": profile_name, "budget_limit": budget_limit,
            "global_context": global_context,
            "initial_constraints": initial_constraints or [],
            "grove": grove,
        })

        active_skills = []
        if skills:
            from ..governance.skills import SkillLoader, SkillError
            loader = SkillLoader(self.runtime.config.skills_dir,
                                 None if not grove or not grove.get("path")
                                 else __import__("os").path.join(
                                     grove["path"],
                                     grove.get("skills_path") or "skills"))
            for name in skills:
                try:
                    sk = loader.load(name)
                    active_skills.append({"name": sk["name"],
                                          "description": sk["description"],
                                          "content": sk["content"]})
                except SkillError:
                    pass
        root_state = AgentState(
            agent_id=ids.agent_id("root"),
            task_id=task_id,
            parent_id=None,
            profile=profile_name,
            model_pool=list(profile.model_pool),
            capability_groups=list(profile.capability_groups),
            max_refinement_rounds=profile.max_refinement_rounds,
            force_reflection=profile.force_reflection,
            role=role,
            cognitive_style=cognitive_style,
            output_style=output_style,
            constraints=list(initial_constraints or []),
            grove=grove,
            system_prompt_fields={"system_prompt": system_prompt}
            if system_prompt else {},
            active_skills=active_skills,
            delegation_strategy=delegation_strategy,
            budget_mode="root" if budget_limit is not None else "na",
            budget_allocated=budget_limit,
        )
        root_state.init_model_maps()
        actor = self.supervisor.start_agent(root_state)

        parts = []
        if global_context:
            parts.append(f"# Global context\n{global_context}")
        parts.append(f"# Task\n{prompt}")
        if success_criteria:
            parts.append(f"# Success criteria\n{success_criteria}")
        if immediate_context:
            parts.append(f"# Immediate context\n{immediate_context}")
        if approach_guidance:
            parts.append(f"# Approach guidance\n{approach_guidance}")
        initial = "\n\n".join(parts)
        await actor.deliver({"type": "user_message", "content": initial})
        return {"task_id": task_id, "root_agent_id": root_state.agent_id}

    # -- user -> task messaging ----------------------------------------------------
    async def send_user_message(self, task_id: str, content: str,
                                agent_id: Optional[str] = None) -> bool:
        targets = [agent_id] if agent_id else [
            a for a in self.runtime.registry.agents_for_task(task_id)
            if self.runtime.registry.parent_of(a) is None]
        delivered = False
        for target in targets:
            entry = self.runtime.registry.lookup(target)
            if entry is not None:
                await entry.actor.deliver({"type": "user_message",
                                           "content": content})
                delivered = True
        return delivered

    # -- pause / restore ---------------------------------------------------------------
    async def pause_task(self, task_id: str) -> None:
        self.runtime.store.update_task_status(task_id, "pausing")
        roots = [a for a in self.runtime.registry.agents_for_task(task_id)
                 if self.runtime.registry.parent_of(a) is None]
        for root in roots:
            await self.supervisor.terminate_tree(root, reason="paused")
        # late registrations sweep
        for agent_id in self.runtime.registry.agents_for_task(task_id):
            await self.supervisor.terminate_agent(agent_id, reason="paused")
        for row in self.runtime.store.agents_for_task(task_id):
            if row["status"] != "terminated":
                self.runtime.store.update_agent_status(row["agent_id"], "paused")
            else:
                self.runtime.store.update_agent_status(row["agent_id"], "paused")
        self.runtime.store.update_task_status(task_id, "paused")

    async def restore_task(self, task_id: str) -> Dict[str, Any]:
        """Rebuild the agent tree from checkpoints, parents before children,
        skipping subtrees whose parent failed to restore."""
        task = self.runtime.store.get_task(task_id)
        if task is None:
            raise TaskError("task_not_found")
        rows = self.runtime.store.agents_for_task(task_id)
        by_id = {r["agent_id"]: r for r in rows}
        # topological order: parents first
        ordered: List[str] = []
        visited = set()

        def visit(agent_id: str):
            if agent_id in visited or agent_id not in by_id:
                return
            parent = by_id[agent_id].get("parent_id")
            if parent and parent in by_id and parent not in visited:
                visit(parent)
            visited.add(agent_id)
            ordered.append(agent_id)

        for agent_id in by_id:
            visit(agent_id)

        restored, failed = [], []
        failed_subtrees = set()
        for agent_id in ordered:
            row = by_id[agent_id]
            parent = row.get("parent_id")
            if parent in failed_subtrees:
                failed_subtrees.add(agent_id)
                continue
            if row.get("state") is None:
                failed.append(agent_id)
                failed_subtrees.add(agent_id)
                continue
            try:
                state = AgentState.from_checkpoint(row["state"])
                # registry-conflict resolution: terminate the impostor first
                if self.runtime.registry.alive(agent_id):
                    await self.supervisor.terminate_agent(agent_id,
                                                          reason="conflict")
                self.supervisor.start_agent(state)
                self.runtime.store.update_agent_status(agent_id, "running")
                restored.append(agent_id)
            except (DuplicateAgentError, Exception) as exc:  # noqa: BLE001
                logger.warning("restore of %s failed: %s", agent_id, exc)
                failed.append(agent_id)
                failed_subtrees.add(agent_id)
        self.runtime.store.update_task_status(task_id, "running")
        return {"restored": restored, "failed": failed}

    async def delete_task(self, task_id: str) -> None:
        await self.pause_task(task_id)
        self.runtime.store.delete_task(task_id)

    # -- boot revival -----------------------------------------------------------------
    async def restore_running_tasks(self) -> Dict[str, Any]:
        """At boot: finalize stuck 'pausing' tasks, restore all 'running' ones
        (reference: boot/agent_revival.ex:27)."""
        results = {}
        for task in self.runtime.store.list_tasks("pausing"):
            self.runtime.store.update_task_status(task["task_id"], "paused")
        for task in self.runtime.store.list_tasks("running"):
            try:
                results[task["task_id"]] = await self.restore_task(task["task_id"])
            except TaskError as exc:
                results[task["task_id"]] = {"error": exc.reason}
        return results
