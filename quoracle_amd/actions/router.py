"""Action router: the gates pipeline wrapped around every action execution.

Behavior-parity with the reference Router/ClientAPI (reference:
lib/quoracle/actions/router.ex:42-168, router/client_api.ex:29-51,
router/execution.ex): validation -> capability ActionGate -> grove hard rules
-> budget enforcement -> secret resolution -> execution (smart sync/async) ->
output scrubbing -> untrusted-content wrapping -> event broadcast + log
persistence.  The reference spawns an ephemeral GenServer per action; here
each execution is an asyncio task owned by the agent, which gives the same
isolation (an action crash never takes the agent down).
"""

from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from ..budget import tracker as budget_mod
from ..governance import groves as groves_mod
from ..governance import profiles as profiles_mod
from ..governance import security as security_mod
from . import executors
from .validator import ValidationError, validate_params


@dataclass
class ActionContext:
    agent: Any                 # agent.core.AgentActor
    runtime: Any               # tasks.runtime.TaskRuntime
    action_id: str
    action: str
    params: Dict[str, Any] = field(default_factory=dict)
    skill_name: Optional[str] = None


class ActionError(Exception):
    def __init__(self, reason: str, detail: Any = None):
        super().__init__(reason)
        self.reason = reason
        self.detail = detail


# Per-action timeout overrides (reference: action_executor.ex:302-312)
ACTION_TIMEOUTS: Dict[str, float] = {
    "execute_shell": 600.0,
    "fetch_web": 120.0,
    "call_api": 120.0,
    "call_mcp": 120.0,
    "answer_engine": 600.0,
    "generate_images": 600.0,
    "spawn_child": 60.0,
}


async def execute_action(ctx: ActionContext) -> Dict[str, Any]:
    """Run the full gate pipeline and the action itself.

    Returns the (scrubbed, wrapped) result dict.  Raises ActionError with a
    machine-readable reason on any gate failure.
    """
    agent = ctx.agent
    runtime = ctx.runtime
    state = agent.state

    # 1. Validation (consensus output is pre-validated; batch sub-actions and
    #    direct API calls are not)
    try:
        ctx.params = validate_params(ctx.action, ctx.params,
                                     profile_optional=agent.spawn_profile_optional())
    except ValidationError as exc:
        raise ActionError(exc.reason) from None

    # 2. Capability gate (profile)
    try:
        profiles_mod.check_action(ctx.action, state.capability_groups
                                  if state.profile is not None else None)
    except profiles_mod.ActionNotAllowedError:
        raise ActionError("action_not_allowed") from None

    # 3. Grove hard rules
    grove = state.grove or {}
    try:
        groves_mod.check_action(ctx.action, grove.get("hard_rules"), ctx.skill_name)
    except groves_mod.HardRuleViolation as exc:
        raise ActionError("hard_rule_violation", exc.detail) from None

    # 4. Budget enforcement
    view = budget_mod.BudgetView(mode=state.budget_mode,
                                 allocated=state.budget_allocated,
                                 spent=state.budget_spent,
                                 committed=state.budget_committed)
    try:
        budget_mod.check_can_spend(view)
    except budget_mod.BudgetError as exc:
        raise ActionError(exc.reason) from None

    # 5. Secret resolution (track usage for the audit table).  The vault is
    # lazy and refuses to exist without key material, so only touch it when
    # the params actually carry {{SECRET:...}} templates.
    used_secrets: set = set()
    resolved_params = ctx.params
    if security_mod.has_secret_templates(ctx.params):
        try:
            resolved_params = security_mod.resolve_params(
                ctx.params, runtime.vault, used_secrets)
        except security_mod.SecretNotFoundError as exc:
            raise ActionError("secret_not_found", str(exc)) from None
        except security_mod.VaultKeyError as exc:
            raise ActionError("vault_key_missing", str(exc)) from None
    for name in used_secrets:
        runtime.store.record_secret_usage(name, state.agent_id, ctx.action)

    # 6. Execute
    runtime.bus.action_event(state.agent_id, "started", ctx.action, ctx.action_id)
    started = time.monotonic()
    executor = executors.EXECUTORS.get(ctx.action)
    if executor is None:
        raise ActionError("unknown_action")
    timeout = ACTION_TIMEOUTS.get(ctx.action, runtime.config.action_timeout_s)
    exec_ctx = ActionContext(agent=agent, runtime=runtime, action_id=ctx.action_id,
                             action=ctx.action, params=resolved_params,
                             skill_name=ctx.skill_name)
    try:
        result = await asyncio.wait_for(executor(exec_ctx), timeout=timeout)
    except asyncio.TimeoutError:
        runtime.bus.action_event(state.agent_id, "error", ctx.action, ctx.action_id,
                                 {"error": "timeout"})
        raise ActionError("action_timeout") from None
    except ActionError:
        raise
    except (groves_mod.HardRuleViolation, groves_mod.ConfinementViolation) as exc:
        raise ActionError("confinement_violation"
                          if isinstance(exc, groves_mod.ConfinementViolation)
                          else "hard_rule_violation", exc.detail) from None
    except groves_mod.SchemaViolation as exc:
        raise ActionError("schema_violation", exc.detail) from None
    except budget_mod.BudgetError as exc:
        raise ActionError(exc.reason) from None

    elapsed_ms = (time.monotonic() - started) * 1000.0

    # 7. Scrub secrets out of the result, then wrap untrusted content.
    # The vault is lazy: open it for scrubbing only when the store actually
    # holds secrets (and a key is available to read them).
    vault = getattr(runtime, "_vault", None)
    if vault is None and runtime.store.list_secret_names():
        try:
            vault = runtime.vault
        except security_mod.VaultKeyError:
            vault = None
    if vault is not None:
        secret_values = {}
        for name in vault.names():
            try:
                secret_values[name] = vault.get(name)
            except security_mod.SecretNotFoundError:
                continue   # unmigrated v0 blob: nothing to scrub with
        result = security_mod.scrub_output(result, secret_values)
    result = security_mod.wrap_untrusted_result(ctx.action, result)

    # 7b. Image payloads become on-disk artifacts + placeholders so binary
    # blobs never enter model histories (reference: agent/image_detector.ex;
    # compression is a documented divergence — no vision model hosted)
    from ..utils import images as images_mod
    result, image_artifacts = images_mod.extract_images(result)
    if image_artifacts:
        runtime.bus.log(state.agent_id, "info",
                        f"{len(image_artifacts)} image artifact(s) detected",
                        {"artifacts": image_artifacts})
        if isinstance(result, dict):
            # artifact metadata rides with the result (the reference's
            # multimodal history entry carries the image reference)
            result = {**result, "image_artifacts": image_artifacts}

    # 8. Broadcast + persist
    runtime.bus.action_event(state.agent_id, "completed", ctx.action, ctx.action_id,
                             {"elapsed_ms": elapsed_ms})
    runtime.store.save_log(state.agent_id, state.task_id, "info",
                           f"action_{ctx.action}",
                           f"{ctx.action} completed in {elapsed_ms:.0f}ms",
                           {"action_id": ctx.action_id})
    return result if isinstance(result, dict) else {"result": result}
