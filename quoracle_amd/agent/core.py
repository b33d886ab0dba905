"""AgentActor: the event-driven agent main loop.

The native rebuild of the reference's Agent.Core GenServer + MessageHandler +
ConsensusHandler + ActionExecutor (reference: lib/quoracle/agent/core.ex,
message_handler.ex:58-148,353-485, consensus_handler.ex:64-333,
consensus_handler/action_executor.ex:16-281).  One asyncio task owns the
agent: a single-owner inbox makes the reference's protocol races (stale
triggers, dismiss-vs-spawn, ack tracking) structurally simple — messages
arriving during a consensus cycle wait in the inbox and are batch-flushed
into history at the next cycle start, and wait timers carry a generation
counter so a stale timeout can't trigger a spurious cycle.

Zero hardcoded decision logic: every step is (flush messages) -> consensus ->
execute winning action -> wait semantics -> repeat.
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, Dict, List, Optional

from ..actions import router as router_mod
from ..consensus import pipeline as pipeline_mod
from ..consensus import prompt_builder
from ..engine.api import GenerateRequest, count_messages_tokens, dynamic_max_tokens
from ..utils import ids
from . import condensation as condensation_mod
from . import context as context_mod
from . import injectors
from .state import AgentState, PendingAction, history_entry

logger = logging.getLogger(__name__)

# Self-contained actions whose results arrive synchronously enough that
# wait=true would stall the agent; auto-corrected to wait=false
# (reference: action_executor.ex:83-97).
SELF_CONTAINED_ACTIONS = {
    "orient", "todo", "file_read", "file_write", "generate_secret",
    "search_secrets", "record_cost", "adjust_budget", "learn_skills",
    "create_skill", "batch_sync",
}

MESSAGE_TYPES = {"user_message", "agent_message", "shell_completed",
                 "child_spawned", "spawn_failed", "budget_adjusted",
                 "child_dismissed"}


class AgentActor:
    def __init__(self, state: AgentState, runtime):
        self.state = state
        self.runtime = runtime
        self.inbox: asyncio.Queue = asyncio.Queue()
        self.shell_commands: Dict[str, Any] = {}
        self.mcp_connections: Dict[str, Any] = {}
        self._task: Optional[asyncio.Task] = None
        self._running = False
        self._trigger = False           # re-run cycle without blocking on inbox
        self._wait_deadline: Optional[float] = None
        self._stopped = asyncio.Event()
        self.steps_completed = 0        # consensus cycles completed (bench metric)

    # -- lifecycle ---------------------------------------------------------------
    def start(self) -> None:
        self.state.init_model_maps()
        self._running = True
        self.state.status = "ready"
        self.runtime.bus.state_change(self.state.agent_id, "ready")
        self._task = asyncio.ensure_future(self._loop())

    async def stop(self, reason: str = "normal") -> None:
        self._running = False
        await self.inbox.put({"type": "stop", "reason": reason})
        await self._stopped.wait()

    async def deliver(self, message: Dict[str, Any]) -> None:
        await self.inbox.put(message)

    # -- helpers used by executors -------------------------------------------------
    def spawn_profile_optional(self) -> bool:
        """True when the grove topology auto-injects the child profile
        (reference: validator.ex spawn_profile_optional?)."""
        grove = self.state.grove or {}
        topology = grove.get("topology") or {}
        edges = topology.get("edges") or []
        for edge in edges:
            inject = (edge.get("auto_inject") or {})
            if isinstance(inject.get("profile"), str) and inject["profile"]:
                return True
        return False

    def skill_loader(self):
        from ..governance.skills import SkillLoader
        grove = self.state.grove or {}
        grove_skills = None
        if grove.get("path"):
            import os
            grove_skills = os.path.join(
                grove["path"], grove.get("skills_path") or "skills")
        return SkillLoader(self.runtime.config.skills_dir, grove_skills)

    async def switch_model_pool(self, new_pool) -> Dict[str, str]:
        """Runtime model-pool switching with history transfer
        (reference: agent/history_transfer.ex:38-240)."""
        from . import history_transfer
        report = await history_transfer.transfer_histories(
            self.state, list(new_pool), self.runtime.engines.engine_for,
            embed_many=self._embed_many())
        self._persist()
        self.runtime.bus.log(self.state.agent_id, "info",
                             "model pool switched", {"report": report})
        return report

    def invalidate_system_prompt(self) -> None:
        self.state.cached_system_prompt = None

    # -- main loop -----------------------------------------------------------------
    async def _loop(self) -> None:
        try:
            while self._running:
                message = await self._next_message()
                if message is None:
                    # wait timer expired
                    self._wait_deadline = None
                    await self._run_cycle()
                    continue
                mtype = message.get("type")
                if mtype == "stop":
                    break
                if mtype == "action_result":
                    cycle = self._handle_action_result(message)
                    if cycle:
                        await self._run_cycle()
                    continue
                if mtype in MESSAGE_TYPES:
                    self.state.message_queue.append(message)
                    # drain everything already waiting so simultaneous
                    # messages batch into ONE cycle (reference:
                    # message_batcher.ex; stale triggers/results folded in)
                    while not self.inbox.empty():
                        extra = self.inbox.get_nowait()
                        etype = extra.get("type")
                        if etype == "stop":
                            self._running = False
                            break
                        if etype == "action_result":
                            self._handle_action_result(extra)
                        elif etype in MESSAGE_TYPES:
                            self.state.message_queue.append(extra)
                    if not self._running:
                        break
                    await self._run_cycle()
                    continue
                logger.warning("%s: unknown inbox message %s",
                               self.state.agent_id, mtype)
        except asyncio.CancelledError:
            pass
        except Exception:
            logger.exception("agent %s crashed", self.state.agent_id)
            self.runtime.bus.log(self.state.agent_id, "error", "agent crashed")
            self._crashed = True
        finally:
            await self._terminate()
            if getattr(self, "_crashed", False) and self.runtime.supervisor:
                # supervision restart policy (reference: dyn_sup.ex 5/60s)
                self.runtime.supervisor.schedule_restart(self.state.agent_id)

    async def _next_message(self) -> Optional[Dict[str, Any]]:
        """Block on the inbox honoring trigger/wait-timer semantics.
        Returns None when a wait timer fires."""
        if self._trigger:
            self._trigger = False
            # Drain anything already queued first (stale-trigger draining)
            while not self.inbox.empty():
                msg = self.inbox.get_nowait()
                if msg.get("type") == "stop":
                    self._running = False
                    return msg
                if msg.get("type") == "action_result":
                    self._handle_action_result(msg)
                elif msg.get("type") in MESSAGE_TYPES:
                    self.state.message_queue.append(msg)
            return None  # proceed straight into a cycle
        if self._wait_deadline is not None:
            remaining = self._wait_deadline - time.monotonic()
            if remaining <= 0:
                return None
            try:
                return await asyncio.wait_for(self.inbox.get(), remaining)
            except asyncio.TimeoutError:
                return None
        return await self.inbox.get()

    def _handle_action_result(self, message: Dict[str, Any]) -> bool:
        """Record the result; return True when consensus should re-trigger.

        Wait semantics (reference: action_result_handler.ex:39-99): wait=false
        -> continuation; wait=true -> stay idle (only a real message wakes the
        agent); wait=N -> the result cancels the timer and triggers."""
        action_id = message.get("action_id")
        pending = self.state.pending_actions.pop(action_id, None)
        result = message.get("result")
        action = message.get("action") or (pending.action if pending else "?")
        entry = history_entry(
            "result",
            f"[Action result: {action}]\n" + _result_text(result),
            action_id=action_id, action_type=action)
        self.state.append_history(entry)
        if pending is None or message.get("batch"):
            # batch_async sub-results behave like events: always re-trigger
            return True
        if pending.wait is True:
            return False
        if pending.wait is False or pending.wait == 0:
            return True
        # timed wait: result beat the timer
        self._wait_deadline = None
        self.state.wait_generation += 1
        return True

    # -- the consensus cycle ---------------------------------------------------------
    async def _run_cycle(self) -> None:
        if self.state.consensus_in_flight:
            return
        self._flush_message_queue()
        if not any(self.state.model_histories.get(m) for m in self.state.model_pool):
            return  # nothing to decide on yet
        self.state.consensus_in_flight = True
        self.state.status = "busy"
        self.runtime.bus.state_change(self.state.agent_id, "busy")
        try:
            outcome = await self._consensus_with_retry()
        finally:
            self.state.consensus_in_flight = False
        if outcome is None:
            self._flush_embedding_costs()
            self.state.status = "waiting"
            self.runtime.bus.state_change(self.state.agent_id, "waiting")
            return
        decision = outcome.decision
        self._flush_embedding_costs()
        self.steps_completed += 1
        self.runtime.bus.consensus_decision(self.state.agent_id, {
            "kind": decision.kind,
            "action": decision.action.get("action"),
            "confidence": decision.confidence,
            "rounds": outcome.rounds_used,
            "temperatures": outcome.temperatures,
            "clusters": [{"count": c.count,
                          "action": c.representative.get("action")}
                         for c in decision.clusters or []],
        })
        await self._execute_decision(decision.action)

    def _flush_message_queue(self) -> None:
        """Batch queued messages into history (reference: message_batcher.ex)."""
        queued = self.state.message_queue
        if not queued:
            return
        self.state.message_queue = []
        if len(queued) == 1:
            content = _format_incoming(queued[0])
        else:
            parts = [f"<message index=\"{i + 1}\">\n{_format_incoming(m)}\n</message>"
                     for i, m in enumerate(queued)]
            content = ("<simultaneous_messages count=\"%d\">\n%s\n</simultaneous_messages>"
                       % (len(queued), "\n".join(parts)))
        self.state.append_history(history_entry("event", content))

    async def _consensus_with_retry(self):
        """Consensus with up to N retries feeding correction context, then a
        stall notification (reference: message_handler.ex:353-485)."""
        retries = self.runtime.config.consensus_retries
        last_errors: Dict[str, str] = {}
        for attempt in range(retries):
            try:
                outcome = await pipeline_mod.run_consensus(
                    self.state.model_pool,
                    self._make_query_fn(last_errors),
                    max_refinement_rounds=self.state.max_refinement_rounds,
                    force_reflection=self.state.force_reflection,
                    embed_many=self._embed_many(),
                    profile_optional_spawn=self.spawn_profile_optional(),
                    prompt=self._last_user_prompt(),
                )
                self.state.consensus_failures = 0
                return outcome
            except pipeline_mod.ConsensusError as exc:
                last_errors = dict(exc.model_errors)
                self.state.consensus_failures += 1
                self.runtime.bus.log(
                    self.state.agent_id, "warning",
                    f"consensus failed ({exc.reason}), attempt {attempt + 1}/{retries}",
                    {"model_errors": exc.model_errors})
        await self._notify_stall(last_errors)
        return None

    async def _notify_stall(self, errors: Dict[str, str]) -> None:
        note = (f"Agent {self.state.agent_id} stalled: consensus failed "
                f"{self.runtime.config.consensus_retries} times "
                f"({json.dumps(errors)[:500]})")
        self.runtime.bus.log(self.state.agent_id, "error", note)
        parent = self.runtime.registry.lookup(self.state.parent_id) \
            if self.state.parent_id else None
        if parent is not None:
            await parent.actor.deliver({"type": "agent_message",
                                        "from": self.state.agent_id,
                                        "content": note})
        else:
            self.runtime.bus.task_message(self.state.task_id, {
                "from": self.state.agent_id, "to": "user", "content": note})

    EMBED_PRICE_PER_MTOK = 0.01    # matches the embed-small serving price

    def _embed_many(self):
        try:
            # one facade per actor: callable embed_many + fused
            # similarity_matrix + the per-cycle cost accumulator
            if getattr(self, "_embed_facade", None) is None:
                self._embed_facade = self.runtime.engines.embed_facade
            return self._embed_facade
        except RuntimeError:
            return None

    def _flush_embedding_costs(self) -> None:
        """Flush the cycle's accumulated embedding cost in ONE record
        (reference: message_handler.ex:133-144)."""
        facade = getattr(self, "_embed_facade", None)
        if facade is None:
            return
        tokens = facade.embedded_tokens - getattr(self, "_embed_flushed", 0)
        if tokens <= 0:
            return
        self._embed_flushed = facade.embedded_tokens
        cost = tokens * self.EMBED_PRICE_PER_MTOK / 1e6
        self.state.budget_spent += cost
        self.runtime.store.save_cost(
            self.state.agent_id, self.state.task_id, "embed-small", cost,
            category="embedding",
            metadata={"tokens": tokens})

    def _last_user_prompt(self) -> str:
        for model in self.state.model_pool:
            for entry in self.state.model_histories.get(model, ()):
                if entry.get("type") in ("prompt", "event", "user"):
                    content = entry.get("content")
                    return content if isinstance(content, str) \
                        else json.dumps(content, default=str)
        return ""

    # -- per-model query (reference: per_model_query.ex) ----------------------------
    def _make_query_fn(self, correction_errors: Dict[str, str]):
        async def query_fn(model_key: str, round_num: int,
                           refinement_prompt: Optional[str]) -> Optional[str]:
            engine = self.runtime.engines.engine_for(model_key)
            from ..consensus.temperature import round_temperature

            # Reactive condensation at 100% of window
            from . import token_manager as tm
            history = self.state.model_histories.get(model_key, [])
            if tm.needs_condensation(engine.count_tokens, history,
                                     engine.context_limit(model_key)):
                await condensation_mod.condense_model_history(
                    self.state, model_key, engine, embed_many=self._embed_many())

            def _build() -> List[Dict[str, str]]:
                convo = context_mod.build_conversation_messages(
                    self.state.model_histories.get(model_key, []))
                convo = injectors.inject_all(
                    convo,
                    todos=self.state.todos,
                    children=self.state.children,
                    budget={"mode": self.state.budget_mode,
                            "allocated": self.state.budget_allocated,
                            "spent": self.state.budget_spent,
                            "committed": self.state.budget_committed},
                    lessons=self.state.context_lessons.get(model_key),
                    model_state=self.state.model_states.get(model_key),
                    correction=injectors.correction_block(correction_errors,
                                                          model_key),
                    refinement_prompt=refinement_prompt,
                    # context-size telemetry block (reference:
                    # consensus_handler/context_injector.ex)
                    used_tokens=tm.history_tokens(
                        engine.count_tokens,
                        self.state.model_histories.get(model_key, [])),
                    context_limit=engine.context_limit(model_key),
                )
                return [{"role": "system", "content": self._system_prompt()}] + convo

            messages = _build()
            input_tokens = await condensation_mod.ensure_fits(
                self.state, model_key, engine,
                lambda: count_messages_tokens(engine, _build()),
                embed_many=self._embed_many())
            messages = _build()

            request = GenerateRequest(
                model_key=model_key,
                messages=messages,
                temperature=round_temperature(model_key, round_num,
                                              self.state.max_refinement_rounds),
                max_tokens=dynamic_max_tokens(engine, model_key, input_tokens),
                seed=hash((self.state.agent_id, model_key, round_num)) & 0x7FFFFFFF,
                action_grammar=True,
                allowed_actions=self._grammar_actions(),
                grammar_context=self._grammar_context(),
                request_id=ids.request_id(),
                # stable per (agent, model): consecutive cycles reuse the KV
                # of the unchanged history prefix (engine prefix cache)
                session_id=f"{self.state.agent_id}:{model_key}",
            )
            try:
                result = await asyncio.wait_for(
                    engine.generate(request),
                    timeout=self.runtime.config.generate_timeout_s)
            except asyncio.TimeoutError:
                raise RuntimeError("generate_timeout") from None
            if self.runtime.config.trace_prompts:
                self.runtime.bus.broadcast(
                    f"agents:{self.state.agent_id}:trace", "llm_exchange", {
                        "model": model_key, "round": round_num,
                        "temperature": request.temperature,
                        "messages": messages,
                        "response": result.text, "error": result.error,
                        "input_tokens": result.input_tokens,
                        "output_tokens": result.output_tokens,
                        "latency_ms": result.latency_ms})
            if result.error == "context_overflow":
                # condense once and retry (reference: per_model_query.ex:93-124)
                await condensation_mod.condense_model_history(
                    self.state, model_key, engine, embed_many=self._embed_many())
                result = await asyncio.wait_for(
                    engine.generate(GenerateRequest(
                        **{**request.__dict__, "messages": _build()})),
                    timeout=self.runtime.config.generate_timeout_s)
            if not result.ok:
                raise RuntimeError(result.error or "query_failed")
            if result.cost:
                self.state.budget_spent += result.cost
                self.runtime.store.save_cost(
                    self.state.agent_id, self.state.task_id, model_key,
                    result.cost, category="model_query")
            # Model-initiated condensation request rides on the response
            from ..utils.jsonx import extract_json
            parsed = extract_json(result.text) or {}
            condense_n = parsed.get("condense")
            if isinstance(condense_n, int) and condense_n > 0:
                await condensation_mod.condense_model_history(
                    self.state, model_key, engine, n_oldest=condense_n,
                    embed_many=self._embed_many())
            return result.text
        return query_fn

    def _forbidden_actions(self) -> List[str]:
        """Actions mechanically blocked by grove hard rules (reference:
        consensus_handler.ex:294-333): excluded from prompts and grammar so
        models never burn rounds proposing them."""
        grove = self.state.grove or {}
        out: List[str] = []
        for rule in grove.get("hard_rules") or []:
            if rule.get("type") == "action_block":
                out.extend(a for a in rule.get("actions", [])
                           if isinstance(a, str))
        return out

    def _grammar_actions(self) -> List[str]:
        """Capability-gated action set for constrained decoding; the engine
        keeps the subset its grammar can template."""
        from ..governance import profiles as profiles_mod
        forbidden = set(self._forbidden_actions())
        return [a for a in
                profiles_mod.allowed_actions(self.state.capability_groups)
                if a not in forbidden]

    def _grammar_context(self) -> Dict[str, Any]:
        grove = self.state.grove or {}
        topology = grove.get("topology") or {}
        spawn_profile = None
        for edge in topology.get("edges") or []:
            inject = edge.get("auto_inject") or {}
            if isinstance(inject.get("profile"), str):
                spawn_profile = inject["profile"]
                break
        ctx: Dict[str, Any] = {
            "spawn_profile": spawn_profile or self.state.profile}
        # live child id so dismiss_child / adjust_budget decisions target a
        # real child instead of a placeholder
        if self.state.children:
            ctx["child_id"] = next(iter(self.state.children))
        wd = (grove.get("confinement") or {}).get("working_dir") \
            or self.runtime.config.default_working_dir
        ctx["file_read_path"] = f"{wd}/notes.txt"
        ctx["file_write_path"] = f"{wd}/scratch.txt"
        return ctx

    def _system_prompt(self) -> str:
        if self.state.cached_system_prompt is not None:
            return self.state.cached_system_prompt
        profile = None
        if self.state.profile and self.runtime.profiles.exists(self.state.profile):
            profile = self.runtime.profiles.resolve(self.state.profile)
        grove = self.state.grove or {}
        governance = grove.get("governance")
        governance_docs = []
        if isinstance(governance, dict):
            for name, doc in governance.items():
                if isinstance(doc, dict):
                    governance_docs.append({"name": name,
                                            "content": doc.get("content", ""),
                                            "priority": doc.get("priority", "normal")})
                elif isinstance(doc, str):
                    governance_docs.append({"name": name, "content": doc,
                                            "priority": "normal"})
        try:
            available_skills = self.skill_loader().list_metadata()
        except Exception:
            available_skills = []
        prompt = prompt_builder.build_system_prompt(
            role=self.state.role,
            cognitive_style=self.state.cognitive_style,
            output_style=self.state.output_style,
            delegation_strategy=self.state.delegation_strategy,
            profile=profile,
            constraints=self.state.constraints,
            capability_groups=self.state.capability_groups
            if self.state.profile is not None else None,
            forbidden_actions=self._forbidden_actions(),
            profile_names=self.runtime.profiles.names(),
            profile_catalog=self.runtime.profiles.catalog(),
            skills=self.state.active_skills,
            available_skills=available_skills,
            governance_docs=governance_docs,
            agent_id=self.state.agent_id,
            extra_system_prompt=self.state.system_prompt_fields.get("system_prompt"),
        )
        self.state.cached_system_prompt = prompt
        return prompt

    # -- decision execution -------------------------------------------------------
    async def _execute_decision(self, action: Dict[str, Any]) -> None:
        action_name = action.get("action")
        params = action.get("params") or {}
        wait = action.get("wait", False)
        # wait auto-correction for self-contained actions
        if wait is True and action_name in SELF_CONTAINED_ACTIONS:
            wait = False

        # record the decision in every model's history
        self.state.append_history(history_entry(
            "decision",
            {"action": action_name, "params": params,
             "reasoning": action.get("reasoning", ""), "wait": wait}))
        self._persist()

        # The wait action is pure timing — no executor dispatch; the merged
        # wait value below does all the work (reference: router.ex:309-311
        # holds a timer instead of executing anything).
        if action_name == "wait":
            # the wait action's own param is unified with the top-level field
            if "wait" in params:
                wait = params["wait"]
            if wait is False or wait == 0:
                self._trigger = True
            elif wait is True:
                self.state.status = "waiting"
                self.runtime.bus.state_change(self.state.agent_id, "waiting")
            else:
                self.state.wait_generation += 1
                self._wait_deadline = time.monotonic() + float(wait)
                self.state.status = "waiting"
                self.runtime.bus.state_change(self.state.agent_id, "waiting")
            return

        action_id = ids.action_id()
        self.state.pending_actions[action_id] = PendingAction(
            action_id=action_id, action=action_name, params=params, wait=wait)

        ctx = router_mod.ActionContext(
            agent=self, runtime=self.runtime, action_id=action_id,
            action=action_name, params=params)

        async def _dispatch():
            try:
                result = await router_mod.execute_action(ctx)
            except router_mod.ActionError as exc:
                result = {"error": exc.reason, "detail": _safe(exc.detail)}
                self.runtime.bus.action_event(
                    self.state.agent_id, "error", action_name, action_id,
                    {"error": exc.reason})
            except Exception as exc:  # noqa: BLE001 — action crash isolation
                logger.exception("action %s crashed", action_name)
                result = {"error": "action_crashed", "detail": str(exc)}
            await self.deliver({"type": "action_result", "action_id": action_id,
                                "action": action_name, "result": result})

        asyncio.ensure_future(_dispatch())

        # wait semantics
        if wait is False or wait == 0:
            self._trigger = True
        elif wait is True:
            self.state.status = "waiting"
            self.runtime.bus.state_change(self.state.agent_id, "waiting")
        else:
            self.state.wait_generation += 1
            self._wait_deadline = time.monotonic() + float(wait)
            self.state.status = "waiting"
            self.runtime.bus.state_change(self.state.agent_id, "waiting")

    # -- persistence ---------------------------------------------------------------
    def _persist(self) -> None:
        self.runtime.store.update_agent_state(
            self.state.agent_id, self.state.to_checkpoint())

    async def _terminate(self) -> None:
        self.state.status = "terminating"
        for cmd in list(self.shell_commands.values()):
            try:
                import os as _os
                _os.killpg(_os.getpgid(cmd.proc.pid), 9)
            except Exception:
                pass
        for conn in list(self.mcp_connections.values()):
            try:
                conn.proc.terminate()
                await asyncio.wait_for(conn.proc.wait(), 5)
            except Exception:
                pass
        # release the engine-side prefix-cache sessions (KV blocks) this
        # agent held; histories are persisted, so a future restore simply
        # re-prefills through the prefix cache
        for model in self.state.model_pool:
            try:
                engine = self.runtime.engines.engine_for(model)
            except KeyError:
                continue
            drop = getattr(engine, "drop_session", None)
            if drop is not None:
                drop(model, f"{self.state.agent_id}:{model}")
        self._persist()
        self.runtime.store.update_agent_status(self.state.agent_id, "terminated")
        self.runtime.registry.unregister(self.state.agent_id)
        self.runtime.bus.agent_terminated(self.state.agent_id, self.state.task_id)
        self._stopped.set()


def _result_text(result: Any) -> str:
    if isinstance(result, str):
        return result
    return json.dumps(result, default=str)


def _safe(value: Any) -> Any:
    try:
        json.dumps(value)
        return value
    except (TypeError, ValueError):
        return str(value)


def _format_incoming(message: Dict[str, Any]) -> str:
    mtype = message.get("type")
    if mtype == "user_message":
        return str(message.get("content", ""))
    if mtype == "agent_message":
        kind = "announcement" if message.get("announcement") else "message"
        return json.dumps({"from": message.get("from"), "kind": kind,
                           "content": message.get("content")}, default=str)
    if mtype == "shell_completed":
        return (f"[Async shell command {message.get('command_id')} completed "
                f"with exit code {message.get('exit_code')}; "
                "poll it with execute_shell check_id to read the output]")
    if mtype == "child_spawned":
        return (f"[Child agent {message.get('child_id')} is now running]")
    if mtype == "spawn_failed":
        return (f"[Spawning child {message.get('child_id')} FAILED: "
                f"{message.get('reason')}]")
    if mtype == "budget_adjusted":
        return f"[Your budget was adjusted to ${message.get('new_budget')}]"
    if mtype == "child_dismissed":
        return (f"[Child agent {message.get('child_id')} was dismissed: "
                f"{message.get('reason') or 'no reason given'}]")
    return json.dumps(message, default=str)
