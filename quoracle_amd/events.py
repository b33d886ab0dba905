"""Event bus: the framework's observability backbone.

Replaces the reference's Phoenix.PubSub + UI.EventHistory ring buffers
(reference: lib/quoracle/pubsub/agent_events.ex:62-309,
lib/quoracle/ui/event_history.ex).  In-process async pub/sub: topics map to
subscriber queues; sync callbacks are supported for tests and the CLI monitor.
Every user-visible thing the system does (agent lifecycle, logs, consensus
decisions with full clusters and temperatures, action start/complete/error,
state changes, budget/cost updates, messages) is a broadcast.

Topic scheme (parity with the reference):
  agents:lifecycle        spawn/terminate events
  agents:<id>:logs        per-agent log entries
  agents:<id>:state       status changes
  agents:<id>:todos       TODO list replacement
  agents:<id>:costs       cost records
  tasks:<id>:messages     inter-agent traffic for a task
"""

from __future__ import annotations

import asyncio
import time
from collections import defaultdict, deque
from dataclasses import dataclass, field
from typing import Any, Callable, Deque, Dict, List, Optional

LOG_HISTORY_LIMIT = 100
MESSAGE_HISTORY_LIMIT = 50


@dataclass
class Event:
    topic: str
    type: str
    payload: Dict[str, Any]
    ts: float = field(default_factory=time.time)


class EventBus:
    """Injected everywhere as a parameter (never a global) so tests isolate
    by construction, mirroring the reference's pubsub-injection style."""

    def __init__(self) -> None:
        self._queues: Dict[str, List[asyncio.Queue]] = defaultdict(list)
        self._callbacks: Dict[str, List[Callable[[Event], None]]] = defaultdict(list)
        self._history: Dict[str, Deque[Event]] = {}

    # -- subscription ---------------------------------------------------------
    def subscribe(self, topic: str, maxsize: int = 0) -> asyncio.Queue:
        queue: asyncio.Queue = asyncio.Queue(maxsize=maxsize)
        self._queues[topic].append(queue)
        return queue

    def unsubscribe(self, topic: str, queue: asyncio.Queue) -> None:
        try:
            self._queues[topic].remove(queue)
        except ValueError:
            pass

    def on(self, topic: str, callback: Callable[[Event], None]) -> None:
        self._callbacks[topic].append(callback)

    # -- broadcast --------------------------------------------------------------
    def broadcast(self, topic: str, type_: str, payload: Dict[str, Any]) -> Event:
        event = Event(topic=topic, type=type_, payload=payload)
        self._record_history(event)
        # "*" subscribers get the firehose (monitor UI event stream)
        for queue in list(self._queues.get("*", ())) if topic != "*" else []:
            try:
                queue.put_nowait(event)
            except asyncio.QueueFull:
                pass
        for cb in list(self._callbacks.get("*", ())) if topic != "*" else []:
            cb(event)
        for queue in list(self._queues.get(topic, ())):
            try:
                queue.put_nowait(event)
            except asyncio.QueueFull:
                pass  # bounded replay consumers may drop
        for cb in list(self._callbacks.get(topic, ())):
            cb(event)
        return event

    def _record_history(self, event: Event) -> None:
        limit = None
        if event.topic.endswith(":logs"):
            limit = LOG_HISTORY_LIMIT
        elif event.topic.endswith(":messages"):
            limit = MESSAGE_HISTORY_LIMIT
        if limit is None:
            return
        buf = self._history.get(event.topic)
        if buf is None or buf.maxlen != limit:
            buf = deque(self._history.get(event.topic) or (), maxlen=limit)
            self._history[event.topic] = buf
        buf.append(event)

    def history(self, topic: str) -> List[Event]:
        """Bounded replay buffer for monitor mounts."""
        return list(self._history.get(topic, ()))

    # -- typed helpers (AgentEvents parity) ------------------------------------
    def agent_spawned(self, agent_id: str, parent_id: Optional[str], task_id: str,
                      config: Optional[dict] = None) -> None:
        self.broadcast("agents:lifecycle", "agent_spawned", {
            "agent_id": agent_id, "parent_id": parent_id, "task_id": task_id,
            "config": config or {}})

    def agent_terminated(self, agent_id: str, task_id: str, reason: str = "normal") -> None:
        self.broadcast("agents:lifecycle", "agent_terminated", {
            "agent_id": agent_id, "task_id": task_id, "reason": reason})

    def log(self, agent_id: str, level: str, message: str,
            metadata: Optional[dict] = None) -> None:
        self.broadcast(f"agents:{agent_id}:logs", "log", {
            "agent_id": agent_id, "level": level, "message": message,
            "metadata": metadata or {}})

    def state_change(self, agent_id: str, status: str) -> None:
        self.broadcast(f"agents:{agent_id}:state", "state_change", {
            "agent_id": agent_id, "status": status})

    def consensus_decision(self, agent_id: str, decision: dict) -> None:
        self.broadcast(f"agents:{agent_id}:logs", "consensus_decision", {
            "agent_id": agent_id, **decision})

    def action_event(self, agent_id: str, phase: str, action: str,
                     action_id: str, payload: Optional[dict] = None) -> None:
        self.broadcast(f"agents:{agent_id}:logs", f"action_{phase}", {
            "agent_id": agent_id, "action": action, "action_id": action_id,
            **(payload or {})})

    def todos_updated(self, agent_id: str, items: list) -> None:
        self.broadcast(f"agents:{agent_id}:todos", "todos_updated", {
            "agent_id": agent_id, "items": items})

    def cost_recorded(self, agent_id: str, amount: float, detail: dict) -> None:
        self.broadcast(f"agents:{agent_id}:costs", "cost_recorded", {
            "agent_id": agent_id, "amount": amount, **detail})

    def task_message(self, task_id: str, payload: dict) -> None:
        self.broadcast(f"tasks:{task_id}:messages", "message", payload)
