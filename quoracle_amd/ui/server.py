"""Web monitor: REST + SSE dashboard over the orchestrator.

The MI355X-native replacement for the reference's Phoenix LiveView layer
(reference: lib/quoracle_web/ — DashboardLive 3-panel task tree / log
viewer / mailbox, SecretManagementLive tabs, health endpoint, router.ex).
Instead of LiveView websockets, a single-page dashboard polls the JSON API
and tails one Server-Sent-Events stream fed by the EventBus (events.py —
the PubSub equivalent), so everything the reference broadcasts is visible
here too.
"""

from __future__ import annotations

import asyncio
import json
from typing import Optional

from fastapi import FastAPI, HTTPException
from fastapi.responses import HTMLResponse, StreamingResponse
from pydantic import BaseModel


class CreateTaskBody(BaseModel):
    prompt: str
    profile: str = "default"
    budget_limit: Optional[float] = None
    global_context: Optional[str] = None
    role: Optional[str] = None
    grove: Optional[str] = None      # grove name under the groves dir
    success_criteria: Optional[str] = None
    immediate_context: Optional[str] = None
    approach_guidance: Optional[str] = None
    skills: Optional[list] = None
    cognitive_style: Optional[str] = None
    output_style: Optional[str] = None
    delegation_strategy: Optional[str] = None


class MessageBody(BaseModel):
    content: str
    target_agent: Optional[str] = None


class SecretBody(BaseModel):
    name: str
    value: str
    description: str = ""


class ProfileBody(BaseModel):
    name: str
    description: str = ""
    model_pool: list
    capability_groups: list = []
    max_refinement_rounds: int = 4
    force_reflection: bool = False


def create_app(manager) -> FastAPI:
    runtime = manager.runtime
    app = FastAPI(title="quoracle-amd monitor")

    @app.get("/health")
    def health():
        return {"status": "ok", "agents": len(runtime.registry.all_ids())}

    # -- tasks ---------------------------------------------------------------

    @app.get("/api/tasks")
    def tasks():
        return runtime.store.list_tasks()

    @app.post("/api/tasks")
    async def create_task(body: CreateTaskBody):
        from ..tasks.manager import TaskError
        grove = None
        if body.grove:
            from ..governance import groves as groves_mod
            import os as _os
            base = runtime.config.groves_dir or "groves"
            # grove is a NAME, never a path: '../' or an absolute path would
            # load a GROVE.md outside groves_dir and anchor confinement there
            target = _os.path.realpath(_os.path.join(base, body.grove))
            if (_os.sep in body.grove or "/" in body.grove
                    or body.grove in (".", "..")
                    or not (target + _os.sep).startswith(
                        _os.path.realpath(base) + _os.sep)):
                raise HTTPException(400, "bad_grove: invalid grove name")
            try:
                grove = groves_mod.load_grove(target)
            except (OSError, ValueError) as exc:
                raise HTTPException(400, f"bad_grove: {exc}")
        try:
            return await manager.create_task(
                body.prompt, body.profile, budget_limit=body.budget_limit,
                global_context=body.global_context, role=body.role,
                grove=grove, success_criteria=body.success_criteria,
                immediate_context=body.immediate_context,
                approach_guidance=body.approach_guidance,
                skills=body.skills, cognitive_style=body.cognitive_style,
                output_style=body.output_style,
                delegation_strategy=body.delegation_strategy)
        except TaskError as exc:
            raise HTTPException(400, exc.reason)

    @app.get("/api/groves")
    def groves():
        from ..governance import groves as groves_mod
        return groves_mod.list_groves(runtime.config.groves_dir or "groves")

    @app.get("/api/tasks/{task_id}/tree")
    def task_tree(task_id: str):
        agents = runtime.store.agents_for_task(task_id)
        nodes = []
        for a in agents:
            entry = runtime.registry.lookup(a["agent_id"])
            status = entry.actor.state.status if entry else a.get("status")
            nodes.append({"agent_id": a["agent_id"],
                          "parent_id": a.get("parent_id"),
                          "status": status,
                          "alive": entry is not None})
        return {"task_id": task_id, "agents": nodes}

    @app.get("/api/tasks/{task_id}/export")
    def task_export(task_id: str):
        """Full machine-readable transcript: task row, agent rows (with
        checkpointed state), logs, messages, cost rollup — the JSON
        counterpart of the show-task CLI renderer."""
        task = runtime.store.get_task(task_id)
        if task is None:
            raise HTTPException(404, "unknown task")
        agents = runtime.store.agents_for_task(task_id)
        return {
            "task": task,
            "agents": [{
                **a,
                "logs": runtime.store.logs_for_agent(a["agent_id"],
                                                     limit=200),
                "costs": runtime.store.cost_rollup(a["agent_id"]),
            } for a in agents],
            "messages": runtime.store.messages_for_task(task_id),
        }

    @app.get("/api/tasks/{task_id}/messages")
    def task_messages(task_id: str):
        return runtime.store.messages_for_task(task_id)

    @app.post("/api/tasks/{task_id}/message")
    async def send_message(task_id: str, body: MessageBody):
        from ..tasks.manager import TaskError
        try:
            await manager.send_user_message(task_id, body.content,
                                            agent_id=body.target_agent)
        except TaskError as exc:
            raise HTTPException(400, exc.reason)
        return {"ok": True}

    @app.post("/api/tasks/{task_id}/pause")
    async def pause(task_id: str):
        await manager.pause_task(task_id)
        return {"ok": True}

    @app.post("/api/tasks/{task_id}/restore")
    async def restore(task_id: str):
        return await manager.restore_task(task_id)

    @app.delete("/api/tasks/{task_id}")
    async def delete(task_id: str):
        await manager.delete_task(task_id)
        return {"ok": True}

    # -- agents --------------------------------------------------------------

    @app.get("/api/agents/{agent_id}/logs")
    def agent_logs(agent_id: str, limit: int = 100):
        return runtime.store.logs_for_agent(agent_id, limit=limit)

    @app.get("/api/agents/{agent_id}/costs")
    def agent_costs(agent_id: str):
        return runtime.store.cost_rollup(agent_id)

    @app.get("/api/agents/{agent_id}/state")
    def agent_state(agent_id: str):
        entry = runtime.registry.lookup(agent_id)
        if entry is None:
            row = runtime.store.get_agent(agent_id)
            if row is None:
                raise HTTPException(404, "unknown agent")
            return {"agent_id": agent_id, "alive": False,
                    "status": row.get("status")}
        st = entry.actor.state
        return {"agent_id": agent_id, "alive": True, "status": st.status,
                "model_pool": st.model_pool, "todos": st.todos,
                "children": list(st.children),
                "pending_actions": list(st.pending_actions),
                "history_lengths": {m: len(h) for m, h in
                                    st.model_histories.items()}}

    @app.get("/api/agents/{agent_id}/history/{kind}")
    def agent_event_history(agent_id: str, kind: str):
        """Ring-buffer replay (the reference's EventHistory mount replay):
        kind in {logs, state, trace, todos, costs}."""
        events = runtime.bus.history(f"agents:{agent_id}:{kind}")
        return [{"type": e.type, "payload": e.payload, "ts": e.ts}
                for e in events]

    # -- config surfaces (SecretManagementLive equivalents) ------------------

    @app.get("/api/profiles")
    def profiles():
        return runtime.store.list_profiles()

    @app.post("/api/profiles")
    def put_profile(body: ProfileBody):
        from ..governance.profiles import Profile
        runtime.profiles.put(Profile(
            name=body.name, description=body.description,
            model_pool=body.model_pool,
            capability_groups=body.capability_groups,
            max_refinement_rounds=body.max_refinement_rounds,
            force_reflection=body.force_reflection))
        return {"ok": True}

    @app.post("/api/admin/reload")
    def admin_reload():
        """Hot-reload analog of the reference's `mix quoracle.reload` /
        `llm_db.hot_reload` dev tasks: groves, skills and profiles are
        already read from disk/DB per use, so the only cached state is
        each live agent's system prompt — invalidate it so prompt-file /
        grove / skill edits take effect on the next consensus cycle."""
        invalidated = 0
        for agent_id in runtime.registry.all_ids():
            entry = runtime.registry.lookup(agent_id)
            if entry is not None:
                entry.actor.state.cached_system_prompt = None
                invalidated += 1
        return {"ok": True, "agents_invalidated": invalidated,
                "hot_sources": ["groves (per-use disk read)",
                                "skills (per-use disk read)",
                                "profiles (DB-backed)"]}

    @app.get("/api/secrets")
    def secrets():
        return {"names": runtime.vault.names()}

    @app.post("/api/secrets")
    def put_secret(body: SecretBody):
        runtime.vault.put(body.name, body.value, body.description)
        return {"ok": True}

    @app.get("/metrics")
    def metrics():
        """Prometheus-format counters (the reference's Telemetry /
        LiveDashboard metrics layer, reference: quoracle_web/telemetry.ex)."""
        from fastapi.responses import PlainTextResponse
        lines = []

        def emit(name, value, help_=""):
            if help_:
                lines.append(f"# HELP {name} {help_}")
            lines.append(f"# TYPE {name} gauge")
            lines.append(f"{name} {value}")

        emit("quoracle_agents_alive", len(runtime.registry.all_ids()),
             "live agents in the registry")
        tasks_ = runtime.store.list_tasks()
        emit("quoracle_tasks_total", len(tasks_))
        emit("quoracle_tasks_running",
             sum(1 for t in tasks_ if t.get("status") == "running"))
        stats_fn = getattr(runtime.engines, "engine_stats", None)
        for key, val in (stats_fn() if stats_fn else {}).items():
            emit(f"quoracle_engine_{key}", val)
        total = runtime.store.total_cost(
            [a for t in tasks_ for a in
             [r["agent_id"] for r in
              runtime.store.agents_for_task(t["task_id"])]])
        emit("quoracle_cost_usd_total", round(total, 6))
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/api/engine/stats")
    def engine_stats():
        stats_fn = getattr(runtime.engines, "engine_stats", None)
        return stats_fn() if stats_fn else {}

    # -- event stream (PubSub tail) ------------------------------------------

    @app.get("/api/events")
    async def events():
        queue = runtime.bus.subscribe("*", maxsize=500)

        async def stream():
            try:
                while True:
                    try:
                        ev = await asyncio.wait_for(queue.get(), timeout=15)
                        data = json.dumps({"topic": ev.topic, "type": ev.type,
                                           "payload": ev.payload,
                                           "ts": ev.ts}, default=str)
                        yield f"data: {data}\n\n"
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                runtime.bus.unsubscribe("*", queue)

        return StreamingResponse(stream(), media_type="text/event-stream")

    @app.get("/", response_class=HTMLResponse)
    def dashboard():
        return _DASHBOARD_HTML

    return app


_DASHBOARD_HTML = """<!doctype html>
<html><head><title>quoracle-amd</title><style>
body{font-family:monospace;background:#111;color:#ddd;margin:0;display:grid;
grid-template-columns:300px 1fr 1fr;height:100vh}
.panel{overflow:auto;border-right:1px solid #333;padding:8px}
h2{font-size:13px;color:#7af;margin:4px 0}
.agent{margin-left:12px;cursor:pointer}.agent.busy{color:#fc6}
.agent.waiting{color:#6cf}.agent.dead{color:#666}
.log{font-size:11px;border-bottom:1px solid #222;padding:2px}
.log.error{color:#f66}.log.warning{color:#fc6}
.msg{font-size:11px;padding:3px;border-bottom:1px solid #222}
textarea,input,select{width:95%;background:#222;color:#ddd;border:1px solid #444}
button{background:#247;color:#fff;border:0;padding:4px 10px;cursor:pointer}
</style></head><body>
<div class=panel id=left><h2>engine</h2><div id=estats style="font-size:11px;color:#9c9"></div>
<h2>tasks</h2><div id=tasks></div>
<h2>new task</h2><textarea id=prompt rows=3></textarea>
<input id=profile value=default placeholder=profile>
<button onclick=createTask()>create</button></div>
<div class=panel><h2>agent logs <span id=sel></span></h2><div id=logs></div></div>
<div class=panel><h2>mailbox / events</h2>
<textarea id=usermsg rows=2 placeholder="message to task"></textarea>
<button onclick=sendMsg()>send</button><div id=events></div></div>
<script>
let selTask=null, selAgent=null;
async function j(u,opt){const r=await fetch(u,opt);return r.json()}
async function refresh(){
 try{const es=await j('/api/engine/stats');
  document.getElementById('estats').textContent=
   Object.entries(es).map(([k,v])=>`${k}=${v}`).join('  ')||'no engine stats';
 }catch(e){}
 const tasks=await j('/api/tasks');
 // agent-generated strings (ids, statuses, log text) reach this DOM: build
 // nodes with textContent, never innerHTML, so they cannot inject markup
 const tdiv=document.getElementById('tasks');tdiv.replaceChildren();
 for(const t of tasks){
  const box=document.createElement('div');
  const b=document.createElement('b');
  b.textContent=`[${t.status}] ${t.task_id}`;
  b.onclick=()=>{selTask=t.task_id};
  box.appendChild(b);
  const tree=await j(`/api/tasks/${encodeURIComponent(t.task_id)}/tree`);
  for(const a of tree.agents){
   const row=document.createElement('div');
   row.className='agent '+(a.alive?a.status:'dead');
   row.textContent=`${a.agent_id} (${a.status})`;
   row.onclick=()=>pick(t.task_id,a.agent_id);
   box.appendChild(row)}
  tdiv.appendChild(box)}
 if(selAgent){
  const logs=await j(`/api/agents/${encodeURIComponent(selAgent)}/logs`);
  const ldiv=document.getElementById('logs');ldiv.replaceChildren();
  for(const l of logs){
   const row=document.createElement('div');
   row.className='log '+(l.level||'');
   row.textContent=`${l.event_type||''} ${l.message||''}`;
   ldiv.appendChild(row)}
  document.getElementById('sel').textContent=selAgent}}
function pick(t,a){selTask=t;selAgent=a;refresh()}
async function createTask(){await j('/api/tasks',{method:'POST',
 headers:{'Content-Type':'application/json'},
 body:JSON.stringify({prompt:document.getElementById('prompt').value,
 profile:document.getElementById('profile').value})});refresh()}
async function sendMsg(){if(!selTask)return;
 await j(`/api/tasks/${selTask}/message`,{method:'POST',
 headers:{'Content-Type':'application/json'},
 body:JSON.stringify({content:document.getElementById('usermsg').value})})}
const es=new EventSource('/api/events');
es.onmessage=e=>{const d=JSON.parse(e.data);
 const el=document.getElementById('events');
 el.insertAdjacentHTML('afterbegin',
  `<div class=msg>[${d.type}] ${d.topic} ${JSON.stringify(d.payload).slice(0,200)}</div>`);
 while(el.children.length>200)el.removeChild(el.lastChild)};
setInterval(refresh,2000);refresh();
</script></body></html>"""
