"""The 22 action executors.

Each executor is `async def (ctx: ActionContext) -> dict`.  Parity map to the
reference (reference: lib/quoracle/actions/*.ex — spawn.ex, shell.ex, web.ex,
api.ex, mcp.ex, send_message.ex, orient.ex, todo.ex, wait.ex,
dismiss_child.ex, adjust_budget.ex, record_cost.ex, generate_secret.ex,
search_secrets.ex, file_read.ex, file_write.ex, learn_skills.ex,
create_skill.ex, batch_sync.ex, batch_async.ex, answer_engine.ex,
generate_images.ex), rebuilt natively: asyncio subprocesses instead of BEAM
Ports, aiohttp instead of Req, asyncio task groups instead of Task.Supervisor.
"""

from __future__ import annotations

import asyncio
import json
import os
import re
import time
from typing import Any, Dict, List, Optional

from ..budget import tracker as budget_mod
from ..engine.api import GenerateRequest
from ..governance import groves as groves_mod
from ..governance import skills as skills_mod
from ..governance.security import generate_secret_value
from ..utils import ids

SYNC_THRESHOLD_S = 0.1  # smart-mode boundary (reference: shell.ex:13)


def _err(reason: str, **extra: Any) -> Dict[str, Any]:
    out = {"error": reason}
    out.update(extra)
    return out


# ---------------------------------------------------------------------------
# wait / orient / todo
# ---------------------------------------------------------------------------

async def execute_wait(ctx) -> Dict[str, Any]:
    # Wait *semantics* (timers, idling) are owned by the agent loop via the
    # merged wait value; the action itself just acknowledges.
    return {"status": "waiting", "wait": ctx.params.get("wait", True)}


async def execute_orient(ctx) -> Dict[str, Any]:
    # Structured self-reflection: the value is the entry in history.
    return {"status": "oriented", "assessment": dict(ctx.params)}


async def execute_todo(ctx) -> Dict[str, Any]:
    items = ctx.params.get("items", [])
    ctx.agent.state.todos = items
    ctx.runtime.bus.todos_updated(ctx.agent.state.agent_id, items)
    return {"status": "updated", "count": len(items)}


# ---------------------------------------------------------------------------
# send_message
# ---------------------------------------------------------------------------

async def execute_send_message(ctx) -> Dict[str, Any]:
    state = ctx.agent.state
    registry = ctx.runtime.registry
    to = ctx.params["to"]
    content = ctx.params["content"]

    if to == "parent":
        targets = [state.parent_id] if state.parent_id else []
        if not targets:
            # root agent: route to the task's user mailbox
            ctx.runtime.bus.task_message(state.task_id, {
                "from": state.agent_id, "to": "user", "content": content})
            ctx.runtime.store.save_message(state.task_id, state.agent_id,
                                           "user", content)
            return {"status": "sent", "delivered_to": ["user"]}
    elif to == "children":
        targets = registry.children_of(state.agent_id)
    elif to == "announcement":
        targets = registry.descendants_of(state.agent_id)
    elif isinstance(to, list):
        targets = list(to)
    else:
        return _err("invalid_recipient")

    delivered = []
    for target in targets:
        entry = registry.lookup(target)
        if entry is None:
            continue
        await entry.actor.deliver({
            "type": "agent_message", "from": state.agent_id, "content": content,
            "announcement": to == "announcement"})
        ctx.runtime.store.save_message(state.task_id, state.agent_id, target, content)
        ctx.runtime.bus.task_message(state.task_id, {
            "from": state.agent_id, "to": target, "content": content})
        delivered.append(target)
    return {"status": "sent", "delivered_to": delivered}


# ---------------------------------------------------------------------------
# execute_shell — smart mode (reference: shell.ex:37-177, router shell state)
# ---------------------------------------------------------------------------

class ShellCommand:
    def __init__(self, command_id: str, command: str, proc):
        self.command_id = command_id
        self.command = command
        self.proc = proc
        self.stdout = b""
        self.stderr = b""
        self.exit_code: Optional[int] = None
        self.started = time.monotonic()
        self.task: Optional[asyncio.Task] = None


async def execute_shell(ctx) -> Dict[str, Any]:
    params = ctx.params
    agent = ctx.agent
    if params.get("check_id"):
        return await _shell_check(ctx, params["check_id"],
                                  terminate=bool(params.get("terminate")))

    command = params["command"]
    working_dir = params.get("working_dir") or ctx.runtime.config.default_working_dir
    grove = agent.state.grove or {}
    groves_mod.check_shell_command(command, grove.get("hard_rules"), ctx.skill_name)
    groves_mod.check_shell_working_dir(
        working_dir, grove.get("confinement"), ctx.skill_name,
        grove.get("confinement_mode"))
    if not os.path.isdir(working_dir):
        return _err("invalid_working_dir", working_dir=working_dir)

    proc = await asyncio.create_subprocess_exec(
        "/bin/bash", "-c", command, cwd=working_dir,
        stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.PIPE,
        start_new_session=True)
    command_id = ids.command_id()
    cmd = ShellCommand(command_id, command, proc)

    async def _collect():
        cmd.stdout, cmd.stderr = await proc.communicate()
        cmd.exit_code = proc.returncode

    collect_task = asyncio.ensure_future(_collect())
    cmd.task = collect_task
    try:
        await asyncio.wait_for(asyncio.shield(collect_task), SYNC_THRESHOLD_S)
        return {"sync": True, "stdout": cmd.stdout.decode(errors="replace"),
                "stderr": cmd.stderr.decode(errors="replace"),
                "exit_code": cmd.exit_code}
    except asyncio.TimeoutError:
        agent.shell_commands[command_id] = cmd

        async def _notify_on_completion():
            await collect_task
            await agent.deliver({
                "type": "shell_completed", "command_id": command_id,
                "exit_code": cmd.exit_code})

        asyncio.ensure_future(_notify_on_completion())
        return {"async": True, "command_id": command_id,
                "status": "running",
                "note": "use execute_shell with check_id to poll"}


async def _shell_check(ctx, check_id: str, *, terminate: bool) -> Dict[str, Any]:
    cmd: Optional[ShellCommand] = ctx.agent.shell_commands.get(check_id)
    if cmd is None:
        return _err("unknown_command_id", command_id=check_id)
    if terminate and cmd.exit_code is None:
        try:
            os.killpg(os.getpgid(cmd.proc.pid), 9)
        except (ProcessLookupError, PermissionError):
            cmd.proc.kill()
        await cmd.task
        ctx.agent.shell_commands.pop(check_id, None)
        return {"command_id": check_id, "status": "terminated",
                "stdout": cmd.stdout.decode(errors="replace"),
                "stderr": cmd.stderr.decode(errors="replace")}
    if cmd.exit_code is None:
        return {"command_id": check_id, "status": "running",
                "elapsed_s": time.monotonic() - cmd.started}
    ctx.agent.shell_commands.pop(check_id, None)
    return {"command_id": check_id, "status": "completed",
            "exit_code": cmd.exit_code,
            "stdout": cmd.stdout.decode(errors="replace"),
            "stderr": cmd.stderr.decode(errors="replace")}


# ---------------------------------------------------------------------------
# file_read / file_write
# ---------------------------------------------------------------------------

async def execute_file_read(ctx) -> Dict[str, Any]:
    params = ctx.params
    path = params["path"]
    grove = ctx.agent.state.grove or {}
    groves_mod.check_file_access(path, "read", grove.get("confinement"),
                                 ctx.skill_name, grove.get("confinement_mode"))
    if not os.path.isfile(path):
        return _err("file_not_found", path=path)
    offset = max(1, int(params.get("offset", 1)))
    limit = params.get("limit")
    with open(path, "r", errors="replace") as f:
        lines = f.readlines()
    selected = lines[offset - 1: offset - 1 + limit if limit else None]
    numbered = "".join(f"{offset + i:6d}\t{line}" for i, line in enumerate(selected))
    return {"path": path, "content": numbered, "total_lines": len(lines)}


async def execute_file_write(ctx) -> Dict[str, Any]:
    params = ctx.params
    path = params["path"]
    mode = params["mode"]
    grove = ctx.agent.state.grove or {}
    groves_mod.check_file_access(path, "write", grove.get("confinement"),
                                 ctx.skill_name, grove.get("confinement_mode"))
    if mode == "write":
        content = params.get("content")
        if content is None:
            return _err("missing_content")
        groves_mod.validate_file_write(grove, path, content)
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            f.write(content)
        return {"path": path, "status": "written", "bytes": len(content)}
    # edit mode
    old = params.get("old_string")
    new = params.get("new_string")
    if old is None or new is None:
        return _err("missing_edit_strings")
    if not os.path.isfile(path):
        return _err("file_not_found", path=path)
    with open(path, "r") as f:
        text = f.read()
    count = text.count(old)
    if count == 0:
        return _err("old_string_not_found", path=path)
    if params.get("replace_all"):
        new_text = text.replace(old, new)
        replaced = count
    else:
        new_text = text.replace(old, new, 1)
        replaced = 1
    groves_mod.validate_file_write(grove, path, new_text)
    with open(path, "w") as f:
        f.write(new_text)
    return {"path": path, "status": "edited", "replacements": replaced}


# ---------------------------------------------------------------------------
# fetch_web / call_api — aiohttp with an injectable transport for tests
# ---------------------------------------------------------------------------

_TAG_STRIP = re.compile(r"<(script|style)[^>]*>.*?</\1>", re.DOTALL | re.IGNORECASE)
_BLOCK_TAGS = re.compile(r"</?(p|div|br|li|tr|h[1-6])[^>]*>", re.IGNORECASE)
_ALL_TAGS = re.compile(r"<[^>]+>")
_HEADING = re.compile(r"<h([1-6])[^>]*>(.*?)</h\1>", re.DOTALL | re.IGNORECASE)
_LINK = re.compile(r'<a[^>]*href="([^"]*)"[^>]*>(.*?)</a>', re.DOTALL | re.IGNORECASE)

MAX_WEB_CONTENT = 100_000


def html_to_markdown(html: str) -> str:
    text = _TAG_STRIP.sub("", html)
    text = _HEADING.sub(lambda m: "\n" + "#" * int(m.group(1)) + " " + m.group(2) + "\n", text)
    text = _LINK.sub(lambda m: f"[{m.group(2).strip()}]({m.group(1)})", text)
    text = _BLOCK_TAGS.sub("\n", text)
    text = _ALL_TAGS.sub("", text)
    text = re.sub(r"\n{3,}", "\n\n", text)
    return text.strip()


def _private_host(host: str) -> bool:
    import ipaddress
    if host in ("localhost",):
        return True
    try:
        return ipaddress.ip_address(host).is_private or \
            ipaddress.ip_address(host).is_loopback
    except ValueError:
        return False


async def _http_request(ctx, method: str, url: str, *, headers=None, data=None,
                        timeout_s: float = 30.0, follow_redirects: bool = True):
    """Returns (status, headers, body_text).  Tests inject
    runtime.extras['http_fn']; production uses aiohttp."""
    http_fn = ctx.runtime.extras.get("http_fn")
    if http_fn is not None:
        return await http_fn(method, url, headers=headers, data=data)
    import aiohttp
    timeout = aiohttp.ClientTimeout(total=timeout_s)
    async with aiohttp.ClientSession(timeout=timeout) as session:
        async with session.request(method, url, headers=headers, data=data,
                                   allow_redirects=follow_redirects) as resp:
            body = await resp.text()
            return resp.status, dict(resp.headers), body


async def execute_fetch_web(ctx) -> Dict[str, Any]:
    from urllib.parse import urlparse
    params = ctx.params
    url = params["url"]
    parsed = urlparse(url)
    if parsed.scheme not in ("http", "https"):
        return _err("invalid_url_scheme", url=url)
    if params.get("security_check") and _private_host(parsed.hostname or ""):
        return _err("ssrf_blocked", url=url)
    headers = {}
    if params.get("user_agent"):
        headers["User-Agent"] = params["user_agent"]
    try:
        status, _resp_headers, body = await _http_request(
            ctx, "GET", url, headers=headers,
            timeout_s=float(params.get("timeout", 30)),
            follow_redirects=params.get("follow_redirects", True))
    except Exception as exc:  # noqa: BLE001 — network failure is a result
        return _err("fetch_failed", detail=str(exc))
    markdown = html_to_markdown(body)
    truncated = len(markdown) > MAX_WEB_CONTENT
    return {"url": url, "status": status,
            "content": markdown[:MAX_WEB_CONTENT], "truncated": truncated}


def _auth_headers(auth: Optional[Dict[str, Any]]) -> Dict[str, str]:
    if not auth:
        return {}
    auth_type = (auth.get("auth_type") or "").lower()
    if auth_type == "bearer":
        return {"Authorization": f"Bearer {auth.get('token', '')}"}
    if auth_type == "basic":
        import base64
        creds = auth.get("credentials") or {}
        raw = f"{creds.get('username', '')}:{creds.get('password', '')}"
        return {"Authorization": "Basic " + base64.b64encode(raw.encode()).decode()}
    if auth_type == "api_key":
        header = auth.get("header", "X-API-Key")
        return {header: auth.get("token", "")}
    if auth_type == "oauth2":
        return {"Authorization": f"Bearer {auth.get('token', '')}"}
    return {}


async def execute_call_api(ctx) -> Dict[str, Any]:
    params = ctx.params
    api_type = params["api_type"]
    url = params["url"]
    headers = dict(params.get("headers") or {})
    headers.update(_auth_headers(params.get("auth")))
    timeout_s = float(params.get("timeout", 30))

    if api_type == "rest":
        method = params.get("method", "GET")
        data = None
        if params.get("body") is not None:
            data = json.dumps(params["body"]) if not isinstance(params["body"], str) \
                else params["body"]
            headers.setdefault("Content-Type", "application/json")
        if params.get("query_params"):
            from urllib.parse import urlencode
            sep = "&" if "?" in url else "?"
            url = url + sep + urlencode(params["query_params"])
    elif api_type == "graphql":
        method = "POST"
        data = json.dumps({"query": params.get("query"),
                           "variables": params.get("variables") or {}})
        headers.setdefault("Content-Type", "application/json")
    else:  # jsonrpc
        method = "POST"
        data = json.dumps({"jsonrpc": "2.0",
                           "method": params.get("rpc_method"),
                           "params": params.get("rpc_params"),
                           "id": params.get("rpc_id") or ids.request_id()})
        headers.setdefault("Content-Type", "application/json")

    max_body = int(params.get("max_body_size", 5 * 1024 * 1024))
    if data is not None and len(data) > max_body:
        return _err("body_too_large")
    try:
        status, _resp_headers, body = await _http_request(
            ctx, method, url, headers=headers, data=data, timeout_s=timeout_s)
    except Exception as exc:  # noqa: BLE001
        return _err("api_call_failed", detail=str(exc))
    try:
        parsed_body = json.loads(body)
    except (json.JSONDecodeError, ValueError):
        parsed_body = body
    return {"status": status, "body": parsed_body}


# ---------------------------------------------------------------------------
# call_mcp — stdio JSON-RPC client (line-delimited)
# ---------------------------------------------------------------------------

class MCPConnection:
    def __init__(self, connection_id: str, proc):
        self.connection_id = connection_id
        self.proc = proc
        self._next_id = 0
        self.tools: List[dict] = []

    async def rpc(self, method: str, params: Optional[dict] = None,
                  timeout_s: float = 30.0) -> Any:
        self._next_id += 1
        msg = {"jsonrpc": "2.0", "id": self._next_id, "method": method,
               "params": params or {}}
        self.proc.stdin.write((json.dumps(msg) + "\n").encode())
        await self.proc.stdin.drain()
        while True:
            line = await asyncio.wait_for(self.proc.stdout.readline(), timeout_s)
            if not line:
                raise ConnectionError("mcp server closed")
            try:
                reply = json.loads(line)
            except json.JSONDecodeError:
                continue
            if reply.get("id") == self._next_id:
                if "error" in reply:
                    raise RuntimeError(json.dumps(reply["error"]))
                return reply.get("result")

    def notify(self, method: str, params: Optional[dict] = None) -> None:
        msg = {"jsonrpc": "2.0", "method": method, "params": params or {}}
        self.proc.stdin.write((json.dumps(msg) + "\n").encode())

    async def close(self) -> None:
        self.proc.terminate()
        try:
            await asyncio.wait_for(self.proc.wait(), 5)
        except (asyncio.TimeoutError, ProcessLookupError):
            self.proc.kill()


class MCPHttpConnection:
    """Streamable-HTTP MCP client (reference: actions/mcp.ex:48-104
    supports stdio AND streamable HTTP).  JSON-RPC over POST; the server
    may answer application/json or a text/event-stream carrying the JSON
    message; an Mcp-Session-Id response header is echoed on later calls
    and released with DELETE on close."""

    def __init__(self, connection_id: str, url: str):
        import httpx
        self.connection_id = connection_id
        self.url = url
        self.proc = None
        self._next_id = 0
        self.tools: List[dict] = []
        self._session_id: Optional[str] = None
        self._client = httpx.AsyncClient()

    def _headers(self) -> dict:
        h = {"Content-Type": "application/json",
             "Accept": "application/json, text/event-stream"}
        if self._session_id:
            h["Mcp-Session-Id"] = self._session_id
        return h

    @staticmethod
    def _from_sse(text: str, want_id: int) -> Any:
        for line in text.splitlines():
            if not line.startswith("data:"):
                continue
            try:
                msg = json.loads(line[5:].strip())
            except json.JSONDecodeError:
                continue
            if msg.get("id") == want_id:
                return msg
        raise ConnectionError("no matching message in SSE stream")

    async def rpc(self, method: str, params: Optional[dict] = None,
                  timeout_s: float = 30.0) -> Any:
        self._next_id += 1
        msg = {"jsonrpc": "2.0", "id": self._next_id, "method": method,
               "params": params or {}}
        resp = await self._client.post(self.url, json=msg,
                                       headers=self._headers(),
                                       timeout=timeout_s)
        if resp.status_code >= 400:
            raise ConnectionError(f"http {resp.status_code}: {resp.text[:200]}")
        sid = resp.headers.get("mcp-session-id")
        if sid:
            self._session_id = sid
        ctype = resp.headers.get("content-type", "")
        if "text/event-stream" in ctype:
            reply = self._from_sse(resp.text, self._next_id)
        else:
            reply = resp.json()
        if "error" in reply:
            raise RuntimeError(json.dumps(reply["error"]))
        return reply.get("result")

    def notify(self, method: str, params: Optional[dict] = None) -> None:
        msg = {"jsonrpc": "2.0", "method": method, "params": params or {}}
        asyncio.ensure_future(self._client.post(
            self.url, json=msg, headers=self._headers(), timeout=10.0))

    async def close(self) -> None:
        try:
            if self._session_id:
                await self._client.delete(self.url, headers=self._headers(),
                                          timeout=5.0)
        except Exception:  # noqa: BLE001 — session release is best-effort
            pass
        await self._client.aclose()


async def execute_call_mcp(ctx) -> Dict[str, Any]:
    params = ctx.params
    agent = ctx.agent
    timeout_s = float(params.get("timeout", 30000)) / 1000.0

    if params.get("connection_id"):
        conn: Optional[MCPConnection] = agent.mcp_connections.get(params["connection_id"])
        if conn is None:
            return _err("unknown_connection_id")
        if params.get("terminate"):
            await conn.close()
            agent.mcp_connections.pop(params["connection_id"], None)
            return {"connection_id": conn.connection_id, "status": "terminated"}
        tool = params.get("tool")
        if not tool:
            return _err("missing_tool")
        try:
            result = await conn.rpc("tools/call",
                                    {"name": tool,
                                     "arguments": params.get("arguments") or {}},
                                    timeout_s)
        except Exception as exc:  # noqa: BLE001
            return _err("mcp_call_failed", detail=str(exc))
        return {"connection_id": conn.connection_id, "tool": tool, "result": result}

    # connect
    transport = params.get("transport")
    if transport == "stdio":
        command = params.get("command")
        if not command:
            return _err("missing_command")
        proc = await asyncio.create_subprocess_exec(
            "/bin/bash", "-c", command,
            cwd=params.get("cwd") or ctx.runtime.config.default_working_dir,
            stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.DEVNULL)
        conn = MCPConnection(ids.connection_id(), proc)
        try:
            await conn.rpc("initialize", {
                "protocolVersion": "2024-11-05",
                "capabilities": {},
                "clientInfo": {"name": "quoracle-amd", "version": "0.1"}},
                timeout_s)
            conn.notify("notifications/initialized")
            tools_result = await conn.rpc("tools/list", {}, timeout_s)
            conn.tools = (tools_result or {}).get("tools", [])
        except Exception as exc:  # noqa: BLE001
            await conn.close()
            return _err("mcp_connect_failed", detail=str(exc))
        agent.mcp_connections[conn.connection_id] = conn
        return {"connection_id": conn.connection_id,
                "tools": conn.tools, "status": "connected"}
    if transport == "http":
        url = params.get("url")
        if not url:
            return _err("missing_url")
        conn = MCPHttpConnection(ids.connection_id(), url)
        try:
            await conn.rpc("initialize", {
                "protocolVersion": "2024-11-05",
                "capabilities": {},
                "clientInfo": {"name": "quoracle-amd", "version": "0.1"}},
                timeout_s)
            conn.notify("notifications/initialized")
            tools_result = await conn.rpc("tools/list", {}, timeout_s)
            conn.tools = (tools_result or {}).get("tools", [])
        except Exception as exc:  # noqa: BLE001
            await conn.close()
            return _err("mcp_connect_failed", detail=str(exc))
        agent.mcp_connections[conn.connection_id] = conn
        return {"connection_id": conn.connection_id,
                "tools": conn.tools, "status": "connected"}
    return _err("invalid_transport")


# ---------------------------------------------------------------------------
# answer_engine / generate_images
# ---------------------------------------------------------------------------

def _answer_corpus_dirs(ctx) -> List[str]:
    """Grounding corpus: an explicit answer_corpus_dir plus the agent's
    grove directory/workspace (the locally-reachable knowledge this
    environment has in place of the web)."""
    dirs: List[str] = []
    extra = ctx.runtime.extras.get("answer_corpus_dir") \
        or getattr(ctx.runtime.config, "answer_corpus_dir", None)
    if extra:
        dirs.append(extra)
    grove = ctx.agent.state.grove or {}
    gdir = grove.get("path")
    if gdir:
        dirs.append(gdir)
    return [d for d in dirs if os.path.isdir(d)]


def _gather_passages(dirs: List[str], max_files: int = 64,
                     chunk_chars: int = 1200) -> List[Dict[str, str]]:
    passages: List[Dict[str, str]] = []
    exts = {".md", ".txt", ".json", ".py", ".yaml", ".yml", ".csv"}
    for base in dirs:
        for root, _dirnames, files in os.walk(base):
            for name in sorted(files):
                if os.path.splitext(name)[1].lower() not in exts:
                    continue
                path = os.path.join(root, name)
                try:
                    with open(path, "r", errors="replace") as f:
                        text = f.read(256 * 1024)
                except OSError:
                    continue
                for off in range(0, len(text), chunk_chars):
                    passages.append({"source": path,
                                     "text": text[off:off + chunk_chars]})
                if len(passages) >= max_files * 4:
                    return passages
    return passages


async def execute_answer_engine(ctx) -> Dict[str, Any]:
    """Grounded Q&A.  The reference's answer engine is web-grounded
    (reference: actions/answer_engine.ex:28); with no egress, grounding
    here is retrieval over the locally-reachable corpus (grove dir +
    configured answer_corpus_dir): passages are ranked by embedding
    cosine against the question (the same GPU embed + cosine path the
    consensus vote uses) and injected as context, with sources reported.
    With no corpus available it degrades to an ungrounded local answer
    (documented divergence, PARITY.md)."""
    prompt = ctx.params["prompt"]
    model = (ctx.runtime.config.model_roles.get("answer_engine")
             or ctx.runtime.extras.get("answer_engine_model")
             or (ctx.agent.state.model_pool[0]
                 if ctx.agent.state.model_pool else None))
    if model is None:
        return _err("no_answer_engine_model")

    sources: List[str] = []
    context_block = ""
    passages = _gather_passages(_answer_corpus_dirs(ctx))
    if passages:
        try:
            facade = ctx.runtime.engines.embed_facade
            vecs = facade([prompt] + [p["text"] for p in passages])
            from ..consensus.rules import cosine_similarity
            scored = sorted(
                ((cosine_similarity(vecs[0], v), p)
                 for v, p in zip(vecs[1:], passages)),
                key=lambda sv: -sv[0])[:4]
            parts = []
            for score, p in scored:
                parts.append(f"[source: {p['source']}]\n{p['text']}")
                if p["source"] not in sources:
                    sources.append(p["source"])
            context_block = ("Grounding passages (local corpus):\n\n"
                             + "\n\n---\n\n".join(parts) + "\n\n")
        except Exception:  # noqa: BLE001 — grounding is best-effort
            sources = []
            context_block = ""

    engine = ctx.runtime.engines.engine_for(model)
    result = await engine.generate(GenerateRequest(
        model_key=model,
        messages=[{"role": "user",
                   "content": context_block
                   + "Answer factually and concisely:\n" + prompt}],
        temperature=0.3, max_tokens=2048))
    if not result.ok:
        return _err("answer_engine_failed", detail=result.error)
    out = {"answer": result.text, "model": model,
           "grounded": bool(sources)}
    if sources:
        out["sources"] = sources
    return out


async def execute_generate_images(ctx) -> Dict[str, Any]:
    """Image generation: an injected image_fn (external model) wins;
    otherwise the locally-hosted procedural model renders a real PNG
    (utils/imagegen.py — the no-network stand-in for a diffusion model,
    reference: models/image_query.ex).  Data-URL payloads flow through
    the image-artifact pipeline into multimodal history entries."""
    import base64 as _b64
    prompt = ctx.params["prompt"]
    source = ctx.params.get("source_image")
    image_fn = ctx.runtime.extras.get("image_fn")
    if image_fn is not None:
        images = await image_fn(prompt, source)
        return {"images": images}
    from ..utils import imagegen
    src_bytes = None
    if isinstance(source, str) and source:
        try:
            src_bytes = _b64.b64decode(source, validate=False)
        except Exception:  # noqa: BLE001
            src_bytes = source.encode()
    png = imagegen.render(prompt, source_image=src_bytes)
    data_url = "data:image/png;base64," + _b64.b64encode(png).decode()
    return {"images": [data_url], "model": imagegen.MODEL_NAME,
            "mode": "edit" if src_bytes else "generate"}


# ---------------------------------------------------------------------------
# secrets / costs / budget
# ---------------------------------------------------------------------------

async def execute_generate_secret(ctx) -> Dict[str, Any]:
    params = ctx.params
    name = params["name"]
    if not re.match(r"^[A-Za-z0-9_]+$", name):
        return _err("invalid_secret_name")
    value = generate_secret_value(
        length=int(params.get("length", 32)),
        include_symbols=bool(params.get("include_symbols", False)),
        include_numbers=bool(params.get("include_numbers", True)))
    ctx.runtime.vault.put(name, value, params.get("description", ""))
    return {"name": name, "status": "created",
            "reference": "{{SECRET:" + name + "}}"}


async def execute_search_secrets(ctx) -> Dict[str, Any]:
    names = ctx.runtime.vault.search(ctx.params["search_terms"])
    return {"matches": names}


async def execute_record_cost(ctx) -> Dict[str, Any]:
    params = ctx.params
    try:
        amount = budget_mod.parse_amount(params["amount"])
    except budget_mod.BudgetError as exc:
        return _err(exc.reason)
    state = ctx.agent.state
    ctx.runtime.store.save_cost(
        state.agent_id, state.task_id, None, amount,
        category=params.get("category") or "manual",
        description=params.get("description") or "",
        metadata=params.get("metadata"))
    state.budget_spent += amount
    ctx.runtime.bus.cost_recorded(state.agent_id, amount,
                                  {"category": params.get("category") or "manual"})
    return {"status": "recorded", "amount": amount}


async def execute_adjust_budget(ctx) -> Dict[str, Any]:
    params = ctx.params
    state = ctx.agent.state
    child_id = params["child_id"]
    if child_id not in state.children:
        return _err("not_a_direct_child", child_id=child_id)
    try:
        new_budget = budget_mod.parse_amount(params["new_budget"])
    except budget_mod.BudgetError as exc:
        return _err(exc.reason)
    child_entry = ctx.runtime.registry.lookup(child_id)
    if child_entry is None:
        return _err("child_not_running", child_id=child_id)
    child_state = child_entry.actor.state
    try:
        budget_mod.validate_decrease(new_budget, child_state.budget_spent,
                                     child_state.budget_committed)
    except budget_mod.BudgetError as exc:
        return _err(exc.reason)
    old_budget = child_state.budget_allocated or 0.0
    delta = new_budget - old_budget
    view = budget_mod.BudgetView(state.budget_mode, state.budget_allocated,
                                 state.budget_spent, state.budget_committed)
    if delta > 0 and state.budget_allocated is not None and \
            (view.available or 0.0) < delta:
        return _err("insufficient_budget")
    state.budget_committed = max(0.0, state.budget_committed + delta)
    child_state.budget_allocated = new_budget
    child_state.budget_mode = "allocated"
    state.children[child_id]["budget"] = new_budget
    await child_entry.actor.deliver({
        "type": "budget_adjusted", "new_budget": new_budget})
    return {"status": "adjusted", "child_id": child_id, "new_budget": new_budget}


# ---------------------------------------------------------------------------
# skills
# ---------------------------------------------------------------------------

async def execute_learn_skills(ctx) -> Dict[str, Any]:
    loader: skills_mod.SkillLoader = ctx.agent.skill_loader()
    loaded, missing = [], []
    for name in ctx.params["skills"]:
        try:
            skill = loader.load(name)
        except skills_mod.SkillError:
            missing.append(name)
            continue
        loaded.append(skill)
    if ctx.params.get("permanent"):
        existing = {s["name"] for s in ctx.agent.state.active_skills}
        for skill in loaded:
            if skill["name"] not in existing:
                ctx.agent.state.active_skills.append(
                    {"name": skill["name"], "description": skill["description"],
                     "content": skill["content"]})
        ctx.agent.invalidate_system_prompt()
        return {"status": "learned_permanently",
                "skills": [s["name"] for s in loaded], "missing": missing}
    return {"status": "loaded",
            "skills": [{"name": s["name"], "content": s["content"]}
                       for s in loaded],
            "missing": missing}


async def execute_create_skill(ctx) -> Dict[str, Any]:
    loader: skills_mod.SkillLoader = ctx.agent.skill_loader()
    params = ctx.params
    try:
        path = loader.create(params["name"], params["description"],
                             params["content"], params.get("metadata"),
                             params.get("attachments"))
    except skills_mod.SkillError as exc:
        return _err(exc.reason)
    return {"status": "created", "name": params["name"], "path": path}


# ---------------------------------------------------------------------------
# spawn_child / dismiss_child — delegated to the supervisor
# ---------------------------------------------------------------------------

async def execute_spawn_child(ctx) -> Dict[str, Any]:
    return await ctx.runtime.supervisor.spawn_child_action(ctx.agent, ctx.params)


async def execute_dismiss_child(ctx) -> Dict[str, Any]:
    return await ctx.runtime.supervisor.dismiss_child_action(
        ctx.agent, ctx.params["child_id"], ctx.params.get("reason"))


# ---------------------------------------------------------------------------
# batch_sync / batch_async
# ---------------------------------------------------------------------------

async def execute_batch_sync(ctx) -> Dict[str, Any]:
    """Sequential, stop-on-first-error, partial results returned
    (reference: batch_sync.ex:33-80)."""
    from . import router as router_mod
    results = []
    for spec in ctx.params["actions"]:
        sub_ctx = router_mod.ActionContext(
            agent=ctx.agent, runtime=ctx.runtime, action_id=ids.action_id(),
            action=spec["action"], params=spec.get("params") or {},
            skill_name=ctx.skill_name)
        try:
            result = await router_mod.execute_action(sub_ctx)
        except router_mod.ActionError as exc:
            results.append({"action": spec["action"], "error": exc.reason})
            return {"status": "stopped_on_error", "results": results,
                    "completed": len(results) - 1}
        results.append({"action": spec["action"], "result": result})
        # executors that report failure as a result payload (file ops,
        # shell exit codes) also stop the batch (reference: batch_sync.ex
        # stop-on-first-error)
        if isinstance(result, dict) and result.get("error"):
            return {"status": "stopped_on_error", "results": results,
                    "completed": len(results) - 1}
    return {"status": "completed", "results": results,
            "completed": len(results)}


async def execute_batch_async(ctx) -> Dict[str, Any]:
    """Concurrent; each sub-action's result arrives as its own action_result
    (reference: batch_async.ex:37)."""
    from . import router as router_mod
    agent = ctx.agent
    started = []
    for spec in ctx.params["actions"]:
        sub_action_id = ids.action_id()
        sub_ctx = router_mod.ActionContext(
            agent=agent, runtime=ctx.runtime, action_id=sub_action_id,
            action=spec["action"], params=spec.get("params") or {},
            skill_name=ctx.skill_name)

        async def _run(sc=sub_ctx):
            try:
                result = await router_mod.execute_action(sc)
            except router_mod.ActionError as exc:
                result = {"error": exc.reason, "detail": exc.detail}
            await agent.deliver({
                "type": "action_result", "action_id": sc.action_id,
                "action": sc.action, "result": result, "batch": True})

        asyncio.ensure_future(_run())
        started.append({"action": spec["action"], "action_id": sub_action_id})
    return {"status": "started", "actions": started}


EXECUTORS = {
    "wait": execute_wait,
    "orient": execute_orient,
    "todo": execute_todo,
    "send_message": execute_send_message,
    "execute_shell": execute_shell,
    "file_read": execute_file_read,
    "file_write": execute_file_write,
    "fetch_web": execute_fetch_web,
    "call_api": execute_call_api,
    "call_mcp": execute_call_mcp,
    "answer_engine": execute_answer_engine,
    "generate_images": execute_generate_images,
    "generate_secret": execute_generate_secret,
    "search_secrets": execute_search_secrets,
    "record_cost": execute_record_cost,
    "adjust_budget": execute_adjust_budget,
    "learn_skills": execute_learn_skills,
    "create_skill": execute_create_skill,
    "spawn_child": execute_spawn_child,
    "dismiss_child": execute_dismiss_child,
    "batch_sync": execute_batch_sync,
    "batch_async": execute_batch_async,
}
