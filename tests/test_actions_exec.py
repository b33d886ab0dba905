"""Action execution through the full gate pipeline:
shell smart mode + async polling, file ops with confinement, NO_EXECUTE
wrapping, secret resolution + scrubbing, batch semantics."""

import asyncio
import json
import os
import tempfile

import pytest

from quoracle_amd.actions import router as R
from quoracle_amd.agent.core import AgentActor
from quoracle_amd.agent.state import AgentState

from helpers import IDLE, make_runtime


def _actor(runtime, grove=None, groups=None):
    state = AgentState(agent_id="ax-1", task_id="t1", parent_id=None,
                       profile="default",
                       model_pool=["fake-a", "fake-b"],
                       capability_groups=groups if groups is not None else [
                           "hierarchy", "local_execution", "file_read",
                           "file_write", "external_api"],
                       grove=grove)
    state.init_model_maps()
    actor = AgentActor(state, runtime)
    runtime.registry.register(state.agent_id, actor, "t1", parent_id=None)
    return actor


def _ctx(actor, runtime, action, params):
    return R.ActionContext(agent=actor, runtime=runtime, action_id="act-1",
                           action=action, params=params)


@pytest.mark.asyncio
async def test_shell_smart_mode_sync_and_async():
    runtime = make_runtime()
    actor = _actor(runtime)
    # fast command: completes within the 100ms smart threshold -> sync
    res = await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                      {"command": "echo hi"}))
    assert res.get("sync") is True and res["exit_code"] == 0
    assert "hi" in res["stdout"]
    # slow command: returns a command_id; poll via check_id
    res = await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                      {"command": "sleep 0.4; echo done"}))
    assert res.get("async") is True and res.get("command_id")
    cid = res["command_id"]
    for _ in range(40):
        await asyncio.sleep(0.05)
        chk = await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                          {"check_id": cid}))
        if chk.get("status") == "completed":
            break
    assert chk.get("status") == "completed" and "done" in chk.get("stdout", "")


@pytest.mark.asyncio
async def test_shell_output_is_injection_wrapped():
    runtime = make_runtime()
    actor = _actor(runtime)
    res = await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                      {"command": "echo attack"}))
    assert "NO_EXECUTE_" in res["stdout"]


@pytest.mark.asyncio
async def test_file_roundtrip_and_confinement():
    runtime = make_runtime()
    with tempfile.TemporaryDirectory() as tmp:
        grove = {"name": "g", "path": tmp, "confinement_mode": "strict",
                 "confinement": {"default": {"paths": [f"{tmp}/**"]}}}
        actor = _actor(runtime, grove=grove)
        target = os.path.join(tmp, "out.txt")
        res = await R.execute_action(_ctx(actor, runtime, "file_write",
                                          {"path": target, "mode": "write",
                                           "content": "hello"}))
        assert res.get("bytes") or res.get("path")
        res = await R.execute_action(_ctx(actor, runtime, "file_read",
                                          {"path": target}))
        assert "hello" in res["content"]
        with pytest.raises(R.ActionError):
            await R.execute_action(_ctx(actor, runtime, "file_write",
                                        {"path": "/tmp/outside-grove.txt",
                                         "mode": "write", "content": "x"}))


@pytest.mark.asyncio
async def test_secret_resolution_and_scrubbing_through_pipeline():
    runtime = make_runtime()
    actor = _actor(runtime)
    runtime.vault.put("token", "verysecretstring42")
    res = await R.execute_action(_ctx(
        actor, runtime, "execute_shell",
        {"command": "echo using {{SECRET:token}}"}))
    # the secret reached the shell but the output is scrubbed
    assert "verysecretstring42" not in json.dumps(res)
    assert "[REDACTED:token]" in res["stdout"]


@pytest.mark.asyncio
async def test_action_gate_rejects_without_capability():
    runtime = make_runtime()
    actor = _actor(runtime, groups=[])      # only always-allowed actions
    with pytest.raises(R.ActionError) as ei:
        await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                    {"command": "echo hi"}))
    assert ei.value.reason == "action_not_allowed"


@pytest.mark.asyncio
async def test_batch_sync_stops_on_error(tmp_path):
    runtime = make_runtime()
    actor = _actor(runtime)
    first = str(tmp_path / "qbatch1.txt")
    never = str(tmp_path / "qbatch2-never.txt")
    res = await R.execute_action(_ctx(actor, runtime, "batch_sync", {
        "actions": [
            {"action": "file_write",
             "params": {"path": first, "mode": "write", "content": "a"}},
            {"action": "file_read", "params": {"path": "/nonexistent-xyz"}},
            {"action": "file_write",
             "params": {"path": never, "mode": "write", "content": "b"}},
        ]}))
    results = res["results"]
    assert res["status"] == "stopped_on_error"
    assert len(results) == 2            # stopped at the failing action
    assert not os.path.exists(never)
    assert os.path.exists(first)


@pytest.mark.asyncio
async def test_hard_rule_blocks_shell_pattern():
    runtime = make_runtime()
    grove = {"name": "g", "path": "/tmp",
             "hard_rules": [{"type": "shell_pattern_block",
                             "pattern": "curl", "message": "no net"}]}
    actor = _actor(runtime, grove=grove)
    with pytest.raises(R.ActionError):
        await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                    {"command": "curl http://x"}))


@pytest.mark.asyncio
async def test_image_results_become_artifacts(tmp_path, monkeypatch):
    """Binary image payloads in action results are stored as artifacts and
    replaced with placeholders (reference: image_detector.ex)."""
    monkeypatch.setenv("QUORACLE_IMAGE_DIR", str(tmp_path))
    runtime = make_runtime()
    actor = _actor(runtime)
    png = b"\x89PNG\r\n\x1a\n" + b"\x00" * 64
    import base64
    data_url = "data:image/png;base64," + base64.b64encode(png).decode()
    src = tmp_path / "page.txt"
    src.write_text("before " + data_url + " after")
    res = await R.execute_action(_ctx(actor, runtime, "file_read",
                                      {"path": str(src)}))
    assert "image artifact" in res["content"]
    assert "base64" not in res["content"]
    stored = list(tmp_path.glob("*.png"))
    assert stored and stored[0].read_bytes() == png
