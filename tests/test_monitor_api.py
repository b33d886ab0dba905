"""Web monitor API (L7 equivalent) over an isolated runtime + FakeEngine."""

import pytest

pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from helpers import IDLE, action_json, make_manager
from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.ui.server import create_app


@pytest.fixture()
def client():
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    app = create_app(manager)
    with TestClient(app) as c:
        yield c, runtime


def test_health_and_task_lifecycle(client):
    c, runtime = client
    assert c.get("/health").json()["status"] == "ok"
    r = c.post("/api/tasks", json={"prompt": "do it", "profile": "default"})
    assert r.status_code == 200
    task_id = r.json()["task_id"]
    root = r.json()["root_agent_id"]

    tasks = c.get("/api/tasks").json()
    assert any(t["task_id"] == task_id for t in tasks)
    tree = c.get(f"/api/tasks/{task_id}/tree").json()
    assert any(a["agent_id"] == root for a in tree["agents"])
    state = c.get(f"/api/agents/{root}/state").json()
    assert state["alive"] and state["model_pool"]
    costs = c.get(f"/api/agents/{root}/costs").json()
    assert "total" in costs and "descendants" in costs
    assert c.get(f"/api/agents/{root}/logs").status_code == 200
    assert c.post(f"/api/tasks/{task_id}/message",
                  json={"content": "status?"}).json()["ok"]
    assert c.post(f"/api/tasks/{task_id}/pause").json()["ok"]


def test_unknown_profile_rejected(client):
    c, _ = client
    r = c.post("/api/tasks", json={"prompt": "x", "profile": "nope"})
    assert r.status_code == 400


def test_profiles_and_secrets(client):
    c, _ = client
    r = c.post("/api/profiles", json={
        "name": "p2", "model_pool": ["fake-a"], "description": "d"})
    assert r.json()["ok"]
    assert any(p["name"] == "p2" for p in c.get("/api/profiles").json())
    assert c.post("/api/secrets", json={
        "name": "tok", "value": "s3cr3tvalue"}).json()["ok"]
    assert "tok" in c.get("/api/secrets").json()["names"]


def test_dashboard_served(client):
    c, _ = client
    r = c.get("/")
    assert r.status_code == 200 and "quoracle" in r.text


def test_grove_task_creation(client, monkeypatch):
    c, runtime = client
    import os
    runtime.config.groves_dir = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "groves")
    names = {g["name"] for g in c.get("/api/groves").json()}
    assert "qa-benchmark" in names
    r = c.post("/api/tasks", json={"prompt": "run it", "profile": "default",
                                   "grove": "qa-benchmark"})
    assert r.status_code == 200
    root = r.json()["root_agent_id"]
    actor = runtime.registry.lookup(root).actor
    assert (actor.state.grove or {}).get("name") == "qa-benchmark"
    r = c.post("/api/tasks", json={"prompt": "x", "grove": "no-such"})
    assert r.status_code == 400


def test_prometheus_metrics(client):
    c, runtime = client
    c.post("/api/tasks", json={"prompt": "m", "profile": "default"})
    body = c.get("/metrics").text
    assert "quoracle_agents_alive" in body
    assert "quoracle_tasks_running 1" in body
    assert "quoracle_cost_usd_total" in body


def test_event_history_replay(client):
    c, runtime = client
    r = c.post("/api/tasks", json={"prompt": "replay me",
                                   "profile": "default"})
    root = r.json()["root_agent_id"]
    import time
    deadline = time.monotonic() + 5
    logs = []
    while time.monotonic() < deadline and not logs:
        logs = c.get(f"/api/agents/{root}/history/logs").json()
        time.sleep(0.05)
    assert logs and all("type" in e and "ts" in e for e in logs)


def test_full_task_work_fields(client, tmp_path):
    c, runtime = client
    import os
    skill_dir = tmp_path / "skills" / "analysis"
    os.makedirs(skill_dir)
    (skill_dir / "SKILL.md").write_text(
        "---\nname: analysis\ndescription: analyze\n---\nLook closely.")
    runtime.config.skills_dir = str(tmp_path / "skills")
    r = c.post("/api/tasks", json={
        "prompt": "build the report",
        "profile": "default",
        "success_criteria": "report.md exists",
        "immediate_context": "data is in /data",
        "approach_guidance": "start with the schema",
        "skills": ["analysis"],
        "cognitive_style": "systematic",
        "delegation_strategy": "parallel"})
    assert r.status_code == 200
    root = r.json()["root_agent_id"]
    actor = runtime.registry.lookup(root).actor
    assert actor.state.cognitive_style == "systematic"
    assert actor.state.delegation_strategy == "parallel"
    assert any(s["name"] == "analysis" for s in actor.state.active_skills)
    import time
    deadline = time.monotonic() + 5
    found = False
    while time.monotonic() < deadline and not found:
        h = actor.state.model_histories[actor.state.model_pool[0]]
        found = any("Success criteria" in str(e.get("content", ""))
                    and "Approach guidance" in str(e.get("content", ""))
                    for e in h)
        time.sleep(0.05)
    assert found


def test_task_export_endpoint(client):
    c, runtime = client
    created = c.post("/api/tasks", json={"prompt": "export me",
                                         "profile": "default"}).json()
    task_id = created["task_id"]
    import time
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline:
        out = c.get(f"/api/tasks/{task_id}/export").json()
        if out["agents"] and any(a["logs"] for a in out["agents"]):
            break
        time.sleep(0.1)
    assert out["task"]["task_id"] == task_id
    assert out["task"]["prompt"] == "export me"
    assert out["agents"][0]["agent_id"] == created["root_agent_id"]
    assert isinstance(out["agents"][0]["costs"], (dict, list))
    assert isinstance(out["messages"], list)
    assert c.get("/api/tasks/t-ghost/export").status_code == 404


def test_admin_reload_invalidates_prompt_caches(client):
    c = client[0] if isinstance(client, tuple) else client
    c.post("/api/tasks", json={"prompt": "reload me",
                               "profile": "default"}).json()
    res = c.post("/api/admin/reload").json()
    assert res["ok"] and res["agents_invalidated"] >= 1
    assert any("groves" in s for s in res["hot_sources"])
