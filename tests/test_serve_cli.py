"""`python -m quoracle_amd serve --fake` boots the full stack (engine
stand-in, runtime, monitor) and serves the API — the user-facing
entrypoint equivalent of the reference's Phoenix endpoint."""

import json
import os
import socket
import subprocess
import sys
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _get(url):
    with urllib.request.urlopen(url, timeout=5) as r:
        return json.loads(r.read())


def _post(url, body):
    req = urllib.request.Request(
        url, data=json.dumps(body).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=10) as r:
        return json.loads(r.read())


def test_serve_fake_end_to_end(tmp_path):
    port = _free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "quoracle_amd", "serve", "--fake",
         "--db", str(tmp_path / "serve.db"), "--port", str(port)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.monotonic() + 60
        last = None
        while time.monotonic() < deadline:
            try:
                last = _get(base + "/health")
                break
            except Exception as exc:  # noqa: BLE001
                last = exc
                if proc.poll() is not None:
                    raise AssertionError(
                        f"serve exited early: {proc.stderr.read()[-1500:]}")
                time.sleep(0.3)
        assert isinstance(last, dict) and last["status"] == "ok", last
        created = _post(base + "/api/tasks",
                        {"prompt": "serve e2e", "profile": "default"})
        assert created["task_id"]
        tasks = _get(base + "/api/tasks")
        assert any(t["task_id"] == created["task_id"] for t in tasks)
        tree = _get(base + f"/api/tasks/{created['task_id']}/tree")
        assert tree["agents"]
        metrics = urllib.request.urlopen(base + "/metrics", timeout=5).read()
        assert b"quoracle_tasks_running" in metrics
        # lifecycle through the HTTP surface: pause -> restore -> delete
        paused = _post(base + f"/api/tasks/{created['task_id']}/pause", {})
        assert paused.get("ok") is True, paused
        tasks = _get(base + "/api/tasks")
        mine = next(t for t in tasks if t["task_id"] == created["task_id"])
        assert mine["status"] == "paused", mine
        restored = _post(base + f"/api/tasks/{created['task_id']}/restore",
                         {})
        assert isinstance(restored, dict), restored
        stats = _get(base + "/api/engine/stats")
        assert isinstance(stats, dict)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
