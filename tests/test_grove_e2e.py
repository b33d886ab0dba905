"""Grove-governed task end-to-end: topology auto-inject on consensus spawn,
parent<->child messaging, schema-validated confined writes — the
qa-benchmark example grove through the whole stack (FakeEngine scripts the
decisions; everything else is the production path)."""

import asyncio
import json
import os
import shutil

import pytest

from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.governance import groves as G

from helpers import IDLE, POOL2, action_json, make_manager, wait_until

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.asyncio
async def test_qa_benchmark_grove_full_cycle(tmp_path):
    # work on a COPY of the shipped grove so writes stay test-local;
    # shrink the 600-question bank so bootstrap immediate_context stays
    # small (the full bank slows scripted cycles enough to flake timing)
    grove_dir = str(tmp_path / "qa-benchmark")
    shutil.copytree(os.path.join(REPO, "groves", "qa-benchmark"), grove_dir)
    bank = json.loads(open(os.path.join(grove_dir, "questions.json")).read())
    small = {s: qs[:2] for s, qs in list(bank.items())[:2]}
    with open(os.path.join(grove_dir, "questions.json"), "w") as f:
        json.dump(small, f)
    grove = G.load_grove(grove_dir)

    engine = FakeEngine(default_response=IDLE)
    # root round 1: spawn one solver (profile OMITTED -> grove topology
    # auto-injects "default", reference: spawn/topology_resolver.ex)
    spawn = action_json("spawn_child", {
        "task_description": "solve the logic subject questions",
        "success_criteria": "answers reported to parent",
        "immediate_context": "see questions.json",
        "approach_guidance": "answer with option letters",
    })
    for m in POOL2:
        engine.push_response(m, spawn)
    # child's first decision: report its answer upward
    child_report = action_json("send_message",
                               {"to": "parent", "content": "logic: B, C"})
    # root after child report: write the (schema-validated) results file
    write_results = action_json("file_write", {
        "path": os.path.join(grove_dir, "results.json"), "mode": "write",
        "content": json.dumps({"subjects": {"logic": {"correct": 2,
                                                      "total": 2}}})})

    manager, runtime = make_manager(engine)
    result = await manager.create_task(
        grove["bootstrap"]["task_description"], "default", grove=grove)
    root_id = result["root_agent_id"]

    # wait for the child to exist
    assert await wait_until(
        lambda: runtime.registry.children_of(root_id), timeout=10)
    child_id = runtime.registry.children_of(root_id)[0]
    child = runtime.registry.lookup(child_id).actor
    # grove topology injected the child profile even though the spawn
    # params omitted it
    assert child.state.profile == "default"
    # grove governance doc reaches the child prompt context
    assert child.state.grove and child.state.grove["name"] == "qa-benchmark"

    # script the next decisions now that both agents exist
    for m in POOL2:
        engine.push_response(m, child_report)
    for m in POOL2:
        engine.push_response(m, write_results)

    # nudge the child so it runs its reporting cycle
    await child.deliver({"type": "user_message", "content": "report now"})
    assert await wait_until(
        lambda: any("logic: B, C" in str(e.get("content"))
                    for e in runtime.registry.lookup(root_id)
                    .actor.state.model_histories[POOL2[0]]), timeout=10)

    # nudge the root; its file_write must pass grove schema validation
    await manager.send_user_message(result["task_id"], "finalize results")
    results_path = os.path.join(grove_dir, "results.json")
    assert await wait_until(lambda: os.path.exists(results_path), timeout=10)
    data = json.loads(open(results_path).read())
    assert "subjects" in data

    # a write that VIOLATES the grove schema is rejected
    bad = action_json("file_write", {
        "path": results_path, "mode": "write",
        "content": json.dumps({"wrong_shape": 1})})
    # push several copies per model: if one model's queue ran a cycle
    # ahead, round-1 unanimity fails and the refinement round re-queries
    # — the duplicates keep every round unanimous on `bad`
    for m in POOL2:
        for _ in range(4):
            engine.push_response(m, bad)
    await manager.send_user_message(result["task_id"], "write bad file")
    root = runtime.registry.lookup(root_id).actor

    def saw_schema_error():
        h = root.state.model_histories[POOL2[0]]
        return any("schema" in str(e.get("content", "")).lower()
                   for e in h if e["type"] == "result")
    assert await wait_until(saw_schema_error, timeout=20)
    data = json.loads(open(results_path).read())
    assert "subjects" in data          # original file untouched
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_qa_benchmark_runner_scores_bank_sample():
    """The mmlu-pro-analog runner (scripts/run_qa_benchmark.py) drives
    real spawn/dismiss + schema-validated confined writes over a bank
    sample and scores exactly: oracle 1.0, noisy below it."""
    import argparse
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "qa_runner", os.path.join(REPO, "scripts", "run_qa_benchmark.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    out = await mod.run(argparse.Namespace(
        subjects=2, per_subject=6, engine="oracle", noise=0.0))
    assert out["overall_accuracy"] == 1.0 and out["questions"] == 12
    assert os.path.exists(out["results_file"])
    data = json.loads(open(out["results_file"]).read())
    assert set(data["subjects"]) == set(out["subjects"])
    out2 = await mod.run(argparse.Namespace(
        subjects=2, per_subject=12, engine="noisy", noise=0.5))
    assert out2["overall_accuracy"] < 1.0
