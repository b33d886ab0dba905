"""The shipped example groves load and enforce their rules end-to-end."""

import os

import pytest

from quoracle_amd.governance import groves as G

ROOT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    "groves")


def test_groves_listed():
    names = {g["name"] for g in G.list_groves(ROOT)}
    assert {"qa-benchmark", "code-sandbox"} <= names


def test_qa_benchmark_grove_contents():
    g = G.load_grove(os.path.join(ROOT, "qa-benchmark"))
    assert g["bootstrap"]["task_description"].startswith("Run the QA")
    # *_file bootstrap fields resolve to file contents — the full
    # 600-question procedurally-generated bank (scripts/gen_qa_bank.py)
    import json as _json
    bank = _json.loads(g["bootstrap"]["immediate_context"])
    assert len(bank) == 6 and all(len(v) == 100 for v in bank.values())
    one = bank["arithmetic"][0]
    assert len(one["options"]) == 10 and one["answer"] in "ABCDEFGHIJ"
    edge = g["topology"]["edges"][0]
    assert edge["auto_inject"]["profile"] == "default"
    assert g["workspace"] == "scratch"


def test_code_sandbox_rules_enforced():
    g = G.load_grove(os.path.join(ROOT, "code-sandbox"))
    with pytest.raises(G.HardRuleViolation):
        G.check_shell_command("curl http://x", g["hard_rules"])
    with pytest.raises(G.HardRuleViolation):
        G.check_shell_command("rm -rf /", g["hard_rules"])
    G.check_shell_command("ls workspace", g["hard_rules"])
    with pytest.raises(G.HardRuleViolation):
        G.check_action("fetch_web", g["hard_rules"])
    base = os.path.join(ROOT, "code-sandbox")
    G.check_file_access(os.path.join(base, "workspace/a.py"), "write",
                        g["confinement"],
                        confinement_mode=g["confinement_mode"])
    G.check_file_access(os.path.join(base, "GROVE.md"), "read",
                        g["confinement"],
                        confinement_mode=g["confinement_mode"])
    with pytest.raises(G.ConfinementViolation):
        G.check_file_access(os.path.join(base, "GROVE.md"), "write",
                            g["confinement"],
                            confinement_mode=g["confinement_mode"])
