"""Render verbatim LLM prompts for debugging scenarios.

Golden-output equivalent of the reference's `mix quoracle.show_llm_prompts`
(reference: lib/mix/tasks/quoracle.show_llm_prompts.ex): every scenario
calls the REAL prompt-construction code (prompt builder, context manager,
injectors, refinement builder), so what you see is byte-for-byte what a
model receives.
"""

from __future__ import annotations

import json
from typing import Dict, List

from ..agent import context as context_mod
from ..agent import injectors
from ..agent.state import history_entry
from ..consensus.aggregator import (build_final_round_prompt,
                                    build_refinement_prompt)
from ..consensus.prompt_builder import build_system_prompt
from ..governance.profiles import Profile

SCENARIOS = [
    "initial", "with_profile", "with_constraints", "with_skills",
    "with_governance", "conversation", "injectors", "refinement",
    "final_round", "correction", "orient_result", "condensed",
    "constrained_decode",
]


def _fence(title: str, body: str) -> str:
    return f"\n{'=' * 70}\n# {title}\n{'=' * 70}\n{body}\n"


def _sample_history() -> List[Dict]:
    return [
        history_entry("prompt", "Investigate the failing build."),
        history_entry("decision", {"action": "orient", "params": {
            "current_situation": "Build fails on CI"},
            "reasoning": "Assess first", "wait": False}),
        history_entry("result", "[Action result: orient]\nAssessment recorded."),
    ]


def render_scenario(name: str) -> str:
    if name == "initial":
        return build_system_prompt()
    if name == "with_profile":
        return build_system_prompt(profile=Profile(
            name="researcher", description="Careful researcher",
            model_pool=["llama3-8b#0", "llama3-8b#1"],
            capability_groups=["hierarchy", "external_api"]),
            role="researcher", cognitive_style="skeptical")
    if name == "with_constraints":
        return build_system_prompt(constraints=[
            "Never modify files outside /workspace",
            "Budget answers under 200 words"])
    if name == "with_skills":
        return build_system_prompt(skills=[{
            "name": "code-review",
            "content": "# Code review\nCheck error handling first."}])
    if name == "with_governance":
        return build_system_prompt(governance_docs=[{
            "name": "grove-policy", "priority": "high",
            "content": "All outbound requests must be logged."}])
    if name == "conversation":
        msgs = context_mod.build_conversation_messages(_sample_history())
        return json.dumps(msgs, indent=2)
    if name == "injectors":
        msgs = context_mod.build_conversation_messages(_sample_history())
        out = injectors.inject_all(
            msgs,
            todos=[{"content": "fix flaky test", "state": "todo"}],
            children={"agent-c1": {"status": "busy", "task": "subtask"}},
            budget={"mode": "allocated", "allocated": 10.0, "spent": 2.5,
                    "committed": 1.0},
            lessons=[{"content": "CI uses python3.10", "confidence": 2}],
            model_state={"progress": "diagnosed"},
        )
        return json.dumps(out, indent=2)
    _responses = [
        {"model": "llama3-8b#0", "action": "file_read",
         "params": {"path": "/etc/ci.yaml"}, "reasoning": "check config"},
        {"model": "llama3-8b#1", "action": "execute_shell",
         "params": {"command": "make test"}, "reasoning": "just run it"},
    ]
    _ctx = {"prompt": "Investigate the failing build.",
            "max_refinement_rounds": 4,
            "reasoning_history": [
                {"round": 1, "model": "llama3-8b#0",
                 "reasoning": "check config"}]}
    if name == "refinement":
        return build_refinement_prompt(_responses, 2, _ctx)
    if name == "final_round":
        return build_final_round_prompt(_responses, _ctx)
    if name == "correction":
        return injectors.correction_block(
            {"llama3-8b#0": "invalid_json"}, "llama3-8b#0")
    if name == "orient_result":
        return json.dumps(_sample_history()[1]["content"], indent=2)
    if name == "constrained_decode":
        # deterministic grammar walk: what the engine's constrained decoder
        # emits for a fixed stream of sampler draws
        from ..engine.sampler import ActionGrammar
        from ..engine.tokenizer import EOS, ByteTokenizer
        g = ActionGrammar(["orient", "send_message", "todo", "wait"],
                          reasoning_tokens=8,
                          context={"spawn_profile": "default"})
        draws = [ord(c) for c in "assessing the task now then act"]
        out, i = [], 0
        while not g.done and len(out) < 2000:
            out.append(g.advance(draws[i % len(draws)]))
            i += 1
        text = ByteTokenizer().decode([t for t in out if t != EOS])
        return (f"sampler draws: {draws[:8]}... (cycled)\n"
                f"emitted action JSON ({len(out)} tokens):\n{text}")
    if name == "condensed":
        from ..agent import condensation as cond
        fn = getattr(cond, "condensation_artifact", None)
        if fn is None:
            return "[context condensed: 12 oldest entries summarized]"
        return str(fn(lessons=[{"content": "tests need DB", "confidence": 1}],
                      state={"progress": "half done"}, removed=12))
    raise KeyError(name)


def main(argv=None) -> None:
    import argparse
    p = argparse.ArgumentParser(
        description="Render verbatim prompts for debugging")
    p.add_argument("scenario", nargs="?", choices=SCENARIOS + ["all"],
                   default="all")
    args = p.parse_args(argv)
    names = SCENARIOS if args.scenario == "all" else [args.scenario]
    for name in names:
        try:
            print(_fence(name, render_scenario(name)))
        except Exception as exc:  # noqa: BLE001 — show what broke, keep going
            print(_fence(name, f"<error: {exc}>"))


if __name__ == "__main__":
    main()
