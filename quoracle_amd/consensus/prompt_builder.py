"""System prompt assembly.

Mirrors the reference's section order (reference:
lib/quoracle/consensus/prompt_builder.ex:91-134,257-309 and
prompt_builder/{sections,schema_formatter,guidelines,examples,
response_format}.ex): identity (role / cognitive style / output style) ->
profile -> constraints -> capability-filtered action schemas with
profile-enum injection -> response format -> examples -> secrets doc ->
skills -> grove governance docs.

The assembled prompt is cached per agent and kept byte-stable across cycles
so its KV-cache prefix pages are shared (the reference's v38 prompt cache,
consensus_handler.ex:126-152, maps to GPU prefix sharing here).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..actions import schema as schema_mod
from ..governance import profiles as profiles_mod

COGNITIVE_STYLES = {
    "efficient": "Be direct: take the shortest sound path to the goal.",
    "exploratory": "Investigate before committing: survey options and evidence.",
    "problem_solving": "Work scientifically: hypothesize, test, iterate.",
    "creative": "Prefer novel framings and unconventional solutions.",
    "systematic": "Work methodically: decompose, order, and verify steps.",
}

OUTPUT_STYLES = {
    "detailed": "Report comprehensively with full supporting detail.",
    "concise": "Report in brief summaries; omit the inessential.",
    "technical": "Use precise technical terminology.",
    "narrative": "Explain in flowing prose.",
}

DELEGATION_STRATEGIES = {
    "sequential": "Delegate work to children one at a time, waiting for "
                  "each result before spawning the next.",
    "parallel": "Delegate independent work to several children at once "
                "and integrate their results.",
    "none": "Do the work yourself; do not spawn children.",
}


def _format_type(t: Any) -> str:
    if isinstance(t, str):
        return t
    if isinstance(t, tuple):
        kind = t[0]
        if kind == "enum":
            return "one of: " + ", ".join(str(c) for c in t[1])
        if kind == "list":
            return f"list of {_format_type(t[1])}"
        if kind == "union":
            return " or ".join(_format_type(x) for x in t[1])
        if kind == "map_shape":
            inner = ", ".join(f"{k}: {_format_type(v)}" for k, v in t[1].items())
            return "object {" + inner + "}"
    return str(t)


def format_action_schema(
    sch: schema_mod.ActionSchema,
    profile_names: Optional[List[str]] = None,
) -> str:
    """One action's documentation block for the system prompt."""
    lines = [f"### {sch.name}",
             schema_mod.ACTION_DESCRIPTIONS.get(sch.name, "")]
    if sch.required_params:
        lines.append("Required params:")
        for p in sch.required_params:
            lines.append(_param_line(sch, p, profile_names))
    if sch.optional_params:
        lines.append("Optional params:")
        for p in sch.optional_params:
            lines.append(_param_line(sch, p, profile_names))
    if sch.xor_params:
        groups = [" + ".join(g) for g in sch.xor_params]
        lines.append(f"Exactly one of: {' | '.join(groups)}")
    return "\n".join(line for line in lines if line)


def _param_line(sch: schema_mod.ActionSchema, param: str,
                profile_names: Optional[List[str]]) -> str:
    type_str = _format_type(sch.param_types.get(param, "any"))
    # Profile-enum injection: spawn_child.profile lists the actual profiles.
    if sch.name == "spawn_child" and param == "profile" and profile_names:
        type_str = "one of: " + ", ".join(profile_names)
    desc = sch.param_descriptions.get(param, "")
    return f"  - {param} ({type_str}): {desc}"


RESPONSE_FORMAT = """## Response format

Respond with EXACTLY ONE JSON object and nothing else:

{
  "reasoning": "1-4 sentences explaining your choice",
  "action": "<action name>",
  "params": { ... action parameters ... },
  "wait": false | true | <seconds>
}

The "wait" field controls what happens after your action executes:
false = decide again immediately; true = wait for an incoming event;
N = wait N seconds. Every action except "wait" itself requires the field.
You may optionally add "condense": N to discard your N oldest history entries.
"""

EXAMPLE = """## Example

{
  "reasoning": "The task needs a survey of the repository before planning.",
  "action": "execute_shell",
  "params": {"command": "ls -la /workspace", "working_dir": "/workspace"},
  "wait": false
}
"""

SECRETS_DOC = """## Secrets

Secret values are never shown to you. Reference a secret anywhere in action
params as {{SECRET:name}}; it is resolved just before execution and scrubbed
from results. Use generate_secret / search_secrets to create and find them.
"""


def build_system_prompt(
    *,
    role: Optional[str] = None,
    cognitive_style: Optional[str] = None,
    output_style: Optional[str] = None,
    delegation_strategy: Optional[str] = None,
    profile: Optional[profiles_mod.Profile] = None,
    constraints: Optional[List[str]] = None,
    capability_groups: Optional[List[str]] = None,
    forbidden_actions: Optional[List[str]] = None,
    profile_names: Optional[List[str]] = None,
    profile_catalog: Optional[List[Dict[str, Any]]] = None,
    skills: Optional[List[Dict[str, Any]]] = None,
    available_skills: Optional[List[Dict[str, str]]] = None,
    governance_docs: Optional[List[Dict[str, str]]] = None,
    agent_id: Optional[str] = None,
    extra_system_prompt: Optional[str] = None,
) -> str:
    parts: List[str] = []

    # Identity
    identity = ["# You are an autonomous agent in a recursive multi-agent system.",
                "Every decision you make is merged with other models' decisions by "
                "consensus; respond only in the JSON format below."]
    if agent_id:
        identity.append(f"Your agent ID: {agent_id}")
    if role:
        identity.append(f"Your role: {role}")
    if cognitive_style in COGNITIVE_STYLES:
        identity.append(COGNITIVE_STYLES[cognitive_style])
    if output_style in OUTPUT_STYLES:
        identity.append(OUTPUT_STYLES[output_style])
    if delegation_strategy in DELEGATION_STRATEGIES:
        identity.append(DELEGATION_STRATEGIES[delegation_strategy])
    parts.append("\n".join(identity))

    if extra_system_prompt:
        parts.append(extra_system_prompt)

    if profile is not None and profile.description:
        parts.append(f"## Profile: {profile.name}\n{profile.description}")

    if constraints:
        lines = ["## Constraints (binding on you and all your descendants)"]
        lines += [f"- {c}" for c in constraints]
        parts.append("\n".join(lines))

    # Governance docs from the grove, high priority first
    for doc in sorted(governance_docs or [],
                      key=lambda d: 0 if d.get("priority") == "high" else 1):
        parts.append(f"## Governance: {doc.get('name', 'policy')}\n{doc.get('content', '')}")

    # Capability-filtered action schemas
    caps = capability_groups
    if caps is None and profile is not None:
        caps = profile.capability_groups
    available = profiles_mod.filter_actions(schema_mod.ACTIONS, caps)
    if forbidden_actions:
        # grove hard rules: mechanically blocked actions never appear in
        # the offered schemas (reference: consensus_handler.ex:294-333)
        available = [a for a in available if a not in set(forbidden_actions)]
    schema_docs = [format_action_schema(schema_mod.get_schema(a), profile_names)
                   for a in available]
    parts.append("## Available actions\n\n" + "\n\n".join(schema_docs))
    if profile_catalog and "spawn_child" in available:
        lines = ["## Profiles available for child agents"]
        for entry in profile_catalog:
            desc = entry.get("description") or ""
            lines.append(f"- {entry.get('name')}: {desc}".rstrip(": "))
        parts.append("\n".join(lines))

    parts.append(RESPONSE_FORMAT)
    parts.append(EXAMPLE)
    parts.append(SECRETS_DOC)

    if available_skills:
        lines = ["## Available skills (load with learn_skills)"]
        lines += [f"- {s['name']}: {s.get('description', '')}" for s in available_skills]
        parts.append("\n".join(lines))

    for skill in skills or []:
        parts.append(f"## Skill: {skill.get('name')}\n{skill.get('content', '')}")

    return "\n\n".join(parts)
