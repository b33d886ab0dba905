"""Action schema registry: the full capability surface of the framework.

Defines the 22 actions, their parameter types, per-parameter consensus rules,
tie-break priorities, and batchability.  Behavior-parity with the reference's
schema layer (reference: lib/quoracle/actions/schema/action_list.ex:6-29,
agent_schemas.ex, api_schemas.ex, metadata.ex) — re-expressed natively:
actions and enums are strings, types are small tuples, rules are
("rule", arg) pairs.

Type grammar (param_types values):
    "string" | "integer" | "number" | "boolean" | "map" | "any"
    ("enum", [..choices..])
    ("list", <type>)
    ("union", [<type>, ...])
    ("map_shape", {field: <type>, ...})
    ("list", "batchable_action_spec") / ("list", "async_action_spec")

Consensus-rule grammar (consensus_rules values):
    "exact_match" | "mode_selection" | "union_merge" | "structural_merge"
    | "first_non_nil" | "merge_maps" | "wait_parameter" | "batch_sequence_merge"
    | ("semantic_similarity", threshold) | ("percentile", n)
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

ACTIONS: List[str] = [
    "spawn_child",
    "wait",
    "send_message",
    "orient",
    "answer_engine",
    "execute_shell",
    "fetch_web",
    "call_api",
    "call_mcp",
    "todo",
    "generate_secret",
    "search_secrets",
    "dismiss_child",
    "generate_images",
    "record_cost",
    "adjust_budget",
    "file_read",
    "file_write",
    "learn_skills",
    "create_skill",
    "batch_sync",
    "batch_async",
]

# Actions allowed inside batch_sync: fast, synchronous, no nesting, no timing.
BATCHABLE_ACTIONS: List[str] = [
    "spawn_child",
    "send_message",
    "orient",
    "todo",
    "generate_secret",
    "search_secrets",
    "dismiss_child",
    "adjust_budget",
    "record_cost",
    "file_read",
    "file_write",
    "learn_skills",
    "create_skill",
]

# Everything except these may appear in batch_async.
ASYNC_EXCLUDED_ACTIONS: List[str] = ["wait", "batch_sync", "batch_async"]


def async_batchable(action: str) -> bool:
    return action not in ASYNC_EXCLUDED_ACTIONS


# Tie-break priorities: lower = more conservative, wins ties.
# (reference: actions/schema/metadata.ex @action_priorities)
ACTION_PRIORITIES: Dict[str, int] = {
    "orient": 1,
    "send_message": 2,
    "batch_sync": 3,
    "batch_async": 4,
    "fetch_web": 5,
    "file_read": 6,
    "search_secrets": 7,
    "learn_skills": 8,
    "answer_engine": 9,
    "todo": 10,
    "adjust_budget": 11,
    "wait": 12,
    "generate_secret": 13,
    "generate_images": 14,
    "record_cost": 15,
    "call_mcp": 16,
    "call_api": 17,
    "execute_shell": 18,
    "file_write": 19,
    "dismiss_child": 20,
    "create_skill": 21,
    "spawn_child": 22,
}


@dataclass
class ActionSchema:
    name: str
    required_params: List[str]
    optional_params: List[str] = field(default_factory=list)
    param_types: Dict[str, Any] = field(default_factory=dict)
    param_descriptions: Dict[str, str] = field(default_factory=dict)
    consensus_rules: Dict[str, Any] = field(default_factory=dict)
    # xor groups: exactly one group of params may be present
    xor_params: Optional[List[List[str]]] = None

    @property
    def all_params(self) -> List[str]:
        return self.required_params + self.optional_params


SEM = lambda t: ("semantic_similarity", t)  # noqa: E731
PCT = lambda n: ("percentile", n)  # noqa: E731

_SCHEMAS: Dict[str, ActionSchema] = {}


def _register(schema: ActionSchema) -> None:
    _SCHEMAS[schema.name] = schema


_register(ActionSchema(
    name="spawn_child",
    required_params=["task_description", "success_criteria", "immediate_context",
                     "approach_guidance", "profile"],
    optional_params=["role", "cognitive_style", "output_style", "delegation_strategy",
                     "sibling_context", "downstream_constraints", "skills", "budget",
                     "grove_vars"],
    param_types={
        "task_description": "string",
        "success_criteria": "string",
        "immediate_context": "string",
        "approach_guidance": "string",
        "profile": "string",
        "role": "string",
        "cognitive_style": ("enum", ["efficient", "exploratory", "problem_solving",
                                     "creative", "systematic"]),
        "output_style": ("enum", ["detailed", "concise", "technical", "narrative"]),
        "delegation_strategy": ("enum", ["sequential", "parallel", "none"]),
        "sibling_context": ("list", "map"),
        "downstream_constraints": "string",
        "skills": ("list", "string"),
        "budget": "string",
        "grove_vars": "map",
    },
    param_descriptions={
        "task_description": "A bounded, specific objective for the child, stating both "
                            "what it owns and what is out of scope. Becomes part of the "
                            "child's first user prompt.",
        "success_criteria": "Measurable completion conditions for the child's task.",
        "immediate_context": "Facts and background the child needs before starting.",
        "approach_guidance": "Suggested strategy or methodology for the work.",
        "profile": "Name of an existing profile controlling the child's model pool "
                   "and permitted actions.",
        "role": "Persona for the child's system prompt (e.g. 'meticulous code reviewer').",
        "cognitive_style": "Thinking pattern: efficient, exploratory, problem_solving, "
                           "creative, or systematic.",
        "output_style": "Result formatting: detailed, concise, technical, or narrative.",
        "delegation_strategy": "How the child should delegate onward: sequential, "
                               "parallel, or none.",
        "sibling_context": "Array of {agent_id, task} entries describing sibling scopes "
                           "the child must treat as off-limits.",
        "downstream_constraints": "Extra constraint inherited by this child and every "
                                  "descendant; accumulates with upstream constraints.",
        "skills": "Skill names preloaded into the child's system prompt.",
        "budget": "USD budget for the child as a positive decimal string; omitted "
                  "means unlimited. Escrowed from the parent's budget.",
        "grove_vars": "Template variables substituted into inherited grove config "
                      "(e.g. {child_workspace} confinement placeholders).",
    },
    consensus_rules={
        "task_description": SEM(0.95),
        "success_criteria": SEM(0.85),
        "immediate_context": SEM(0.85),
        "approach_guidance": SEM(0.85),
        "profile": "exact_match",
        "role": SEM(0.85),
        "cognitive_style": "mode_selection",
        "output_style": "mode_selection",
        "delegation_strategy": "exact_match",
        "sibling_context": "structural_merge",
        "downstream_constraints": SEM(0.90),
        "skills": "union_merge",
        "budget": "exact_match",
        "grove_vars": "exact_match",
    },
))

_register(ActionSchema(
    name="wait",
    required_params=[],
    optional_params=["wait"],
    param_types={"wait": ("union", ["boolean", "number"])},
    param_descriptions={
        "wait": "true = wait indefinitely for an event, false/0 = continue "
                "immediately, N = wait N seconds.",
    },
    consensus_rules={"wait": PCT(50)},
))

_register(ActionSchema(
    name="send_message",
    required_params=["to", "content"],
    param_types={
        "to": ("union", ["string", ("list", "string")]),
        "content": "string",
    },
    param_descriptions={
        "to": "'parent' (status and results go here), 'children' (direct children), "
              "'announcement' (broadcast a directive to all descendants — never "
              "status updates), or a list of agent IDs.",
        "content": "Message body.",
    },
    consensus_rules={"to": "exact_match", "content": SEM(0.85)},
))

_ORIENT_FIELDS = [
    "current_situation", "goal_clarity", "available_resources", "key_challenges",
    "delegation_consideration", "assumptions", "unknowns", "approach_options",
    "parallelization_opportunities", "risk_factors", "success_criteria",
    "next_steps", "constraints_impact",
]
_register(ActionSchema(
    name="orient",
    required_params=["current_situation", "goal_clarity", "available_resources",
                     "key_challenges", "delegation_consideration"],
    optional_params=["assumptions", "unknowns", "approach_options",
                     "parallelization_opportunities", "risk_factors",
                     "success_criteria", "next_steps", "constraints_impact"],
    param_types={f: "string" for f in _ORIENT_FIELDS},
    param_descriptions={
        "current_situation": "Present state of the task.",
        "goal_clarity": "How well the objective is understood.",
        "available_resources": "Tools, data and capabilities at hand.",
        "key_challenges": "Main obstacles between here and the goal.",
        "delegation_consideration": "Whether spawning child agents would help, and "
                                    "what kind.",
        "assumptions": "Premises in play that may need validation.",
        "unknowns": "Information gaps that could change the approach.",
        "approach_options": "Candidate strategies.",
        "parallelization_opportunities": "Work that could run concurrently in children.",
        "risk_factors": "Failure modes and edge cases.",
        "success_criteria": "How completion will be recognized.",
        "next_steps": "Immediate actions implied by this assessment.",
        "constraints_impact": "How current constraints shape the options.",
    },
    consensus_rules={f: SEM(0.8) for f in _ORIENT_FIELDS},
))

_register(ActionSchema(
    name="todo",
    required_params=["items"],
    param_types={
        "items": ("list", ("map_shape", {
            "content": "string",
            "state": ("enum", ["todo", "pending", "done"]),
        })),
    },
    param_descriptions={
        "items": "Full replacement TODO list: [{content, state}] with state one of "
                 "todo / pending / done.",
    },
    consensus_rules={"items": SEM(0.85)},
))

_register(ActionSchema(
    name="dismiss_child",
    required_params=["child_id"],
    optional_params=["reason"],
    param_types={"child_id": "string", "reason": "string"},
    param_descriptions={
        "child_id": "Direct child to terminate (recursively, with its descendants).",
        "reason": "Optional reason, logged and broadcast.",
    },
    consensus_rules={"child_id": "exact_match", "reason": "first_non_nil"},
))

_register(ActionSchema(
    name="adjust_budget",
    required_params=["child_id", "new_budget"],
    param_types={"child_id": "string", "new_budget": "string"},
    param_descriptions={
        "child_id": "Direct child whose allocation changes.",
        "new_budget": "New positive decimal budget.",
    },
    consensus_rules={"child_id": "exact_match", "new_budget": "exact_match"},
))

_register(ActionSchema(
    name="answer_engine",
    required_params=["prompt"],
    param_types={"prompt": "string"},
    param_descriptions={
        "prompt": "Question to answer with web-grounded search.",
    },
    consensus_rules={"prompt": SEM(0.95)},
))

_register(ActionSchema(
    name="execute_shell",
    required_params=[],
    optional_params=["command", "check_id", "working_dir", "terminate"],
    xor_params=[["command"], ["check_id"]],
    param_types={
        "command": "string",
        "check_id": "string",
        "working_dir": "string",
        "terminate": "boolean",
    },
    param_descriptions={
        "command": "Shell command to start. Commands finishing under the sync "
                   "threshold return immediately; longer ones run async.",
        "check_id": "ID of a running command to poll (mutually exclusive with "
                    "'command').",
        "working_dir": "Absolute working directory (default /tmp).",
        "terminate": "With check_id: kill the running command.",
    },
    consensus_rules={
        "command": "exact_match",
        "check_id": "exact_match",
        "working_dir": "exact_match",
        "terminate": "exact_match",
    },
))

_register(ActionSchema(
    name="fetch_web",
    required_params=["url"],
    optional_params=["security_check", "timeout", "user_agent", "follow_redirects"],
    param_types={
        "url": "string",
        "security_check": "boolean",
        "timeout": "number",
        "user_agent": "string",
        "follow_redirects": "boolean",
    },
    param_descriptions={
        "url": "Page to fetch; content is converted to markdown.",
        "security_check": "Block private IPs / localhost (SSRF guard).",
        "timeout": "Request timeout in seconds (default 30).",
        "user_agent": "Custom User-Agent header.",
        "follow_redirects": "Follow HTTP redirects (default true).",
    },
    consensus_rules={
        "url": "exact_match",
        "security_check": "mode_selection",
        "timeout": PCT(50),
        "user_agent": "exact_match",
        "follow_redirects": "mode_selection",
    },
))

_register(ActionSchema(
    name="call_api",
    required_params=["api_type", "url"],
    optional_params=["method", "query_params", "body", "headers", "auth", "query",
                     "variables", "rpc_method", "rpc_params", "rpc_id", "timeout",
                     "max_body_size"],
    param_types={
        "api_type": ("enum", ["rest", "graphql", "jsonrpc"]),
        "url": "string",
        "timeout": "integer",
        "headers": "map",
        "auth": "map",
        "max_body_size": "integer",
        "method": "string",
        "query_params": "map",
        "body": "any",
        "query": "string",
        "variables": "map",
        "rpc_method": "string",
        "rpc_params": "any",
        "rpc_id": "string",
    },
    param_descriptions={
        "api_type": "rest, graphql, or jsonrpc.",
        "url": "Endpoint URL.",
        "timeout": "Request timeout in seconds (default 30).",
        "headers": "Extra HTTP headers.",
        "auth": "Auth config: {auth_type, token | credentials}.",
        "method": "REST verb: GET/POST/PUT/DELETE/PATCH.",
        "query_params": "REST query string parameters.",
        "body": "REST request body.",
        "query": "GraphQL query/mutation string.",
        "variables": "GraphQL variables.",
        "rpc_method": "JSON-RPC method name.",
        "rpc_params": "JSON-RPC params (map or array).",
        "rpc_id": "JSON-RPC id (auto-generated if omitted).",
        "max_body_size": "Max request body bytes (default 5 MB).",
    },
    consensus_rules={
        "api_type": "exact_match", "url": "exact_match", "method": "exact_match",
        "timeout": PCT(100), "auth": "exact_match", "query_params": "exact_match",
        "body": "exact_match", "headers": "exact_match", "query": "exact_match",
        "variables": "exact_match", "rpc_method": "exact_match",
        "rpc_params": "exact_match", "rpc_id": "exact_match",
        "max_body_size": PCT(100),
    },
))

_register(ActionSchema(
    name="call_mcp",
    required_params=[],
    optional_params=["transport", "command", "url", "cwd", "connection_id", "tool",
                     "arguments", "terminate", "timeout"],
    xor_params=[["transport"], ["connection_id"]],
    param_types={
        "transport": ("enum", ["stdio", "http"]),
        "command": "string",
        "url": "string",
        "cwd": "string",
        "connection_id": "string",
        "tool": "string",
        "arguments": "map",
        "terminate": "boolean",
        "timeout": "number",
    },
    param_descriptions={
        "transport": "stdio (subprocess server) or http (remote server).",
        "command": "Command that launches a stdio MCP server.",
        "url": "HTTP MCP server URL.",
        "cwd": "Working directory for the stdio command.",
        "connection_id": "Existing connection (returned by connect).",
        "tool": "Tool to invoke on the connection.",
        "arguments": "Tool arguments.",
        "terminate": "Close the connection.",
        "timeout": "Timeout in milliseconds (default 30000).",
    },
    consensus_rules={
        "transport": "exact_match", "command": "exact_match", "url": "exact_match",
        "cwd": "exact_match", "connection_id": "exact_match", "tool": "exact_match",
        "arguments": "exact_match", "terminate": "exact_match", "timeout": PCT(50),
    },
))

_register(ActionSchema(
    name="generate_secret",
    required_params=["name"],
    optional_params=["length", "include_symbols", "include_numbers", "description"],
    param_types={
        "name": "string",
        "length": "integer",
        "include_symbols": "boolean",
        "include_numbers": "boolean",
        "description": "string",
    },
    param_descriptions={
        "name": "Identifier (alphanumeric + underscore); reference later with "
                "{{SECRET:name}}. The value itself is never shown to models.",
        "length": "Characters (default 32, min 8, max 128).",
        "include_symbols": "Allow punctuation characters (default false).",
        "include_numbers": "Allow digits (default true).",
        "description": "Operator-facing note about the secret's purpose.",
    },
    consensus_rules={
        "name": "exact_match", "length": PCT(50),
        "include_symbols": "mode_selection", "include_numbers": "mode_selection",
        "description": SEM(0.8),
    },
))

_register(ActionSchema(
    name="search_secrets",
    required_params=["search_terms"],
    param_types={"search_terms": ("list", "string")},
    param_descriptions={
        "search_terms": "Case-insensitive substrings; returns secret names "
                        "matching any term.",
    },
    consensus_rules={"search_terms": "union_merge"},
))

_register(ActionSchema(
    name="generate_images",
    required_params=["prompt"],
    optional_params=["source_image"],
    param_types={"prompt": "string", "source_image": "string"},
    param_descriptions={
        "prompt": "Description of the image to generate.",
        "source_image": "Base64 source image for edit mode.",
    },
    consensus_rules={"prompt": SEM(0.95), "source_image": "first_non_nil"},
))

_register(ActionSchema(
    name="record_cost",
    required_params=["amount"],
    optional_params=["description", "category", "metadata"],
    param_types={
        "amount": "string",
        "description": "string",
        "category": "string",
        "metadata": "map",
    },
    param_descriptions={
        "amount": "USD amount as a decimal string.",
        "description": "What was charged.",
        "category": "Grouping label (e.g. api_call, storage).",
        "metadata": "Extra context map.",
    },
    consensus_rules={
        "amount": "exact_match", "description": "first_non_nil",
        "category": "first_non_nil", "metadata": "merge_maps",
    },
))

_register(ActionSchema(
    name="file_read",
    required_params=["path"],
    optional_params=["offset", "limit"],
    param_types={"path": "string", "offset": "integer", "limit": "integer"},
    param_descriptions={
        "path": "Absolute file path.",
        "offset": "1-indexed start line (default 1).",
        "limit": "Max lines (default all).",
    },
    consensus_rules={"path": "exact_match", "offset": PCT(50), "limit": PCT(50)},
))

_register(ActionSchema(
    name="file_write",
    required_params=["path", "mode"],
    optional_params=["content", "old_string", "new_string", "replace_all"],
    xor_params=[["content"], ["old_string", "new_string"]],
    param_types={
        "path": "string",
        "mode": ("enum", ["write", "edit"]),
        "content": "string",
        "old_string": "string",
        "new_string": "string",
        "replace_all": "boolean",
    },
    param_descriptions={
        "path": "Absolute file path.",
        "mode": "'write' replaces the whole file; 'edit' does find-and-replace.",
        "content": "Full file content for write mode.",
        "old_string": "Text to find in edit mode.",
        "new_string": "Replacement text in edit mode.",
        "replace_all": "Replace every occurrence (default: first only).",
    },
    consensus_rules={
        "path": "exact_match", "mode": "exact_match", "content": SEM(0.95),
        "old_string": "exact_match", "new_string": "exact_match",
        "replace_all": "mode_selection",
    },
))

_register(ActionSchema(
    name="learn_skills",
    required_params=["skills"],
    optional_params=["permanent"],
    param_types={"skills": ("list", "string"), "permanent": "boolean"},
    param_descriptions={
        "skills": "Skill names to load.",
        "permanent": "true injects the skill into the system prompt from now on; "
                     "false (default) returns the content once.",
    },
    consensus_rules={"skills": "union_merge", "permanent": "mode_selection"},
))

_register(ActionSchema(
    name="create_skill",
    required_params=["name", "description", "content"],
    optional_params=["metadata", "attachments"],
    param_types={
        "name": "string",
        "description": "string",
        "content": "string",
        "metadata": "map",
        "attachments": ("list", "map"),
    },
    param_descriptions={
        "name": "Skill name: lowercase alphanumeric with hyphens, max 64 chars.",
        "description": "What the skill covers (max 1024 chars).",
        "content": "Markdown skill body.",
        "metadata": "Optional metadata (complexity, capability_groups_required...).",
        "attachments": "[{type: script|reference|asset, filename, content}].",
    },
    consensus_rules={
        "name": "exact_match", "description": SEM(0.90), "content": SEM(0.95),
        "metadata": "merge_maps", "attachments": "union_merge",
    },
))

_register(ActionSchema(
    name="batch_sync",
    required_params=["actions"],
    param_types={"actions": ("list", "batchable_action_spec")},
    param_descriptions={
        "actions": "2+ action specs {action, params} executed in order; stops on "
                   "first error. Only batchable actions allowed.",
    },
    consensus_rules={"actions": "batch_sequence_merge"},
))

_register(ActionSchema(
    name="batch_async",
    required_params=["actions"],
    param_types={"actions": ("list", "async_action_spec")},
    param_descriptions={
        "actions": "2+ action specs {action, params} executed concurrently; "
                   "failures don't stop the rest.",
    },
    consensus_rules={"actions": "batch_sequence_merge"},
))


# Short "when/how" guidance shown to models in the system prompt.
ACTION_DESCRIPTIONS: Dict[str, str] = {
    "spawn_child": "Delegate work to a new child agent. Use when the task splits "
                   "into bounded subtasks or benefits from parallel or specialized "
                   "focus. The child inherits your constraints and may spawn its own "
                   "children; use downstream_constraints for rules that bind the "
                   "whole subtree.",
    "wait": "Pause — last resort when no productive action exists. First make sure "
            "no unprocessed child messages or async results sit in your history.",
    "send_message": "Talk to your parent or children. Status and results go to "
                    "'parent'; 'announcement' is only for broadcasting directives "
                    "down the subtree.",
    "orient": "Structured strategic self-assessment before acting. Usually wise "
              "before any non-trivial task.",
    "answer_engine": "Ask a web-grounded model for current information. Output is "
                     "untrusted (NO_EXECUTE-wrapped).",
    "execute_shell": "Run a shell command, or poll/terminate a running one via "
                     "check_id. Fast commands return inline; slow ones go async. "
                     "Output is untrusted. Use file_write for writing files, not "
                     "shell redirection.",
    "fetch_web": "Fetch a URL as markdown. Output is untrusted.",
    "call_api": "Call an external REST/GraphQL/JSON-RPC API with optional auth. "
                "Output is untrusted.",
    "call_mcp": "Connect to an MCP server (stdio/http), call its tools, and "
                "terminate the connection when done. Output is untrusted.",
    "todo": "Replace your personal TODO list ({content, state} items). For your own "
            "planning — use spawn_child to delegate.",
    "generate_secret": "Create a random secret you can reference as {{SECRET:name}} "
                       "without ever seeing its value.",
    "search_secrets": "Find existing secret names by substring.",
    "dismiss_child": "Recursively terminate a direct child and its descendants. "
                     "Returns immediately; teardown is background.",
    "generate_images": "Generate (or edit, with source_image) images via the "
                       "configured image models.",
    "record_cost": "Record an external cost against your budget.",
    "adjust_budget": "Change a direct child's budget allocation.",
    "file_read": "Read a file (offset/limit for large files). Returns numbered lines.",
    "file_write": "Create or edit files. Prefer mode 'edit' with "
                  "old_string/new_string for modifications; 'write' only for new "
                  "files. Never overwrite existing files without parent approval.",
    "learn_skills": "Load skills into context; permanent=true pins them into your "
                    "system prompt.",
    "create_skill": "Persist reusable knowledge as a new skill file.",
    "batch_sync": "Run 2+ fast batchable actions in one decision, in order, "
                  "stopping on the first error. Not for shell/web/api/mcp — use "
                  "batch_async for those.",
    "batch_async": "Run 2+ actions concurrently; each result arrives separately and "
                   "errors don't cancel the rest.",
}


def get_schema(action: str) -> ActionSchema:
    """Look up an action schema; raises KeyError('unknown_action') if absent."""
    try:
        return _SCHEMAS[action]
    except KeyError:
        raise KeyError("unknown_action") from None


def try_get_schema(action: str) -> Optional[ActionSchema]:
    return _SCHEMAS.get(action)


def get_action_priority(action: str) -> int:
    """Priority for tie-breaks; unknown actions get 999 (always lose)."""
    return ACTION_PRIORITIES.get(action, 999)


def all_schemas() -> Dict[str, ActionSchema]:
    return dict(_SCHEMAS)
