"""Groves: directory-based governance manifests.

A grove is a directory with a GROVE.md whose YAML frontmatter declares
bootstrap fields, spawn topology, governance docs, filesystem confinement
(+ mode), file-write schemas, and a workspace.  Behavior-parity with the
reference (reference: lib/quoracle/groves/loader.ex:12-70,
hard_rule_enforcer.ex:45-188, path_security.ex:15-28,
schema_validator.ex:20-79,236-317).
"""

from __future__ import annotations

import json
import os
import re
from typing import Any, Dict, List, Optional, Tuple

import yaml


class HardRuleViolation(Exception):
    def __init__(self, detail: Dict[str, Any]):
        super().__init__(detail.get("message", "hard rule violation"))
        self.detail = detail


class ConfinementViolation(Exception):
    def __init__(self, detail: Dict[str, Any]):
        super().__init__(detail.get("message", "confinement violation"))
        self.detail = detail


class SchemaViolation(Exception):
    def __init__(self, detail: Any):
        super().__init__(str(detail))
        self.detail = detail


# ---------------------------------------------------------------------------
# Glob path matching (reference: schema_validator.ex:262-317)
#   segments match exactly; "*" matches one segment; "**" matches >= 0
#   segments; "*" inside a segment is a within-segment wildcard.
# ---------------------------------------------------------------------------

def _segments(path: str) -> List[str]:
    return [s for s in path.split("/") if s]


def _segment_match(segment: str, pattern: str) -> bool:
    if pattern == "*":
        return True
    if "*" not in pattern:
        return segment == pattern
    regex = "^" + ".*".join(re.escape(p) for p in pattern.split("*")) + "$"
    # within-segment wildcard must not cross "/" (segments never contain "/")
    return re.match(regex, segment) is not None


def _match_segments(path_segs: List[str], pat_segs: List[str]) -> bool:
    if not pat_segs:
        return not path_segs
    if pat_segs[0] == "**":
        if _match_segments(path_segs, pat_segs[1:]):
            return True
        return bool(path_segs) and _match_segments(path_segs[1:], pat_segs)
    if not path_segs:
        return False
    return (_segment_match(path_segs[0], pat_segs[0])
            and _match_segments(path_segs[1:], pat_segs[1:]))


def path_matches_pattern(path: str, pattern: str) -> bool:
    return _match_segments(_segments(path), _segments(pattern))


def pattern_specificity(pattern: str) -> Tuple[int, int]:
    """(wildcard_count, -length): lower sorts first = most specific wins."""
    wildcards = sum(1 for s in _segments(pattern) if "*" in s)
    return (wildcards, -len(pattern))


# ---------------------------------------------------------------------------
# Hard rules: shell_pattern_block + action_block
# ---------------------------------------------------------------------------

def _rule_applies(rule: Dict[str, Any], skill_name: Optional[str]) -> bool:
    scope = rule.get("scope")
    if scope == "all":
        return True
    if isinstance(scope, list):
        return skill_name in scope
    return True


def _rule_message(rule: Dict[str, Any]) -> str:
    msg = rule.get("message")
    return msg if isinstance(msg, str) and msg else "Action blocked by grove hard rule"


def check_shell_command(command: str, hard_rules: Optional[List[Dict[str, Any]]],
                        skill_name: Optional[str] = None) -> None:
    for rule in hard_rules or []:
        if not isinstance(rule, dict) or rule.get("type") != "shell_pattern_block":
            continue
        pattern = rule.get("pattern")
        if not isinstance(pattern, str) or not _rule_applies(rule, skill_name):
            continue
        try:
            regex = re.compile(pattern)
        except re.error:
            continue  # invalid rule regex: warn-and-allow, like the reference
        if regex.search(command):
            raise HardRuleViolation({
                "type": "shell_pattern_block", "pattern": pattern,
                "command": command, "message": _rule_message(rule)})


def check_action(action: str, hard_rules: Optional[List[Dict[str, Any]]],
                 skill_name: Optional[str] = None) -> None:
    for rule in hard_rules or []:
        if not isinstance(rule, dict) or rule.get("type") != "action_block":
            continue
        actions = [a for a in rule.get("actions", []) if isinstance(a, str)]
        if _rule_applies(rule, skill_name) and action in actions:
            raise HardRuleViolation({
                "type": "action_block", "actions": actions,
                "action": action, "message": _rule_message(rule)})


# ---------------------------------------------------------------------------
# Confinement
# ---------------------------------------------------------------------------

def _confinement_entry(confinement: Dict[str, Any], skill_name: Optional[str],
                       mode: Optional[str]):
    entry = confinement.get(skill_name)
    if isinstance(entry, dict):
        return entry
    default = confinement.get("default")
    if isinstance(default, dict):
        return default
    if mode == "strict":
        return None  # unlisted skill in strict mode: deny
    return "allow"


def _expand_home(pattern: str) -> str:
    if pattern == "~":
        return os.path.expanduser("~")
    if pattern.startswith("~/"):
        return os.path.join(os.path.expanduser("~"), pattern[2:])
    return pattern


def _path_allowed(path: str, patterns: List[str]) -> bool:
    expanded = os.path.abspath(path)
    return any(path_matches_pattern(expanded, _expand_home(p))
               for p in patterns if isinstance(p, str))


def check_shell_working_dir(working_dir: str, confinement: Optional[Dict[str, Any]],
                            skill_name: Optional[str] = None,
                            confinement_mode: Optional[str] = None) -> None:
    if not confinement:
        return
    entry = _confinement_entry(confinement, skill_name, confinement_mode)
    if entry == "allow":
        return
    if entry is None:
        raise ConfinementViolation({
            "working_dir": working_dir, "skill": skill_name, "allowed_paths": [],
            "message": f"Strict confinement: no entry for skill {skill_name}"})
    allowed = entry.get("paths") or []
    if not _path_allowed(working_dir, allowed):
        raise ConfinementViolation({
            "working_dir": working_dir, "skill": skill_name,
            "allowed_paths": allowed,
            "message": "Working directory is outside allowed confinement paths"})


def check_file_access(path: str, access_type: str,
                      confinement: Optional[Dict[str, Any]],
                      skill_name: Optional[str] = None,
                      confinement_mode: Optional[str] = None) -> None:
    if not confinement:
        return
    entry = _confinement_entry(confinement, skill_name, confinement_mode)
    if entry == "allow":
        return
    if entry is None:
        raise ConfinementViolation({
            "path": path, "skill": skill_name, "access_type": access_type,
            "allowed_paths": [],
            "message": f"Strict confinement: no entry for skill {skill_name}"})
    allowed = list(entry.get("paths") or [])
    if access_type == "read":
        allowed += list(entry.get("read_only_paths") or [])
    if not _path_allowed(path, allowed):
        raise ConfinementViolation({
            "path": path, "skill": skill_name, "access_type": access_type,
            "allowed_paths": allowed,
            "message": f"File {access_type} outside allowed confinement paths"})


# ---------------------------------------------------------------------------
# Path security (reference: groves/path_security.ex)
# ---------------------------------------------------------------------------

def path_traversal(filename: str) -> bool:
    return filename.startswith("/") or ".." in _segments(filename)


def symlink_outside_root(full_path: str, root: str) -> bool:
    """True when the target or any intermediate directory symlinks outside root."""
    canonical_root = os.path.abspath(root)
    try:
        resolved = os.path.realpath(full_path)
    except OSError:
        return True
    return not (resolved + "/").startswith(canonical_root + "/")


def safe_read_file(source_path: str, grove_path: str) -> str:
    if path_traversal(source_path):
        raise ConfinementViolation({"path": source_path,
                                    "message": "path traversal rejected"})
    full = os.path.join(grove_path, source_path)
    if symlink_outside_root(full, grove_path):
        raise ConfinementViolation({"path": source_path,
                                    "message": "symlink escapes grove"})
    with open(full, "r") as f:
        return f.read().strip()


# ---------------------------------------------------------------------------
# Grove loader (reference: groves/loader.ex)
# ---------------------------------------------------------------------------

_FRONTMATTER_RE = re.compile(r"\A---\s*\n(.*?)\n---\s*(\n|\Z)", re.DOTALL)

BOOTSTRAP_FIELDS = [
    # inline values and *_file indirections are both accepted; a _file field
    # resolves to the file's contents at load (path-security checked)
    "global_context", "task_description", "success_criteria",
    "immediate_context", "approach_guidance",
    "global_context_file", "task_description_file", "success_criteria_file",
    "immediate_context_file", "approach_guidance_file", "global_constraints",
    "output_style", "role", "cognitive_style", "delegation_strategy",
    "skills", "profile", "budget_limit",
]


def parse_grove_markdown(text: str, path: str = "") -> Dict[str, Any]:
    match = _FRONTMATTER_RE.match(text)
    if not match:
        raise ValueError("missing GROVE.md frontmatter")
    meta = yaml.safe_load(match.group(1)) or {}
    if not isinstance(meta, dict):
        raise ValueError("malformed frontmatter")
    grove = {
        "name": str(meta.get("name", os.path.basename(path) or "grove")),
        "description": str(meta.get("description", "")),
        "version": str(meta.get("version", "0")),
        "path": path,
        "bootstrap": {k: meta.get(k) for k in BOOTSTRAP_FIELDS if k in meta},
        "topology": meta.get("topology") or {},
        "governance": meta.get("governance"),
        "confinement": meta.get("confinement"),
        "confinement_mode": meta.get("confinement_mode"),
        # non-dict rule entries are dropped at parse time so the enforcement
        # hot path (check_shell_command/check_action) can trust the shape
        "hard_rules": [r for r in (meta.get("hard_rules") or [])
                       if isinstance(r, dict)]
        if isinstance(meta.get("hard_rules"), list) else [],
        "schemas": meta.get("schemas"),
        "workspace": meta.get("workspace"),
        "skills_path": meta.get("skills_path"),
    }
    return grove


def load_grove(grove_dir: str) -> Dict[str, Any]:
    manifest = os.path.join(grove_dir, "GROVE.md")
    with open(manifest, "r") as f:
        grove = parse_grove_markdown(f.read(), path=grove_dir)
    # Confinement path patterns are grove-relative: anchor them here so
    # enforcement compares absolute paths.
    conf = grove.get("confinement")
    if isinstance(conf, dict):
        for entry in conf.values():
            if not isinstance(entry, dict):
                continue
            for key in ("paths", "read_only_paths"):
                pats = entry.get(key)
                if isinstance(pats, list):
                    entry[key] = [
                        p if (not isinstance(p, str) or os.path.isabs(p)
                              or p.startswith("~"))
                        else os.path.join(grove_dir, p)
                        for p in pats]
    # Bootstrap *_file fields resolve to file contents (path-security checked).
    bootstrap = grove["bootstrap"]
    for key in list(bootstrap):
        if key.endswith("_file") and isinstance(bootstrap[key], str):
            target = key[: -len("_file")]
            try:
                bootstrap[target] = safe_read_file(bootstrap[key], grove_dir)
            except (OSError, ConfinementViolation):
                bootstrap[target] = None
    return grove


def list_groves(groves_dir: str) -> List[Dict[str, Any]]:
    out = []
    if not os.path.isdir(groves_dir):
        return out
    for entry in sorted(os.listdir(groves_dir)):
        grove_dir = os.path.join(groves_dir, entry)
        if not os.path.isfile(os.path.join(grove_dir, "GROVE.md")):
            continue
        try:
            grove = load_grove(grove_dir)
        except (ValueError, OSError, yaml.YAMLError):
            continue  # malformed groves are skipped
        out.append({"name": grove["name"], "description": grove["description"],
                    "version": grove["version"], "path": grove_dir})
    return out


def substitute_grove_vars(value: Any, grove_vars: Dict[str, str]) -> Any:
    """Replace {var} placeholders in confinement paths etc."""
    if isinstance(value, str):
        for key, replacement in grove_vars.items():
            value = value.replace("{" + key + "}", str(replacement))
        return value
    if isinstance(value, dict):
        return {k: substitute_grove_vars(v, grove_vars) for k, v in value.items()}
    if isinstance(value, list):
        return [substitute_grove_vars(v, grove_vars) for v in value]
    return value


# ---------------------------------------------------------------------------
# File-write schema validation (reference: groves/schema_validator.ex:20-79)
# ---------------------------------------------------------------------------

def select_schema(schemas: Optional[List[Dict[str, Any]]], file_path: str,
                  workspace: Optional[str],
                  grove_path: Optional[str] = None) -> Optional[Dict[str, Any]]:
    """Most-specific matching path_pattern wins.  Patterns are relative to
    the workspace when the file lives there, else to the grove root."""
    if not schemas:
        return None
    expanded = os.path.abspath(file_path)
    bases = []
    if workspace:
        ws = workspace if os.path.isabs(workspace) else \
            os.path.join(grove_path or ".", workspace)
        bases.append(os.path.abspath(ws))
    if grove_path:
        bases.append(os.path.abspath(grove_path))
    relative = None
    for base in bases:
        prefix = base.rstrip("/") + "/"
        if expanded.startswith(prefix):
            relative = expanded[len(prefix):]
            break
    if relative is None:
        return None
    matching = [s for s in schemas
                if isinstance(s.get("path_pattern"), str)
                and path_matches_pattern(relative, s["path_pattern"])]
    if not matching:
        return None
    return min(matching, key=lambda s: pattern_specificity(s["path_pattern"]))


def validate_json_subset(instance: Any, schema: Dict[str, Any], path: str = "$") -> None:
    """Minimal JSON-Schema (Draft 2020-12 subset) validator: type, properties,
    required, items, enum, additionalProperties, const, min/max bounds."""
    if "const" in schema and instance != schema["const"]:
        raise SchemaViolation(f"{path}: expected const {schema['const']!r}")
    if "enum" in schema and instance not in schema["enum"]:
        raise SchemaViolation(f"{path}: not in enum")
    stype = schema.get("type")
    if stype:
        types = stype if isinstance(stype, list) else [stype]
        if not any(_json_type_ok(instance, t) for t in types):
            raise SchemaViolation(f"{path}: expected type {stype}")
    if isinstance(instance, dict):
        for req in schema.get("required", []):
            if req not in instance:
                raise SchemaViolation(f"{path}: missing required '{req}'")
        props = schema.get("properties", {})
        for key, sub in props.items():
            if key in instance:
                validate_json_subset(instance[key], sub, f"{path}.{key}")
        if schema.get("additionalProperties") is False:
            extra = [k for k in instance if k not in props]
            if extra:
                raise SchemaViolation(f"{path}: unexpected keys {extra}")
    if isinstance(instance, list) and "items" in schema:
        for i, item in enumerate(instance):
            validate_json_subset(item, schema["items"], f"{path}[{i}]")
    if isinstance(instance, (int, float)) and not isinstance(instance, bool):
        if "minimum" in schema and instance < schema["minimum"]:
            raise SchemaViolation(f"{path}: below minimum")
        if "maximum" in schema and instance > schema["maximum"]:
            raise SchemaViolation(f"{path}: above maximum")
    if isinstance(instance, str):
        if "minLength" in schema and len(instance) < schema["minLength"]:
            raise SchemaViolation(f"{path}: too short")
        if "maxLength" in schema and len(instance) > schema["maxLength"]:
            raise SchemaViolation(f"{path}: too long")


def _json_type_ok(instance: Any, t: str) -> bool:
    return {
        "object": isinstance(instance, dict),
        "array": isinstance(instance, list),
        "string": isinstance(instance, str),
        "integer": isinstance(instance, int) and not isinstance(instance, bool),
        "number": isinstance(instance, (int, float)) and not isinstance(instance, bool),
        "boolean": isinstance(instance, bool),
        "null": instance is None,
    }.get(t, False)


def validate_file_write(grove: Optional[Dict[str, Any]], file_path: str,
                        content: str) -> None:
    """Validate file content against the grove schema matching its path."""
    if not grove:
        return
    schema_entry = select_schema(grove.get("schemas"), file_path,
                                 grove.get("workspace"), grove.get("path"))
    if schema_entry is None:
        return
    definition = schema_entry.get("definition") or schema_entry.get("schema")
    if isinstance(definition, str):
        schema_json = safe_read_file(definition, grove.get("path", "."))
        schema = json.loads(schema_json)
    elif isinstance(definition, dict):
        schema = definition
    else:
        raise SchemaViolation("invalid schema definition")
    try:
        instance = json.loads(content)
    except json.JSONDecodeError as exc:
        raise SchemaViolation(f"content is not valid JSON: {exc}") from None
    validate_json_subset(instance, schema)
