"""Schema-driven action parameter validation and coercion.

Behavior-parity with the reference validator (reference:
lib/quoracle/actions/validator.ex, validator/type_validation.ex,
validator/batch_sync.ex, validator/batch_async.ex): required params, unknown
params, XOR groups, typed coercion with LLM leniency ({} -> [], "true" ->
true, enum strings), protocol-specific call_api checks, and batch spec
validation (min 2, batchable subset, no nesting).
"""

from __future__ import annotations

import math

from typing import Any, Dict, List, Optional
from urllib.parse import urlparse

from . import schema as schema_mod
from .schema import ASYNC_EXCLUDED_ACTIONS, BATCHABLE_ACTIONS


class ValidationError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


def validate_action(action_json: Dict[str, Any], *,
                    profile_optional: bool = False) -> Dict[str, Any]:
    """Validate a full {action, params[, reasoning]} map.

    Returns {action, params[, reasoning]} with coerced params.
    Raises ValidationError.
    """
    if "action" not in action_json:
        raise ValidationError("missing_action_field")
    if "params" not in action_json:
        raise ValidationError("missing_params_field")
    action = action_json["action"]
    if action not in schema_mod.ACTIONS:
        raise ValidationError("unknown_action")
    params = validate_params(action, action_json["params"],
                             profile_optional=profile_optional)
    result = {"action": action, "params": params}
    if action_json.get("reasoning") is not None:
        result["reasoning"] = action_json["reasoning"]
    return result


def validate_params(action: str, params: Any, *,
                    profile_optional: bool = False) -> Dict[str, Any]:
    sch = schema_mod.try_get_schema(action)
    if sch is None:
        raise ValidationError("unknown_action")
    if not isinstance(params, dict):
        raise ValidationError("invalid_param_type")

    if action == "spawn_child":
        required = [p for p in sch.required_params
                    if not (profile_optional and p == "profile")]
        _check_required(params, required)
    elif action == "call_api":
        _check_call_api(params, sch)
    elif action == "batch_sync":
        _check_batch(params, sync=True)
    elif action == "batch_async":
        _check_batch(params, sync=False)
    else:
        _check_required(params, sch.required_params)

    _check_unknown(params, sch)
    _check_xor(params, sch.xor_params)
    validated = _validate_types(params, sch.param_types)

    if action == "call_api" and isinstance(validated.get("method"), str):
        validated["method"] = validated["method"].upper()
    return validated


def _check_required(params: Dict[str, Any], required: List[str]) -> None:
    missing = [p for p in required if p not in params]
    if missing:
        raise ValidationError("missing_required_param")


def _check_unknown(params: Dict[str, Any], sch: schema_mod.ActionSchema) -> None:
    allowed = set(sch.all_params)
    unknown = [k for k in params if k not in allowed]
    if unknown:
        raise ValidationError("unknown_parameter")


def _check_xor(params: Dict[str, Any], xor_groups: Optional[List[List[str]]]) -> None:
    if not xor_groups:
        return
    present = [g for g in xor_groups if any(p in params for p in g)]
    if len(present) == 0:
        raise ValidationError("xor_params_required")
    if len(present) > 1:
        raise ValidationError("xor_params_conflict")


def _check_call_api(params: Dict[str, Any], sch: schema_mod.ActionSchema) -> None:
    _check_required(params, sch.required_params)
    url = params.get("url")
    if url is None:
        raise ValidationError("missing_required_param")
    if not isinstance(url, str) or urlparse(url).scheme not in ("http", "https"):
        raise ValidationError("invalid_url_scheme")
    api_type = params.get("api_type")
    needs = {"rest": "method", "graphql": "query", "jsonrpc": "rpc_method"}
    needed = needs.get(api_type)
    if needed and needed not in params:
        raise ValidationError("missing_required_param")
    method = params.get("method")
    if isinstance(method, str) and \
            method.upper() not in ("GET", "POST", "PUT", "DELETE", "PATCH"):
        raise ValidationError("invalid_http_method")


def _check_batch(params: Dict[str, Any], *, sync: bool) -> None:
    actions = params.get("actions")
    if not isinstance(actions, list):
        raise ValidationError("missing_required_param")
    if len(actions) < 2:
        raise ValidationError("batch_too_small")
    for spec in actions:
        if not isinstance(spec, dict):
            raise ValidationError("invalid_param_type")
        action = spec.get("action")
        if action is None:
            raise ValidationError("invalid_param_type")
        if action in ("batch_sync", "batch_async"):
            raise ValidationError("nested_batch_not_allowed")
        if sync:
            if action not in BATCHABLE_ACTIONS:
                raise ValidationError("action_not_batchable")
        else:
            if action in ASYNC_EXCLUDED_ACTIONS:
                raise ValidationError("action_not_batchable")


def _validate_types(params: Dict[str, Any], param_types: Dict[str, Any]) -> Dict[str, Any]:
    out: Dict[str, Any] = {}
    for key, value in params.items():
        expected = param_types.get(key)
        if expected is None:
            out[key] = value
            continue
        out[key] = _validate_type(value, expected)
    return out


def _validate_type(value: Any, expected: Any) -> Any:
    if expected == "string":
        if isinstance(value, str):
            return value
        raise ValidationError("invalid_param_type")
    if expected == "integer":
        if isinstance(value, int) and not isinstance(value, bool):
            return value
        raise ValidationError("invalid_param_type")
    if expected == "number":
        if isinstance(value, (int, float)) and not isinstance(value, bool):
            # Python's json accepts NaN/Infinity literals; they would poison
            # merges (median/percentile) and timers downstream
            if isinstance(value, float) and not math.isfinite(value):
                raise ValidationError("invalid_param_type")
            return value
        raise ValidationError("invalid_param_type")
    if expected == "boolean":
        if isinstance(value, bool):
            return value
        # LLM leniency: "true"/"false" strings
        if value == "true":
            return True
        if value == "false":
            return False
        raise ValidationError("invalid_param_type")
    if expected == "map":
        if isinstance(value, dict):
            return value
        raise ValidationError("invalid_param_type")
    if expected == "any":
        return value

    if isinstance(expected, tuple):
        kind = expected[0]
        if kind == "enum":
            choices = expected[1]
            if value in choices:
                return value
            raise ValidationError("invalid_enum_value")
        if kind == "list":
            item_type = expected[1]
            # LLM leniency: {} treated as []
            if value == {}:
                return []
            if not isinstance(value, list):
                raise ValidationError("invalid_param_type")
            if item_type in ("batchable_action_spec", "async_action_spec"):
                return [_validate_action_spec(v) for v in value]
            return [_validate_type(v, item_type) for v in value]
        if kind == "union":
            for t in expected[1]:
                try:
                    return _validate_type(value, t)
                except ValidationError:
                    continue
            raise ValidationError("invalid_param_type")
        if kind == "map_shape":
            if not isinstance(value, dict):
                raise ValidationError("invalid_param_type")
            props = expected[1]
            missing = [k for k in props if k not in value]
            if missing:
                raise ValidationError("missing_required_field")
            extra = [k for k in value if k not in props]
            if extra:
                raise ValidationError("unknown_field")
            return {k: _validate_type(value[k], t) for k, t in props.items()}
    raise ValidationError("invalid_param_type")


def _validate_action_spec(spec: Any) -> Dict[str, Any]:
    if not isinstance(spec, dict):
        raise ValidationError("invalid_param_type")
    action = spec.get("action")
    if not isinstance(action, str):
        raise ValidationError("invalid_param_type")
    if action not in schema_mod.ACTIONS:
        raise ValidationError("unknown_action")
    params = validate_params(action, spec.get("params") or {})
    return {"action": action, "params": params}
