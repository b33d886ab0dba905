"""Validator + parser behavior tests (parity targets:
lib/quoracle/actions/validator.ex, lib/quoracle/consensus/action_parser.ex)."""

import pytest

from quoracle_amd.actions.validator import ValidationError, validate_action, validate_params
from quoracle_amd.consensus.parser import ParseError, parse_pool_responses, parse_response
from quoracle_amd.consensus.temperature import round_temperature


class TestParser:
    def test_plain_json(self):
        r = parse_response('{"action": "wait", "params": {}, "reasoning": "idle", "wait": 5}')
        assert r["action"] == "wait" and r["wait"] == 5

    def test_fenced_json(self):
        text = 'Sure!\n```json\n{"action": "wait", "params": {}, "reasoning": "x"}\n```'
        assert parse_response(text)["action"] == "wait"

    def test_embedded_object(self):
        text = 'prefix {"action": "wait", "params": {}, "reasoning": "x"} suffix'
        assert parse_response(text)["action"] == "wait"

    def test_invalid_json(self):
        with pytest.raises(ParseError) as e:
            parse_response("not json at all")
        assert e.value.reason == "invalid_json"

    def test_missing_reasoning(self):
        with pytest.raises(ParseError) as e:
            parse_response('{"action": "wait", "params": {}}')
        assert e.value.reason == "missing_fields"

    def test_unknown_action(self):
        with pytest.raises(ParseError) as e:
            parse_response('{"action": "frobnicate", "params": {}, "reasoning": "x"}')
        assert e.value.reason == "unknown_action"

    def test_condense_extracted(self):
        r = parse_response('{"action": "wait", "params": {}, "reasoning": "x", "condense": 3}')
        assert r["condense"] == 3

    def test_pool_partial(self):
        pool = parse_pool_responses({
            "m1": '{"action": "wait", "params": {}, "reasoning": "x"}',
            "m2": "garbage",
        })
        assert len(pool.valid) == 1
        assert pool.errors == {"m2": "invalid_json"}
        assert pool.valid[0]["model"] == "m1"


class TestValidator:
    def test_ok(self):
        v = validate_action({"action": "file_read", "params": {"path": "/x"}})
        assert v["params"]["path"] == "/x"

    def test_missing_required(self):
        with pytest.raises(ValidationError) as e:
            validate_params("file_read", {})
        assert e.value.reason == "missing_required_param"

    def test_unknown_param(self):
        with pytest.raises(ValidationError) as e:
            validate_params("file_read", {"path": "/x", "bogus": 1})
        assert e.value.reason == "unknown_parameter"

    def test_enum_validation(self):
        with pytest.raises(ValidationError) as e:
            validate_params("file_write",
                            {"path": "/x", "mode": "append", "content": "c"})
        assert e.value.reason == "invalid_enum_value"

    def test_xor_required(self):
        with pytest.raises(ValidationError) as e:
            validate_params("execute_shell", {})
        assert e.value.reason == "xor_params_required"

    def test_xor_conflict(self):
        with pytest.raises(ValidationError) as e:
            validate_params("execute_shell", {"command": "ls", "check_id": "c1"})
        assert e.value.reason == "xor_params_conflict"

    def test_bool_coercion(self):
        v = validate_params("execute_shell", {"check_id": "c1", "terminate": "true"})
        assert v["terminate"] is True

    def test_empty_map_as_list(self):
        v = validate_params("search_secrets", {"search_terms": {}})
        assert v["search_terms"] == []

    def test_call_api_rest_needs_method(self):
        with pytest.raises(ValidationError):
            validate_params("call_api", {"api_type": "rest", "url": "https://x.test"})
        v = validate_params("call_api", {"api_type": "rest", "url": "https://x.test",
                                         "method": "get"})
        assert v["method"] == "GET"

    def test_call_api_bad_scheme(self):
        with pytest.raises(ValidationError) as e:
            validate_params("call_api", {"api_type": "rest", "url": "ftp://x",
                                         "method": "GET"})
        assert e.value.reason == "invalid_url_scheme"

    def test_call_api_bad_method(self):
        with pytest.raises(ValidationError) as e:
            validate_params("call_api", {"api_type": "rest", "url": "https://x.test",
                                         "method": "BREW"})
        assert e.value.reason == "invalid_http_method"

    def test_batch_sync_min_two(self):
        with pytest.raises(ValidationError) as e:
            validate_params("batch_sync",
                            {"actions": [{"action": "todo", "params": {"items": []}}]})
        assert e.value.reason == "batch_too_small"

    def test_batch_sync_not_batchable(self):
        with pytest.raises(ValidationError) as e:
            validate_params("batch_sync", {"actions": [
                {"action": "execute_shell", "params": {"command": "ls"}},
                {"action": "todo", "params": {"items": []}}]})
        assert e.value.reason == "action_not_batchable"

    def test_batch_async_allows_shell(self):
        v = validate_params("batch_async", {"actions": [
            {"action": "execute_shell", "params": {"command": "ls"}},
            {"action": "fetch_web", "params": {"url": "https://x.test"}}]})
        assert len(v["actions"]) == 2

    def test_no_nested_batch(self):
        with pytest.raises(ValidationError) as e:
            validate_params("batch_async", {"actions": [
                {"action": "batch_sync", "params": {"actions": []}},
                {"action": "todo", "params": {"items": []}}]})
        assert e.value.reason == "nested_batch_not_allowed"

    def test_spawn_profile_optional(self):
        params = {"task_description": "t", "success_criteria": "s",
                  "immediate_context": "i", "approach_guidance": "a"}
        with pytest.raises(ValidationError):
            validate_params("spawn_child", params)
        v = validate_params("spawn_child", params, profile_optional=True)
        assert v["task_description"] == "t"

    def test_todo_items_shape(self):
        v = validate_params("todo", {"items": [{"content": "x", "state": "todo"}]})
        assert v["items"][0]["state"] == "todo"
        with pytest.raises(ValidationError):
            validate_params("todo", {"items": [{"content": "x", "state": "nah"}]})


class TestTemperature:
    def test_low_family_schedule(self):
        temps = [round_temperature("llama-3-8b", r) for r in (1, 2, 3, 4, 5)]
        assert temps == [1.0, 0.7, 0.5, 0.2, 0.2]

    def test_high_family_schedule(self):
        temps = [round_temperature("openai:gpt-4o", r) for r in (1, 2, 3, 4)]
        assert temps == [2.0, 1.5, 0.9, 0.4]

    def test_adaptive_rounds(self):
        assert round_temperature("llama", 2, max_refinement_rounds=2) == 0.2
        assert round_temperature("llama", 1, max_refinement_rounds=2) == 1.0

    def test_invalid_round(self):
        assert round_temperature("llama", 0) == 1.0


def test_nan_and_infinity_rejected_in_number_params():
    """Python json parses NaN/Infinity literals; the validator must reject
    them before they poison merges or timers."""
    import pytest as _pytest
    from quoracle_amd.actions.validator import ValidationError, validate_params
    for bad in (float("nan"), float("inf"), float("-inf")):
        with _pytest.raises(ValidationError):
            validate_params("wait", {"wait": bad})
    assert validate_params("wait", {"wait": 5.0}) == {"wait": 5.0}


def test_wait_merge_filters_nonfinite():
    from quoracle_amd.consensus.rules import merge_wait_values
    assert merge_wait_values([float("nan"), 10, 20]) == 15
    assert merge_wait_values([float("nan"), float("inf")]) is False


def test_tokenizer_edges():
    from quoracle_amd.engine.tokenizer import EOS, ByteTokenizer
    tok = ByteTokenizer()
    assert tok.count("") == 0
    assert tok.encode("") == []
    assert tok.decode([]) == ""
    assert tok.decode([EOS]) == ""          # EOS renders empty
    assert tok.count("héllo") == len("héllo".encode())


def test_grammar_covers_all_22_actions_and_validates():
    """The constrained-decoding grammar templates every action in the
    schema registry; a forced walk through each choice yields JSON that
    parses and passes schema validation."""
    import json
    import random
    from quoracle_amd.engine.sampler import ActionGrammar, _PARAM_PLANS
    from quoracle_amd.engine.tokenizer import EOS
    from quoracle_amd.actions.schema import ACTIONS
    from quoracle_amd.actions.validator import validate_params

    assert set(_PARAM_PLANS) == set(ACTIONS)
    rng = random.Random(11)
    for action in ACTIONS:
        idx = ACTIONS.index(action)
        for _ in range(3):
            g = ActionGrammar(ACTIONS, context={"child_id": "kid-1"})
            out = []
            while not g.done and len(out) < 4000:
                op = g.current()
                tok = g.advance(idx if op[0] == "choice"
                                else rng.randrange(0, 256))
                out.append(tok)
            assert g.done, f"{action}: grammar never terminated"
            doc = json.loads(bytes(b for b in out if b != EOS).decode())
            assert doc["action"] == action
            validate_params(action, doc.get("params", {}))
