"""Property-based tests (hypothesis) — the reference's stream_data strategy
(SURVEY.md §4): consensus merge-rule invariants, tokenizer roundtrip,
grammar robustness under arbitrary sampler draws."""

import json

from hypothesis import given, settings, strategies as st

from quoracle_amd.consensus import rules as R
from quoracle_amd.engine.sampler import ActionGrammar
from quoracle_amd.engine.tokenizer import EOS, ByteTokenizer

numbers = st.floats(min_value=-1e6, max_value=1e6,
                    allow_nan=False, allow_infinity=False)


@given(st.lists(numbers, min_size=1, max_size=8),
       st.integers(min_value=0, max_value=100))
@settings(max_examples=60, deadline=None)
def test_percentile_rule_bounds_and_interpolation(values, pct):
    out = R.apply_rule(("percentile", pct), list(values))
    # the rule rounds to int (reference: Elixir round/1 semantics)
    import math
    assert math.floor(min(values)) - 1 <= out <= math.ceil(max(values)) + 1
    assert out == int(out)
    if pct == 0:
        assert abs(out - min(values)) <= 0.5 + 1e-6
    if pct == 100:
        assert abs(out - max(values)) <= 0.5 + 1e-6


@given(st.lists(st.sampled_from(["a", "b", "c"]), min_size=1, max_size=9))
@settings(max_examples=40, deadline=None)
def test_mode_selection_returns_most_common(values):
    out = R.apply_rule("mode_selection", list(values))
    counts = {v: values.count(v) for v in set(values)}
    assert counts[out] == max(counts.values())


@given(st.lists(st.lists(st.sampled_from(["x", "y", "z", "w"]), max_size=4),
                min_size=1, max_size=5))
@settings(max_examples=40, deadline=None)
def test_union_merge_is_duplicate_free_superset(lists):
    out = R.apply_rule("union_merge", [list(l) for l in lists])
    assert len(out) == len(set(map(str, out)))
    for l in lists:
        for item in l:
            assert item in out


@given(st.lists(st.one_of(st.booleans(),
                          st.integers(min_value=0, max_value=600)),
                min_size=1, max_size=7))
@settings(max_examples=60, deadline=None)
def test_wait_parameter_rule_type_and_range(values):
    out = R.apply_rule("wait_parameter", list(values))
    assert isinstance(out, (bool, int, float))
    if all(v is True for v in values):
        assert out is True
    if all(v is False for v in values):
        assert out is False
    nums = [v for v in values if not isinstance(v, bool)]
    if nums and not isinstance(out, bool) and not any(
            isinstance(v, bool) for v in values):
        # pure-numeric pools: median stays within range (booleans mix in
        # as 0/inf sentinels per the special-case table)
        assert min(nums) - 0.5 <= out <= max(nums) + 0.5


@given(st.text(max_size=300))
@settings(max_examples=80, deadline=None)
def test_tokenizer_roundtrip_unicode(text):
    tok = ByteTokenizer()
    assert tok.decode(tok.encode(text)) == text.encode(
        "utf-8", "replace").decode("utf-8", "replace")
    assert tok.count(text) == len(text.encode("utf-8", "replace"))


@given(st.lists(st.integers(min_value=0, max_value=2 ** 31 - 1),
                min_size=1, max_size=20),
       st.sampled_from([["orient"], ["wait", "todo"],
                        ["send_message", "orient", "todo", "wait"],
                        ["spawn_child", "file_read"]]))
@settings(max_examples=60, deadline=None)
def test_grammar_always_yields_valid_action_json(draws, allowed):
    """ANY sequence of sampler draws produces parseable action JSON with a
    permitted action — the invariant the consensus parser depends on."""
    g = ActionGrammar(allowed, reasoning_tokens=4)
    out = []
    i = 0
    for _ in range(2000):
        if g.done:
            break
        out.append(g.advance(draws[i % len(draws)]))
        i += 1
    assert g.done
    text = ByteTokenizer().decode([t for t in out if t != EOS])
    parsed = json.loads(text)
    assert parsed["action"] in allowed
    assert isinstance(parsed["params"], dict)


from quoracle_amd.actions import validator as V


@given(st.dictionaries(st.sampled_from(["path", "mode", "content",
                                        "old_string", "new_string",
                                        "bogus_param"]),
                       st.one_of(st.text(max_size=12), st.integers(),
                                 st.booleans()),
                       max_size=5))
@settings(max_examples=80, deadline=None)
def test_validator_never_crashes_and_gates_required(params):
    """Arbitrary param dicts either validate or raise ValidationError —
    never crash; file_write's required params stay enforced."""
    try:
        out = V.validate_params("file_write", dict(params))
    except V.ValidationError as exc:
        assert isinstance(exc.reason, str) and exc.reason
        return
    assert "path" in out and "mode" in out


@given(st.lists(st.dictionaries(
    st.sampled_from(["content", "state", "extra"]),
    st.text(max_size=8), max_size=3), max_size=4))
@settings(max_examples=60, deadline=None)
def test_validator_todo_items_shape(items):
    try:
        out = V.validate_params("todo", {"items": items})
    except V.ValidationError:
        return
    for item in out["items"]:
        assert isinstance(item, dict)
        assert item.get("state") in (None, "todo", "pending", "done")


from quoracle_amd.utils.jsonx import extract_json


@given(st.text(max_size=200))
@settings(max_examples=80, deadline=None)
def test_extract_json_never_crashes(noise):
    """Arbitrary junk around a valid object: extraction still finds it (or
    returns None) without ever raising."""
    assert extract_json(noise) is None or isinstance(extract_json(noise), dict)
    embedded = noise + '{"action": "wait", "params": {}, "reasoning": "r"}'
    out = extract_json(embedded)
    # the balanced-span scan finds the object unless the noise itself
    # contains an earlier '{' that breaks balance — never an exception
    assert out is None or isinstance(out, dict)


def test_extract_json_fenced_and_prefixed():
    body = '{"action": "todo", "params": {"items": []}, "reasoning": "x"}'
    assert extract_json(body)["action"] == "todo"
    assert extract_json(f"Sure! Here:\n```json\n{body}\n```")["action"] == "todo"
    assert extract_json(f"preamble {body} trailing")["action"] == "todo"
    assert extract_json("no json here") is None
    assert extract_json('{"broken": ') is None


# -- validator fuzz: no input shape may crash it ----------------------------

_JSON_LEAF = st.one_of(st.none(), st.booleans(), st.integers(),
                       st.floats(allow_nan=False), st.text(max_size=20))
_JSON = st.recursive(
    _JSON_LEAF,
    lambda inner: st.one_of(st.lists(inner, max_size=4),
                            st.dictionaries(st.text(max_size=8), inner,
                                            max_size=4)),
    max_leaves=10)


@settings(max_examples=200, deadline=None)
@given(params=_JSON, action_idx=st.integers(min_value=0, max_value=21))
def test_validator_never_crashes_on_junk(params, action_idx):
    from quoracle_amd.actions import schema as S
    from quoracle_amd.actions.validator import ValidationError, validate_params
    action = sorted(S.ACTIONS)[action_idx]
    try:
        out = validate_params(action, params)
    except ValidationError:
        return                     # rejection is a valid outcome
    assert isinstance(out, dict)   # acceptance must yield coerced dict params


@settings(max_examples=60, deadline=None)
@given(draws=st.lists(st.integers(min_value=0, max_value=2 ** 31 - 1),
                      min_size=1, max_size=16),
       subset=st.sets(st.integers(min_value=0, max_value=21), max_size=6),
       ctx_depth=st.integers(min_value=0, max_value=3))
def test_grammar_valid_over_any_action_subset(draws, subset, ctx_depth):
    """The grammar invariant holds for EVERY subset of the 22 actions
    (including the empty set -> wait fallback) and any context dict."""
    from quoracle_amd.actions import schema as S
    all_actions = sorted(S.ACTIONS)
    allowed = [all_actions[i] for i in sorted(subset)]
    context = {"task_description": "t" * ctx_depth,
               "profile": "default"} if ctx_depth else None
    g = ActionGrammar(allowed, reasoning_tokens=3, context=context)
    out, i = [], 0
    for _ in range(4000):
        if g.done:
            break
        out.append(g.advance(draws[i % len(draws)]))
        i += 1
    assert g.done
    parsed = json.loads(ByteTokenizer().decode([t for t in out if t != EOS]))
    assert parsed["action"] in (allowed or ["wait"]) or \
        parsed["action"] in g.candidates


@settings(max_examples=120, deadline=None)
@given(action_idx=st.integers(min_value=0, max_value=21), params=_JSON)
def test_fingerprint_total_and_deterministic(action_idx, params):
    """action_fingerprint never crashes on junk responses, is deterministic,
    and is insensitive to unknown params (schema-normalized signature)."""
    from quoracle_amd.actions import schema as S
    from quoracle_amd.consensus.aggregator import action_fingerprint
    action = sorted(S.ACTIONS)[action_idx]
    resp = {"action": action, "params": params}
    fp1 = action_fingerprint(resp)
    fp2 = action_fingerprint(dict(resp))
    assert fp1 == fp2
    assert fp1[0] == action
    if isinstance(params, dict):
        noisy = dict(params)
        noisy["__totally_unknown_param__"] = 123
        assert action_fingerprint(
            {"action": action, "params": noisy}) == fp1


def test_fingerprint_unknown_action_is_invalid_cluster():
    from quoracle_amd.consensus.aggregator import action_fingerprint
    assert action_fingerprint({"action": "no_such", "params": {}}) == \
        ("no_such", "invalid")


_RULES = ["exact_match", ("semantic_similarity", 0.9), "mode_selection",
          "union_merge", "structural_merge", ("percentile", 50),
          "wait_parameter", "batch_sequence_merge"]


@settings(max_examples=150, deadline=None)
@given(rule_idx=st.integers(min_value=0, max_value=len(_RULES) - 1),
       values=st.lists(_JSON, max_size=5))
def test_merge_rules_total_over_junk(rule_idx, values):
    """Every consensus rule either merges or raises NoConsensus on ANY
    value shapes — a junk model response must never crash the merge."""
    from quoracle_amd.consensus.rules import NoConsensus, apply_rule
    from quoracle_amd.engine.fake import deterministic_embedding
    def embed(texts):
        return [deterministic_embedding(str(t)) for t in texts]
    try:
        apply_rule(_RULES[rule_idx], values, embed_many=embed)
    except NoConsensus:
        pass


@settings(max_examples=60, deadline=None)
@given(body=st.text(max_size=400))
def test_grove_loader_total_over_junk_frontmatter(body, tmp_path_factory):
    """load_grove on arbitrary GROVE.md content: parse or ValueError,
    never a crash (governance input is user-editable)."""
    import os
    from quoracle_amd.governance.groves import load_grove
    d = tmp_path_factory.mktemp("grove")
    with open(os.path.join(str(d), "GROVE.md"), "w") as f:
        f.write(body)
    try:
        grove = load_grove(str(d))
        assert isinstance(grove, dict)
    except (ValueError, KeyError):
        pass


@settings(max_examples=60, deadline=None)
@given(rules=_JSON, confinement=_JSON)
def test_grove_loader_total_over_junk_yaml_types(rules, confinement,
                                                 tmp_path_factory):
    """Structurally valid YAML frontmatter with arbitrarily WRONG types for
    hard_rules/confinement must load-or-reject, never crash."""
    import os
    import yaml
    from quoracle_amd.governance.groves import load_grove
    d = tmp_path_factory.mktemp("grove")
    front = yaml.safe_dump({"name": "g", "hard_rules": rules,
                            "confinement": confinement},
                           default_flow_style=True)
    with open(os.path.join(str(d), "GROVE.md"), "w") as f:
        f.write(f"---\n{front}---\nbody\n")
    try:
        grove = load_grove(str(d))
        assert isinstance(grove, dict)
    except (ValueError, KeyError):
        pass


@settings(max_examples=120, deadline=None)
@given(hard_rules=_JSON, confinement=_JSON)
def test_grove_enforcement_total_over_junk_rules(hard_rules, confinement):
    """The gate-chain checks (shell/action/file/working-dir) must treat a
    malformed grove as deny-or-allow, never crash at dispatch time."""
    from quoracle_amd.governance import groves as G
    rules = hard_rules if isinstance(hard_rules, list) else None
    conf = confinement if isinstance(confinement, dict) else None
    for fn, args in [
        (G.check_shell_command, ("rm -rf /", rules)),
        (G.check_action, ("spawn_child", rules)),
        (G.check_shell_working_dir, ("/tmp/x", conf)),
        (G.check_file_access, ("/tmp/y.txt", "write", conf)),
    ]:
        try:
            fn(*args)
        except (G.HardRuleViolation, G.ConfinementViolation):
            pass


_ENTRY = st.one_of(
    _JSON,
    st.fixed_dictionaries({"type": st.text(max_size=10)},
                          optional={"content": _JSON,
                                    "ts": st.one_of(st.none(), st.floats(
                                        allow_nan=False))}))


@settings(max_examples=100, deadline=None)
@given(history=st.lists(_ENTRY, max_size=8))
def test_context_build_total_over_corrupt_history(history):
    """A corrupted checkpoint (arbitrary JSON in model_histories) must not
    crash message building — the restore path renders whatever it finds."""
    from quoracle_amd.agent.context import (build_conversation_messages,
                                            merge_consecutive)
    entries = [e for e in history if isinstance(e, dict)]
    msgs = build_conversation_messages(entries)
    assert isinstance(msgs, list)
    for m in msgs:
        assert m["role"] in ("system", "user", "assistant")
        assert isinstance(m["content"], str)
    merged = merge_consecutive(msgs)
    # alternation guard: no two consecutive same-role messages
    for a, b in zip(merged, merged[1:]):
        assert not (a["role"] == b["role"] and a["role"] != "system")


@settings(max_examples=100, deadline=None)
@given(history=st.lists(_ENTRY, max_size=8))
def test_token_manager_total_over_corrupt_history(history):
    from quoracle_amd.agent.token_manager import (entry_tokens,
                                                  history_tokens,
                                                  split_for_condensation)
    from quoracle_amd.engine.tokenizer import ByteTokenizer
    count = ByteTokenizer().count
    entries = [e for e in history if isinstance(e, dict)]
    total = history_tokens(count, entries)
    assert total >= 0
    keep, discarded = split_for_condensation(count, entries)
    assert len(keep) + len(discarded) == len(entries)


@settings(max_examples=100, deadline=None)
@given(extra=st.dictionaries(st.text(max_size=12), _JSON, max_size=6))
def test_checkpoint_roundtrip_tolerates_extra_junk(extra):
    """from_checkpoint over a valid checkpoint polluted with junk keys
    (schema drift between versions) restores the known fields and ignores
    the rest."""
    from quoracle_amd.agent.state import AgentState
    state = AgentState(agent_id="a1", task_id="t1", profile="default",
                       model_pool=["m"])
    state.init_model_maps()
    ckpt = state.to_checkpoint()
    polluted = {**extra, **ckpt}
    restored = AgentState.from_checkpoint(polluted)
    assert restored.agent_id == "a1" and restored.task_id == "t1"
    assert restored.model_pool == ["m"]
    assert restored.to_checkpoint() == ckpt


@settings(max_examples=200, deadline=None)
@given(allocated=st.floats(min_value=0.01, max_value=1e6),
       ops=st.lists(st.tuples(st.sampled_from(["spend", "lock", "release"]),
                               st.floats(min_value=0.001, max_value=1e6)),
                    max_size=20))
def test_budget_escrow_invariant(allocated, ops):
    """Property: through any sequence of spend/escrow/release operations the
    tracker never lets available go negative without raising, and committed
    never underflows below zero."""
    from quoracle_amd.budget.tracker import (BudgetError, BudgetView,
                                             check_can_spend,
                                             lock_allocation,
                                             release_allocation)
    view = BudgetView(mode="allocated", allocated=allocated,
                      spent=0.0, committed=0.0)
    for op, amount in ops:
        try:
            if op == "spend":
                check_can_spend(view, amount)
                view.spent += amount
            elif op == "lock":
                view.committed = lock_allocation(view, amount)
            else:
                view.committed = release_allocation(view.committed, amount)
        except BudgetError:
            continue
        assert view.committed >= 0
        assert view.available >= -1e-6, (op, amount, view)


def test_parse_amount_rejects_junk():
    from quoracle_amd.budget.tracker import BudgetError, parse_amount
    import pytest as _pytest
    assert parse_amount("50.00") == 50.0
    for bad in ("nope", None, -1, 0, [], {}, "nan", "inf", "-inf",
                float("nan"), float("inf")):
        with _pytest.raises(BudgetError):
            parse_amount(bad)


@settings(max_examples=150, deadline=None)
@given(text=st.one_of(st.text(max_size=300),
                      _JSON.map(lambda v: __import__("json").dumps(v))))
def test_parse_response_total(text):
    """parse_response over arbitrary model text: a parsed dict or
    ParseError, never a crash."""
    from quoracle_amd.consensus.parser import ParseError, parse_response
    try:
        out = parse_response(text)
        assert isinstance(out, dict) and "action" in out
    except ParseError:
        pass


@settings(max_examples=80, deadline=None)
@given(text=st.text(max_size=400))
def test_skill_parser_total(text):
    """SKILL.md content is user-authored (and agent-authored via
    create_skill): parse-or-reject, never crash."""
    from quoracle_amd.governance.skills import SkillError, parse_skill_markdown
    try:
        out = parse_skill_markdown(text)
        assert isinstance(out, dict)
    except (SkillError, ValueError, KeyError):
        pass


@settings(max_examples=60, deadline=None)
@given(ids=st.lists(st.integers(min_value=0, max_value=10 ** 6),
                    min_size=0, max_size=4200),
       subset=st.sets(st.sampled_from(
           __import__("quoracle_amd.actions.schema",
                      fromlist=["ACTIONS"]).ACTIONS), min_size=0, max_size=22),
       child=st.one_of(st.none(), st.text(max_size=12)))
def test_grammar_walk_total_over_all_actions(ids, subset, child):
    """Any sampled-id stream through the 22-action grammar terminates in
    bounded steps and emits parseable JSON (or the stream simply ran
    short) — no crashes for any action subset or context."""
    ctx = {"child_id": child} if child is not None else {}
    g = ActionGrammar(sorted(subset) or ["wait"], context=ctx)
    out = []
    for sid in ids:
        if g.done:
            break
        out.append(g.advance(sid))
    if g.done:
        text = bytes(b for b in out if b != EOS).decode()
        doc = json.loads(text)
        assert doc["action"] in (sorted(subset) or ["wait"])


@settings(max_examples=80, deadline=None)
@given(text=st.text(max_size=400),
       want=st.integers(min_value=0, max_value=5))
def test_mcp_sse_parser_total(text, want):
    """SSE bodies from an HTTP MCP server are untrusted bytes: parse the
    matching message or raise ConnectionError, never crash."""
    from quoracle_amd.actions.executors import MCPHttpConnection
    try:
        msg = MCPHttpConnection._from_sse(text, want)
        assert isinstance(msg, dict) and msg.get("id") == want
    except ConnectionError:
        pass


@settings(max_examples=25, deadline=None)
@given(prompt=st.text(max_size=200),
       src=st.one_of(st.none(), st.binary(max_size=64)))
def test_imagegen_always_valid_png(prompt, src):
    """The procedural image model renders a structurally valid PNG for
    ANY prompt/source, deterministically."""
    from quoracle_amd.utils import imagegen
    png = imagegen.render(prompt, size=(32, 24), source_image=src)
    assert png.startswith(b"\x89PNG\r\n\x1a\n")
    assert png == imagegen.render(prompt, size=(32, 24), source_image=src)


@settings(max_examples=40, deadline=None)
@given(histories=st.dictionaries(
    st.sampled_from(["m1", "m2", "m3"]),
    st.lists(st.one_of(
        st.fixed_dictionaries({"type": st.sampled_from(
            ["user", "decision", "result", "junk"]),
            "content": st.one_of(st.text(max_size=40), st.integers(),
                                 st.none(),
                                 st.dictionaries(st.text(max_size=5),
                                                 st.integers(), max_size=2))}),
        st.dictionaries(st.text(max_size=6), st.integers(), max_size=2)),
        max_size=6),
    max_size=3),
    new_pool=st.lists(st.sampled_from(["m2", "m3", "m4", "m5"]),
                      min_size=1, max_size=3, unique=True))
def test_history_transfer_total_over_corrupt_state(histories, new_pool):
    """Runtime pool switching over arbitrary corrupt per-model history
    shapes: never crashes, always leaves exactly the new pool's maps."""
    import asyncio as _asyncio
    from quoracle_amd.agent.history_transfer import transfer_histories
    from quoracle_amd.agent.state import AgentState
    from quoracle_amd.engine.fake import FakeEngine
    state = AgentState(agent_id="f", task_id="t",
                       model_pool=sorted(histories) or ["m1"])
    state.init_model_maps()
    for m, h in histories.items():
        state.model_histories[m] = h
    eng = FakeEngine(default_response='{"action": "wait", "params": {}}')
    report = _asyncio.new_event_loop().run_until_complete(
        transfer_histories(state, new_pool, lambda m: eng))
    assert set(state.model_histories) == set(new_pool)
    assert set(state.model_pool) == set(new_pool)
    assert set(report) == set(new_pool)


@settings(max_examples=50, deadline=None)
@given(existing=st.lists(st.one_of(
    st.fixed_dictionaries({"text": st.text(max_size=30),
                           "confidence": st.integers(-5, 500)}),
    st.dictionaries(st.text(max_size=5), st.integers(), max_size=2)),
    max_size=8),
    new=st.lists(st.one_of(
        st.fixed_dictionaries({"text": st.text(max_size=30)}),
        st.fixed_dictionaries({"confidence": st.integers()}),
        st.dictionaries(st.text(max_size=4), st.text(max_size=4),
                        max_size=2)), max_size=8))
def test_lesson_merge_total_over_junk(existing, new):
    """Lesson merge over malformed lesson dicts (missing text/confidence,
    junk keys): bounded output, never crashes."""
    from quoracle_amd.agent.lessons import MAX_LESSONS, merge_lessons
    out = merge_lessons(existing, new)
    assert isinstance(out, list) and len(out) <= MAX_LESSONS


@settings(max_examples=25, deadline=None)
@given(path_bit=st.text(
    alphabet=st.characters(blacklist_characters="/\x00",
                           blacklist_categories=("Cs",)),
    min_size=1, max_size=24))
def test_ui_paths_never_500(path_bit):
    """Arbitrary ids in UI paths produce 2xx/4xx, never a server error
    (the ids reach SQL parameters and registry lookups)."""
    from fastapi.testclient import TestClient
    from quoracle_amd.engine.fake import FakeEngine
    from quoracle_amd.ui.server import create_app
    from helpers import IDLE, make_manager
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    app = create_app(manager)
    from urllib.parse import quote
    pb = quote(path_bit, safe="")
    with TestClient(app, raise_server_exceptions=False) as client:
        for url in (f"/api/tasks/{pb}/tree",
                    f"/api/agents/{pb}/logs",
                    f"/api/agents/{pb}/state",
                    f"/api/tasks/{pb}/export"):
            r = client.get(url)
            assert r.status_code < 500, (url, r.status_code, r.text[:200])


_ACTION_JSONS = st.sampled_from([
    '{"reasoning": "r", "action": "wait", "params": {"wait": true}}',
    '{"reasoning": "r", "action": "orient", "params": {"current_situation":'
    ' "s", "goal_clarity": "g", "available_resources": "a",'
    ' "key_challenges": "k", "delegation_consideration": "d"}}',
    '{"reasoning": "r", "action": "send_message", "params":'
    ' {"to": "parent", "content": "hello"}}',
    '{"action": "todo", "params": {"items": [{"content": "x",'
    ' "state": "todo"}]}}',
    '{"action": "record_cost", "params": {"amount": "3"}}',
])


@settings(max_examples=40, deadline=None)
@given(scripts=st.lists(
    st.lists(st.one_of(_ACTION_JSONS, st.text(max_size=60), st.none()),
             min_size=1, max_size=6),
    min_size=1, max_size=3),
    rounds=st.integers(min_value=0, max_value=4))
def test_consensus_pipeline_total_over_arbitrary_responses(scripts, rounds):
    """The full consensus loop (clustering -> unanimity/majority ->
    refinement -> forced decision) over ARBITRARY per-model response
    streams (valid actions, junk text, model failures): terminates within
    the round budget with exactly one decision or a structured
    ConsensusError — never hangs, never crashes."""
    import asyncio as _asyncio
    from quoracle_amd.consensus import pipeline as P

    pool = [f"m{i}" for i in range(len(scripts))]
    calls = {m: 0 for m in pool}

    async def query_fn(model_key, round_num, refinement_prompt):
        i = pool.index(model_key)
        seq = scripts[i]
        out = seq[min(calls[model_key], len(seq) - 1)]
        calls[model_key] += 1
        if out is None:
            raise RuntimeError("model failure")
        return out

    async def drive():
        try:
            outcome = await _asyncio.wait_for(P.run_consensus(
                pool, query_fn, max_refinement_rounds=rounds), timeout=30)
            assert outcome.decision.action
            assert 1 <= outcome.rounds_used <= rounds + 1
            return True
        except P.ConsensusError as exc:
            assert exc.reason in ("all_models_failed",
                                  "all_responses_invalid")
            return False

    _asyncio.new_event_loop().run_until_complete(drive())
    # no model was queried more than once per round (+1 forced round)
    assert all(c <= rounds + 1 for c in calls.values()), calls
