"""Merge-rule behavior tests (parity targets cited from the reference:
lib/quoracle/actions/consensus_rules.ex)."""

import pytest

from quoracle_amd.consensus import rules
from quoracle_amd.consensus.rules import NoConsensus, apply_rule, merge_param


def fake_embed_many(texts):
    """Deterministic embedding: bag-of-chars, so similar strings embed close."""
    vecs = []
    for t in texts:
        v = [0.0] * 32
        for ch in t.lower():
            v[ord(ch) % 32] += 1.0
        vecs.append(v)
    return vecs


class TestExactMatch:
    def test_single_value(self):
        assert apply_rule("exact_match", ["a"]) == "a"

    def test_identical(self):
        assert apply_rule("exact_match", ["a", "a", "a"]) == "a"

    def test_mismatch(self):
        with pytest.raises(NoConsensus):
            apply_rule("exact_match", ["a", "b"])

    def test_empty(self):
        with pytest.raises(NoConsensus):
            apply_rule("exact_match", [])

    def test_dicts(self):
        assert apply_rule("exact_match", [{"x": 1}, {"x": 1}]) == {"x": 1}


class TestModeSelection:
    def test_mode(self):
        assert apply_rule("mode_selection", ["a", "b", "a"]) == "a"

    def test_tie_first_wins(self):
        assert apply_rule("mode_selection", ["b", "a"]) == "b"

    def test_bool_vs_int_distinct(self):
        # True and 1 must not collapse into one bucket
        assert apply_rule("mode_selection", [True, 1, 1]) == 1


class TestUnionMerge:
    def test_flatten_dedupe(self):
        assert apply_rule("union_merge", [["a", "b"], ["b", "c"]]) == ["a", "b", "c"]

    def test_order_preserved(self):
        assert apply_rule("union_merge", [["z"], ["a", "z"]]) == ["z", "a"]


class TestStructuralMerge:
    def test_deep_merge_later_wins(self):
        merged = apply_rule("structural_merge",
                            [{"a": {"x": 1}, "b": 1}, {"a": {"y": 2}, "b": 2}])
        assert merged == {"a": {"x": 1, "y": 2}, "b": 2}


class TestPercentile:
    def test_median_even_interpolates(self):
        # ref: 75th pct of [100,200,300,400] -> 325 (consensus_rules.ex:342-358)
        assert apply_rule(("percentile", 75), [100, 200, 300, 400]) == 325

    def test_median(self):
        assert apply_rule(("percentile", 50), [1, 2, 3]) == 2

    def test_median_two(self):
        assert apply_rule(("percentile", 50), [10, 20]) == 15

    def test_non_numeric_falls_back_to_mode(self):
        assert apply_rule(("percentile", 50), [True, True, False]) is True

    def test_rounding_half_up(self):
        assert apply_rule(("percentile", 50), [1, 2]) == 2  # 1.5 rounds to 2


class TestWaitParameter:
    def test_all_false(self):
        assert apply_rule("wait_parameter", [False, False]) is False

    def test_all_true(self):
        assert apply_rule("wait_parameter", [True, True]) is True

    def test_mixed_bools_3plus_any_true(self):
        assert apply_rule("wait_parameter", [True, False, False]) is True

    def test_integers_median(self):
        assert apply_rule("wait_parameter", [10, 20, 30]) == 20

    def test_integers_median_even(self):
        # reference uses integer division for even medians
        assert apply_rule("wait_parameter", [10, 21]) == 15

    def test_mixed_true_converts_to_max_int(self):
        # true -> max(integers)=60; values [60, 30] -> median 45
        assert apply_rule("wait_parameter", [True, 30, 60]) == 60

    def test_two_mixed_bools(self):
        # [true, false] -> converted [30, 0] -> median 15
        assert apply_rule("wait_parameter", [True, False]) == 15


class TestSemanticSimilarity:
    def test_identical_short_circuits_without_embedder(self):
        assert apply_rule(("semantic_similarity", 0.9), ["x", "x"]) == "x"

    def test_similar_strings_agree(self):
        v = apply_rule(("semantic_similarity", 0.8),
                       ["list the files", "list the files now"],
                       embed_many=fake_embed_many)
        assert v == "list the files"

    def test_dissimilar_strings_disagree(self):
        with pytest.raises(NoConsensus):
            apply_rule(("semantic_similarity", 0.99),
                       ["alpha beta gamma", "zzzzzz qqqq"],
                       embed_many=fake_embed_many)

    def test_no_embedder_fails(self):
        with pytest.raises(NoConsensus):
            apply_rule(("semantic_similarity", 0.9), ["a", "b"])


class TestFirstNonNilAndMergeMaps:
    def test_first_non_nil(self):
        assert apply_rule("first_non_nil", ["a", "b"]) == "a"

    def test_merge_maps(self):
        assert apply_rule("merge_maps", [{"a": 1}, {"b": 2}, {"a": 3}]) == \
            {"a": 3, "b": 2}


class TestBatchSequenceMerge:
    def test_empty(self):
        assert apply_rule("batch_sequence_merge", []) == []

    def test_single_sequence_passthrough(self):
        seq = [{"action": "todo", "params": {"items": []}}]
        assert apply_rule("batch_sequence_merge", [seq]) == seq

    def test_length_mismatch(self):
        with pytest.raises(NoConsensus) as exc:
            apply_rule("batch_sequence_merge", [
                [{"action": "file_read", "params": {"path": "/a"}}],
                [{"action": "file_read", "params": {"path": "/a"}},
                 {"action": "file_read", "params": {"path": "/b"}}],
            ])
        assert exc.value.reason == "sequence_length_mismatch"

    def test_action_type_mismatch(self):
        with pytest.raises(NoConsensus) as exc:
            apply_rule("batch_sequence_merge", [
                [{"action": "file_read", "params": {"path": "/a"}}],
                [{"action": "file_write", "params": {"path": "/a", "mode": "write"}}],
            ])
        assert exc.value.reason == "sequence_mismatch"

    def test_positionwise_merge(self):
        merged = apply_rule("batch_sequence_merge", [
            [{"action": "file_read", "params": {"path": "/a", "offset": 10}}],
            [{"action": "file_read", "params": {"path": "/a", "offset": 20}}],
        ])
        assert merged == [{"action": "file_read",
                           "params": {"path": "/a", "offset": 15}}]


class TestMergeParam:
    def test_wait_always_wait_rule(self):
        assert merge_param("execute_shell", "wait", [5, 15]) == 10

    def test_schema_rule_lookup(self):
        assert merge_param("file_read", "offset", [1, 3]) == 2

    def test_unknown_param(self):
        with pytest.raises(NoConsensus) as exc:
            merge_param("file_read", "nope", [1])
        assert exc.value.reason == "unknown_param"


class TestCosine:
    def test_identical(self):
        assert rules.cosine_similarity([1, 2, 3], [1, 2, 3]) == pytest.approx(1.0)

    def test_orthogonal(self):
        assert rules.cosine_similarity([1, 0], [0, 1]) == pytest.approx(0.0)

    def test_zero_vector(self):
        assert rules.cosine_similarity([0, 0], [1, 1]) == 0.0

    def test_length_mismatch_raises(self):
        with pytest.raises(ValueError):
            rules.cosine_similarity([1], [1, 2])
