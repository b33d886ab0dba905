"""Clustering + winner selection tests (parity targets:
lib/quoracle/consensus/aggregator.ex, result.ex, result/scoring.ex)."""

import pytest

from quoracle_amd.consensus import aggregator, result
from quoracle_amd.consensus.aggregator import (action_fingerprint,
                                               cluster_responses,
                                               find_majority_cluster)
from quoracle_amd.consensus.result import (break_tie, calculate_confidence,
                                           cluster_wait_score, format_result,
                                           wait_score)


def resp(action, params=None, reasoning="r", wait=None, model="m"):
    return {"action": action, "params": params or {}, "reasoning": reasoning,
            "wait": wait, "model": model}


class TestFingerprint:
    def test_same_exact_params_same_fingerprint(self):
        a = resp("file_read", {"path": "/a"})
        b = resp("file_read", {"path": "/a"})
        assert action_fingerprint(a) == action_fingerprint(b)

    def test_different_exact_params_differ(self):
        a = resp("file_read", {"path": "/a"})
        b = resp("file_read", {"path": "/b"})
        assert action_fingerprint(a) != action_fingerprint(b)

    def test_percentile_params_mergeable(self):
        a = resp("file_read", {"path": "/a", "offset": 1})
        b = resp("file_read", {"path": "/a", "offset": 100})
        assert action_fingerprint(a) == action_fingerprint(b)

    def test_mode_params_mergeable(self):
        a = resp("fetch_web", {"url": "u", "security_check": True})
        b = resp("fetch_web", {"url": "u", "security_check": False})
        assert action_fingerprint(a) == action_fingerprint(b)

    def test_semantic_normalization_clusters_similar(self):
        a = resp("send_message", {"to": "parent", "content": "Finished analyzing the logs"})
        b = resp("send_message", {"to": "parent", "content": "finished analyzing the logs!"})
        assert action_fingerprint(a) == action_fingerprint(b)

    def test_batch_sync_order_matters(self):
        a = resp("batch_sync", {"actions": [{"action": "file_read", "params": {}},
                                            {"action": "todo", "params": {}}]})
        b = resp("batch_sync", {"actions": [{"action": "todo", "params": {}},
                                            {"action": "file_read", "params": {}}]})
        assert action_fingerprint(a) != action_fingerprint(b)

    def test_batch_async_order_independent(self):
        a = resp("batch_async", {"actions": [{"action": "file_read", "params": {}},
                                             {"action": "todo", "params": {}}]})
        b = resp("batch_async", {"actions": [{"action": "todo", "params": {}},
                                             {"action": "file_read", "params": {}}]})
        assert action_fingerprint(a) == action_fingerprint(b)

    def test_unknown_action_invalid(self):
        assert action_fingerprint({"action": "bogus", "params": {}})[1] == "invalid"

    def test_union_merge_sorted(self):
        a = resp("search_secrets", {"search_terms": ["b", "a"]})
        b = resp("search_secrets", {"search_terms": ["a", "b"]})
        assert action_fingerprint(a) == action_fingerprint(b)


class TestClustering:
    def test_clusters_sorted_by_count(self):
        rs = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/a"}),
              resp("file_read", {"path": "/b"})]
        clusters = cluster_responses(rs)
        assert clusters[0].count == 2
        assert clusters[1].count == 1

    def test_round1_unanimity(self):
        rs = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/b"})]
        clusters = cluster_responses(rs)
        assert find_majority_cluster(clusters, 2, round_num=1) is None
        rs2 = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/a"})]
        assert find_majority_cluster(cluster_responses(rs2), 2, round_num=1) is not None

    def test_round2_majority(self):
        rs = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/a"}),
              resp("file_read", {"path": "/b"})]
        clusters = cluster_responses(rs)
        winner = find_majority_cluster(clusters, 3, round_num=2)
        assert winner is not None and winner.count == 2

    def test_half_is_not_majority(self):
        rs = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/a"}),
              resp("file_read", {"path": "/b"}), resp("file_read", {"path": "/b"})]
        assert find_majority_cluster(cluster_responses(rs), 4, round_num=2) is None


class TestFormatResult:
    def test_majority_consensus(self):
        rs = [resp("file_read", {"path": "/a", "offset": 10}, wait=False),
              resp("file_read", {"path": "/a", "offset": 20}, wait=False),
              resp("file_read", {"path": "/b"}, wait=False)]
        decision = format_result(cluster_responses(rs), 3, 2)
        assert decision.kind == "consensus"
        assert decision.action["action"] == "file_read"
        assert decision.action["params"]["path"] == "/a"
        assert decision.action["params"]["offset"] == 15
        assert decision.action["wait"] is False

    def test_plurality_forced_decision(self):
        rs = [resp("file_read", {"path": "/a"}), resp("file_read", {"path": "/b"}),
              resp("file_write", {"path": "/c", "mode": "write", "content": "x"})]
        decision = format_result(cluster_responses(rs), 3, 5)
        assert decision.kind == "forced_decision"

    def test_tie_break_prefers_lower_priority_action(self):
        # orient (1) beats execute_shell (18)
        rs = [resp("execute_shell", {"command": "ls"}),
              resp("orient", {f: "x" for f in
                              ["current_situation", "goal_clarity",
                               "available_resources", "key_challenges",
                               "delegation_consideration"]})]
        decision = format_result(cluster_responses(rs), 2, 3)
        assert decision.action["action"] == "orient"

    def test_wait_default_false_when_omitted(self):
        rs = [resp("file_read", {"path": "/a"})]
        decision = format_result(cluster_responses(rs), 1, 1)
        assert decision.action["wait"] is False

    def test_wait_merged(self):
        rs = [resp("file_read", {"path": "/a"}, wait=10),
              resp("file_read", {"path": "/a"}, wait=30)]
        decision = format_result(cluster_responses(rs), 2, 1)
        assert decision.action["wait"] == 20

    def test_reasoning_first_nonempty(self):
        rs = [resp("file_read", {"path": "/a"}, reasoning=""),
              resp("file_read", {"path": "/a"}, reasoning="because")]
        decision = format_result(cluster_responses(rs), 2, 1)
        assert decision.action["reasoning"] == "because"


class TestConfidence:
    def test_unanimous(self):
        assert calculate_confidence(3, 3, 1) == pytest.approx(1.0)

    def test_majority_bonus_tiers(self):
        assert calculate_confidence(2, 3, 2) == pytest.approx(2 / 3 + 0.10)
        assert calculate_confidence(3, 5, 2) == pytest.approx(0.6 + 0.05)
        assert calculate_confidence(5, 6, 2) == pytest.approx(5 / 6 + 0.15)

    def test_round_penalty(self):
        assert calculate_confidence(2, 3, 6, 4) == pytest.approx(2 / 3 + 0.10 - 0.2)

    def test_clamped_low(self):
        assert calculate_confidence(1, 10, 9, 4) == pytest.approx(0.1)


class TestWaitScore:
    def test_scores(self):
        assert wait_score(True) == (0, 0)
        assert wait_score(None) == (0, 1)
        assert wait_score(5) == (0, 6)
        assert wait_score(0) == (1, 0)
        assert wait_score(False) == (1, 0)

    def test_tie_break_wait(self):
        # same action, same priority: lower wait score wins (more conservative)
        a = aggregator.Cluster(count=1, actions=[resp("wait", {}, wait=True)],
                               representative=resp("wait", {}, wait=True))
        b = aggregator.Cluster(count=1, actions=[resp("wait", {}, wait=False)],
                               representative=resp("wait", {}, wait=False))
        assert break_tie([b, a]) is a  # wait=True scores (0,0) < (1,0)


class TestBatchSyncMerge:
    def test_merge_positionwise(self):
        mk = lambda off: resp("batch_sync", {"actions": [
            {"action": "file_read", "params": {"path": "/a", "offset": off}},
            {"action": "todo", "params": {"items": []}},
        ]})
        decision = format_result(cluster_responses([mk(10), mk(30)]), 2, 1)
        acts = decision.action["params"]["actions"]
        assert acts[0]["params"]["offset"] == 20
        assert acts[1]["action"] == "todo"
