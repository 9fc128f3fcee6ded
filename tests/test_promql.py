"""Unit tests for the PromQL-subset evaluator, centered on the reference
recording rule (cuda-test-prometheusrule.yaml:13)."""

import pytest

from mi355x_gpu_hpa.control import (
    REFERENCE_RULE_EXPR,
    PromQLError,
    Sample,
    evaluate,
    evaluate_scalar,
)


def gpu(pod, node, value, gpu_id="0", ns="default"):
    return Sample(
        "dcgm_gpu_utilization",
        {"gpu": gpu_id, "pod": pod, "node": node, "namespace": ns},
        value,
    )


def pod_label(pod, app="cuda-test"):
    return Sample("kube_pod_labels", {"pod": pod, "label_app": app}, 1.0)


class TestReferenceRule:
    def test_single_pod(self):
        samples = [gpu("cuda-test-1", "n0", 42.0), pod_label("cuda-test-1")]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == 42.0

    def test_avg_across_pods(self):
        samples = [
            gpu("cuda-test-1", "n0", 40.0),
            gpu("cuda-test-2", "n0", 20.0),
            pod_label("cuda-test-1"),
            pod_label("cuda-test-2"),
        ]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == 30.0

    def test_filters_other_apps(self):
        samples = [
            gpu("cuda-test-1", "n0", 40.0),
            gpu("other-1", "n0", 99.0),
            pod_label("cuda-test-1"),
            pod_label("other-1", app="other"),
        ]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == 40.0

    def test_multi_gpu_pod_takes_max(self):
        # inner max by(node,pod,namespace) collapses a multi-GPU pod to its
        # busiest GPU
        samples = [
            gpu("cuda-test-1", "n0", 10.0, gpu_id="0"),
            gpu("cuda-test-1", "n0", 70.0, gpu_id="1"),
            pod_label("cuda-test-1"),
        ]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == 70.0

    def test_duplicate_scrape_target_deduped(self):
        # the same exporter seen via two service endpoints produces duplicate
        # series differing only in scrape-level labels; max-by dedupes
        s1 = gpu("cuda-test-1", "n0", 55.0)
        s2 = gpu("cuda-test-1", "n0", 55.0)
        s2.labels["instance"] = "10.0.0.2:9400"
        samples = [s1, s2, pod_label("cuda-test-1")]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == 55.0

    def test_no_pods_yields_empty(self):
        samples = [gpu("idle-1", "n0", 5.0)]
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) is None

    def test_eight_gpu_node(self):
        # config 3: 8 per-GPU series on one node, one pod per GPU
        samples = []
        for i in range(8):
            samples.append(gpu(f"cuda-test-{i}", "n0", 10.0 * i, gpu_id=str(i)))
            samples.append(pod_label(f"cuda-test-{i}"))
        assert evaluate_scalar(REFERENCE_RULE_EXPR, samples) == pytest.approx(35.0)


class TestPrimitives:
    def test_selector_matchers(self):
        samples = [
            Sample("m", {"a": "x"}, 1.0),
            Sample("m", {"a": "y"}, 2.0),
            Sample("n", {"a": "x"}, 3.0),
        ]
        assert [s.value for s in evaluate('m{a="x"}', samples)] == [1.0]
        assert [s.value for s in evaluate('m{a!="x"}', samples)] == [2.0]
        assert sorted(s.value for s in evaluate('m{a=~"x|y"}', samples)) == [1.0, 2.0]
        assert [s.value for s in evaluate('m{a!~"x"}', samples)] == [2.0]

    def test_aggregations(self):
        samples = [Sample("m", {"g": str(i % 2), "i": str(i)}, float(i))
                   for i in range(4)]
        by_g = {tuple(s.labels.items()): s.value
                for s in evaluate("sum by(g) (m)", samples)}
        assert by_g == {(("g", "0"),): 2.0, (("g", "1"),): 4.0}
        assert evaluate_scalar("avg(m)", samples) == 1.5
        assert evaluate_scalar("max(m)", samples) == 3.0
        assert evaluate_scalar("min(m)", samples) == 0.0
        assert evaluate_scalar("count(m)", samples) == 4.0

    def test_suffix_by(self):
        samples = [Sample("m", {"g": "a"}, 1.0), Sample("m", {"g": "b"}, 3.0)]
        res = evaluate("sum(m) by (g)", samples)
        assert sorted(s.value for s in res) == [1.0, 3.0]

    def test_scalar_arithmetic(self):
        samples = [Sample("m", {}, 4.0)]
        assert evaluate_scalar("m * 100", samples) == 400.0
        assert evaluate_scalar("m / 4", samples) == 1.0
        assert evaluate_scalar("m + 1 - 2", samples) == 3.0

    def test_many_to_one_requires_group_left(self):
        samples = [
            Sample("l", {"pod": "p", "gpu": "0"}, 1.0),
            Sample("l", {"pod": "p", "gpu": "1"}, 2.0),
            Sample("r", {"pod": "p"}, 1.0),
        ]
        with pytest.raises(PromQLError):
            evaluate("l * on(pod) r", samples)
        res = evaluate("l * on(pod) group_left() r", samples)
        assert sorted(s.value for s in res) == [1.0, 2.0]

    def test_group_left_copies_labels(self):
        samples = [
            Sample("l", {"pod": "p"}, 2.0),
            Sample("r", {"pod": "p", "label_app": "z"}, 3.0),
        ]
        res = evaluate("l * on(pod) group_left(label_app) r", samples)
        assert res[0].value == 6.0
        assert res[0].labels["label_app"] == "z"

    def test_unmatched_left_dropped(self):
        samples = [Sample("l", {"pod": "p"}, 2.0), Sample("r", {"pod": "q"}, 3.0)]
        assert evaluate("l * on(pod) r", samples) == []

    def test_operator_precedence(self):
        """* and / bind tighter than + and -
        (Prometheus precedence), so `a + b * c` is `a + (b * c)`."""
        samples = [
            Sample("a", {"k": "x"}, 2.0),
            Sample("b", {"k": "x"}, 3.0),
            Sample("c", {"k": "x"}, 4.0),
        ]
        assert evaluate("a + b * c", samples)[0].value == 14.0
        assert evaluate("a - b / c", samples)[0].value == 2.0 - 3.0 / 4.0
        assert evaluate("(a + b) * c", samples)[0].value == 20.0
        # scalar forms too
        assert evaluate("2 + 3 * 4", [])[0].value == 14.0
        assert evaluate("a * 100 - b", samples)[0].value == 197.0

    def test_parse_errors(self):
        with pytest.raises(PromQLError):
            evaluate("avg(", [])
        with pytest.raises(PromQLError):
            evaluate("m{a=}", [])
        with pytest.raises(PromQLError):
            evaluate("m n", [])


class TestIgnoring:
    def test_ignoring_drops_named_labels_from_key(self):
        samples = [
            Sample("l", {"pod": "p", "gpu": "0"}, 3.0),
            Sample("r", {"pod": "p"}, 2.0),
        ]
        # default matching fails (label sets differ); ignoring(gpu) matches
        assert evaluate("l * r", samples) == []
        res = evaluate("l * ignoring(gpu) r", samples)
        assert len(res) == 1 and res[0].value == 6.0

    def test_ignoring_many_to_many_rejected(self):
        samples = [
            Sample("l", {"pod": "p", "gpu": "0"}, 1.0),
            Sample("r", {"pod": "p", "gpu": "0"}, 1.0),
            Sample("r", {"pod": "p", "gpu": "1"}, 1.0),
        ]
        with pytest.raises(PromQLError, match="many-to-many"):
            evaluate("l * ignoring(gpu) r", samples)

    def test_ignoring_with_group_left(self):
        samples = [
            Sample("l", {"pod": "p", "gpu": "0"}, 2.0),
            Sample("l", {"pod": "p", "gpu": "1"}, 3.0),
            Sample("r", {"pod": "p"}, 10.0),
        ]
        res = evaluate("l * ignoring(gpu) group_left() r", samples)
        assert sorted(s.value for s in res) == [20.0, 30.0]
