"""The prometheus-adapter seam, executable offline.

The reference verifies adapter registration by hand:
`kubectl get --raw /apis/custom.metrics.k8s.io/v1beta1 | jq -r . | grep
cuda_test_gpu_avg` (reference README.md:98-102). These tests are the
automated equivalent over the default-rule discovery model
(mi355x_gpu_hpa/control/adapter.py): they fail if the recording rule's
output stops satisfying the adapter's default discovery (wrong name shape,
missing static labels) — the exact silent-breakage mode of the drop-in
contract.
"""

import pytest

from mi355x_gpu_hpa.control import (
    Adapter,
    AdapterError,
    ControlLoop,
    HpaSpec,
    REFERENCE_RULE_EXPR,
    REFERENCE_RULE_NAME,
    RecordingRule,
    Sample,
    discover,
    evaluate,
    synth_pod_labels,
)


def recorded_reference_series(value=12.5):
    """cuda_test_gpu_avg exactly as the recording rule emits it
    (deploy/cuda-test-prometheusrule.yaml: static namespace/deployment
    labels, reference cuda-test-prometheusrule.yaml:14-16)."""
    return Sample("cuda_test_gpu_avg",
                  {"namespace": "default", "deployment": "cuda-test"}, value)


def raw_exporter_series():
    return [
        Sample("dcgm_gpu_utilization",
               {"gpu": "0", "node": "n0", "pod": "cuda-test-abc",
                "namespace": "default"}, 42.0),
        Sample("dcgm_gpu_utilization",
               {"gpu": "1", "node": "n0", "pod": "cuda-test-def",
                "namespace": "default"}, 10.0),
        # exporter self-metrics carry no namespace => not discoverable
        Sample("dcgm_exporter_samples", {"node": "n0"}, 100.0),
    ]


class TestDefaultRuleDiscovery:
    def test_recorded_metric_binds_to_deployment_and_namespace(self):
        d = discover([recorded_reference_series()])
        m = d["cuda_test_gpu_avg"]
        assert m.resources == {"namespace", "deployment"}
        assert not m.is_counter

    def test_registration_probe_equivalent(self):
        # the automated analog of `kubectl get --raw ... | grep`
        a = Adapter([recorded_reference_series()] + raw_exporter_series())
        names = a.metric_names()
        assert "deployments.apps/cuda_test_gpu_avg" in names
        assert "namespaces/cuda_test_gpu_avg" in names

    def test_raw_series_bind_to_pods(self):
        # the real adapter also exposes the raw per-GPU series on pods —
        # parity with prometheus-adapter's template resource association
        a = Adapter(raw_exporter_series())
        assert "pods/dcgm_gpu_utilization" in a.metric_names()

    def test_without_static_labels_metric_does_not_bind(self):
        """The could-fail test: drop the rule's static labels (the mistake
        the reference's design guards against) and the adapter must NOT
        serve the metric for the deployment — the HPA would read
        <unknown>."""
        bare = Sample("cuda_test_gpu_avg", {}, 12.5)
        assert "cuda_test_gpu_avg" not in discover([bare])
        a = Adapter([bare])
        assert a.metric_names() == set()
        assert a.get_object_metric_value(
            "default", "deployments", "cuda-test", "cuda_test_gpu_avg") is None

    def test_namespace_only_does_not_bind_to_deployment(self):
        s = Sample("cuda_test_gpu_avg", {"namespace": "default"}, 12.5)
        a = Adapter([s])
        names = a.metric_names()
        assert "namespaces/cuda_test_gpu_avg" in names
        assert "deployments.apps/cuda_test_gpu_avg" not in names

    def test_container_series_excluded(self):
        s = Sample("container_cpu_usage",
                   {"namespace": "default", "pod": "p"}, 1.0)
        assert discover([s]) == {}

    def test_counter_suffix_stripped(self):
        s = Sample("http_requests_total",
                   {"namespace": "default", "service": "svc"}, 100.0)
        d = discover([s])
        assert "http_requests" in d and d["http_requests"].is_counter
        # rate() needs range data; the instant adapter refuses loudly
        a = Adapter([s])
        with pytest.raises(AdapterError, match="rate"):
            a.get_object_metric("default", "services", "svc", "http_requests")

    def test_seconds_total_suffix_stripped(self):
        s = Sample("work_seconds_total", {"namespace": "default"}, 5.0)
        assert "work" in discover([s])


class TestObjectMetricGet:
    def test_get_returns_recorded_value(self):
        a = Adapter([recorded_reference_series(17.0)] + raw_exporter_series())
        resp = a.get_object_metric("default", "deployments", "cuda-test",
                                   "cuda_test_gpu_avg")
        item = resp["items"][0]
        assert item["value"] == 17.0
        assert item["describedObject"] == {
            "kind": "Deployment", "namespace": "default",
            "name": "cuda-test", "apiVersion": "apps/v1",
        }
        assert item["metricName"] == "cuda_test_gpu_avg"

    def test_wrong_namespace_404s(self):
        a = Adapter([recorded_reference_series()])
        with pytest.raises(AdapterError, match="no samples"):
            a.get_object_metric("prod", "deployments", "cuda-test",
                                "cuda_test_gpu_avg")

    def test_wrong_object_404s(self):
        a = Adapter([recorded_reference_series()])
        assert a.get_object_metric_value(
            "default", "deployments", "other", "cuda_test_gpu_avg") is None

    def test_unknown_resource_rejected(self):
        a = Adapter([recorded_reference_series()])
        with pytest.raises(AdapterError, match="unknown resource"):
            a.get_object_metric("default", "widgets", "w", "cuda_test_gpu_avg")

    def test_sum_by_semantics_match_metrics_query(self):
        # default metricsQuery is sum(...) by (resource); with the rule's
        # single recorded series this is the identity — assert via the
        # promql engine directly for parity
        series = [recorded_reference_series(9.0)]
        resp = Adapter(series).get_object_metric(
            "default", "deployments", "cuda-test", "cuda_test_gpu_avg")
        direct = evaluate(
            'sum(cuda_test_gpu_avg{namespace="default",'
            'deployment="cuda-test"}) by (deployment)', series)
        assert resp["items"][0]["value"] == direct[0].value == 9.0


class TestLoopThroughAdapter:
    def _loop(self, store, **kw):
        class StaticScraper:
            def scrape_once(self):
                return list(store)
        return ControlLoop(StaticScraper(), use_adapter=True,
                           hpa_spec=HpaSpec(max_replicas=8), **kw)

    def _store(self, util=40.0):
        return [
            Sample("dcgm_gpu_utilization",
                   {"gpu": "0", "node": "n0", "pod": "cuda-test-abc",
                    "namespace": "default"}, util),
        ] + synth_pod_labels(["cuda-test-abc"])

    def test_full_l1_to_l5_path(self):
        """scrape -> rule eval -> adapter discovery + Object GET -> HPA:
        every seam of SURVEY.md §3.1-3.4 exercised in one step."""
        loop = self._loop(self._store(util=40.0))
        res = loop.step(now_s=0.0)
        assert res.metric_value == 40.0
        assert res.replicas == 8          # ratio 40/5 => clamped to max
        assert res.adapter_s > 0.0

    def test_adapter_path_detects_missing_static_labels(self):
        """Break the rule (no static labels) and the adapter-routed loop
        must see <unknown> (None) — the direct-read harness of round 1
        could not catch this."""
        rules = [RecordingRule(REFERENCE_RULE_NAME, REFERENCE_RULE_EXPR,
                               static_labels={})]
        loop = self._loop(self._store(util=40.0), rules=rules)
        res = loop.step(now_s=0.0)
        assert res.metric_value is None   # HPA reads <unknown>
        assert res.replicas == 1          # no change


class TestInputValidation:
    def test_query_injection_blocked(self):
        a = Adapter([recorded_reference_series()])
        for bad in ('cuda-test"} or up{x="', "a\\b", "", "x" * 300):
            with pytest.raises(AdapterError, match="invalid object name"):
                a.get_object_metric("default", "deployments", bad,
                                    "cuda_test_gpu_avg")
        with pytest.raises(AdapterError, match="invalid object name"):
            a.get_object_metric('d"efault', "deployments", "cuda-test",
                                "cuda_test_gpu_avg")
