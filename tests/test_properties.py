"""Property-based tests (hypothesis) for the control-plane math."""

import math

from hypothesis import given, settings
from hypothesis import strategies as st

from mi355x_gpu_hpa.control import HpaSpec, Sample, desired_replicas, evaluate

values = st.floats(min_value=0, max_value=1000, allow_nan=False,
                   allow_infinity=False)


class TestHpaProperties:
    @given(st.lists(values, min_size=2, max_size=20),
           st.integers(min_value=1, max_value=8))
    @settings(max_examples=200, deadline=None)
    def test_desired_monotone_in_metric(self, vals, replicas):
        spec = HpaSpec(min_replicas=1, max_replicas=64, target_value=5.0)
        got = [desired_replicas(spec, replicas, v) for v in sorted(vals)]
        assert got == sorted(got)  # non-decreasing in the metric

    @given(values, st.integers(min_value=1, max_value=64))
    @settings(max_examples=200, deadline=None)
    def test_desired_within_bounds(self, v, replicas):
        spec = HpaSpec(min_replicas=2, max_replicas=16, target_value=5.0)
        d = desired_replicas(spec, replicas, v)
        assert 2 <= d <= 16

    @given(st.floats(min_value=0.91, max_value=1.09),
           st.integers(min_value=1, max_value=16))
    @settings(max_examples=100, deadline=None)
    def test_tolerance_band_no_change(self, ratio, replicas):
        spec = HpaSpec(min_replicas=1, max_replicas=64, target_value=10.0)
        assert desired_replicas(spec, replicas, 10.0 * ratio) == replicas


class TestPromqlProperties:
    @given(st.lists(st.tuples(st.sampled_from(["a", "b", "c"]), values),
                    min_size=1, max_size=30))
    @settings(max_examples=200, deadline=None)
    def test_aggregations_match_model(self, pairs):
        samples = [Sample("m", {"g": g, "i": str(i)}, v)
                   for i, (g, v) in enumerate(pairs)]
        vals = [v for _, v in pairs]
        assert evaluate("sum(m)", samples)[0].value == sum(vals)
        assert evaluate("max(m)", samples)[0].value == max(vals)
        assert evaluate("min(m)", samples)[0].value == min(vals)
        assert math.isclose(evaluate("avg(m)", samples)[0].value,
                            sum(vals) / len(vals), rel_tol=1e-9)
        # group sums partition the total
        by_g = evaluate("sum by(g) (m)", samples)
        assert math.isclose(sum(s.value for s in by_g), sum(vals),
                            rel_tol=1e-9)

    @given(st.lists(st.tuples(st.sampled_from(["p0", "p1", "p2"]), values),
                    min_size=1, max_size=20))
    @settings(max_examples=100, deadline=None)
    def test_join_filters_to_labeled_pods(self, pairs):
        samples = [Sample("util", {"pod": p, "i": str(i)}, v)
                   for i, (p, v) in enumerate(pairs)]
        samples.append(Sample("kube_pod_labels",
                              {"pod": "p0", "label_app": "x"}, 1.0))
        expr = ('avg(max by(pod) (util) * on(pod) group_left(label_app) '
                'max by(pod, label_app) (kube_pod_labels{label_app="x"}))')
        res = evaluate(expr, samples)
        p0 = [v for p, v in pairs if p == "p0"]
        if not p0:
            assert res == []
        else:
            assert math.isclose(res[0].value, max(p0), rel_tol=1e-9)


class TestRoundTrip:
    @given(st.lists(values, min_size=1, max_size=8))
    @settings(max_examples=50, deadline=None)
    def test_render_parse_roundtrip(self, vals):
        """Values written in the exporter's %.6g format parse back to within
        6 significant digits."""
        from mi355x_gpu_hpa.control import parse_prometheus_text

        text = "\n".join(
            f'dcgm_gpu_utilization{{gpu="{i}"}} {v:.6g}'
            for i, v in enumerate(vals))
        parsed = parse_prometheus_text(text)
        assert len(parsed) == len(vals)
        for s, v in zip(parsed, vals):
            assert math.isclose(s.value, v, rel_tol=1e-5, abs_tol=1e-4)


class TestNativeRendererProperties:
    """Hypothesis through the C++ renderer (capi, mock backend): injected
    busy% must come back exactly through render -> parse."""

    @given(st.floats(min_value=0, max_value=100, allow_nan=False))
    @settings(max_examples=50, deadline=None)
    def test_injected_busy_roundtrip(self, busy):
        import ctypes
        import os

        from mi355x_gpu_hpa import NATIVE_BUILD
        from mi355x_gpu_hpa.control import parse_prometheus_text

        lib_path = NATIVE_BUILD / "libmi355x_sampler.so"
        if not lib_path.exists():
            import pytest

            pytest.skip("sampler lib not built")
        lib = ctypes.CDLL(str(lib_path))
        lib.mi355x_render_mock_metrics.argtypes = [
            ctypes.c_int, ctypes.c_char_p, ctypes.c_char_p,
            ctypes.c_char_p, ctypes.c_int]
        os.environ["MI355X_MOCK_BUSY"] = f"{busy!r}"
        try:
            buf = ctypes.create_string_buffer(1 << 18)
            rc = lib.mi355x_render_mock_metrics(2, b"", b"", buf, len(buf))
            assert rc > 0
            samples = parse_prometheus_text(buf.value.decode())
            utils = [s for s in samples if s.name == "dcgm_gpu_utilization"]
            assert len(utils) == 2
            for s in utils:
                assert math.isclose(s.value, busy, rel_tol=1e-5, abs_tol=1e-4)
        finally:
            del os.environ["MI355X_MOCK_BUSY"]


class TestSchemaValidatorRobustness:
    """validate_manifest must never crash: any JSON-ish document yields a
    (possibly long) error list, not an exception."""

    json_scalars = st.one_of(st.none(), st.booleans(),
                             st.integers(-10**6, 10**6),
                             st.floats(allow_nan=False, allow_infinity=False),
                             st.text(max_size=20))
    json_values = st.recursive(
        json_scalars,
        lambda children: st.one_of(
            st.lists(children, max_size=4),
            st.dictionaries(st.text(max_size=12), children, max_size=4)),
        max_leaves=25)

    @given(st.dictionaries(st.text(max_size=12), json_values, max_size=6))
    @settings(max_examples=300, deadline=None)
    def test_never_raises_on_random_docs(self, doc):
        from mi355x_gpu_hpa.k8s_schema import validate_manifest

        errors = validate_manifest(doc)
        assert isinstance(errors, list)

    @given(st.sampled_from(["DaemonSet", "Deployment", "Service",
                            "HorizontalPodAutoscaler", "PrometheusRule",
                            "ConfigMap"]),
           st.dictionaries(st.text(max_size=12), json_values, max_size=5))
    @settings(max_examples=300, deadline=None)
    def test_never_raises_on_random_typed_docs(self, kind, extra):
        from mi355x_gpu_hpa.k8s_schema import validate_manifest

        doc = dict(extra)
        doc["kind"] = kind
        errors = validate_manifest(doc)
        assert isinstance(errors, list)
        assert errors  # a random doc of a pinned kind cannot be valid


class TestAdapterDiscoveryProperties:
    labels = st.dictionaries(
        st.sampled_from(["namespace", "pod", "deployment", "node", "zzz",
                         "service", "job", "gpu"]),
        st.text(alphabet="abcdefg-", min_size=0, max_size=8), max_size=5)
    samples = st.lists(
        st.builds(lambda n, l, v: __import__(
            "mi355x_gpu_hpa.control", fromlist=["Sample"]).Sample(n, l, v),
            st.sampled_from(["m1", "m2_total", "container_x", "",
                             "dcgm_gpu_utilization"]),
            labels,
            st.floats(allow_nan=False, allow_infinity=False)),
        max_size=12)

    @given(samples)
    @settings(max_examples=300, deadline=None)
    def test_discovery_invariants(self, samples):
        from mi355x_gpu_hpa.control import discover

        d = discover(samples)
        for metric, m in d.items():
            # only namespace-labeled, non-container series are discovered
            assert not m.series.startswith("container_")
            assert any(s.name == m.series and s.labels.get("namespace")
                       for s in samples)
            # name mangling strips _total exactly when is_counter
            assert m.is_counter == m.series.endswith("_total")
            if m.is_counter:
                assert not metric.endswith("_total")
            # resource bindings only from non-empty known labels
            for r in m.resources:
                assert any(s.labels.get(r) for s in samples
                           if s.name == m.series)
