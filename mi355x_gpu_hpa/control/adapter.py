"""adapter.py — prometheus-adapter's default discovery rules, offline.

The reference installs prometheus-adapter with ONLY ``prometheus.url`` set
(reference README.md:91-95), which means the adapter runs its *default*
rule set (Helm chart ``rules.default: true``). Those defaults are what make
``cuda_test_gpu_avg`` appear on ``/apis/custom.metrics.k8s.io/v1beta1`` and
bind to ``namespaces/default/deployments/cuda-test`` — the registration the
reference verifies by hand (README.md:98-102). That discovery step is
exactly where drop-in compatibility can break *silently* (a missing static
label and the HPA reads ``<unknown>`` forever), so this module reimplements
the default rules faithfully enough that the registration probe has an
automated equivalent that can fail.

Modeled behaviors (prometheus-adapter default rule set, i.e. the chart's
``rules.default: true`` config):

1. series discovery ``{namespace!="", __name__!~"^container_.*"}`` — only
   series that carry a non-empty ``namespace`` label and are not cAdvisor
   ``container_*`` series are considered;
2. the non-counter rule: names NOT ending in ``_total`` are exposed under
   their own name, query ``sum(<series>{<matchers>}) by (<group-by>)``;
3. the counter rule: names ending in ``_total`` are exposed with the suffix
   stripped and a ``rate(...[window])`` query (rate needs range data an
   instant store does not have — `get_object_metric` refuses it loudly);
4. resource association via the ``resources: template: <<.Resource>>``
   idiom: EVERY label whose name equals a known resource's singular name
   associates the series with that resource kind. This is why the recording
   rule's static ``namespace``/``deployment`` labels
   (deploy/cuda-test-prometheusrule.yaml; reference
   cuda-test-prometheusrule.yaml:14-16) are required.

The cAdvisor ``container_*`` rules exist upstream but can never match this
stack's series (no exporter series is named ``container_*``); they are
represented by the discovery exclusion in (1).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

from .promql import PromQLError, Sample, evaluate

# Singular label name -> (plural resource, API group, kind), as
# prometheus-adapter's resource lister resolves <<.Resource>> templates
# against the cluster's discovery info. Pinned to the stock k8s >=1.26
# resource set the stack touches.
RESOURCE_LABELS: Dict[str, Tuple[str, str, str]] = {
    "namespace": ("namespaces", "", "Namespace"),
    "pod": ("pods", "", "Pod"),
    "node": ("nodes", "", "Node"),
    "service": ("services", "", "Service"),
    "deployment": ("deployments", "apps", "Deployment"),
    "statefulset": ("statefulsets", "apps", "StatefulSet"),
    "daemonset": ("daemonsets", "apps", "DaemonSet"),
    "replicaset": ("replicasets", "apps", "ReplicaSet"),
    "job": ("jobs", "batch", "Job"),
    "ingress": ("ingresses", "networking.k8s.io", "Ingress"),
}

_PLURAL_TO_SINGULAR = {v[0]: k for k, v in RESOURCE_LABELS.items()}


class AdapterError(ValueError):
    pass


@dataclass
class DiscoveredMetric:
    """One metric the adapter would serve: its API name, the underlying
    series, the query shape, and the resources it binds to."""

    metric: str                  # name on the custom metrics API
    series: str                  # underlying Prometheus series name
    is_counter: bool             # True => rate() query (suffix stripped)
    resources: Set[str] = field(default_factory=set)  # singular label names


def _discoverable(s: Sample) -> bool:
    """Default seriesQuery: {namespace!="", __name__!~"^container_.*"}."""
    if not s.name or s.name.startswith("container_"):
        return False
    return bool(s.labels.get("namespace"))


def _metric_name(series: str) -> Tuple[str, bool]:
    """Default name mangling: strip a _total suffix (counter rule), and
    the _seconds_total suffix first (seconds-counter rule)."""
    m = re.match(r"^(.*)_seconds_total$", series)
    if m:
        return m.group(1), True
    m = re.match(r"^(.*)_total$", series)
    if m:
        return m.group(1), True
    return series, False


def discover(samples: List[Sample]) -> Dict[str, DiscoveredMetric]:
    """Run default-rule discovery over an instant sample set (the stand-in
    for the adapter's periodic series-list query against Prometheus)."""
    out: Dict[str, DiscoveredMetric] = {}
    for s in samples:
        if not _discoverable(s):
            continue
        metric, is_counter = _metric_name(s.name)
        d = out.get(metric)
        if d is None:
            d = DiscoveredMetric(metric=metric, series=s.name,
                                 is_counter=is_counter)
            out[metric] = d
        for label, value in s.labels.items():
            if value and label in RESOURCE_LABELS:
                d.resources.add(label)
    return out


class Adapter:
    """In-process stand-in for the prometheus-adapter APIService
    (reused-as-is in a cluster; modeled here so the
    L4 seam is executable offline)."""

    def __init__(self, samples: Optional[List[Sample]] = None):
        self._samples: List[Sample] = list(samples or [])

    def update(self, samples: List[Sample]) -> None:
        """Replace the sample view (one 'scrape' of Prometheus state)."""
        self._samples = list(samples)

    # -- the /apis/custom.metrics.k8s.io/v1beta1 discovery document -------

    def api_resource_list(self) -> dict:
        """The APIResourceList a `kubectl get --raw
        /apis/custom.metrics.k8s.io/v1beta1` returns (reference
        README.md:98-102 greps this for cuda_test_gpu_avg)."""
        resources = []
        for d in sorted(discover(self._samples).values(),
                        key=lambda d: d.metric):
            for label in sorted(d.resources):
                plural, group, _kind = RESOURCE_LABELS[label]
                qualified = plural if not group else f"{plural}.{group}"
                resources.append({
                    "name": f"{qualified}/{d.metric}",
                    "singularName": "",
                    "namespaced": label != "namespace",
                    "kind": "MetricValueList",
                    "verbs": ["get"],
                })
        return {
            "kind": "APIResourceList",
            "apiVersion": "v1",
            "groupVersion": "custom.metrics.k8s.io/v1beta1",
            "resources": resources,
        }

    def metric_names(self) -> Set[str]:
        return {r["name"] for r in self.api_resource_list()["resources"]}

    # -- GET namespaces/{ns}/{resource}/{name}/{metric} --------------------

    def get_object_metric(self, namespace: str, resource: str,
                          object_name: str, metric: str) -> dict:
        """Serve one Object-metric GET — the call the HPA controller makes
        every sync period. Returns a MetricValueList-
        shaped dict; raises AdapterError when the metric/object is not
        served (the HPA then reports <unknown>)."""
        singular = _PLURAL_TO_SINGULAR.get(resource)
        if singular is None:
            raise AdapterError(f"unknown resource {resource!r}")
        # object/namespace names are interpolated into the PromQL matcher:
        # restrict to the RFC-1123 charset k8s enforces (also blocks any
        # quote/backslash injection into the query)
        for val in (namespace, object_name):
            if not re.fullmatch(r"[A-Za-z0-9._-]{1,253}", val or ""):
                raise AdapterError(f"invalid object name {val!r}")
        discovered = discover(self._samples)
        d = discovered.get(metric)
        if d is None or singular not in d.resources:
            raise AdapterError(
                f"metric {metric!r} not served for resource {resource!r}")
        if d.is_counter:
            raise AdapterError(
                f"metric {metric!r} uses a rate() query; an instant store "
                "has no range data (counters are out of this stack's path)")
        # default metricsQuery: sum(<series>{<matchers>}) by (<group-by>)
        expr = (f'sum({d.series}{{namespace="{namespace}",'
                f'{singular}="{object_name}"}}) by ({singular})')
        try:
            vec = evaluate(expr, self._samples)
        except PromQLError as e:
            raise AdapterError(f"query failed: {e}") from e
        if not vec:
            raise AdapterError(
                f"no samples for {metric!r} on {resource}/{object_name} "
                f"in namespace {namespace!r}")
        plural, group, kind = RESOURCE_LABELS[singular]
        api_version = f"{group}/v1" if group else "v1"
        return {
            "kind": "MetricValueList",
            "apiVersion": "custom.metrics.k8s.io/v1beta1",
            "items": [{
                "describedObject": {
                    "kind": kind,
                    "namespace": namespace,
                    "name": object_name,
                    "apiVersion": api_version,
                },
                "metricName": metric,
                "value": vec[0].value,
            }],
        }

    def get_object_metric_value(self, namespace: str, resource: str,
                                object_name: str, metric: str
                                ) -> Optional[float]:
        """The value the HPA controller extracts, or None (= <unknown>)
        when the adapter would 404."""
        try:
            resp = self.get_object_metric(namespace, resource,
                                          object_name, metric)
        except AdapterError:
            return None
        return float(resp["items"][0]["value"])
