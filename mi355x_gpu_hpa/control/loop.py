"""loop.py — the closed-loop metric pipeline, in-process.

One object wires the whole reference data path so it
can run — and be timed — without a cluster:

    exporter /metrics  --scrape-->  sample store   (L1->L2)
    recording rules (PromQL subset) over the store  (L3)
    HPA reconcile on the recorded metric            (L4->L5 collapsed: the
        adapter is a pass-through of the recorded series in-process)

`step()` performs exactly one scrape -> rule-eval -> HPA-decision cycle and
reports per-stage latencies; bench.py's "p50 scrape->HPA-scale latency" is
the distribution of `LoopResult.total_s` over steps. In a real cluster each
hop is a separate component (Prometheus, prometheus-adapter, controller
manager — all reused stock); this harness preserves the
metric/label contracts between them.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .adapter import Adapter
from .hpa import HpaSpec, HpaState, reconcile
from .promql import PromQLError, Sample, evaluate
from .scraper import Scraper

# The reference recording rule, verbatim semantics
# (cuda-test-prometheusrule.yaml:12-16; static labels added on record).
REFERENCE_RULE_NAME = "cuda_test_gpu_avg"
REFERENCE_RULE_EXPR = (
    'avg(max by(node, pod, namespace) (dcgm_gpu_utilization) '
    '* on(pod) group_left(label_app) '
    'max by(pod, label_app) (kube_pod_labels{label_app="cuda-test"}))'
)
REFERENCE_RULE_STATIC_LABELS = {"namespace": "default", "deployment": "cuda-test"}


@dataclass
class RecordingRule:
    record: str
    expr: str
    static_labels: Dict[str, str] = field(default_factory=dict)


@dataclass
class LoopResult:
    scrape_s: float
    rule_eval_s: float
    hpa_s: float
    total_s: float
    recorded: Dict[str, Optional[float]]
    replicas: int
    metric_value: Optional[float]
    adapter_s: float = 0.0


class ControlLoop:
    def __init__(
        self,
        scraper: Scraper,
        rules: Optional[List[RecordingRule]] = None,
        hpa_spec: Optional[HpaSpec] = None,
        hpa_metric: str = REFERENCE_RULE_NAME,
        extra_samples: Optional[Callable[[], List[Sample]]] = None,
        use_adapter: bool = False,
        adapter_target: tuple = ("default", "deployments", "cuda-test"),
    ):
        self.scraper = scraper
        self.rules = rules if rules is not None else [
            RecordingRule(REFERENCE_RULE_NAME, REFERENCE_RULE_EXPR,
                          dict(REFERENCE_RULE_STATIC_LABELS))
        ]
        self.hpa_spec = hpa_spec or HpaSpec()
        self.hpa_state = HpaState()
        self.hpa_metric = hpa_metric
        self.extra_samples = extra_samples
        # use_adapter=True routes the HPA's metric fetch through the
        # prometheus-adapter default-rule model (adapter.py) — the full
        # L4 hop, discovery included — instead of reading the recorded
        # value directly. adapter_target is the HPA scaleTargetRef
        # (namespace, resource, name), reference cuda-test-hpa.yaml:13-20.
        self.adapter = Adapter() if use_adapter else None
        self.adapter_target = adapter_target
        self.recorded_series: List[Sample] = []
        # rule name -> first error message (a failing rule is reported once,
        # like Prometheus's unhealthy-rule state)
        self._rule_errors: Dict[str, str] = {}

    def step(self, now_s: Optional[float] = None) -> LoopResult:
        t0 = time.monotonic()
        samples = list(self.scraper.scrape_once())
        if self.extra_samples:
            samples.extend(self.extra_samples())
        t1 = time.monotonic()

        recorded: Dict[str, Optional[float]] = {}
        self.recorded_series = []
        for rule in self.rules:
            try:
                vec = evaluate(rule.expr, samples)
            except PromQLError as e:
                # Prometheus marks a failing rule as unhealthy and keeps
                # evaluating the group; mirror that instead of crashing the
                # loop on a bad custom rule.
                if rule.record not in self._rule_errors:
                    self._rule_errors[rule.record] = str(e)
                recorded[rule.record] = None
                continue
            if not vec:
                recorded[rule.record] = None
                continue
            for s in vec:
                labels = dict(s.labels)
                labels.update(rule.static_labels)
                self.recorded_series.append(Sample(rule.record, labels, s.value))
            # scalar rules (like the reference's) produce one series
            recorded[rule.record] = vec[0].value if len(vec) == 1 else None
        t2 = time.monotonic()

        if self.adapter is not None:
            # L4 for real: adapter discovery + Object-metric GET over the
            # recorded series, exactly what the HPA controller reads
            # 
            ns, resource, name = self.adapter_target
            self.adapter.update(self.recorded_series)
            metric_value = self.adapter.get_object_metric_value(
                ns, resource, name, self.hpa_metric)
        else:
            metric_value = recorded.get(self.hpa_metric)
        t2b = time.monotonic()
        replicas = reconcile(
            self.hpa_spec, self.hpa_state, metric_value,
            now_s if now_s is not None else time.time(),
        )
        t3 = time.monotonic()

        return LoopResult(
            scrape_s=t1 - t0,
            rule_eval_s=t2 - t1,
            hpa_s=t3 - t2b,
            total_s=t3 - t0,
            recorded=recorded,
            replicas=replicas,
            metric_value=metric_value,
            adapter_s=t2b - t2,
        )


def synth_pod_labels(pods: List[str], app: str = "cuda-test") -> List[Sample]:
    """Synthesize the kube-state-metrics `kube_pod_labels` join series
     for a set of pod names."""
    return [
        Sample("kube_pod_labels", {"pod": p, "label_app": app, "namespace": "default"},
               1.0)
        for p in pods
    ]
