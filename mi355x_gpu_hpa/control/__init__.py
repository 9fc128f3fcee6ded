from .adapter import Adapter, AdapterError, DiscoveredMetric, discover
from .hpa import (
    HpaSpec,
    HpaState,
    MetricTarget,
    desired_replicas,
    desired_replicas_multi,
    reconcile,
    reconcile_multi,
)
from .loop import (
    REFERENCE_RULE_EXPR,
    REFERENCE_RULE_NAME,
    ControlLoop,
    LoopResult,
    RecordingRule,
    synth_pod_labels,
)
from .promql import PromQLError, Sample, evaluate, evaluate_scalar
from .scraper import Scraper, ScrapeTarget, parse_prometheus_text

__all__ = [
    "Adapter", "AdapterError", "DiscoveredMetric", "discover",
    "HpaSpec", "HpaState", "MetricTarget", "desired_replicas",
    "desired_replicas_multi", "reconcile", "reconcile_multi",
    "ControlLoop", "LoopResult", "RecordingRule", "synth_pod_labels",
    "REFERENCE_RULE_EXPR", "REFERENCE_RULE_NAME",
    "PromQLError", "Sample", "evaluate", "evaluate_scalar",
    "Scraper", "ScrapeTarget", "parse_prometheus_text",
]
