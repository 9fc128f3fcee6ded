"""hpa.py — the HorizontalPodAutoscaler reconciliation algorithm.

Implements what the stock kube-controller-manager HPA loop does for an
``Object``-type custom metric (the reference's cuda-test-hpa.yaml:13-21,
``metricName: cuda_test_gpu_avg, targetValue: 5``; reconcile math per
SURVEY.md §3.4):

    desired = ceil(current / target * currentReplicas)
    clamped to [minReplicas, maxReplicas]

plus the controller's behavior details that matter for the scale-up curve:
  * 10% tolerance band around ratio 1.0 (no scaling inside it)
  * scale-down stabilization window (default 300 s: the highest desired
    within the window wins, which is what delays downscale)
  * missing metric => no change

In a real cluster the stock controller does this (SURVEY.md C12 "reuse
as-is"); this implementation drives the in-process control-loop harness and
the latency bench, and is unit-tested against the documented edge cases.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Optional, Tuple


@dataclass
class HpaSpec:
    min_replicas: int = 1
    max_replicas: int = 3          # reference default (cuda-test-hpa.yaml:11-12)
    target_value: float = 5.0      # reference targetValue (cuda-test-hpa.yaml:21)
    tolerance: float = 0.1         # upstream default
    downscale_stabilization_s: float = 300.0  # upstream default
    # v2 behavior.scaleUp policy (deploy/cuda-test-hpa.yaml's fix for the
    # reference's documented overshoot, README.md:123): at most
    # `scale_up_pods` added per `scale_up_period_s`. 0 = unlimited
    # (v2beta1 / reference behavior).
    scale_up_pods: int = 0
    scale_up_period_s: float = 15.0
    # v2 behavior.scaleDown Pods policy: at most `scale_down_pods` removed
    # per `scale_down_period_s` (applied AFTER stabilization, upstream
    # ordering). 0 = unlimited.
    scale_down_pods: int = 0
    scale_down_period_s: float = 15.0


@dataclass
class HpaState:
    current_replicas: int = 1
    # (timestamp_s, desired) recommendations within the stabilization window
    recommendations: List[Tuple[float, int]] = field(default_factory=list)
    # (period_start_s, replicas_at_period_start) for the scaleUp policy
    scaleup_window: Tuple[Optional[float], Optional[int]] = (None, None)
    # same for the scaleDown policy
    scaledown_window: Tuple[Optional[float], Optional[int]] = (None, None)


def desired_replicas(
    spec: HpaSpec, current_replicas: int, metric_value: Optional[float]
) -> int:
    """Raw desired-replica computation (no stabilization). The min/max
    clamp applies even on the no-change paths (a missing metric or the
    tolerance band): the controller always enforces the replica bounds."""
    clamp = lambda r: max(spec.min_replicas, min(spec.max_replicas, r))  # noqa: E731
    if metric_value is None:
        return clamp(current_replicas)
    if spec.target_value <= 0:
        raise ValueError("target_value must be positive")
    ratio = metric_value / spec.target_value
    if abs(ratio - 1.0) <= spec.tolerance:
        return clamp(current_replicas)
    return clamp(math.ceil(ratio * current_replicas))


@dataclass
class MetricTarget:
    """One Object-metric entry of a multi-metric HPA (deploy/multi-metric/
    cuda-test-hpa-multi.yaml): metric name + target value."""

    name: str
    target_value: float


def desired_replicas_multi(
    spec: HpaSpec,
    metrics: List[MetricTarget],
    current_replicas: int,
    values: dict,
) -> int:
    """Multi-metric desired count: the HPA controller computes a desired
    replica count per metric and takes the MAX (upstream semantics). A
    metric with no value contributes the current count (no-change)."""
    desired = spec.min_replicas
    any_value = False
    for m in metrics:
        v = values.get(m.name)
        if v is None:
            d = current_replicas
        else:
            any_value = True
            one = HpaSpec(
                min_replicas=spec.min_replicas,
                max_replicas=spec.max_replicas,
                target_value=m.target_value,
                tolerance=spec.tolerance,
            )
            d = desired_replicas(one, current_replicas, v)
        desired = max(desired, d)
    if not any_value:
        return current_replicas
    return desired


def reconcile(
    spec: HpaSpec, state: HpaState, metric_value: Optional[float], now_s: float
) -> int:
    """One HPA sync: returns the new replica count (and updates state).

    Scale-ups apply immediately; scale-downs are stabilized — the new count
    is the MAX of desired values recommended within the stabilization
    window, so a transient dip can't flap the deployment.
    """
    desired = desired_replicas(spec, state.current_replicas, metric_value)
    return _stabilize(spec, state, desired, now_s)


def reconcile_multi(
    spec: HpaSpec, state: HpaState, metrics: List[MetricTarget],
    values: dict, now_s: float
) -> int:
    """Multi-metric sync (max-of-desireds), with the same stabilization."""
    desired = desired_replicas_multi(spec, metrics, state.current_replicas, values)
    return _stabilize(spec, state, desired, now_s)


def _stabilize(spec: HpaSpec, state: HpaState, desired: int, now_s: float) -> int:
    state.recommendations.append((now_s, desired))
    cutoff = now_s - spec.downscale_stabilization_s
    state.recommendations = [(t, d) for (t, d) in state.recommendations if t >= cutoff]

    # kube-controller-manager ordering: (1) stabilization — the highest
    # recommendation within the window wins, which is what delays
    # downscale; (2) behavior rate limits apply AFTER stabilization, so a
    # replayed high recommendation that was rate-limited earlier cannot
    # bypass the scaleUp policy on a later dip.
    stabilized = max(d for (_, d) in state.recommendations)
    new = stabilized
    if spec.scale_up_pods > 0 and new > state.current_replicas:
        # v2 scaleUp policy: cap growth to scale_up_pods per period,
        # measured against the replica count at the period's start
        base_t, base_r = state.scaleup_window
        if base_t is None or now_s - base_t >= spec.scale_up_period_s:
            base_t, base_r = now_s, state.current_replicas
        new = min(new, base_r + spec.scale_up_pods)
        state.scaleup_window = (base_t, base_r)
    if spec.scale_down_pods > 0 and new < state.current_replicas:
        # v2 scaleDown Pods policy, symmetric to scaleUp
        base_t, base_r = state.scaledown_window
        if base_t is None or now_s - base_t >= spec.scale_down_period_s:
            base_t, base_r = now_s, state.current_replicas
        new = max(new, base_r - spec.scale_down_pods)
        state.scaledown_window = (base_t, base_r)
    new = max(spec.min_replicas, min(spec.max_replicas, new))
    state.current_replicas = new
    return new
