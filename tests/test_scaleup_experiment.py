"""The 1->8 scale-up curve experiment (tools/scaleup_experiment.py) on CPU:
reproduces the reference's documented overshoot behavior (README.md:123)
and produces the scale-up curve (BASELINE.json config 4)."""

import os

import pytest

from mi355x_gpu_hpa.exporter import EXPORTER_BIN

needs_bin = pytest.mark.skipif(
    not os.path.exists(EXPORTER_BIN), reason="native exporter not built"
)


@needs_bin
def test_native_profile_scales_to_max():
    from tools.scaleup_experiment import run

    r = run("native", max_replicas=8, per_replica_busy=40.0, target=5.0,
            duration_s=90.0, pod_start_s=20.0)
    assert r["peak_desired"] == 8
    assert r["time_to_max_replicas_s"] is not None
    assert r["time_to_max_replicas_s"] <= 60.0
    # the reference's documented overshoot: desired hits max before any new
    # replica has landed (metric lag + pod start delay)
    assert r["overshoot"] is True


@needs_bin
def test_moderate_load_partial_scale():
    from tools.scaleup_experiment import run

    # 7.5% per replica vs target 5: ratio 1.5 on 1 replica -> 2, then the
    # averaged metric stays 7.5 (each replica equally busy) -> 3, settles
    # when ceil(1.5 * r) == r is impossible -> climbs to max; use busy 6
    # (ratio 1.2): 1->2 ... still climbs; tolerance stops it only within 10%.
    # busy 5.2 (ratio 1.04, inside tolerance): no scaling at all.
    r = run("native", max_replicas=8, per_replica_busy=5.2, target=5.0,
            duration_s=60.0, pod_start_s=10.0)
    assert r["peak_desired"] == 1


@needs_bin
def test_native_v2_no_overshoot():
    """The v2 behavior block removes the overshoot: replicas ramp one pod
    per sync instead of the desired count jumping to max at once."""
    from tools.scaleup_experiment import run

    r = run("native-v2", max_replicas=8, per_replica_busy=40.0, target=5.0,
            duration_s=200.0, pod_start_s=10.0)
    assert r["overshoot"] is False
    assert r["time_to_max_replicas_s"] is not None
    desired_seq = [c["desired"] for c in r["curve"]]
    assert max(desired_seq[:2]) <= 3  # stepped, not pinned at 8 immediately
