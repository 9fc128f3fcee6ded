"""CPU regression tests over RECORDED real-MI355X exporter output
(tests/fixtures/real_mi355x_{idle,loaded}.prom, captured from the rsmi
backend on a live gpurun box under the MFMA GEMM burn). This is SURVEY.md
§4's "unit tests cover the exporter's counter-to-metric math against
recorded outputs": the contract and the reference recording rule are
validated against genuine gfx950 counter values with no GPU present."""

from pathlib import Path

import pytest

from mi355x_gpu_hpa.control import (
    REFERENCE_RULE_EXPR,
    evaluate_scalar,
    parse_prometheus_text,
    synth_pod_labels,
)

FIXTURES = Path(__file__).resolve().parent / "fixtures"


@pytest.fixture(params=["idle", "loaded"])
def samples(request):
    text = (FIXTURES / f"real_mi355x_{request.param}.prom").read_text()
    return request.param, parse_prometheus_text(text)


def test_schema_on_real_output(samples):
    _, s = samples
    names = {x.name for x in s}
    for fam in ["dcgm_gpu_utilization", "dcgm_gpu_temp", "dcgm_power_usage",
                "dcgm_fb_used", "dcgm_fb_free", "dcgm_sm_clock",
                "amd_vram_total_bytes", "amd_gpu_hotspot_temp"]:
        assert fam in names, fam
    util = [x for x in s if x.name == "dcgm_gpu_utilization"][0]
    assert set(util.labels) >= {"gpu", "uuid", "device", "modelName"}
    assert util.labels["device"].startswith("renderD")
    assert "MI355" in util.labels["modelName"]


def test_value_invariants(samples):
    kind, s = samples
    by = {x.name: x.value for x in s}
    assert 0 <= by["dcgm_gpu_utilization"] <= 100
    # MI355X: 288 GiB HBM3E = 3.09e11 bytes
    assert 300e9 < by["amd_vram_total_bytes"] < 320e9
    assert by["dcgm_fb_used"] + by["dcgm_fb_free"] == pytest.approx(
        by["amd_vram_total_bytes"] / (1 << 20), rel=0.02)
    assert 20 < by["dcgm_gpu_temp"] < 110
    assert 50 < by["dcgm_power_usage"] < 2500
    if kind == "loaded":
        assert by["dcgm_gpu_utilization"] > 60  # captured at 90% duty
        assert by["dcgm_power_usage"] > 400
    else:
        assert by["dcgm_gpu_utilization"] < 20


def test_reference_rule_on_real_series(samples):
    """The shipped recording rule evaluates over genuine exporter output
    once pod attribution labels are present (as the kubelet path adds)."""
    kind, s = samples
    for x in s:
        if "gpu" in x.labels:
            x.labels.setdefault("pod", f"cuda-test-{x.labels['gpu']}")
            x.labels.setdefault("namespace", "default")
            x.labels.setdefault("node", "n0")
    s = s + synth_pod_labels(["cuda-test-0"])
    v = evaluate_scalar(REFERENCE_RULE_EXPR, s)
    assert v is not None
    util = [x.value for x in s if x.name == "dcgm_gpu_utilization"][0]
    assert v == util


class TestRound2Fixture:
    """Fresh round-2 capture (auto backend, full surface incl. partitions,
    probes and the freshness counter) — validates the new families'
    counter-to-metric math against genuine gfx950 output, CPU-only."""

    @pytest.fixture()
    def r2(self):
        text = (FIXTURES / "real_mi355x_idle_r2.prom").read_text()
        return parse_prometheus_text(text)

    def test_partition_families_on_real_hw(self, r2):
        info = [s for s in r2 if s.name == "amd_compute_partition_info"]
        assert info and info[0].labels["compute"] == "SPX"
        assert info[0].labels["memory"].startswith("NPS")
        xcp = [s for s in r2 if s.name == "amd_xcp_busy_percent"]
        assert xcp and all(0 <= s.value <= 100 for s in xcp)
        assert xcp[0].labels["partition"] == "0"  # SPX = one partition

    def test_probe_family_on_real_hw(self, r2):
        un = {s.labels["counter"]: s.labels["reason"] for s in r2
              if s.name == "amd_counter_unavailable"}
        assert "mfma_activity" in un  # probed, honestly absent

    def test_freshness_and_overhead_families(self, r2):
        ticks = [s for s in r2 if s.name == "amd_exporter_samples_total"]
        assert ticks and ticks[0].value >= 1
        dur = [s for s in r2 if s.name == "amd_exporter_sample_duration_ms"]
        assert dur and 0 <= dur[0].value < 100

    def test_reference_rule_on_real_output(self, r2):
        for s in r2:
            if s.labels.get("gpu") == "0" and "pod" not in s.labels:
                s.labels["pod"] = "cuda-test-0"
                s.labels.setdefault("namespace", "default")
        v = evaluate_scalar(REFERENCE_RULE_EXPR,
                            r2 + synth_pod_labels(["cuda-test-0"]))
        assert v is not None and 0 <= v <= 100
