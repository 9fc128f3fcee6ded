"""GPU tests: the native exporter sampling a real MI355X via rocm_smi,
validated against rocm-smi (the BASELINE.json metric-error check), under
real MFMA GEMM load from our CDNA4 kernels."""

import ctypes
import json
import subprocess
import threading
import time

import pytest

from mi355x_gpu_hpa.control import parse_prometheus_text
from mi355x_gpu_hpa.exporter import ExporterProcess

pytestmark = pytest.mark.gpu


def rocm_smi_busy():
    out = subprocess.run(["rocm-smi", "--showuse", "--json"],
                         capture_output=True, timeout=10)
    data = json.loads(out.stdout.decode())
    busy = {}
    for card, vals in data.items():
        if card.startswith("card"):
            for k, v in vals.items():
                if "GPU use" in k:
                    busy[int(card[4:])] = float(v)
    return busy


def test_real_backend_serves_metrics(gpu):
    with ExporterProcess(interval_ms=200) as exp:
        text = exp.scrape()
    samples = parse_prometheus_text(text)
    util = [s for s in samples if s.name == "dcgm_gpu_utilization"]
    assert util, "no dcgm_gpu_utilization from real backend"
    for s in util:
        assert 0 <= s.value <= 100
        assert s.labels["device"].startswith("renderD")
        assert len(s.labels["uuid"]) > 0
    names = {s.name for s in samples}
    assert "dcgm_gpu_temp" in names
    assert "dcgm_power_usage" in names
    assert "dcgm_fb_used" in names
    assert "amd_vram_total_bytes" in names
    vram = [s for s in samples if s.name == "amd_vram_total_bytes"][0]
    assert vram.value > 200e9  # MI355X: 288 GB HBM3E


def test_util_error_vs_rocm_smi_under_load(gpu):
    """The north-star validation: exporter busy% within a few % of
    rocm-smi's own reading while the MFMA GEMM load runs."""
    from mi355x_gpu_hpa import loadgen

    stop = ctypes.c_int(0)

    def burn():
        loadgen._load().lg_gemm_burn(
            0, ctypes.c_double(100.0), ctypes.c_double(20.0),
            4096, 4096, 4096, ctypes.c_double(50.0), ctypes.byref(stop))

    t = threading.Thread(target=burn, daemon=True)
    t.start()
    try:
        time.sleep(2.0)  # let utilization settle at 100%
        with ExporterProcess(interval_ms=200) as exp:
            time.sleep(0.5)
            samples = parse_prometheus_text(exp.scrape())
            ours = {int(s.labels["gpu"]): s.value for s in samples
                    if s.name == "dcgm_gpu_utilization"}
            oracle = rocm_smi_busy()
    finally:
        stop.value = 1
        t.join(timeout=15)

    assert 0 in ours and 0 in oracle, (ours, oracle)
    # device 0 is under continuous GEMM load: both must read high
    assert ours[0] > 80, f"exporter busy {ours[0]}% under full load"
    # tightened from 15pp (round-1 verdict item 8): under a CONTINUOUS
    # (100% duty) load there is no window jitter; bench runs measured
    # 2-5pp paired error (BENCH_r01/r02)
    assert abs(ours[0] - oracle[0]) <= 8, (
        f"exporter {ours[0]}% vs rocm-smi {oracle[0]}%"
    )


def test_duty_cycle_tracks_target(gpu):
    """lg_gemm_burn's duty cycle must produce roughly the requested busy%
    (the tunable-load requirement)."""
    from mi355x_gpu_hpa import loadgen

    stop = ctypes.c_int(0)

    def burn():
        loadgen._load().lg_gemm_burn(
            0, ctypes.c_double(50.0), ctypes.c_double(25.0),
            4096, 4096, 4096, ctypes.c_double(100.0), ctypes.byref(stop))

    t = threading.Thread(target=burn, daemon=True)
    t.start()
    try:
        time.sleep(3.0)
        with ExporterProcess(interval_ms=250) as exp:
            vals = []
            for _ in range(8):
                time.sleep(0.5)
                samples = parse_prometheus_text(exp.scrape())
                for s in samples:
                    if s.name == "dcgm_gpu_utilization" and s.labels["gpu"] == "0":
                        vals.append(s.value)
    finally:
        stop.value = 1
        t.join(timeout=15)
    mean = sum(vals) / len(vals)
    # +/-10pp band (round-1 verdict item 8): the closed-loop duty
    # controller (event-measured GPU-active feedback) landed 48.6/78.0/22.4
    # for targets 50/80/20 across boxes (profiles/duty_closed_loop.md);
    # the band covers instantaneous-sample jitter (stdev ~7pp, n=8+).
    assert 40 <= mean <= 60, f"mean busy {mean}% for 50% duty target ({vals})"


def test_live_schema_superset_of_fixtures(gpu):
    """The committed CPU fixtures must never claim families the live
    exporter no longer serves (fixture drift guard)."""
    from pathlib import Path

    fixture_families = set()
    for name in ("real_mi355x_idle.prom", "real_mi355x_idle_r2.prom"):
        text = (Path(__file__).parent / "fixtures" / name).read_text()
        fixture_families |= {s.name for s in parse_prometheus_text(text)}
    with ExporterProcess(interval_ms=200) as exp:
        time.sleep(0.5)
        live_families = {s.name for s in parse_prometheus_text(exp.scrape())}
    missing = fixture_families - live_families
    assert not missing, f"live exporter dropped families: {missing}"


def test_readyz_real_backend(gpu):
    import urllib.request

    with ExporterProcess(interval_ms=200) as exp:
        with urllib.request.urlopen(
            f"http://127.0.0.1:{exp.port}/readyz", timeout=2
        ) as r:
            assert r.status == 200


def test_hbm_bandwidth_metric_tracks_memory_load(gpu):
    """Config 5's bandwidth axis on real counters: the streaming-triad
    burn (~4.8 TB/s bursts) must drive amd_hbm_bandwidth_utilization (UMC
    activity) far above idle — this is the series the multi-metric HPA
    rule (deploy/multi-metric/) scales on."""
    from mi355x_gpu_hpa import loadgen

    stop = ctypes.c_int(0)
    gbps = ctypes.c_double()

    def burn():
        loadgen._load().lg_bw_burn(
            0, ctypes.c_double(100.0), ctypes.c_double(20.0),
            ctypes.c_double(6.0), ctypes.c_double(100.0),
            ctypes.byref(stop), ctypes.byref(gbps))

    t = threading.Thread(target=burn, daemon=True)
    t.start()
    vals = []
    try:
        time.sleep(2.0)
        with ExporterProcess(interval_ms=250) as exp:
            for _ in range(8):
                time.sleep(0.5)
                for s in parse_prometheus_text(exp.scrape()):
                    if (s.name == "amd_hbm_bandwidth_utilization"
                            and s.labels["gpu"] == "0"):
                        vals.append(s.value)
    finally:
        stop.value = 1
        t.join(timeout=15)
    assert vals, "amd_hbm_bandwidth_utilization not served"
    mean = sum(vals) / len(vals)
    # triad at ~60% of the 8 TB/s peak must push UMC activity high;
    # exact % depends on firmware accounting — require well above idle
    assert mean > 30, f"UMC activity only {mean}% under ~4.8 TB/s triad"
