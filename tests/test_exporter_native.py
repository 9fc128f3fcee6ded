"""CPU tests for the native mi355x-exporter daemon (mock backend).

These verify the exporter's drop-in contract with the reference
(dcgm-exporter.yaml): metric family names, the label schema the recording
rule joins on, the -c/-f/-l/-k config surface, and the readiness probe the
reference lacks."""

import os
import subprocess
import time
import urllib.request

import pytest

from mi355x_gpu_hpa.exporter import EXPORTER_BIN, ExporterProcess
from mi355x_gpu_hpa.control import parse_prometheus_text


needs_bin = pytest.mark.skipif(
    not os.path.exists(EXPORTER_BIN), reason="native exporter not built"
)


@needs_bin
class TestMetricsContract:
    def test_dcgm_schema(self):
        with ExporterProcess(mock_devices=2, interval_ms=100) as exp:
            text = exp.scrape()
        samples = parse_prometheus_text(text)
        util = [s for s in samples if s.name == "dcgm_gpu_utilization"]
        assert len(util) == 2
        for s in util:
            # label contract the rule + dashboards depend on
            assert set(s.labels) >= {"gpu", "uuid", "device", "modelName"}
            assert 0 <= s.value <= 100
        # README.md:46 verification probe greps dcgm_gpu_temp
        assert any(s.name == "dcgm_gpu_temp" for s in samples)
        # families that must exist for parity with the 1.x metric set
        names = {s.name for s in samples}
        for fam in [
            "dcgm_mem_copy_utilization", "dcgm_power_usage", "dcgm_sm_clock",
            "dcgm_memory_clock", "dcgm_fb_used", "dcgm_fb_free",
        ]:
            assert fam in names, f"missing {fam}"

    def test_ras_and_violation_families(self):
        """ECC counters, PCIe replay, and windowed PVIOL/TVIOL (the mock
        backend's residency accumulators advance at 10%/2% of wall)."""
        with ExporterProcess(mock_devices=2, interval_ms=50) as exp:
            time.sleep(0.3)  # windowed metrics need 2 samples
            samples = parse_prometheus_text(exp.scrape())
        by = {}
        for s in samples:
            by.setdefault(s.name, {})[s.labels.get("gpu")] = s.value
        assert by["dcgm_ecc_sbe_aggregate_total"] == {"0": 0.0, "1": 1.0}
        assert by["dcgm_ecc_dbe_aggregate_total"]["0"] == 0.0
        assert by["dcgm_pcie_replay_counter"]["0"] == 3.0
        assert by["dcgm_power_violation"]["0"] == pytest.approx(10.0, abs=1.5)
        assert by["dcgm_thermal_violation"]["0"] == pytest.approx(2.0, abs=1.0)

    def test_amd_native_families(self):
        with ExporterProcess(mock_devices=1, interval_ms=50) as exp:
            time.sleep(0.3)  # need 2 samples for windowed rates
            samples = parse_prometheus_text(exp.scrape())
        names = {s.name for s in samples}
        assert "amd_hbm_bandwidth_utilization" in names
        assert "amd_xgmi_link_read_bytes_per_second" in names
        links = [s for s in samples
                 if s.name == "amd_xgmi_link_read_bytes_per_second"]
        assert len(links) == 7  # 7 xGMI links per MI355X
        assert all("link" in s.labels for s in links)

    def test_xgmi_rate_derivation(self, tmp_path):
        """Windowed xGMI rates = accumulator delta / dt. The mock backend
        advances each link's read/write accumulators at busy*10 KB per
        elapsed ms, so at busy=50 the derived rate must be ~512 MB/s."""
        busy = tmp_path / "busy"
        busy.write_text("50\n")
        with ExporterProcess(mock_devices=1, interval_ms=100,
                             mock_busy_file=str(busy)) as exp:
            time.sleep(0.45)  # several windows
            samples = parse_prometheus_text(exp.scrape())
        rates = [s.value for s in samples
                 if s.name == "amd_xgmi_link_read_bytes_per_second"]
        assert len(rates) == 7
        expected = 50 * 10 * 1024 * 1000.0  # 512e6 B/s
        for r in rates:
            assert abs(r - expected) / expected < 0.25, (r, expected)
        total = [s.value for s in samples
                 if s.name == "amd_xgmi_total_bytes_per_second"][0]
        assert abs(total - expected * 14) / (expected * 14) < 0.25

    def test_metric_set_file_filters(self, tmp_path):
        f = tmp_path / "metrics.csv"
        f.write_text("# only two families\ndcgm_gpu_utilization\ndcgm_gpu_temp\n")
        with ExporterProcess(mock_devices=1, interval_ms=100,
                             metric_file=str(f)) as exp:
            samples = parse_prometheus_text(exp.scrape())
        names = {s.name for s in samples}
        assert names == {"dcgm_gpu_utilization", "dcgm_gpu_temp"}

    def test_mock_busy_file_step_change(self, tmp_path):
        busy = tmp_path / "busy"
        busy.write_text("12.5\n")
        with ExporterProcess(mock_devices=1, interval_ms=50,
                             mock_busy_file=str(busy)) as exp:
            time.sleep(0.15)
            s1 = [s for s in parse_prometheus_text(exp.scrape())
                  if s.name == "dcgm_gpu_utilization"][0]
            busy.write_text("80\n")
            time.sleep(0.15)
            s2 = [s for s in parse_prometheus_text(exp.scrape())
                  if s.name == "dcgm_gpu_utilization"][0]
        assert s1.value == 12.5
        assert s2.value == 80.0

    def test_per_device_busy(self, tmp_path):
        busy = tmp_path / "busy"
        busy.write_text("0:10\n1:90\n")
        with ExporterProcess(mock_devices=2, interval_ms=50,
                             mock_busy_file=str(busy)) as exp:
            time.sleep(0.1)
            vals = {s.labels["gpu"]: s.value
                    for s in parse_prometheus_text(exp.scrape())
                    if s.name == "dcgm_gpu_utilization"}
        assert vals == {"0": 10.0, "1": 90.0}


@needs_bin
class TestEndpoints:
    def test_health_and_ready(self):
        with ExporterProcess(mock_devices=1, interval_ms=100) as exp:
            for path in ("/healthz", "/readyz"):
                with urllib.request.urlopen(
                    f"http://127.0.0.1:{exp.port}{path}", timeout=2
                ) as r:
                    assert r.status == 200
            with pytest.raises(urllib.error.HTTPError) as ei:
                urllib.request.urlopen(
                    f"http://127.0.0.1:{exp.port}/nope", timeout=2
                )
            assert ei.value.code == 404

    def test_scrape_latency_under_load(self):
        # Prometheus scrapes at 1 s; ours must answer far faster than that
        with ExporterProcess(mock_devices=8, interval_ms=100) as exp:
            t0 = time.monotonic()
            n = 50
            for _ in range(n):
                exp.scrape()
            per_scrape_ms = (time.monotonic() - t0) / n * 1e3
        assert per_scrape_ms < 50, f"scrape too slow: {per_scrape_ms:.1f} ms"


@needs_bin
class TestConfig:
    def test_bad_flag_rejected(self):
        rc = subprocess.run(
            [EXPORTER_BIN, "--definitely-not-a-flag"], capture_output=True
        )
        assert rc.returncode == 2
        assert b"unknown flag" in rc.stderr

    def test_help(self):
        rc = subprocess.run([EXPORTER_BIN, "--help"], capture_output=True)
        assert rc.returncode == 0
        assert b"-c <ms>" in rc.stdout

    def test_reference_env_vars_honored(self, tmp_path):
        # DCGM_EXPORTER_LISTEN / DCGM_EXPORTER_KUBERNETES are the reference's
        # env config (dcgm-exporter.yaml:31-34)
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        env = dict(os.environ)
        env["DCGM_EXPORTER_LISTEN"] = f"127.0.0.1:{port}"
        p = subprocess.Popen(
            [EXPORTER_BIN, "--mock", "1", "-c", "100"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
        )
        try:
            deadline = time.monotonic() + 10
            ok = False
            while time.monotonic() < deadline:
                try:
                    with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/metrics", timeout=1
                    ) as r:
                        ok = b"dcgm_gpu_utilization" in r.read()
                        break
                except Exception:
                    time.sleep(0.05)
            assert ok
        finally:
            p.terminate()
            p.wait(timeout=5)


@needs_bin
class TestConfigEdgeCases:
    def run(self, *args):
        return subprocess.run([EXPORTER_BIN, *args], capture_output=True,
                              timeout=15)

    def test_version_flag(self):
        r = self.run("--version")
        assert r.returncode == 0
        assert b"mi355x-exporter" in r.stdout

    def test_bad_listen_spec(self):
        r = self.run("--mock", "1", "-c", "50", "-l", "not a spec")
        assert r.returncode == 2
        assert b"bad listen spec" in r.stderr

    def test_interval_floor(self):
        r = self.run("--mock", "1", "-c", "1")
        assert r.returncode == 2
        assert b"below 10 ms" in r.stderr

    def test_bad_gpu_id_type(self):
        r = self.run("--mock", "1", "--kubernetes-gpu-id-type", "bogus")
        assert r.returncode == 2

    def test_missing_metric_file(self):
        r = self.run("--mock", "1", "-f", "/nonexistent/metrics.csv")
        assert r.returncode == 2
        assert b"cannot open" in r.stderr
