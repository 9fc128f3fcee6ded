"""Failure-detection / recovery semantics: the stack
delegates restart to Kubernetes, so what OUR components must guarantee is
(a) the control loop degrades to stale data, never crashes, when the
exporter dies, and (b) a restarted exporter (what the DaemonSet's liveness
probe produces) is picked up without scraper reconfiguration."""

import time

import pytest

from mi355x_gpu_hpa.control import (
    ControlLoop,
    HpaSpec,
    Scraper,
    ScrapeTarget,
    synth_pod_labels,
)
from mi355x_gpu_hpa.exporter import EXPORTER_BIN, ExporterProcess
import os

needs_bin = pytest.mark.skipif(
    not os.path.exists(EXPORTER_BIN), reason="native exporter not built"
)


@needs_bin
def test_exporter_crash_and_restart(tmp_path):
    busy = tmp_path / "busy"
    busy.write_text("30\n")
    exp = ExporterProcess(mock_devices=1, interval_ms=50,
                          mock_busy_file=str(busy))
    exp.__enter__()
    port = exp.port
    try:
        scraper = Scraper(
            [ScrapeTarget(exp.url, node="n0",
                          extra_labels={"pod": "cuda-test-r",
                                        "namespace": "default"})],
            timeout_s=0.5,
        )
        loop = ControlLoop(
            scraper,
            hpa_spec=HpaSpec(max_replicas=3),
            extra_samples=lambda: synth_pod_labels(["cuda-test-r"]),
        )
        time.sleep(0.12)
        r1 = loop.step()
        assert r1.metric_value == 30.0
        replicas_before = r1.replicas

        # crash the exporter (liveness failure)
        exp.proc.kill()
        exp.proc.wait()
        r2 = loop.step()
        # stale data keeps serving; HPA holds its decision
        assert r2.metric_value == 30.0
        assert r2.replicas == replicas_before

        # "kubelet restarts the container": same port, fresh process
        busy.write_text("60\n")
        exp2 = ExporterProcess(mock_devices=1, interval_ms=50, port=port,
                               mock_busy_file=str(busy))
        exp2.__enter__()
        try:
            time.sleep(0.12)
            r3 = loop.step()
            assert r3.metric_value == 60.0  # fresh data, no reconfiguration
        finally:
            exp2.terminate()
    finally:
        exp.terminate()
