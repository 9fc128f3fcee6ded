"""exporter — Python-side helpers around the native mi355x-exporter daemon.

The daemon itself is C++ (native/exporter/, built to
native/build/mi355x-exporter); this module spawns/manages it for tests,
the bench harness, and smoke():

  * ExporterProcess: run the binary (real rsmi backend or --mock) on an
    ephemeral port and scrape it.
  * stub_exporter: a pure-Python stand-in (see stub.py) emitting the same
    dcgm_* schema, for environments where even the C++ binary can't run
    (BASELINE.json config 1 kind-cluster pods).
"""

from __future__ import annotations

import os
import subprocess
import time
import urllib.request
from typing import List, Optional

from .. import NATIVE_BUILD

EXPORTER_BIN = os.environ.get(
    "MI355X_EXPORTER_BIN", str(NATIVE_BUILD / "mi355x-exporter")
)


class ExporterError(RuntimeError):
    pass


class ExporterProcess:
    """Context manager around one mi355x-exporter daemon."""

    def __init__(
        self,
        mock_devices: int = 0,
        interval_ms: float = 1000,
        port: int = 0,
        kubernetes: bool = False,
        pod_resources_socket: Optional[str] = None,
        metric_file: Optional[str] = None,
        mock_busy_file: Optional[str] = None,
        gpu_id_type: Optional[str] = None,
        backend: Optional[str] = None,
    ):
        self.args = [EXPORTER_BIN, "-c", str(interval_ms)]
        # port 0 would race; pick a free one ourselves
        if port == 0:
            import socket

            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
            s.close()
        self.port = port
        self.args += ["-l", f"127.0.0.1:{port}"]
        if mock_devices:
            self.args += ["--mock", str(mock_devices)]
        if kubernetes:
            self.args += ["-k"]
        if pod_resources_socket:
            self.args += ["--pod-resources-socket", pod_resources_socket]
        if metric_file:
            self.args += ["-f", metric_file]
        if mock_busy_file:
            self.args += ["--mock-busy-file", mock_busy_file]
        if gpu_id_type:
            self.args += ["--kubernetes-gpu-id-type", gpu_id_type]
        if backend:
            self.args += ["--backend", backend]
        self.proc: Optional[subprocess.Popen] = None

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}/metrics"

    def __enter__(self) -> "ExporterProcess":
        if not os.path.exists(EXPORTER_BIN):
            raise ExporterError(
                f"{EXPORTER_BIN} missing; run `make -C native exporter`"
            )
        self.proc = subprocess.Popen(
            self.args, stdout=subprocess.PIPE, stderr=subprocess.PIPE
        )
        # wait for readiness
        deadline = time.monotonic() + 10
        last = ""
        while time.monotonic() < deadline:
            if self.proc.poll() is not None:
                raise ExporterError(
                    f"exporter exited rc={self.proc.returncode}: "
                    f"{self.proc.stderr.read().decode()[:500]}"
                )
            try:
                with urllib.request.urlopen(
                    f"http://127.0.0.1:{self.port}/readyz", timeout=1
                ) as r:
                    if r.status == 200:
                        return self
            except Exception as e:  # noqa: BLE001
                last = str(e)
            time.sleep(0.05)
        self.terminate()
        raise ExporterError(f"exporter not ready in 10s (last: {last})")

    def scrape(self) -> str:
        with urllib.request.urlopen(self.url, timeout=2) as r:
            return r.read().decode()

    def terminate(self):
        if self.proc and self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                self.proc.kill()
                self.proc.wait()

    def __exit__(self, *exc):
        self.terminate()
        return False


def native_exporter_smoke() -> None:
    """One real-backend sample served over HTTP (used by __graft_entry__.smoke
    on a GPU box). Requires an AMD GPU visible to rocm_smi."""
    with ExporterProcess(interval_ms=200) as exp:
        text = exp.scrape()
    assert "dcgm_gpu_utilization{" in text, text[:400]
    assert "dcgm_gpu_temp{" in text, "dcgm_gpu_temp missing (README.md:46 probe)"
