"""Race detection (SURVEY.md §5.2 — the reference has none): run the
exporter's threaded core (sampler tick thread + HTTP handlers + config) under
ThreadSanitizer with concurrent scrapes and assert no data races."""

import os
import shutil
import subprocess
import time
import urllib.request
from pathlib import Path

import pytest

NATIVE = Path(__file__).resolve().parent.parent / "native"
TSAN_BIN = NATIVE / "build" / "mi355x-exporter-tsan"


@pytest.fixture(scope="module")
def tsan_bin():
    if not shutil.which("g++"):
        pytest.skip("no g++")
    r = subprocess.run(["make", "-C", str(NATIVE), "tsan"], capture_output=True)
    if r.returncode != 0 or not TSAN_BIN.exists():
        pytest.skip(f"tsan build unavailable: {r.stderr.decode()[-300:]}")
    return str(TSAN_BIN)


def test_no_races_under_concurrent_scrapes(tsan_bin, tmp_path):
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    log = tmp_path / "tsan.log"
    env = dict(os.environ)
    env["TSAN_OPTIONS"] = f"log_path={log} exitcode=66"
    p = subprocess.Popen(
        [tsan_bin, "--mock", "4", "-c", "20", "-l", f"127.0.0.1:{port}"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
    )
    try:
        deadline = time.monotonic() + 20
        up = False
        while time.monotonic() < deadline:
            try:
                urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/readyz", timeout=1
                )
                up = True
                break
            except Exception:
                time.sleep(0.1)
        assert up, "tsan exporter never became ready"
        # hammer /metrics while the 20 ms sampler tick runs
        for _ in range(50):
            with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/metrics", timeout=5
            ) as r:
                assert b"dcgm_gpu_utilization" in r.read()
    finally:
        p.terminate()
        try:
            p.wait(timeout=10)
        except subprocess.TimeoutExpired:
            p.kill()
            p.wait()

    tsan_reports = list(tmp_path.glob("tsan.log*"))
    report = "".join(f.read_text() for f in tsan_reports)
    assert "WARNING: ThreadSanitizer" not in report, report[:3000]
    assert p.returncode != 66, "TSAN flagged races at exit"
