"""Memory-safety: the exporter's full mock pipeline under
AddressSanitizer + UBSan + LeakSanitizer (SURVEY.md §5.2 exceeds the
reference, which ships no sanitizer coverage at all)."""

import os
import shutil
import socket
import subprocess
import time
import urllib.request
from pathlib import Path

import pytest

NATIVE = Path(__file__).resolve().parent.parent / "native"
ASAN_BIN = NATIVE / "build" / "mi355x-exporter-asan"


@pytest.fixture(scope="module")
def asan_bin():
    if not shutil.which("g++"):
        pytest.skip("no g++")
    r = subprocess.run(["make", "-C", str(NATIVE), "asan"], capture_output=True)
    if r.returncode != 0 or not ASAN_BIN.exists():
        pytest.skip(f"asan build unavailable: {r.stderr.decode()[-300:]}")
    return str(ASAN_BIN)


def test_no_leaks_or_ub(asan_bin, tmp_path):
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    log = tmp_path / "asan.log"
    env = dict(os.environ)
    env["ASAN_OPTIONS"] = f"detect_leaks=1:log_path={log}:exitcode=66"
    env["UBSAN_OPTIONS"] = f"log_path={log}:halt_on_error=0"
    p = subprocess.Popen(
        [asan_bin, "--mock", "4", "-c", "20", "-l", f"127.0.0.1:{port}"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
    )
    try:
        deadline = time.monotonic() + 20
        up = False
        while time.monotonic() < deadline:
            try:
                urllib.request.urlopen(f"http://127.0.0.1:{port}/readyz",
                                       timeout=1)
                up = True
                break
            except Exception:
                time.sleep(0.1)
        assert up
        for _ in range(40):
            with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/metrics", timeout=5
            ) as r:
                assert b"dcgm_gpu_utilization" in r.read()
    finally:
        p.terminate()  # SIGTERM -> clean shutdown path -> LSAN runs at exit
        try:
            p.wait(timeout=15)
        except subprocess.TimeoutExpired:
            p.kill()
            p.wait()

    report = "".join(f.read_text() for f in tmp_path.glob("asan.log*"))
    assert "ERROR: " not in report, report[:3000]
    assert "runtime error" not in report, report[:3000]
    assert p.returncode == 0, (p.returncode, report[:2000])
