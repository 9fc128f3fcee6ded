"""Robustness: the native protobuf walker must never crash on malformed
bytes (it parses kubelet responses in the exporter daemon), and the HTTP
server must survive abusive clients."""

import ctypes
import os
import random
import socket
import time

import pytest

from mi355x_gpu_hpa import NATIVE_BUILD
from mi355x_gpu_hpa.exporter import ExporterProcess

SAMPLER_LIB = str(NATIVE_BUILD / "libmi355x_sampler.so")

needs_lib = pytest.mark.skipif(
    not os.path.exists(SAMPLER_LIB), reason="sampler lib not built"
)


@needs_lib
class TestProtobufFuzz:
    def test_random_bytes_never_crash(self):
        lib = ctypes.CDLL(SAMPLER_LIB)
        lib.mi355x_parse_list_response_json.argtypes = [
            ctypes.c_char_p, ctypes.c_int, ctypes.c_char_p, ctypes.c_int
        ]
        buf = ctypes.create_string_buffer(1 << 16)
        rng = random.Random(0)
        for trial in range(2000):
            n = rng.randrange(0, 200)
            data = bytes(rng.randrange(256) for _ in range(n))
            rc = lib.mi355x_parse_list_response_json(data, n, buf, len(buf))
            assert rc >= -1  # either parsed or clean error; no crash

    def test_mutated_valid_message(self):
        from tests.test_podresources import SAMPLE_RESPONSE

        lib = ctypes.CDLL(SAMPLER_LIB)
        lib.mi355x_parse_list_response_json.argtypes = [
            ctypes.c_char_p, ctypes.c_int, ctypes.c_char_p, ctypes.c_int
        ]
        buf = ctypes.create_string_buffer(1 << 16)
        rng = random.Random(1)
        base = bytearray(SAMPLE_RESPONSE)
        for trial in range(500):
            data = bytearray(base)
            for _ in range(rng.randrange(1, 4)):
                data[rng.randrange(len(data))] = rng.randrange(256)
            rc = lib.mi355x_parse_list_response_json(
                bytes(data), len(data), buf, len(buf))
            assert rc >= -1


class TestHttpRobustness:
    @pytest.fixture()
    def exporter(self):
        from mi355x_gpu_hpa.exporter import EXPORTER_BIN

        if not os.path.exists(EXPORTER_BIN):
            pytest.skip("exporter not built")
        with ExporterProcess(mock_devices=1, interval_ms=100) as exp:
            yield exp

    def _raw(self, port, payload: bytes, read=True):
        s = socket.create_connection(("127.0.0.1", port), timeout=3)
        try:
            s.sendall(payload)
            if read:
                s.settimeout(3)
                return s.recv(4096)
        finally:
            s.close()
        return b""

    def test_garbage_request(self, exporter):
        resp = self._raw(exporter.port, b"\x00\x01\x02garbage\r\n\r\n")
        assert b"405" in resp or resp == b""
        # server still alive
        assert "dcgm_gpu_utilization" in exporter.scrape()

    def test_half_open_connection(self, exporter):
        s = socket.create_connection(("127.0.0.1", exporter.port), timeout=3)
        # send nothing, close after a beat
        time.sleep(0.2)
        s.close()
        assert "dcgm_gpu_utilization" in exporter.scrape()

    def test_oversized_request_line(self, exporter):
        resp = self._raw(exporter.port,
                         b"GET /" + b"a" * 8000 + b" HTTP/1.1\r\n\r\n")
        assert b"404" in resp or b"405" in resp or resp
        assert "dcgm_gpu_utilization" in exporter.scrape()

    def test_many_sequential_connections(self, exporter):
        for _ in range(100):
            self._raw(exporter.port, b"GET /healthz HTTP/1.1\r\n\r\n")
        assert "dcgm_gpu_utilization" in exporter.scrape()


@needs_lib
def test_fuzz_under_asan():
    """Deeper fuzz against an ASAN+UBSAN build of the parser: any OOB read
    or UB aborts the interpreter (detected as a crash here)."""
    import shutil
    import subprocess
    from pathlib import Path

    native = Path(__file__).resolve().parent.parent / "native"
    if not shutil.which("g++"):
        pytest.skip("no g++")
    r = subprocess.run(["make", "-C", str(native), "asan-lib"],
                       capture_output=True)
    asan_lib = native / "build" / "libmi355x_sampler_asan.so"
    if r.returncode != 0 or not asan_lib.exists():
        pytest.skip("asan lib unavailable")
    # run the fuzz loop in a subprocess so an ASAN abort fails cleanly
    code = f"""
import ctypes, random
lib = ctypes.CDLL({str(asan_lib)!r})
lib.mi355x_parse_list_response_json.argtypes = [
    ctypes.c_char_p, ctypes.c_int, ctypes.c_char_p, ctypes.c_int]
buf = ctypes.create_string_buffer(1 << 16)
rng = random.Random(7)
for _ in range(20000):
    n = rng.randrange(0, 300)
    data = bytes(rng.randrange(256) for _ in range(n))
    rc = lib.mi355x_parse_list_response_json(data, n, buf, len(buf))
    assert rc >= -1
print("fuzz clean")
"""
    import os

    asan_rt = subprocess.run(["g++", "-print-file-name=libasan.so"],
                             capture_output=True).stdout.decode().strip()
    env = dict(os.environ)
    env["LD_PRELOAD"] = asan_rt
    env["ASAN_OPTIONS"] = "detect_leaks=0"  # python itself leaks by design
    p = subprocess.run(["python3", "-c", code], capture_output=True,
                       timeout=300, env=env)
    assert p.returncode == 0, (p.stdout.decode()[-500:], p.stderr.decode()[-2000:])
    assert b"fuzz clean" in p.stdout
