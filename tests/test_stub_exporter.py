"""Tests for the pure-Python stub exporter (the config-1 kind-harness pod)."""

import threading
import urllib.request

import pytest

from mi355x_gpu_hpa.control import parse_prometheus_text
from mi355x_gpu_hpa.exporter.stub import serve


@pytest.fixture()
def stub():
    srv, state = serve(port=0, n_gpus=2, busy=7.5)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{port}", state
    srv.shutdown()


def _get(url):
    with urllib.request.urlopen(url, timeout=2) as r:
        return r.status, r.read().decode()


def test_schema_matches_native(stub):
    url, _ = stub
    _, text = _get(url + "/metrics")
    samples = parse_prometheus_text(text)
    util = [s for s in samples if s.name == "dcgm_gpu_utilization"]
    assert len(util) == 2
    for s in util:
        assert set(s.labels) >= {"gpu", "uuid", "device", "modelName"}
        assert s.value == 7.5
    assert any(s.name == "dcgm_gpu_temp" for s in samples)


def test_busy_post_step_change(stub):
    url, _ = stub
    req = urllib.request.Request(url + "/busy", data=b"42.5", method="POST")
    with urllib.request.urlopen(req, timeout=2) as r:
        assert r.status == 200
    _, text = _get(url + "/metrics")
    vals = [s.value for s in parse_prometheus_text(text)
            if s.name == "dcgm_gpu_utilization"]
    assert vals == [42.5, 42.5]


def test_pod_attribution_env(monkeypatch):
    monkeypatch.setenv("POD_NAME", "cuda-test-zzz")
    monkeypatch.setenv("POD_NAMESPACE", "default")
    srv, _ = serve(port=0, n_gpus=1, busy=1.0)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        _, text = _get(f"http://127.0.0.1:{port}/metrics")
    finally:
        srv.shutdown()
    s = [x for x in parse_prometheus_text(text)
         if x.name == "dcgm_gpu_utilization"][0]
    assert s.labels["pod"] == "cuda-test-zzz"
    assert s.labels["namespace"] == "default"


def test_health_endpoints(stub):
    url, _ = stub
    assert _get(url + "/healthz")[0] == 200
    assert _get(url + "/readyz")[0] == 200
