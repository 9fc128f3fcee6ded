"""Real-GPU pod-attribution end-to-end (round-1 verdict item 7).

The exporter runs with -k on a real MI355X against the grpcio fake kubelet
serving the REAL device's identifiers (its renderD* node, then its UUID)
over the v1.PodResourcesLister protocol — the same wire a kubelet speaks
(reference attribution machinery: dcgm-exporter.yaml:33-37,49-52). The
scraped series must carry the pod/namespace/container labels. This closes
the last mock-only seam: the CPU tests prove protocol + mapping on mock
ids, this proves the real backend's device identity feeds the same map.
"""

import time
from concurrent import futures

import pytest

from mi355x_gpu_hpa.podresources_wire import (
    container,
    container_devices,
    list_response,
    pod,
)

pytestmark = pytest.mark.gpu


def _serve(sock_path, response):
    grpc = pytest.importorskip("grpc")

    class Handler(grpc.GenericRpcHandler):
        def service(self, hcd):
            if hcd.method == "/v1.PodResourcesLister/List":
                return grpc.unary_unary_rpc_method_handler(
                    lambda req, ctx: response,
                    request_deserializer=None,
                    response_serializer=None,
                )
            return None

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    server.add_generic_rpc_handlers((Handler(),))
    server.add_insecure_port(f"unix:{sock_path}")
    server.start()
    return server


def _real_device_identity():
    """Scrape the real backend once for device 0's renderD*/uuid labels."""
    from mi355x_gpu_hpa.control import parse_prometheus_text
    from mi355x_gpu_hpa.exporter import ExporterProcess

    with ExporterProcess(interval_ms=200) as exp:
        time.sleep(0.5)
        samples = parse_prometheus_text(exp.scrape())
    s = next(x for x in samples if x.name == "dcgm_gpu_utilization"
             and x.labels["gpu"] == "0")
    return s.labels["device"], s.labels["uuid"]


@pytest.mark.parametrize("id_type", ["device-name", "uuid"])
def test_attribution_with_real_device_ids(gpu, tmp_path, id_type):
    from mi355x_gpu_hpa.control import parse_prometheus_text
    from mi355x_gpu_hpa.exporter import ExporterProcess

    render, uuid = _real_device_identity()
    assert render.startswith("renderD"), render
    device_id = render if id_type == "device-name" else uuid
    assert device_id, (id_type, render, uuid)

    response = list_response([
        pod("cuda-test-real", "default", [
            container("main", [container_devices("amd.com/gpu", [device_id])]),
        ]),
    ])
    sock = tmp_path / "kubelet.sock"
    server = _serve(str(sock), response)
    try:
        with ExporterProcess(
            interval_ms=100, kubernetes=True,
            pod_resources_socket=str(sock), gpu_id_type=id_type,
        ) as exp:
            deadline = time.monotonic() + 5
            attributed = []
            while time.monotonic() < deadline and not attributed:
                samples = parse_prometheus_text(exp.scrape())
                attributed = [s for s in samples
                              if s.name == "dcgm_gpu_utilization"
                              and s.labels.get("pod") == "cuda-test-real"]
                time.sleep(0.1)
            assert attributed, (
                f"no pod-attributed series within 5s ({id_type}: {device_id})")
            s = attributed[0]
            assert s.labels["namespace"] == "default"
            assert s.labels["container"] == "main"
            assert s.labels["gpu"] == "0"
    finally:
        server.stop(0)
