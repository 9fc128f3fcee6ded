"""Tests for the hand-rolled kubelet pod-resources gRPC client.

The C++ client (native/exporter/podresources.cpp) speaks raw HTTP/2 + gRPC
framing + protobuf wire format. Here it is exercised two ways:
  * the protobuf walker on hand-encoded ListPodResourcesResponse bytes
    (pure unit test, via libmi355x_sampler.so ctypes)
  * the full transport against a REAL gRPC server (python grpcio) serving
    the v1.PodResourcesLister/List method over a unix socket — the same
    protocol a kubelet speaks.
"""

import ctypes
import json
import os
import struct
from concurrent import futures

import pytest

from mi355x_gpu_hpa import NATIVE_BUILD

SAMPLER_LIB = str(NATIVE_BUILD / "libmi355x_sampler.so")

needs_lib = pytest.mark.skipif(
    not os.path.exists(SAMPLER_LIB), reason="libmi355x_sampler.so not built"
)


def _lib():
    lib = ctypes.CDLL(SAMPLER_LIB)
    lib.mi355x_list_pod_resources_json.argtypes = [
        ctypes.c_char_p, ctypes.c_char_p, ctypes.c_int
    ]
    lib.mi355x_parse_list_response_json.argtypes = [
        ctypes.c_char_p, ctypes.c_int, ctypes.c_char_p, ctypes.c_int
    ]
    return lib


# --- protobuf encoding helpers: the shared kubelet wire encoder ----------

from mi355x_gpu_hpa.podresources_wire import (  # noqa: E402
    _ld,
    _tag,
    _varint,
    container,
    container_devices,
    list_response,
    pod,
)

SAMPLE_RESPONSE = list_response([
    pod("cuda-test-abc", "default", [
        container("main", [container_devices("amd.com/gpu", ["renderD128"])]),
    ]),
    pod("other-pod", "kube-system", [
        container("sidecar", [container_devices("example.com/nic", ["nic0"])]),
    ]),
    pod("cuda-test-def", "default", [
        container("main", [container_devices("amd.com/gpu",
                                             ["renderD129", "renderD130"])]),
    ]),
])


@needs_lib
class TestProtobufWalker:
    def run_parse(self, data: bytes):
        lib = _lib()
        buf = ctypes.create_string_buffer(65536)
        rc = lib.mi355x_parse_list_response_json(data, len(data), buf, len(buf))
        assert rc >= 0, buf.value
        return json.loads(buf.value.decode())

    def test_parses_pods_and_devices(self):
        allocs = self.run_parse(SAMPLE_RESPONSE)
        gpu_allocs = [a for a in allocs if a["resource"] == "amd.com/gpu"]
        assert len(gpu_allocs) == 2
        assert gpu_allocs[0]["pod"] == "cuda-test-abc"
        assert gpu_allocs[0]["device_ids"] == ["renderD128"]
        assert gpu_allocs[1]["device_ids"] == ["renderD129", "renderD130"]
        assert gpu_allocs[1]["namespace"] == "default"
        assert gpu_allocs[0]["container"] == "main"

    def test_skips_unknown_fields(self):
        # add an unknown field 9 (varint) and 8 (length-delimited) at top level
        data = _tag(9, 0) + _varint(12345) + _ld(8, b"whatever") + SAMPLE_RESPONSE
        allocs = self.run_parse(data)
        assert len([a for a in allocs if a["resource"] == "amd.com/gpu"]) == 2

    def test_empty_response(self):
        assert self.run_parse(b"") == []

    def test_malformed_rejected(self):
        lib = _lib()
        buf = ctypes.create_string_buffer(4096)
        bad = _tag(1, 2) + _varint(1000) + b"short"
        rc = lib.mi355x_parse_list_response_json(bad, len(bad), buf, len(buf))
        assert rc == -1
        assert buf.value.startswith(b"ERR")

    def test_huge_varint_length_rejected(self):
        # a near-2^64 length-delimited size used to
        # overflow `p + l` pointer arithmetic before the bound check; the
        # guard now compares lengths. Must reject, not crash or accept.
        lib = _lib()
        for huge in (2**64 - 1, 2**63, 2**32 + 7):
            buf = ctypes.create_string_buffer(4096)
            bad = _tag(1, 2) + _varint(huge) + b"x"
            rc = lib.mi355x_parse_list_response_json(bad, len(bad),
                                                     buf, len(buf))
            assert rc == -1, huge

    def test_truncated_fixed_width_rejected(self):
        # wire types 1 (64-bit) and 5 (32-bit) with fewer payload bytes
        # than the width must fail cleanly (skip() bound check)
        lib = _lib()
        for wire, payload in ((1, b"\x01\x02"), (5, b"\x01")):
            buf = ctypes.create_string_buffer(4096)
            bad = _tag(7, wire) + payload
            rc = lib.mi355x_parse_list_response_json(bad, len(bad),
                                                     buf, len(buf))
            assert rc == -1, wire


@needs_lib
class TestAgainstRealGrpcServer:
    @pytest.fixture()
    def grpc_socket(self, tmp_path):
        grpc = pytest.importorskip("grpc")

        class Handler(grpc.GenericRpcHandler):
            def service(self, handler_call_details):
                if handler_call_details.method == "/v1.PodResourcesLister/List":
                    return grpc.unary_unary_rpc_method_handler(
                        lambda req, ctx: SAMPLE_RESPONSE,
                        request_deserializer=None,   # raw bytes
                        response_serializer=None,    # raw bytes
                    )
                return None

        sock = tmp_path / "kubelet.sock"
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        server.add_generic_rpc_handlers((Handler(),))
        server.add_insecure_port(f"unix:{sock}")
        server.start()
        yield str(sock)
        server.stop(0)

    def test_unary_list_over_unix_socket(self, grpc_socket):
        lib = _lib()
        buf = ctypes.create_string_buffer(65536)
        rc = lib.mi355x_list_pod_resources_json(
            grpc_socket.encode(), buf, len(buf)
        )
        assert rc >= 0, buf.value
        allocs = json.loads(buf.value.decode())
        gpu_allocs = [a for a in allocs if a["resource"] == "amd.com/gpu"]
        assert {a["pod"] for a in gpu_allocs} == {"cuda-test-abc", "cuda-test-def"}

    def test_missing_socket_fails_cleanly(self, tmp_path):
        lib = _lib()
        buf = ctypes.create_string_buffer(4096)
        rc = lib.mi355x_list_pod_resources_json(
            str(tmp_path / "nope.sock").encode(), buf, len(buf)
        )
        assert rc == -1
        assert b"connect" in buf.value


@needs_lib
class TestExporterK8sMode:
    """Full integration: exporter -k against the fake kubelet; pod labels
    appear on the metrics (the DCGM_EXPORTER_KUBERNETES analog)."""

    def test_pod_labels_on_metrics(self, tmp_path):
        grpc = pytest.importorskip("grpc")
        import time as _time

        from mi355x_gpu_hpa.control import parse_prometheus_text
        from mi355x_gpu_hpa.exporter import ExporterProcess

        # mock backend device 0 is renderD128 (mock_backend.cpp)
        response = list_response([
            pod("cuda-test-xyz", "default", [
                container("main",
                          [container_devices("amd.com/gpu", ["renderD128"])]),
            ]),
        ])

        class Handler(grpc.GenericRpcHandler):
            def service(self, hcd):
                if hcd.method == "/v1.PodResourcesLister/List":
                    return grpc.unary_unary_rpc_method_handler(
                        lambda req, ctx: response,
                        request_deserializer=None,
                        response_serializer=None,
                    )
                return None

        state = {"response": response}

        sock = tmp_path / "kubelet.sock"
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))

        class MutableHandler(grpc.GenericRpcHandler):
            def service(self, hcd):
                if hcd.method == "/v1.PodResourcesLister/List":
                    return grpc.unary_unary_rpc_method_handler(
                        lambda req, ctx: state["response"],
                        request_deserializer=None,
                        response_serializer=None,
                    )
                return None

        server.add_generic_rpc_handlers((MutableHandler(),))
        server.add_insecure_port(f"unix:{sock}")
        server.start()
        try:
            with ExporterProcess(
                mock_devices=2, interval_ms=100, kubernetes=True,
                pod_resources_socket=str(sock),
            ) as exp:
                deadline = _time.monotonic() + 5
                attributed = []
                while _time.monotonic() < deadline and not attributed:
                    samples = parse_prometheus_text(exp.scrape())
                    attributed = [s for s in samples
                                  if s.name == "dcgm_gpu_utilization"
                                  and s.labels.get("pod") == "cuda-test-xyz"]
                    _time.sleep(0.1)
                assert attributed, "no pod-attributed series within 5s"
                s = attributed[0]
                assert s.labels["namespace"] == "default"
                assert s.labels["container"] == "main"
                assert s.labels["gpu"] == "0"
                # device 1 (renderD129) is unallocated: no pod label
                samples_by_gpu = {x.labels["gpu"]: x for x in samples
                                  if x.name == "dcgm_gpu_utilization"}
                assert "pod" not in samples_by_gpu["1"].labels

                # --- pod churn: the old pod dies, a new one gets the GPU;
                # the attribution cache must invalidate (SURVEY.md §7
                # "cache invalidation as pods churn") ---
                state["response"] = list_response([
                    pod("cuda-test-new", "default", [
                        container("main", [container_devices(
                            "amd.com/gpu", ["renderD129"])]),
                    ]),
                ])
                deadline = _time.monotonic() + 5
                churned = False
                while _time.monotonic() < deadline and not churned:
                    samples = parse_prometheus_text(exp.scrape())
                    by_gpu = {x.labels["gpu"]: x for x in samples
                              if x.name == "dcgm_gpu_utilization"}
                    churned = (by_gpu["1"].labels.get("pod") == "cuda-test-new"
                               and "pod" not in by_gpu["0"].labels)
                    _time.sleep(0.1)
                assert churned, "attribution did not follow the pod churn"
        finally:
            server.stop(0)
