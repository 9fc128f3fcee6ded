"""podresources_wire.py — kubelet v1 PodResources protobuf wire encoder.

Hand-encodes `v1.ListPodResourcesResponse` messages (the reply of the
kubelet's `/v1.PodResourcesLister/List` gRPC method the exporter's
attribution client consumes — native/exporter/podresources.cpp; reference
attribution machinery: dcgm-exporter.yaml:33-37,49-52). Used by the fake
kubelets in tests/ and tools/ (served verbatim over a real grpcio server,
so the full transport + framing + proto path is exercised).

Message shapes (kubelet pod-resources v1 API):
    ListPodResourcesResponse { repeated PodResources pod_resources = 1; }
    PodResources   { string name=1; string namespace=2;
                     repeated ContainerResources containers=3; }
    ContainerResources { string name=1; repeated ContainerDevices devices=2; }
    ContainerDevices   { string resource_name=1; repeated string device_ids=2; }
"""

from __future__ import annotations

__all__ = ["container", "container_devices", "list_response", "pod"]


def _tag(field: int, wire: int) -> bytes:
    return bytes([(field << 3) | wire])


def _varint(n: int) -> bytes:
    out = b""
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            out += bytes([b])
            return out


def _ld(field: int, payload: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(payload)) + payload


def _s(field: int, s: str) -> bytes:
    return _ld(field, s.encode())


def container_devices(resource: str, ids) -> bytes:
    out = _s(1, resource)
    for i in ids:
        out += _s(2, i)
    return out


def container(name: str, devices) -> bytes:
    out = _s(1, name)
    for d in devices:
        out += _ld(2, d)
    return out


def pod(name: str, ns: str, containers) -> bytes:
    out = _s(1, name) + _s(2, ns)
    for c in containers:
        out += _ld(3, c)
    return out


def list_response(pods) -> bytes:
    out = b""
    for p in pods:
        out += _ld(1, p)
    return out
