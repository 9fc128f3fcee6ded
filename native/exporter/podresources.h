// podresources.h — kubelet pod-resources API client for GPU->pod attribution.
//
// The reference exporter stamps each GPU's metrics with {pod, namespace,
// container} by querying the kubelet's pod-resources gRPC API over the unix
// socket mounted at /var/lib/kubelet/pod-resources (dcgm-exporter.yaml:33-34
// DCGM_EXPORTER_KUBERNETES=true, :49-52,:56-59 socket mounts).
//
// This is a from-scratch minimal gRPC client for exactly one unary call —
// `/v1.PodResourcesLister/List` — speaking HTTP/2 + gRPC framing + protobuf
// wire format directly (the image carries no grpc++/protobuf C++ libs, and
// a full gRPC stack would be absurd overkill for one call on a unix socket).
// Tested against a real grpcio server (tests/test_podresources.py).

#pragma once

#include "prom_render.h"

#include <string>
#include <vector>

namespace mi355x {

struct DeviceAllocation {
    std::string pod;
    std::string ns;
    std::string container;
    std::string resource_name; // e.g. "amd.com/gpu"
    std::vector<std::string> device_ids;
};

// One unary List() call; returns false + err on transport/parse failure.
bool list_pod_resources(const std::string& socket_path,
                        std::vector<DeviceAllocation>* out, std::string* err);

// Build the metric-label attribution map from kubelet allocations.
// Accepts any GPU-plugin resource ending in "/gpu" (amd.com/gpu primarily).
// A device id is matched to a GPU by equality/substring against the id-type
// key (drm render name / uuid / index) plus, as fallbacks, the PCI BDF —
// AMD device-plugin forks differ in id scheme, so matching is permissive.
AttributionMap build_attribution(const std::vector<DeviceAllocation>& allocs,
                                 const std::vector<GpuInfo>& gpus,
                                 const std::string& id_type);

// --- exposed for unit tests (CPU, no kubelet) ---
namespace wire {
// minimal protobuf walker: parse a serialized ListPodResourcesResponse
bool parse_list_response(const uint8_t* data, size_t len,
                         std::vector<DeviceAllocation>* out, std::string* err);
// HPACK header block for the List request (static-table + literals)
std::vector<uint8_t> build_request_headers(const std::string& authority,
                                           const std::string& path);
} // namespace wire

} // namespace mi355x
