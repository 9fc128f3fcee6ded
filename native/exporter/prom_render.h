// prom_render.h — Prometheus text-format rendering with the reference's
// dcgm_* 1.x-compat metric-name/label contract.
//
// Contract being preserved (SURVEY.md §1 "label-schema coupling"):
//   * metric family names: the dcgm 1.x legacy set the reference selects via
//     `-f .../1.x-compatibility-metrics.csv` (dcgm-exporter.yaml:37), most
//     importantly `dcgm_gpu_utilization` (the recording-rule input,
//     cuda-test-prometheusrule.yaml:13) and `dcgm_gpu_temp` (the README's
//     verification probe, README.md:46);
//   * labels: {gpu, uuid, device, modelName} always; {pod, namespace,
//     container} when Kubernetes attribution is on (DCGM_EXPORTER_KUBERNETES
//     analog, dcgm-exporter.yaml:33-34).
// Plus the MI355X-native amd_* family (HBM bandwidth, xGMI per-link rates,
// windowed busy%) the reference has no counterpart for (BASELINE config 5).

#pragma once

#include "backend.h"
#include "sampler.h"

#include <map>
#include <set>
#include <string>

namespace mi355x {

struct PodAttribution {
    std::string pod;
    std::string ns;
    std::string container;
};

// device-key -> attribution; key form depends on --kubernetes-gpu-id-type:
//   "device-name": drm render node name ("renderD128")
//   "uuid":        GPU unique id hex
//   "index":       decimal device index
using AttributionMap = std::map<std::string, PodAttribution>;

struct RenderOptions {
    bool kubernetes = false;
    std::string hostname;  // dcgm-exporter's Hostname label (empty = omit)
    std::string gpu_id_type = "device-name";
    // empty set = all metrics; otherwise only families named here (the
    // reference's `-f` metric-set file, one name per line, '#' comments).
    std::set<std::string> metric_set;
    // counter-availability probe results (Backend::probes()), rendered as
    // amd_counter_unavailable meta-metrics so degradation is observable.
    std::vector<CounterProbe> probes;
};

std::string attribution_key(const GpuInfo& info, const std::string& id_type);

std::string render_metrics(const std::vector<DeviceMetrics>& devs,
                           const AttributionMap& attr, const RenderOptions& opt);

} // namespace mi355x
