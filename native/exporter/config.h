// config.h — exporter configuration, mirroring the reference exporter's
// surface (dcgm-exporter.yaml:30-37):
//   -c <ms>             collect interval (reference: `-c 10000`)
//   -f <file>           metric-set file, one family per line ('#' comments)
//                       (reference: `-f .../1.x-compatibility-metrics.csv`)
//   -l/--listen <addr>  listen address ":9400" / "0.0.0.0:9400"
//                       (reference env DCGM_EXPORTER_LISTEN)
//   -k/--kubernetes     enable pod attribution
//                       (reference env DCGM_EXPORTER_KUBERNETES=true)
//   --kubernetes-gpu-id-type {device-name|uuid|index}
//   --pod-resources-socket <path>   kubelet pod-resources unix socket
// MI355X additions:
//   --mock <n>          mock backend with n synthetic GPUs (CPU-only tests)
//   --mock-busy-file <p> scriptable busy% source for the mock backend
//   --backend <auto|amdsmi|rsmi>  counter library selection; auto prefers
//                       amd-smi (rocm_smi_lib is in maintenance mode) and
//                       falls back to rocm_smi
//
// Env fallbacks honored (so the reference's env-style config keeps working):
//   DCGM_EXPORTER_LISTEN, DCGM_EXPORTER_KUBERNETES, DCGM_EXPORTER_INTERVAL,
//   and the native MI355X_EXPORTER_* equivalents.

#pragma once

#include <set>
#include <string>

namespace mi355x {

struct Config {
    double interval_ms = 10000; // reference default cadence (-c 10000)
    std::string listen_host = "0.0.0.0";
    int listen_port = 9400;
    bool kubernetes = false;
    std::string gpu_id_type = "device-name";
    std::string pod_resources_socket =
        "/var/lib/kubelet/pod-resources/kubelet.sock";
    std::string metric_file;
    std::set<std::string> metric_set; // parsed from metric_file
    int mock_devices = 0;             // >0 => mock backend
    std::string mock_busy_file;
    std::string backend = "auto";     // auto|amdsmi|rsmi
    bool show_help = false;
    bool show_version = false;
};

// Returns false + err message on a bad flag. argv-style parsing.
bool parse_config(int argc, char** argv, Config* cfg, std::string* err);

// Parse a "host:port" / ":port" listen spec into cfg fields.
bool parse_listen(const std::string& spec, Config* cfg);

// Load the -f metric-set file into cfg->metric_set.
bool load_metric_file(const std::string& path, std::set<std::string>* out,
                      std::string* err);

const char* config_usage();

} // namespace mi355x
