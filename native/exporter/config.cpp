#include "config.h"

#include <cstdlib>
#include <cstring>
#include <fstream>

namespace mi355x {

const char* config_usage()
{
    return "mi355x-exporter — MI355X-native GPU metrics exporter (dcgm_* compat)\n"
           "\n"
           "  -c <ms>                     collect interval in ms (default 10000)\n"
           "  -f <file>                   metric-set file (one family per line)\n"
           "  -l, --listen <[host]:port>  listen address (default :9400)\n"
           "  -k, --kubernetes            enable kubelet pod attribution\n"
           "  --kubernetes-gpu-id-type <device-name|uuid|index>\n"
           "  --pod-resources-socket <path>\n"
           "  --backend <auto|amdsmi|rsmi> counter library (default auto)\n"
           "  --mock <n>                  mock backend with n synthetic GPUs\n"
           "  --mock-busy-file <path>     scriptable busy%% for the mock backend\n"
           "  -v, --version\n"
           "  -h, --help\n";
}

bool parse_listen(const std::string& spec, Config* cfg)
{
    auto colon = spec.rfind(':');
    if (colon == std::string::npos) {
        char* end = nullptr;
        long p = std::strtol(spec.c_str(), &end, 10);
        if (end && *end == 0 && p >= 0 && p < 65536) {
            cfg->listen_port = (int)p;
            return true;
        }
        return false;
    }
    std::string host = spec.substr(0, colon);
    std::string port = spec.substr(colon + 1);
    char* end = nullptr;
    long p = std::strtol(port.c_str(), &end, 10);
    if (!end || *end != 0 || p < 0 || p >= 65536) return false;
    cfg->listen_port = (int)p;
    if (!host.empty()) cfg->listen_host = host;
    return true;
}

bool load_metric_file(const std::string& path, std::set<std::string>* out,
                      std::string* err)
{
    std::ifstream f(path);
    if (!f) {
        if (err) *err = "cannot open metric-set file " + path;
        return false;
    }
    std::string line;
    while (std::getline(f, line)) {
        // trim + drop comments (also tolerates the dcgm CSV style
        // "DCGM_FI..., gauge, help text" by taking the first token)
        auto hash = line.find('#');
        if (hash != std::string::npos) line = line.substr(0, hash);
        auto comma = line.find(',');
        if (comma != std::string::npos) line = line.substr(0, comma);
        size_t b = line.find_first_not_of(" \t\r");
        if (b == std::string::npos) continue;
        size_t e = line.find_last_not_of(" \t\r");
        out->insert(line.substr(b, e - b + 1));
    }
    return true;
}

bool parse_config(int argc, char** argv, Config* cfg, std::string* err)
{
    // env fallbacks first; flags override
    if (const char* v = std::getenv("DCGM_EXPORTER_LISTEN")) parse_listen(v, cfg);
    if (const char* v = std::getenv("MI355X_EXPORTER_LISTEN")) parse_listen(v, cfg);
    if (const char* v = std::getenv("DCGM_EXPORTER_INTERVAL"))
        cfg->interval_ms = std::atof(v);
    if (const char* v = std::getenv("MI355X_EXPORTER_INTERVAL"))
        cfg->interval_ms = std::atof(v);
    auto truthy = [](const char* v) {
        return !std::strcmp(v, "true") || !std::strcmp(v, "1") ||
               !std::strcmp(v, "yes");
    };
    if (const char* v = std::getenv("DCGM_EXPORTER_KUBERNETES"))
        cfg->kubernetes = truthy(v);
    if (const char* v = std::getenv("MI355X_EXPORTER_KUBERNETES"))
        cfg->kubernetes = truthy(v);
    if (const char* v = std::getenv("MI355X_EXPORTER_BACKEND"))
        cfg->backend = v;

    auto need = [&](int i) -> const char* {
        if (i + 1 >= argc) return nullptr;
        return argv[i + 1];
    };
    for (int i = 1; i < argc; ++i) {
        std::string a = argv[i];
        const char* v;
        if (a == "-h" || a == "--help") {
            cfg->show_help = true;
            return true;
        } else if (a == "-v" || a == "--version") {
            cfg->show_version = true;
            return true;
        } else if (a == "-c") {
            if (!(v = need(i))) goto missing;
            cfg->interval_ms = std::atof(v);
            ++i;
        } else if (a == "-f") {
            if (!(v = need(i))) goto missing;
            cfg->metric_file = v;
            ++i;
        } else if (a == "-l" || a == "--listen") {
            if (!(v = need(i))) goto missing;
            if (!parse_listen(v, cfg)) {
                if (err) *err = "bad listen spec: " + std::string(v);
                return false;
            }
            ++i;
        } else if (a == "-k" || a == "--kubernetes") {
            cfg->kubernetes = true;
        } else if (a == "--kubernetes-gpu-id-type") {
            if (!(v = need(i))) goto missing;
            cfg->gpu_id_type = v;
            if (cfg->gpu_id_type != "device-name" && cfg->gpu_id_type != "uuid" &&
                cfg->gpu_id_type != "index") {
                if (err) *err = "bad --kubernetes-gpu-id-type " + cfg->gpu_id_type;
                return false;
            }
            ++i;
        } else if (a == "--pod-resources-socket") {
            if (!(v = need(i))) goto missing;
            cfg->pod_resources_socket = v;
            ++i;
        } else if (a == "--backend") {
            if (!(v = need(i))) goto missing;
            cfg->backend = v;
            if (cfg->backend != "auto" && cfg->backend != "amdsmi" &&
                cfg->backend != "rsmi") {
                if (err) *err = "bad --backend " + cfg->backend;
                return false;
            }
            ++i;
        } else if (a == "--mock") {
            if (!(v = need(i))) goto missing;
            cfg->mock_devices = std::atoi(v);
            ++i;
        } else if (a == "--mock-busy-file") {
            if (!(v = need(i))) goto missing;
            cfg->mock_busy_file = v;
            ++i;
        } else {
            if (err) *err = "unknown flag " + a;
            return false;
        }
        continue;
    missing:
        if (err) *err = "flag " + a + " needs a value";
        return false;
    }
    if (cfg->backend != "auto" && cfg->backend != "amdsmi" &&
        cfg->backend != "rsmi") {
        if (err) *err = "bad backend " + cfg->backend;
        return false;
    }
    if (cfg->interval_ms < 10) {
        if (err) *err = "collect interval below 10 ms";
        return false;
    }
    if (!cfg->metric_file.empty() &&
        !load_metric_file(cfg->metric_file, &cfg->metric_set, err))
        return false;
    return true;
}

} // namespace mi355x
