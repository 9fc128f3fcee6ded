// capi.cpp — C ABI for libmi355x_sampler.so (Python/ctypes test surface).
//
// Lets the CPU test-suite drive the native pieces in isolation:
//   * mi355x_list_pod_resources_json(): the hand-rolled gRPC client against
//     a fake kubelet (tests/test_podresources.py runs a real grpcio server)
//   * mi355x_parse_list_response_json(): the protobuf walker on raw bytes
//   * mi355x_render_mock_metrics(): renderer end-to-end on the mock backend

#include "backend.h"
#include "podresources.h"
#include "prom_render.h"
#include "sampler.h"

#include <algorithm>
#include <chrono>
#include <cstring>
#include <vector>
#include <sstream>
#include <string>

namespace {

std::string jesc(const std::string& s)
{
    std::string o;
    for (char c : s) {
        if (c == '"' || c == '\\') o += '\\';
        o += c;
    }
    return o;
}

std::string allocs_to_json(const std::vector<mi355x::DeviceAllocation>& allocs)
{
    std::ostringstream js;
    js << "[";
    for (size_t i = 0; i < allocs.size(); ++i) {
        const auto& a = allocs[i];
        if (i) js << ",";
        js << "{\"pod\":\"" << jesc(a.pod) << "\",\"namespace\":\"" << jesc(a.ns)
           << "\",\"container\":\"" << jesc(a.container) << "\",\"resource\":\""
           << jesc(a.resource_name) << "\",\"device_ids\":[";
        for (size_t j = 0; j < a.device_ids.size(); ++j) {
            if (j) js << ",";
            js << "\"" << jesc(a.device_ids[j]) << "\"";
        }
        js << "]}";
    }
    js << "]";
    return js.str();
}

int copy_out(const std::string& s, char* buf, int buflen)
{
    if ((int)s.size() + 1 > buflen) return -(int)s.size() - 1;
    std::memcpy(buf, s.c_str(), s.size() + 1);
    return (int)s.size();
}

} // namespace

extern "C" {

// Returns bytes written (>=0), or negative required size on overflow, or
// -1 with err copied into buf on transport failure.
int mi355x_list_pod_resources_json(const char* socket_path, char* buf, int buflen)
{
    std::vector<mi355x::DeviceAllocation> allocs;
    std::string err;
    if (!mi355x::list_pod_resources(socket_path, &allocs, &err)) {
        std::snprintf(buf, buflen, "ERR %s", err.c_str());
        return -1;
    }
    return copy_out(allocs_to_json(allocs), buf, buflen);
}

int mi355x_parse_list_response_json(const unsigned char* data, int len, char* buf,
                                    int buflen)
{
    std::vector<mi355x::DeviceAllocation> allocs;
    std::string err;
    if (!mi355x::wire::parse_list_response(data, len, &allocs, &err)) {
        std::snprintf(buf, buflen, "ERR %s", err.c_str());
        return -1;
    }
    return copy_out(allocs_to_json(allocs), buf, buflen);
}

// Render one sampling pass of the mock backend (n devices) with optional
// kubernetes attribution of device key -> pod (single mapping, id_type
// device-name), for renderer contract tests.
int mi355x_render_mock_metrics(int n_devices, const char* attr_key,
                               const char* attr_pod, char* buf, int buflen)
{
    auto backend = mi355x::make_mock_backend(n_devices, "");
    mi355x::Sampler sampler(backend.get(), 1000);
    sampler.sample_once();
    mi355x::AttributionMap attr;
    mi355x::RenderOptions opt;
    if (attr_key && *attr_key) {
        opt.kubernetes = true;
        attr[attr_key] =
            mi355x::PodAttribution{attr_pod ? attr_pod : "", "default", "main"};
    }
    return copy_out(mi355x::render_metrics(sampler.snapshot(), attr, opt), buf,
                    buflen);
}

// Benchmark a real sampling path: n full-device sampling passes.
// backend_name: "rsmi" / "amdsmi". Returns 0 and fills stats (mean/p50/max
// microseconds per pass + device count), or -1 with the error in errbuf.
int mi355x_sample_benchmark_backend(const char* backend_name, int n,
                                    double* mean_us, double* p50_us,
                                    double* max_us, int* n_devices,
                                    char* errbuf, int errlen)
{
    std::string err;
    auto backend = (backend_name && !std::strcmp(backend_name, "amdsmi"))
                       ? mi355x::make_amdsmi_backend(&err)
                       : mi355x::make_rsmi_backend(&err);
    if (!backend) {
        std::snprintf(errbuf, errlen, "%s", err.c_str());
        return -1;
    }
    mi355x::Sampler sampler(backend.get(), 1e9);
    std::vector<double> us;
    us.reserve(n);
    for (int i = 0; i < n; ++i) {
        auto t0 = std::chrono::steady_clock::now();
        sampler.sample_once();
        auto t1 = std::chrono::steady_clock::now();
        us.push_back(std::chrono::duration<double, std::micro>(t1 - t0).count());
    }
    std::sort(us.begin(), us.end());
    double sum = 0;
    for (double v : us) sum += v;
    if (mean_us) *mean_us = sum / us.size();
    if (p50_us) *p50_us = us[us.size() / 2];
    if (max_us) *max_us = us.back();
    if (n_devices) *n_devices = backend->device_count();
    return 0;
}

int mi355x_sample_benchmark(int n, double* mean_us, double* p50_us,
                            double* max_us, int* n_devices, char* errbuf,
                            int errlen)
{
    return mi355x_sample_benchmark_backend("rsmi", n, mean_us, p50_us, max_us,
                                           n_devices, errbuf, errlen);
}

} // extern "C"
