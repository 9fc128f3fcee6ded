// main.cpp — mi355x-exporter daemon.
//
// The MI355X-native, from-scratch replacement for the reference's
// dcgm-exporter container (dcgm-exporter.yaml:28-41): samples
// gfx950 counters via rocm_smi on the `-c` tick, attributes GPUs to pods via
// the kubelet pod-resources API when -k is set, and serves dcgm_*-compatible
// Prometheus text on :9400/metrics (+ /healthz, /readyz probes).

#include "backend.h"
#include "config.h"
#include "http_server.h"
#include "podresources.h"
#include "prom_render.h"
#include "sampler.h"

#include <atomic>
#include <csignal>
#include <cstdio>
#include <memory>
#include <mutex>
#include <thread>
#include <unistd.h>

using namespace mi355x;

static std::atomic<bool> g_stop{false};
static void on_signal(int) { g_stop = true; }

int main(int argc, char** argv)
{
    Config cfg;
    std::string err;
    if (!parse_config(argc, argv, &cfg, &err)) {
        std::fprintf(stderr, "mi355x-exporter: %s\n%s", err.c_str(), config_usage());
        return 2;
    }
    if (cfg.show_help) {
        std::fputs(config_usage(), stdout);
        return 0;
    }
    if (cfg.show_version) {
        std::puts("mi355x-exporter 0.2.0 (gfx950; amd_smi/rocm_smi backends)");
        return 0;
    }

    std::unique_ptr<Backend> backend;
    if (cfg.mock_devices > 0) {
        backend = make_mock_backend(cfg.mock_devices, cfg.mock_busy_file);
    } else {
        // auto selection is CADENCE-AWARE: libamd_smi caches gpu_metrics
        // internally for ~100-200 ms, so at sub-250 ms ticks every other
        // sample would repeat stale accumulator values (measured:
        // update-interval p90 206 ms via amd-smi vs 102 ms via rocm_smi
        // at a 100 ms tick — profiles/exporter_cadence_jitter.md). Fast
        // cadences therefore prefer rocm_smi; the DaemonSet-default 1 s
        // tick prefers amd-smi (rocm_smi_lib is in maintenance mode
        // upstream). --backend pins one explicitly.
        std::string err_amdsmi, err_rsmi;
        bool fast_tick = cfg.interval_ms < 250.0;
        if (cfg.backend == "rsmi" || (cfg.backend == "auto" && fast_tick))
            backend = make_rsmi_backend(&err_rsmi);
        if (!backend && (cfg.backend == "auto" || cfg.backend == "amdsmi"))
            backend = make_amdsmi_backend(&err_amdsmi);
        if (!backend && cfg.backend == "auto" && !fast_tick)
            backend = make_rsmi_backend(&err_rsmi);
        if (!backend) {
            std::fprintf(stderr,
                         "mi355x-exporter: no GPU backend available "
                         "(amd_smi: %s; rocm_smi: %s); "
                         "use --mock N for a GPU-less stub\n",
                         err_amdsmi.empty() ? "not tried" : err_amdsmi.c_str(),
                         err_rsmi.empty() ? "not tried" : err_rsmi.c_str());
            return 3;
        }
    }
    std::fprintf(stderr, "mi355x-exporter: backend=%s devices=%d interval=%.0fms\n",
                 backend->name().c_str(), backend->device_count(), cfg.interval_ms);

    Sampler sampler(backend.get(), cfg.interval_ms);
    sampler.start();

    // pod attribution state, refreshed on the sampling cadence
    std::mutex attr_mu;
    AttributionMap attr;
    std::thread attr_thread;
    std::vector<GpuInfo> infos;
    for (int i = 0; i < backend->device_count(); ++i) infos.push_back(backend->info(i));
    if (cfg.kubernetes) {
        attr_thread = std::thread([&] {
            while (!g_stop) {
                std::vector<DeviceAllocation> allocs;
                std::string perr;
                if (list_pod_resources(cfg.pod_resources_socket, &allocs, &perr)) {
                    auto m = build_attribution(allocs, infos, cfg.gpu_id_type);
                    std::lock_guard<std::mutex> lk(attr_mu);
                    attr.swap(m);
                } else {
                    static int logged = 0;
                    if (logged++ < 5)
                        std::fprintf(stderr, "mi355x-exporter: pod-resources: %s\n",
                                     perr.c_str());
                }
                for (int i = 0; i < 10 && !g_stop; ++i)
                    std::this_thread::sleep_for(
                        std::chrono::duration<double, std::milli>(cfg.interval_ms / 10));
            }
        });
    }

    RenderOptions ropt;
    {
        char host[256] = {0};
        if (gethostname(host, sizeof(host) - 1) == 0) ropt.hostname = host;
    }
    ropt.kubernetes = cfg.kubernetes;
    ropt.gpu_id_type = cfg.gpu_id_type;
    ropt.metric_set = cfg.metric_set;
    ropt.probes = backend->probes();
    for (auto& p : ropt.probes)
        if (!p.available)
            std::fprintf(stderr, "mi355x-exporter: counter %s unavailable: %s\n",
                         p.counter.c_str(), p.reason.c_str());

    HttpServer server(
        cfg.listen_host, cfg.listen_port,
        [&]() {
            auto devs = sampler.snapshot();
            std::lock_guard<std::mutex> lk(attr_mu);
            return render_metrics(devs, attr, ropt);
        },
        [&]() { return sampler.ready(); });
    if (!server.start(&err)) {
        std::fprintf(stderr, "mi355x-exporter: %s\n", err.c_str());
        return 4;
    }
    std::fprintf(stderr, "mi355x-exporter: listening on %s:%d\n",
                 cfg.listen_host.c_str(), server.bound_port());

    std::signal(SIGINT, on_signal);
    std::signal(SIGTERM, on_signal);
    while (!g_stop) std::this_thread::sleep_for(std::chrono::milliseconds(200));

    server.stop();
    sampler.stop();
    if (attr_thread.joinable()) attr_thread.join();
    return 0;
}
