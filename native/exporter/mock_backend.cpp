// mock_backend.cpp — deterministic synthetic GPU backend.
//
// This is BASELINE.json config 1's "stub /metrics" made first-class: the
// whole exporter pipeline (sampler, renderer, HTTP server, k8s attribution)
// runs against it with no GPU and no ROCm — that's what lets the L2-L5
// cluster plumbing (Prometheus scrape -> rule -> adapter -> HPA) be
// integration-tested on a CPU-only kind cluster, and what the CPU unit
// tests drive.
//
// busy% source, in priority order:
//   1. `busy_file` (one float per line, or "dev:val" pairs) — re-read each
//      sample, so tests can script a utilization step-change;
//   2. env MI355X_MOCK_BUSY (single float, all devices);
//   3. a deterministic per-device square wave (dev*10 + 5, toggling).

#include "backend.h"

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <sstream>

namespace mi355x {

namespace {

double now_ms()
{
    return std::chrono::duration<double, std::milli>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

class MockBackend final : public Backend {
  public:
    MockBackend(int n, std::string busy_file)
        : n_(n), busy_file_(std::move(busy_file)), t0_(now_ms())
    {
    }

    int device_count() override { return n_; }
    std::string name() const override { return "mock"; }

    GpuInfo info(int dev) override
    {
        GpuInfo gi;
        gi.index = dev;
        gi.name = "AMD Instinct MI355X (mock)";
        char u[32];
        std::snprintf(u, sizeof(u), "mock-%016x", dev);
        gi.uuid = u;
        char b[32];
        std::snprintf(b, sizeof(b), "0000:%02x:00.0", 0x10 + dev);
        gi.pci_bdf = b;
        gi.drm_render = "renderD" + std::to_string(128 + dev);
        const char* np = std::getenv("MI355X_MOCK_PARTITIONS");
        if (np && *np && std::atoi(np) > 0) {
            int n_parts = std::atoi(np);
            gi.compute_partition = n_parts >= 8   ? "CPX"
                                   : n_parts == 4 ? "QPX"
                                   : n_parts == 3 ? "TPX"
                                   : n_parts == 2 ? "DPX"
                                                  : "SPX";
            gi.memory_partition = "NPS1";
            gi.partition_id = 0;
        }
        return gi;
    }

    GpuSample sample(int dev) override
    {
        GpuSample s;
        s.ok = true;
        s.ts_ms = now_ms();
        s.busy_pct = busy_for(dev);
        s.mem_busy_pct = s.busy_pct * 0.6;
        s.gfx_activity_pct = s.busy_pct;
        s.umc_activity_pct = s.mem_busy_pct;
        s.vram_total_bytes = 288.0 * (1ull << 30); // 288 GB HBM3E
        s.vram_used_bytes = (4.0 + dev) * (1ull << 30);
        s.temp_edge_c = 40 + dev;
        s.temp_hotspot_c = 55 + dev;
        s.temp_mem_c = 50 + dev;
        s.power_w = 150 + 10 * s.busy_pct;
        s.sclk_mhz = 2400;
        s.mclk_mhz = 1600;
        s.pcie_tx_bps = 1e9;
        s.pcie_rx_bps = 1e9;
        // 7 xGMI links; accumulators advance with busy% so rate derivation
        // is testable: busy% * 1 MB/ms per link.
        s.xgmi_num_links = 7;
        s.xgmi_link_width = 16;
        s.xgmi_link_speed_gbps = 32;
        double elapsed = s.ts_ms - t0_;
        for (int i = 0; i < 7; ++i) {
            s.xgmi_read_acc_kb[i] = elapsed * s.busy_pct * 10.0;
            s.xgmi_write_acc_kb[i] = elapsed * s.busy_pct * 10.0;
        }
        // busy-time accumulators: ~busy% of wall, in ms units
        s.gfx_activity_acc = elapsed * s.busy_pct / 100.0;
        s.mem_activity_acc = elapsed * s.mem_busy_pct / 100.0;
        // RAS / throttle / replay synthetics
        s.ecc_correctable = dev;       // stable per-device counter
        s.ecc_uncorrectable = 0;
        s.accumulation_counter = elapsed;        // 1 unit per ms
        s.ppt_residency_acc = elapsed * 0.10;    // 10% PVIOL
        s.thm_residency_acc = elapsed * 0.02;    // 2% TVIOL
        s.pcie_replay_count = 3;
        // partitions: MI355X_MOCK_PARTITIONS=n (1..8) => per-XCP busy
        // spread around the device busy% (partition p leans +/- p)
        const char* np = std::getenv("MI355X_MOCK_PARTITIONS");
        if (np && *np) {
            int n_parts = std::atoi(np);
            if (n_parts > kMaxXcp) n_parts = kMaxXcp;
            for (int p = 0; p < n_parts; ++p) {
                double b = s.busy_pct + (p - n_parts / 2.0);
                if (b < 0) b = 0;
                if (b > 100) b = 100;
                s.xcp_busy_pct[p] = b;
            }
            s.num_partitions = n_parts > 0 ? n_parts : 0;
        }
        return s;
    }

    std::vector<CounterProbe> probes() override
    {
        // mirror the real backends' probe surface so the renderer's
        // meta-metric path is CPU-testable
        std::vector<CounterProbe> out;
        out.push_back({"mfma_activity", false,
                       "no MFMA/matrix-pipe field in gpu_metrics v1.8; "
                       "requires rocprofiler-sdk PMC (perturbs workloads)"});
        const char* np = std::getenv("MI355X_MOCK_PARTITIONS");
        out.push_back({"xcp_busy", np && *np,
                       np && *np ? "" : "mock: MI355X_MOCK_PARTITIONS unset"});
        return out;
    }

  private:
    double busy_for(int dev)
    {
        if (!busy_file_.empty()) {
            std::ifstream f(busy_file_);
            if (f) {
                std::string line;
                double fallback = -1;
                while (std::getline(f, line)) {
                    if (line.empty()) continue;
                    auto colon = line.find(':');
                    if (colon == std::string::npos) {
                        if (fallback < 0) fallback = std::atof(line.c_str());
                    } else if (std::atoi(line.substr(0, colon).c_str()) == dev) {
                        return std::atof(line.c_str() + colon + 1);
                    }
                }
                if (fallback >= 0) return fallback;
            }
        }
        const char* env = std::getenv("MI355X_MOCK_BUSY");
        if (env && *env) return std::atof(env);
        // deterministic square wave, period 20 s
        double phase = std::fmod((now_ms() - t0_) / 1000.0, 20.0);
        return (phase < 10.0 ? 1.0 : 0.0) * 50.0 + dev * 5.0 + 5.0;
    }

    int n_;
    std::string busy_file_;
    double t0_;
};

} // namespace

std::unique_ptr<Backend> make_mock_backend(int n_devices, const std::string& busy_file)
{
    return std::make_unique<MockBackend>(n_devices, busy_file);
}

} // namespace mi355x
