// rsmi_backend.cpp — gfx950 counter sampling via rocm_smi_lib.
//
// Replaces the reference's DCGM+NVML layer (host libs mounted
// at dcgm-exporter.yaml:53-62, SYS_ADMIN at :42-48). No DCGM, no NVML: we
// talk to the amdgpu KFD interfaces through librocm_smi64, which needs only
// /dev/kfd + /dev/dri access — no privileged container.
//
// librocm_smi64 is resolved with dlopen so the exporter binary starts (and
// can serve its mock backend) on boxes without ROCm installed; on a GPU node
// it must resolve, and every sampling failure is surfaced per-metric rather
// than crashing the daemon.

#include "backend.h"
#include "gpu_metrics_parse.h"

#include <dlfcn.h>
#include <rocm_smi/rocm_smi.h>

#include <chrono>
#include <cstdio>
#include <cstring>

namespace mi355x {

namespace {

double now_ms()
{
    return std::chrono::duration<double, std::milli>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

struct RsmiApi {
    void* handle = nullptr;
#define DECL(fn) decltype(&::fn) fn = nullptr
    DECL(rsmi_init);
    DECL(rsmi_shut_down);
    DECL(rsmi_num_monitor_devices);
    DECL(rsmi_dev_name_get);
    DECL(rsmi_dev_unique_id_get);
    DECL(rsmi_dev_pci_id_get);
    DECL(rsmi_dev_drm_render_minor_get);
    DECL(rsmi_dev_busy_percent_get);
    DECL(rsmi_dev_memory_busy_percent_get);
    DECL(rsmi_dev_memory_usage_get);
    DECL(rsmi_dev_memory_total_get);
    DECL(rsmi_dev_temp_metric_get);
    DECL(rsmi_dev_power_ave_get);
    DECL(rsmi_dev_energy_count_get);
    DECL(rsmi_dev_gpu_clk_freq_get);
    DECL(rsmi_dev_gpu_metrics_info_get);
    DECL(rsmi_dev_ecc_count_get);
    // partition APIs (optional: absent in very old librocm_smi64 builds;
    // resolved lazily, null => per-metric degradation)
    DECL(rsmi_dev_compute_partition_get);
    DECL(rsmi_dev_memory_partition_get);
    DECL(rsmi_dev_partition_id_get);
#undef DECL

    bool load(std::string* err)
    {
        const char* names[] = {"librocm_smi64.so.1", "librocm_smi64.so",
                               "/opt/rocm/lib/librocm_smi64.so"};
        for (const char* n : names) {
            handle = dlopen(n, RTLD_NOW | RTLD_GLOBAL);
            if (handle) break;
        }
        if (!handle) {
            if (err) *err = std::string("dlopen librocm_smi64 failed: ") + dlerror();
            return false;
        }
#define RESOLVE(fn)                                                     \
    fn = reinterpret_cast<decltype(&::fn)>(dlsym(handle, #fn));         \
    if (!fn) {                                                          \
        if (err) *err = "dlsym " #fn " failed";                         \
        return false;                                                   \
    }
        RESOLVE(rsmi_init)
        RESOLVE(rsmi_shut_down)
        RESOLVE(rsmi_num_monitor_devices)
        RESOLVE(rsmi_dev_name_get)
        RESOLVE(rsmi_dev_unique_id_get)
        RESOLVE(rsmi_dev_pci_id_get)
        RESOLVE(rsmi_dev_drm_render_minor_get)
        RESOLVE(rsmi_dev_busy_percent_get)
        RESOLVE(rsmi_dev_memory_busy_percent_get)
        RESOLVE(rsmi_dev_memory_usage_get)
        RESOLVE(rsmi_dev_memory_total_get)
        RESOLVE(rsmi_dev_temp_metric_get)
        RESOLVE(rsmi_dev_power_ave_get)
        RESOLVE(rsmi_dev_energy_count_get)
        RESOLVE(rsmi_dev_gpu_clk_freq_get)
        RESOLVE(rsmi_dev_gpu_metrics_info_get)
        RESOLVE(rsmi_dev_ecc_count_get)
#undef RESOLVE
        // optional symbols: no failure when missing
        rsmi_dev_compute_partition_get =
            reinterpret_cast<decltype(&::rsmi_dev_compute_partition_get)>(
                dlsym(handle, "rsmi_dev_compute_partition_get"));
        rsmi_dev_memory_partition_get =
            reinterpret_cast<decltype(&::rsmi_dev_memory_partition_get)>(
                dlsym(handle, "rsmi_dev_memory_partition_get"));
        rsmi_dev_partition_id_get =
            reinterpret_cast<decltype(&::rsmi_dev_partition_id_get)>(
                dlsym(handle, "rsmi_dev_partition_id_get"));
        return true;
    }
};

class RsmiBackend final : public Backend {
  public:
    RsmiBackend(RsmiApi api, uint32_t n) : api_(api), n_(n) {}
    ~RsmiBackend() override { api_.rsmi_shut_down(); }

    int device_count() override { return (int)n_; }
    std::string name() const override { return "rocm_smi"; }

    GpuInfo info(int dev) override
    {
        GpuInfo gi;
        gi.index = dev;
        char buf[256] = {0};
        if (api_.rsmi_dev_name_get(dev, buf, sizeof(buf)) == RSMI_STATUS_SUCCESS)
            gi.name = buf;
        uint64_t uid = 0;
        if (api_.rsmi_dev_unique_id_get(dev, &uid) == RSMI_STATUS_SUCCESS) {
            char h[32];
            std::snprintf(h, sizeof(h), "%016lx", (unsigned long)uid);
            gi.uuid = h;
        }
        uint64_t bdf = 0;
        if (api_.rsmi_dev_pci_id_get(dev, &bdf) == RSMI_STATUS_SUCCESS) {
            // bdfid: [63:32] domain, [15:8] bus, [7:3] device, [2:0] function
            char b[32];
            std::snprintf(b, sizeof(b), "%04lx:%02x:%02x.%x",
                          (unsigned long)(bdf >> 32) & 0xffffffff,
                          (unsigned)(bdf >> 8) & 0xff, (unsigned)(bdf >> 3) & 0x1f,
                          (unsigned)bdf & 0x7);
            gi.pci_bdf = b;
        }
        uint32_t minor = 0;
        if (api_.rsmi_dev_drm_render_minor_get(dev, &minor) == RSMI_STATUS_SUCCESS) {
            gi.drm_render = "renderD" + std::to_string(minor);
        }
        char pbuf[32] = {0};
        if (api_.rsmi_dev_compute_partition_get &&
            api_.rsmi_dev_compute_partition_get(dev, pbuf, sizeof(pbuf)) ==
                RSMI_STATUS_SUCCESS)
            gi.compute_partition = pbuf;
        std::memset(pbuf, 0, sizeof(pbuf));
        if (api_.rsmi_dev_memory_partition_get &&
            api_.rsmi_dev_memory_partition_get(dev, pbuf, sizeof(pbuf)) ==
                RSMI_STATUS_SUCCESS)
            gi.memory_partition = pbuf;
        uint32_t pid = 0;
        if (api_.rsmi_dev_partition_id_get &&
            api_.rsmi_dev_partition_id_get(dev, &pid) == RSMI_STATUS_SUCCESS &&
            pid != 0xffffffffu)
            gi.partition_id = (int)pid;
        return gi;
    }

    std::vector<CounterProbe> probes() override
    {
        std::vector<CounterProbe> out;
        // MFMA / matrix-pipe activity (BASELINE.json north star names it):
        // probed against the full gpu_metrics layout — no MFMA/matrix-pipe
        // field exists through gpu_metrics v1.8 (ROCm 7.2 headers); the
        // only source is rocprofiler-sdk PMC sampling, which perturbs
        // co-running workloads and conflicts with attached profilers
        // (docs/METRICS.md), so the counter degrades to "unavailable"
        // rather than shipping a perturbing default-on sampler.
        out.push_back({"mfma_activity", false,
                       "no MFMA/matrix-pipe field in gpu_metrics v1.8; "
                       "requires rocprofiler-sdk PMC (perturbs workloads)"});
        rsmi_gpu_metrics_t gm;
        std::memset(&gm, 0, sizeof(gm));
        bool gm_ok = n_ > 0 && api_.rsmi_dev_gpu_metrics_info_get(0, &gm) ==
                                   RSMI_STATUS_SUCCESS;
        if (!gm_ok) {
            out.push_back({"xcp_busy", false, "gpu_metrics read failed"});
        } else if (gm.num_partition == 0xffff) {
            out.push_back({"xcp_busy", false,
                           "gpu_metrics reports no partition count "
                           "(pre-v1.6 firmware)"});
        } else {
            out.push_back({"xcp_busy", true, ""});
        }
        if (!api_.rsmi_dev_compute_partition_get)
            out.push_back({"compute_partition", false,
                           "librocm_smi64 lacks "
                           "rsmi_dev_compute_partition_get"});
        else
            out.push_back({"compute_partition", true, ""});
        return out;
    }

    GpuSample sample(int dev) override
    {
        GpuSample s;
        s.ts_ms = now_ms();

        uint32_t pct = 0;
        if (api_.rsmi_dev_busy_percent_get(dev, &pct) == RSMI_STATUS_SUCCESS) {
            s.busy_pct = pct;
            s.ok = true;
        }
        if (api_.rsmi_dev_memory_busy_percent_get(dev, &pct) == RSMI_STATUS_SUCCESS)
            s.mem_busy_pct = pct;

        uint64_t v = 0;
        if (api_.rsmi_dev_memory_usage_get(dev, RSMI_MEM_TYPE_VRAM, &v) ==
            RSMI_STATUS_SUCCESS)
            s.vram_used_bytes = (double)v;
        if (api_.rsmi_dev_memory_total_get(dev, RSMI_MEM_TYPE_VRAM, &v) ==
            RSMI_STATUS_SUCCESS)
            s.vram_total_bytes = (double)v;

        int64_t t = 0;
        if (api_.rsmi_dev_temp_metric_get(dev, RSMI_TEMP_TYPE_EDGE, RSMI_TEMP_CURRENT,
                                          &t) == RSMI_STATUS_SUCCESS)
            s.temp_edge_c = t / 1000.0;
        if (api_.rsmi_dev_temp_metric_get(dev, RSMI_TEMP_TYPE_JUNCTION,
                                          RSMI_TEMP_CURRENT, &t) == RSMI_STATUS_SUCCESS)
            s.temp_hotspot_c = t / 1000.0;
        if (api_.rsmi_dev_temp_metric_get(dev, RSMI_TEMP_TYPE_MEMORY, RSMI_TEMP_CURRENT,
                                          &t) == RSMI_STATUS_SUCCESS)
            s.temp_mem_c = t / 1000.0;

        uint64_t uw = 0;
        if (api_.rsmi_dev_power_ave_get(dev, 0, &uw) == RSMI_STATUS_SUCCESS)
            s.power_w = uw / 1e6;

        uint64_t energy = 0, ets = 0;
        float res = 0;
        if (api_.rsmi_dev_energy_count_get(dev, &energy, &res, &ets) ==
            RSMI_STATUS_SUCCESS)
            s.energy_j = energy * (double)res / 1e6; // counter*resolution = uJ

        rsmi_frequencies_t f;
        std::memset(&f, 0, sizeof(f));
        if (api_.rsmi_dev_gpu_clk_freq_get(dev, RSMI_CLK_TYPE_SYS, &f) ==
                RSMI_STATUS_SUCCESS &&
            f.current < f.num_supported)
            s.sclk_mhz = f.frequency[f.current] / 1e6;
        std::memset(&f, 0, sizeof(f));
        if (api_.rsmi_dev_gpu_clk_freq_get(dev, RSMI_CLK_TYPE_MEM, &f) ==
                RSMI_STATUS_SUCCESS &&
            f.current < f.num_supported)
            s.mclk_mhz = f.frequency[f.current] / 1e6;

        // RAS ECC totals over the blocks that exist on MI3xx
        {
            static const rsmi_gpu_block_t blocks[] = {
                RSMI_GPU_BLOCK_UMC, RSMI_GPU_BLOCK_SDMA, RSMI_GPU_BLOCK_GFX,
                RSMI_GPU_BLOCK_MMHUB, RSMI_GPU_BLOCK_XGMI_WAFL,
                RSMI_GPU_BLOCK_PCIE_BIF};
            double ce = 0, ue = 0;
            bool any = false;
            for (auto b : blocks) {
                rsmi_error_count_t ec{};
                if (api_.rsmi_dev_ecc_count_get(dev, b, &ec) ==
                    RSMI_STATUS_SUCCESS) {
                    ce += (double)ec.correctable_err;
                    ue += (double)ec.uncorrectable_err;
                    any = true;
                }
            }
            if (any) {
                s.ecc_correctable = ce;
                s.ecc_uncorrectable = ue;
            }
        }

        rsmi_gpu_metrics_t gm;
        std::memset(&gm, 0, sizeof(gm));
        if (api_.rsmi_dev_gpu_metrics_info_get(dev, &gm) == RSMI_STATUS_SUCCESS)
            parse_gpu_metrics(gm, s);

        // dcgm_gpu_temp must exist (README.md:46 probe): some MI3xx SKUs
        // expose no edge sensor — fall back to junction/hotspot.
        if (s.temp_edge_c < 0 && s.temp_hotspot_c >= 0)
            s.temp_edge_c = s.temp_hotspot_c;
        return s;
    }

  private:
    RsmiApi api_;
    uint32_t n_;
};

} // namespace

std::unique_ptr<Backend> make_rsmi_backend(std::string* err)
{
    RsmiApi api;
    if (!api.load(err)) return nullptr;
    rsmi_status_t st = api.rsmi_init(0);
    if (st != RSMI_STATUS_SUCCESS) {
        if (err) *err = "rsmi_init failed (status " + std::to_string((int)st) + ")";
        return nullptr;
    }
    uint32_t n = 0;
    if (api.rsmi_num_monitor_devices(&n) != RSMI_STATUS_SUCCESS || n == 0) {
        api.rsmi_shut_down();
        if (err) *err = "no AMD GPU devices visible to rocm_smi";
        return nullptr;
    }
    return std::make_unique<RsmiBackend>(api, n);
}

} // namespace mi355x
