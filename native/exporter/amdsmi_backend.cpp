// amdsmi_backend.cpp — gfx950 counter sampling via libamd_smi.
//
// Second native backend behind backend.h (round-1 verdict item 6):
// rocm_smi_lib is in maintenance mode upstream, amd-smi is its successor —
// the exporter auto-selects amd-smi when the library resolves and falls
// back to rocm_smi (`--backend` overrides). Same design rules as
// rsmi_backend.cpp: dlopen (the daemon must start on ROCm-less boxes for
// the mock path), per-metric degradation (a failing getter skips that
// metric, never the sample), and the shared gpu_metrics field mapping
// (gpu_metrics_parse.h) so both backends emit identical semantics.
//
// Replaces the reference's DCGM+NVML layer (SURVEY.md C7/C8,
// dcgm-exporter.yaml:53-62) the same way the rsmi backend does.

#include "backend.h"
#include "gpu_metrics_parse.h"

#include <amd_smi/amdsmi.h>
#include <dlfcn.h>

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

struct AmdSmiApi {
    void* handle = nullptr;
#define DECL(fn) decltype(&::fn) fn = nullptr
    DECL(amdsmi_init);
    DECL(amdsmi_shut_down);
    DECL(amdsmi_get_socket_handles);
    DECL(amdsmi_get_processor_handles);
    DECL(amdsmi_get_gpu_asic_info);
    DECL(amdsmi_get_gpu_device_uuid);
    DECL(amdsmi_get_gpu_device_bdf);
    DECL(amdsmi_get_gpu_enumeration_info);
    DECL(amdsmi_get_gpu_activity);
    DECL(amdsmi_get_gpu_memory_total);
    DECL(amdsmi_get_gpu_memory_usage);
    DECL(amdsmi_get_temp_metric);
    DECL(amdsmi_get_power_info);
    DECL(amdsmi_get_energy_count);
    DECL(amdsmi_get_clock_info);
    DECL(amdsmi_get_gpu_metrics_info);
    DECL(amdsmi_get_gpu_total_ecc_count);
    // optional
    DECL(amdsmi_get_gpu_compute_partition);
    DECL(amdsmi_get_gpu_memory_partition);
#undef DECL

    bool load(std::string* err)
    {
        const char* names[] = {"libamd_smi.so.25", "libamd_smi.so",
                               "/opt/rocm/lib/libamd_smi.so"};
        for (const char* n : names) {
            handle = dlopen(n, RTLD_NOW | RTLD_LOCAL);
            if (handle) break;
        }
        if (!handle) {
            if (err) *err = std::string("dlopen libamd_smi failed: ") + dlerror();
            return false;
        }
#define RESOLVE(fn)                                                     \
    fn = reinterpret_cast<decltype(&::fn)>(dlsym(handle, #fn));         \
    if (!fn) {                                                          \
        if (err) *err = "dlsym " #fn " failed";                         \
        return false;                                                   \
    }
        RESOLVE(amdsmi_init)
        RESOLVE(amdsmi_shut_down)
        RESOLVE(amdsmi_get_socket_handles)
        RESOLVE(amdsmi_get_processor_handles)
        RESOLVE(amdsmi_get_gpu_asic_info)
        RESOLVE(amdsmi_get_gpu_device_uuid)
        RESOLVE(amdsmi_get_gpu_device_bdf)
        RESOLVE(amdsmi_get_gpu_activity)
        RESOLVE(amdsmi_get_gpu_memory_total)
        RESOLVE(amdsmi_get_gpu_memory_usage)
        RESOLVE(amdsmi_get_temp_metric)
        RESOLVE(amdsmi_get_power_info)
        RESOLVE(amdsmi_get_energy_count)
        RESOLVE(amdsmi_get_clock_info)
        RESOLVE(amdsmi_get_gpu_metrics_info)
        RESOLVE(amdsmi_get_gpu_total_ecc_count)
#undef RESOLVE
        // optional symbols (per-metric degradation when absent)
        amdsmi_get_gpu_enumeration_info =
            reinterpret_cast<decltype(&::amdsmi_get_gpu_enumeration_info)>(
                dlsym(handle, "amdsmi_get_gpu_enumeration_info"));
        amdsmi_get_gpu_compute_partition =
            reinterpret_cast<decltype(&::amdsmi_get_gpu_compute_partition)>(
                dlsym(handle, "amdsmi_get_gpu_compute_partition"));
        amdsmi_get_gpu_memory_partition =
            reinterpret_cast<decltype(&::amdsmi_get_gpu_memory_partition)>(
                dlsym(handle, "amdsmi_get_gpu_memory_partition"));
        return true;
    }
};

class AmdSmiBackend final : public Backend {
  public:
    AmdSmiBackend(AmdSmiApi api, std::vector<amdsmi_processor_handle> procs)
        : api_(api), procs_(std::move(procs))
    {
    }
    ~AmdSmiBackend() override { api_.amdsmi_shut_down(); }

    int device_count() override { return (int)procs_.size(); }
    std::string name() const override { return "amd_smi"; }

    GpuInfo info(int dev) override
    {
        GpuInfo gi;
        gi.index = dev;
        auto h = procs_[dev];
        amdsmi_asic_info_t ai;
        std::memset(&ai, 0, sizeof(ai));
        if (api_.amdsmi_get_gpu_asic_info(h, &ai) == AMDSMI_STATUS_SUCCESS) {
            gi.name = ai.market_name;
            if (ai.asic_serial[0]) gi.uuid = ai.asic_serial;
        }
        unsigned int ulen = 0;
        char ubuf[128] = {0};
        ulen = sizeof(ubuf);
        if (api_.amdsmi_get_gpu_device_uuid(h, &ulen, ubuf) ==
                AMDSMI_STATUS_SUCCESS &&
            ubuf[0])
            gi.uuid = ubuf;
        amdsmi_bdf_t bdf;
        std::memset(&bdf, 0, sizeof(bdf));
        if (api_.amdsmi_get_gpu_device_bdf(h, &bdf) == AMDSMI_STATUS_SUCCESS) {
            char b[32];
            std::snprintf(b, sizeof(b), "%04lx:%02x:%02x.%x",
                          (unsigned long)bdf.domain_number,
                          (unsigned)bdf.bus_number,
                          (unsigned)bdf.device_number,
                          (unsigned)bdf.function_number);
            gi.pci_bdf = b;
        }
        if (api_.amdsmi_get_gpu_enumeration_info) {
            amdsmi_enumeration_info_t en;
            std::memset(&en, 0, sizeof(en));
            if (api_.amdsmi_get_gpu_enumeration_info(h, &en) ==
                AMDSMI_STATUS_SUCCESS)
                gi.drm_render = "renderD" + std::to_string(en.drm_render);
        }
        char pbuf[32] = {0};
        if (api_.amdsmi_get_gpu_compute_partition &&
            api_.amdsmi_get_gpu_compute_partition(h, pbuf, sizeof(pbuf)) ==
                AMDSMI_STATUS_SUCCESS)
            gi.compute_partition = pbuf;
        std::memset(pbuf, 0, sizeof(pbuf));
        if (api_.amdsmi_get_gpu_memory_partition &&
            api_.amdsmi_get_gpu_memory_partition(h, pbuf, sizeof(pbuf)) ==
                AMDSMI_STATUS_SUCCESS)
            gi.memory_partition = pbuf;
        amdsmi_kfd_info_t kfd;  // partition id lives in kfd info
        std::memset(&kfd, 0, sizeof(kfd));
        if (auto f = reinterpret_cast<amdsmi_status_t (*)(
                amdsmi_processor_handle, amdsmi_kfd_info_t*)>(
                dlsym(api_.handle, "amdsmi_get_gpu_kfd_info"))) {
            if (f(procs_[dev], &kfd) == AMDSMI_STATUS_SUCCESS &&
                kfd.current_partition_id != 0xffffffffu)
                gi.partition_id = (int)kfd.current_partition_id;
        }
        return gi;
    }

    GpuSample sample(int dev) override
    {
        GpuSample s;
        s.ts_ms = now_ms();
        auto h = procs_[dev];

        amdsmi_engine_usage_t eu;
        std::memset(&eu, 0, sizeof(eu));
        if (api_.amdsmi_get_gpu_activity(h, &eu) == AMDSMI_STATUS_SUCCESS) {
            if (eu.gfx_activity != 0xffffffffu) {
                s.busy_pct = eu.gfx_activity;
                s.ok = true;
            }
            if (eu.umc_activity != 0xffffffffu) s.mem_busy_pct = eu.umc_activity;
        }

        uint64_t v = 0;
        if (api_.amdsmi_get_gpu_memory_usage(h, AMDSMI_MEM_TYPE_VRAM, &v) ==
            AMDSMI_STATUS_SUCCESS)
            s.vram_used_bytes = (double)v;
        if (api_.amdsmi_get_gpu_memory_total(h, AMDSMI_MEM_TYPE_VRAM, &v) ==
            AMDSMI_STATUS_SUCCESS)
            s.vram_total_bytes = (double)v;

        int64_t t = 0;
        if (api_.amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_EDGE,
                                        AMDSMI_TEMP_CURRENT, &t) ==
            AMDSMI_STATUS_SUCCESS)
            s.temp_edge_c = (double)t;
        if (api_.amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_JUNCTION,
                                        AMDSMI_TEMP_CURRENT, &t) ==
            AMDSMI_STATUS_SUCCESS)
            s.temp_hotspot_c = (double)t;
        if (api_.amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_VRAM,
                                        AMDSMI_TEMP_CURRENT, &t) ==
            AMDSMI_STATUS_SUCCESS)
            s.temp_mem_c = (double)t;

        amdsmi_power_info_t pi;
        std::memset(&pi, 0, sizeof(pi));
        if (api_.amdsmi_get_power_info(h, &pi) == AMDSMI_STATUS_SUCCESS) {
            if (pi.socket_power && pi.socket_power != ~0ull)
                s.power_w = (double)pi.socket_power;
            else if (pi.current_socket_power != 0xffffffffu)
                s.power_w = pi.current_socket_power;
        }

        uint64_t energy = 0, ets = 0;
        float res = 0;
        if (api_.amdsmi_get_energy_count(h, &energy, &res, &ets) ==
            AMDSMI_STATUS_SUCCESS)
            s.energy_j = energy * (double)res / 1e6;

        amdsmi_clk_info_t ci;
        std::memset(&ci, 0, sizeof(ci));
        if (api_.amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_GFX, &ci) ==
            AMDSMI_STATUS_SUCCESS)
            s.sclk_mhz = ci.clk;
        std::memset(&ci, 0, sizeof(ci));
        if (api_.amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_MEM, &ci) ==
            AMDSMI_STATUS_SUCCESS)
            s.mclk_mhz = ci.clk;

        amdsmi_error_count_t ec;
        std::memset(&ec, 0, sizeof(ec));
        if (api_.amdsmi_get_gpu_total_ecc_count(h, &ec) == AMDSMI_STATUS_SUCCESS) {
            s.ecc_correctable = (double)ec.correctable_count;
            s.ecc_uncorrectable = (double)ec.uncorrectable_count;
        }

        amdsmi_gpu_metrics_t gm;
        std::memset(&gm, 0, sizeof(gm));
        if (api_.amdsmi_get_gpu_metrics_info(h, &gm) == AMDSMI_STATUS_SUCCESS)
            parse_gpu_metrics(gm, s);

        // dcgm_gpu_temp must exist (reference README.md:46 probe): fall
        // back to junction when the SKU has no edge sensor
        if (s.temp_edge_c < 0 && s.temp_hotspot_c >= 0)
            s.temp_edge_c = s.temp_hotspot_c;
        return s;
    }

    std::vector<CounterProbe> probes() override
    {
        std::vector<CounterProbe> out;
        out.push_back({"mfma_activity", false,
                       "no MFMA/matrix-pipe field in gpu_metrics v1.8; "
                       "requires rocprofiler-sdk PMC (perturbs workloads)"});
        amdsmi_gpu_metrics_t gm;
        std::memset(&gm, 0, sizeof(gm));
        bool gm_ok = !procs_.empty() &&
                     api_.amdsmi_get_gpu_metrics_info(procs_[0], &gm) ==
                         AMDSMI_STATUS_SUCCESS;
        if (!gm_ok)
            out.push_back({"xcp_busy", false, "gpu_metrics read failed"});
        else if (gm.num_partition == 0xffff)
            out.push_back({"xcp_busy", false,
                           "gpu_metrics reports no partition count "
                           "(pre-v1.6 firmware)"});
        else
            out.push_back({"xcp_busy", true, ""});
        out.push_back({"compute_partition",
                       api_.amdsmi_get_gpu_compute_partition != nullptr,
                       api_.amdsmi_get_gpu_compute_partition
                           ? ""
                           : "libamd_smi lacks amdsmi_get_gpu_compute_partition"});
        return out;
    }

  private:
    AmdSmiApi api_;
    std::vector<amdsmi_processor_handle> procs_;
};

} // namespace

std::unique_ptr<Backend> make_amdsmi_backend(std::string* err)
{
    AmdSmiApi api;
    if (!api.load(err)) return nullptr;
    amdsmi_status_t st = api.amdsmi_init(AMDSMI_INIT_AMD_GPUS);
    if (st != AMDSMI_STATUS_SUCCESS) {
        if (err) *err = "amdsmi_init failed (status " + std::to_string((int)st) + ")";
        return nullptr;
    }
    uint32_t n_sockets = 0;
    if (api.amdsmi_get_socket_handles(&n_sockets, nullptr) !=
            AMDSMI_STATUS_SUCCESS ||
        n_sockets == 0) {
        api.amdsmi_shut_down();
        if (err) *err = "no AMD GPU sockets visible to amd_smi";
        return nullptr;
    }
    std::vector<amdsmi_socket_handle> sockets(n_sockets);
    if (api.amdsmi_get_socket_handles(&n_sockets, sockets.data()) !=
        AMDSMI_STATUS_SUCCESS) {
        api.amdsmi_shut_down();
        if (err) *err = "amdsmi_get_socket_handles failed";
        return nullptr;
    }
    std::vector<amdsmi_processor_handle> procs;
    for (auto sock : sockets) {
        uint32_t np = 0;
        if (api.amdsmi_get_processor_handles(sock, &np, nullptr) !=
                AMDSMI_STATUS_SUCCESS ||
            np == 0)
            continue;
        std::vector<amdsmi_processor_handle> ph(np);
        if (api.amdsmi_get_processor_handles(sock, &np, ph.data()) !=
            AMDSMI_STATUS_SUCCESS)
            continue;
        for (auto p : ph) procs.push_back(p);
    }
    if (procs.empty()) {
        api.amdsmi_shut_down();
        if (err) *err = "no AMD GPU processors visible to amd_smi";
        return nullptr;
    }
    return std::make_unique<AmdSmiBackend>(api, std::move(procs));
}

} // namespace mi355x
