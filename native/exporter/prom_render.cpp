#include "prom_render.h"

#include <cstdio>
#include <sstream>

namespace mi355x {

namespace {

std::string esc(const std::string& s)
{
    std::string out;
    for (char c : s) {
        if (c == '\\' || c == '"') out += '\\';
        if (c == '\n') {
            out += "\\n";
            continue;
        }
        out += c;
    }
    return out;
}

class Writer {
  public:
    Writer(const RenderOptions& opt) : opt_(opt) {}

    bool enabled(const std::string& family) const
    {
        return opt_.metric_set.empty() || opt_.metric_set.count(family) > 0;
    }

    void family(const std::string& name, const std::string& help,
                const std::string& type)
    {
        cur_family_ = name;
        cur_emitted_ = false;
        cur_help_ = "# HELP " + name + " " + help + "\n# TYPE " + name + " " + type +
                    "\n";
    }

    void sample(const DeviceMetrics& d, const AttributionMap& attr, double value,
                const std::string& extra_labels = "")
    {
        if (!enabled(cur_family_) || value < 0) return;
        if (!cur_emitted_) {
            out_ << cur_help_;
            cur_emitted_ = true;
        }
        out_ << cur_family_ << "{gpu=\"" << d.info.index << "\",uuid=\""
             << esc(d.info.uuid) << "\",device=\"" << esc(d.info.drm_render)
             << "\",modelName=\"" << esc(d.info.name) << "\"";
        if (!opt_.hostname.empty())
            out_ << ",Hostname=\"" << esc(opt_.hostname) << "\"";
        if (opt_.kubernetes) {
            auto it = attr.find(attribution_key(d.info, opt_.gpu_id_type));
            if (it != attr.end()) {
                out_ << ",container=\"" << esc(it->second.container) << "\",namespace=\""
                     << esc(it->second.ns) << "\",pod=\"" << esc(it->second.pod)
                     << "\"";
            }
        }
        if (!extra_labels.empty()) out_ << "," << extra_labels;
        char buf[64];
        std::snprintf(buf, sizeof(buf), "%.6g", value);
        out_ << "} " << buf << "\n";
    }

    // family-filtered series with caller-provided labels only (no device)
    void raw(const std::string& labels, double value)
    {
        if (!enabled(cur_family_)) return;
        if (!cur_emitted_) {
            out_ << cur_help_;
            cur_emitted_ = true;
        }
        char buf[64];
        std::snprintf(buf, sizeof(buf), "%.6g", value);
        out_ << cur_family_ << "{" << labels << "} " << buf << "\n";
    }

    std::string str() const { return out_.str(); }

  private:
    const RenderOptions& opt_;
    std::ostringstream out_;
    std::string cur_family_;
    std::string cur_help_;
    bool cur_emitted_ = false;
};

} // namespace

std::string attribution_key(const GpuInfo& info, const std::string& id_type)
{
    if (id_type == "uuid") return info.uuid;
    if (id_type == "index") return std::to_string(info.index);
    return info.drm_render; // "device-name"
}

std::string render_metrics(const std::vector<DeviceMetrics>& devs,
                           const AttributionMap& attr, const RenderOptions& opt)
{
    Writer w(opt);

    // --- dcgm 1.x-compat families (the drop-in contract) ---
    w.family("dcgm_gpu_utilization", "GPU utilization (%).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.busy_pct);

    w.family("dcgm_mem_copy_utilization", "Memory utilization (%).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.mem_busy_pct);

    w.family("dcgm_gpu_temp", "GPU temperature (in C).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.temp_edge_c);

    w.family("dcgm_memory_temp", "Memory temperature (in C).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.temp_mem_c);

    w.family("dcgm_power_usage", "Power draw (in W).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.power_w);

    w.family("dcgm_total_energy_consumption",
             "Total energy consumption since boot (in mJ).", "counter");
    for (auto& d : devs)
        w.sample(d, attr, d.sample.energy_j < 0 ? -1 : d.sample.energy_j * 1e3);

    w.family("dcgm_sm_clock", "SM clock frequency (in MHz).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.sclk_mhz);

    w.family("dcgm_memory_clock", "Memory clock frequency (in MHz).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.mclk_mhz);

    w.family("dcgm_fb_used", "Framebuffer memory used (in MiB).", "gauge");
    for (auto& d : devs)
        w.sample(d, attr,
                 d.sample.vram_used_bytes < 0 ? -1
                                              : d.sample.vram_used_bytes / (1 << 20));

    w.family("dcgm_fb_free", "Framebuffer memory free (in MiB).", "gauge");
    for (auto& d : devs)
        w.sample(d, attr,
                 (d.sample.vram_total_bytes < 0 || d.sample.vram_used_bytes < 0)
                     ? -1
                     : (d.sample.vram_total_bytes - d.sample.vram_used_bytes) /
                           (1 << 20));

    w.family("dcgm_pcie_tx_throughput", "PCIe TX throughput (in KB/s).", "gauge");
    for (auto& d : devs)
        w.sample(d, attr, d.sample.pcie_tx_bps < 0 ? -1 : d.sample.pcie_tx_bps / 1e3);

    w.family("dcgm_pcie_rx_throughput", "PCIe RX throughput (in KB/s).", "gauge");
    for (auto& d : devs)
        w.sample(d, attr, d.sample.pcie_rx_bps < 0 ? -1 : d.sample.pcie_rx_bps / 1e3);

    w.family("dcgm_ecc_sbe_aggregate_total",
             "Accumulated correctable (single-bit) ECC errors, all blocks.",
             "counter");
    for (auto& d : devs) w.sample(d, attr, d.sample.ecc_correctable);

    w.family("dcgm_ecc_dbe_aggregate_total",
             "Accumulated uncorrectable (double-bit) ECC errors, all blocks.",
             "counter");
    for (auto& d : devs) w.sample(d, attr, d.sample.ecc_uncorrectable);

    w.family("dcgm_pcie_replay_counter", "Accumulated PCIe replay events.",
             "counter");
    for (auto& d : devs) w.sample(d, attr, d.sample.pcie_replay_count);

    w.family("dcgm_power_violation",
             "Power-throttle (PPT) residency %% over the sampling window.",
             "gauge");
    for (auto& d : devs) w.sample(d, attr, d.pviol_pct);

    w.family("dcgm_thermal_violation",
             "Thermal-throttle residency %% over the sampling window.", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.tviol_pct);

    // --- MI355X-native amd_* families (no reference counterpart) ---
    w.family("amd_gpu_busy_percent_windowed",
             "GPU busy % derived from the gfx activity accumulator over the "
             "sampling window (finer than the instantaneous busy%).",
             "gauge");
    for (auto& d : devs) w.sample(d, attr, d.busy_windowed_pct);

    w.family("amd_hbm_bandwidth_utilization",
             "HBM3E memory-controller (UMC) activity (%) — bandwidth proxy.",
             "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.umc_activity_pct);

    w.family("amd_hbm_busy_percent_windowed",
             "Memory-controller busy % from the activity accumulator over the "
             "sampling window.",
             "gauge");
    for (auto& d : devs) w.sample(d, attr, d.mem_busy_windowed_pct);

    w.family("amd_gfx_activity_accumulated",
             "Monotonic gfx busy-time accumulator (firmware units).", "counter");
    for (auto& d : devs) w.sample(d, attr, d.sample.gfx_activity_acc);

    w.family("amd_xgmi_link_read_bytes_per_second",
             "Per-xGMI-link read throughput (bytes/s) over the sampling window.",
             "gauge");
    for (auto& d : devs)
        for (int l = 0; l < d.sample.xgmi_num_links; ++l)
            w.sample(d, attr, d.xgmi_read_bps[l], "link=\"" + std::to_string(l) + "\"");

    w.family("amd_xgmi_link_write_bytes_per_second",
             "Per-xGMI-link write throughput (bytes/s) over the sampling window.",
             "gauge");
    for (auto& d : devs)
        for (int l = 0; l < d.sample.xgmi_num_links; ++l)
            w.sample(d, attr, d.xgmi_write_bps[l],
                     "link=\"" + std::to_string(l) + "\"");

    w.family("amd_xgmi_total_bytes_per_second",
             "Total xGMI traffic (read+write, all links, bytes/s).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.xgmi_total_bps);

    w.family("amd_xgmi_link_utilization",
             "Per-xGMI-link utilization (%) of the link's peak (width x speed).",
             "gauge");
    for (auto& d : devs) {
        double peak_bps = -1;
        if (d.sample.xgmi_link_speed_gbps > 0)
            // xgmi_link_speed is reported in GB/s per link on MI3xx
            peak_bps = d.sample.xgmi_link_speed_gbps * 1e9;
        if (peak_bps <= 0) continue;
        for (int l = 0; l < d.sample.xgmi_num_links; ++l)
            w.sample(d, attr,
                     (d.xgmi_read_bps[l] + d.xgmi_write_bps[l]) / peak_bps * 100.0,
                     "link=\"" + std::to_string(l) + "\"");
    }

    w.family("amd_vram_total_bytes", "Total HBM3E capacity (bytes).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.vram_total_bytes);

    w.family("amd_gpu_hotspot_temp", "Junction/hotspot temperature (C).", "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample.temp_hotspot_c);

    w.family("amd_xcp_busy_percent",
             "Per-compute-partition (XCP) instantaneous gfx busy % - mean "
             "over the partition's XCCs (gpu_metrics xcp_stats).",
             "gauge");
    for (auto& d : devs)
        for (int p = 0; p < d.sample.num_partitions && p < kMaxXcp; ++p)
            w.sample(d, attr, d.sample.xcp_busy_pct[p],
                     "partition=\"" + std::to_string(p) + "\"");

    w.family("amd_compute_partition_info",
             "Compute/memory partition mode of this device (value is "
             "always 1; the information is in the labels).",
             "gauge");
    for (auto& d : devs) {
        if (d.info.compute_partition.empty() &&
            d.info.memory_partition.empty() && d.info.partition_id < 0)
            continue;
        std::string extra;
        if (!d.info.compute_partition.empty())
            extra += "compute=\"" + d.info.compute_partition + "\"";
        if (!d.info.memory_partition.empty()) {
            if (!extra.empty()) extra += ",";
            extra += "memory=\"" + d.info.memory_partition + "\"";
        }
        if (d.info.partition_id >= 0) {
            if (!extra.empty()) extra += ",";
            extra += "partition_id=\"" + std::to_string(d.info.partition_id) +
                     "\"";
        }
        w.sample(d, attr, 1.0, extra);
    }

    w.family("amd_exporter_sample_duration_ms",
             "Wall time of the last per-device sampling pass (ms) — the "
             "exporter's own overhead, for observer-effect monitoring.",
             "gauge");
    for (auto& d : devs) w.sample(d, attr, d.sample_pass_ms);

    w.family("amd_exporter_samples_total",
             "Sampling ticks taken for this device since exporter start "
             "(freshness counter: advances once per -c interval).",
             "counter");
    for (auto& d : devs) w.sample(d, attr, (double)d.samples_taken);

    // counter-availability meta-metrics (probe, don't
    // assume): one series per counter the backend cannot serve, with the
    // reason as a label — what "graceful degradation" looks like on the
    // wire. Emitted without device labels (the probe is backend-global).
    w.family("amd_counter_unavailable",
             "A probed counter family this exporter cannot serve on this "
             "node, with the reason (value always 1).",
             "gauge");
    for (auto& p : opt.probes)
        if (!p.available)
            w.raw("counter=\"" + esc(p.counter) + "\",reason=\"" +
                      esc(p.reason) + "\"",
                  1.0);

    return w.str();
}

} // namespace mi355x
