// gpu_metrics_parse.h — shared gpu_metrics -> GpuSample mapping.
//
// rocm_smi's rsmi_gpu_metrics_t and amd-smi's amdsmi_gpu_metrics_t are both
// copies of the kernel's amdgpu gpu_metrics v1.x layout with identical field
// names, so one template serves both backends (rsmi_backend.cpp and
// amdsmi_backend.cpp) — the per-field sentinel handling and the xcp_stats
// partition math live here exactly once.

#pragma once

#include "backend.h"

namespace mi355x {

// Merge one gpu_metrics snapshot into `s`. Fields already set (>= 0) by the
// backend's dedicated getters are kept; gpu_metrics fills the gaps. Sentinel
// convention per the kernel: 0xffff / 0xffffffff / ~0ull = unsupported.
template <typename GM>
inline void parse_gpu_metrics(const GM& gm, GpuSample& s)
{
    auto u16ok = [](uint16_t x) { return x != 0xffff; };
    if (u16ok(gm.average_gfx_activity)) {
        s.gfx_activity_pct = gm.average_gfx_activity;
        if (s.busy_pct < 0) s.busy_pct = gm.average_gfx_activity;
        s.ok = true;
    }
    if (u16ok(gm.average_umc_activity)) {
        s.umc_activity_pct = gm.average_umc_activity;
        if (s.mem_busy_pct < 0) s.mem_busy_pct = gm.average_umc_activity;
    }
    if (gm.gfx_activity_acc != 0xffffffffu) s.gfx_activity_acc = gm.gfx_activity_acc;
    if (gm.mem_activity_acc != 0xffffffffu) s.mem_activity_acc = gm.mem_activity_acc;
    if (u16ok(gm.current_socket_power) && s.power_w < 0)
        s.power_w = gm.current_socket_power;
    if (u16ok(gm.current_gfxclk) && s.sclk_mhz < 0) s.sclk_mhz = gm.current_gfxclk;
    if (u16ok(gm.current_uclk) && s.mclk_mhz < 0) s.mclk_mhz = gm.current_uclk;
    if (u16ok(gm.temperature_mem) && s.temp_mem_c < 0)
        s.temp_mem_c = gm.temperature_mem;
    if (u16ok(gm.temperature_edge) && gm.temperature_edge != 0 && s.temp_edge_c < 0)
        s.temp_edge_c = gm.temperature_edge;
    if (u16ok(gm.temperature_hotspot) && s.temp_hotspot_c < 0)
        s.temp_hotspot_c = gm.temperature_hotspot;

    // PCIe: instantaneous bandwidth when the firmware reports it
    if (gm.pcie_bandwidth_inst && gm.pcie_bandwidth_inst != ~0ull) {
        s.pcie_tx_bps = gm.pcie_bandwidth_inst * 1e9 / 2.0;
        s.pcie_rx_bps = gm.pcie_bandwidth_inst * 1e9 / 2.0;
    }

    if (gm.accumulation_counter && gm.accumulation_counter != ~0ull) {
        s.accumulation_counter = (double)gm.accumulation_counter;
        if (gm.ppt_residency_acc != ~0ull)
            s.ppt_residency_acc = (double)gm.ppt_residency_acc;
        if (gm.socket_thm_residency_acc != ~0ull)
            s.thm_residency_acc = (double)gm.socket_thm_residency_acc;
    }
    if (gm.pcie_replay_count_acc != ~0ull)
        s.pcie_replay_count = (double)gm.pcie_replay_count_acc;

    if (u16ok(gm.xgmi_link_width)) s.xgmi_link_width = gm.xgmi_link_width;
    if (u16ok(gm.xgmi_link_speed)) s.xgmi_link_speed_gbps = gm.xgmi_link_speed;
    int nl = 0;
    for (int i = 0; i < kMaxXgmiLinks; ++i) {
        uint64_t r = gm.xgmi_read_data_acc[i];
        uint64_t w = gm.xgmi_write_data_acc[i];
        if (r == ~0ull) r = 0;
        if (w == ~0ull) w = 0;
        s.xgmi_read_acc_kb[i] = (double)r;
        s.xgmi_write_acc_kb[i] = (double)w;
        if (gm.xgmi_link_status[i] == 1 || r || w) nl = i + 1;
    }
    s.xgmi_num_links = nl;

    // per-XCP busy: mean of the partition's valid XCC instantaneous busy
    // values (gpu_metrics v1.6+; num_partition 0xffff = unsupported)
    if (gm.num_partition != 0xffff && gm.num_partition > 0) {
        int np = gm.num_partition;
        if (np > kMaxXcp) np = kMaxXcp;
        for (int p = 0; p < np; ++p) {
            double sum = 0;
            int cnt = 0;
            for (unsigned x = 0;
                 x < sizeof(gm.xcp_stats[p].gfx_busy_inst) /
                         sizeof(gm.xcp_stats[p].gfx_busy_inst[0]);
                 ++x) {
                uint32_t v = gm.xcp_stats[p].gfx_busy_inst[x];
                if (v == 0xffffffffu) continue;
                sum += v;
                ++cnt;
            }
            if (cnt) {
                s.xcp_busy_pct[p] = sum / cnt;
                s.num_partitions = p + 1;
            }
        }
    }
}

} // namespace mi355x
