// backend.h — GPU counter-sampling backend interface for the MI355X exporter.
//
// The reference's exporter (dcgm-exporter.yaml:29) sits on DCGM+NVML
// (SURVEY.md C7/C8). This stack replaces that whole layer with a direct
// rocm_smi_lib backend over the amdgpu KFD interfaces (rsmi_backend.cpp) and
// a deterministic mock backend (mock_backend.cpp) so the full exporter path
// is testable with no GPU (BASELINE.json config 1's "stub /metrics").

#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

namespace mi355x {

constexpr int kMaxXgmiLinks = 8;

constexpr int kMaxXcp = 8;   // compute partitions per GPU (SPX..CPX)

struct GpuSample {
    bool ok = false;              // sample succeeded for this device
    double ts_ms = 0;             // host steady-clock timestamp

    // core utilization (dcgm_gpu_utilization contract)
    double busy_pct = -1;         // GRBM GUI-active derived busy %
    double mem_busy_pct = -1;     // memory-controller (UMC) busy % -> HBM BW proxy

    // memory
    double vram_used_bytes = -1;
    double vram_total_bytes = -1;

    // thermals / power / clocks
    double temp_edge_c = -1;
    double temp_mem_c = -1;
    double temp_hotspot_c = -1;
    double power_w = -1;
    double energy_j = -1;         // accumulated energy, if available
    double sclk_mhz = -1;         // current gfx clock
    double mclk_mhz = -1;         // current memory clock

    // PCIe throughput (bytes/s, already rated by the backend)
    double pcie_tx_bps = -1;
    double pcie_rx_bps = -1;

    // xGMI per-link accumulated traffic counters (KB, monotonically
    // increasing as reported by gpu_metrics xgmi_{read,write}_data_acc).
    // The sampler turns deltas into bytes/s. width/speed <= 0 => link absent.
    int xgmi_num_links = 0;
    double xgmi_read_acc_kb[kMaxXgmiLinks] = {0};
    double xgmi_write_acc_kb[kMaxXgmiLinks] = {0};
    double xgmi_link_width = -1;
    double xgmi_link_speed_gbps = -1;

    // activity accumulators (gpu_metrics v1): monotonic busy-time counters;
    // the sampler derives a windowed busy% from deltas when present.
    double gfx_activity_acc = -1;
    double mem_activity_acc = -1;

    // instantaneous activity percentages from gpu_metrics (may be finer
    // grained than busy_pct on some firmware)
    double gfx_activity_pct = -1;
    double umc_activity_pct = -1;

    // RAS: accumulated ECC error counts summed over GPU blocks
    double ecc_correctable = -1;
    double ecc_uncorrectable = -1;

    // throttle residency accumulators (gpu_metrics v1.6): PVIOL/TVIOL % =
    // delta(residency)/delta(accumulation_counter) * 100 over the window
    double accumulation_counter = -1;
    double ppt_residency_acc = -1;
    double thm_residency_acc = -1;

    // PCIe replay events (accumulated)
    double pcie_replay_count = -1;

    // per-XCP (compute partition) instantaneous gfx busy %, mean over the
    // partition's XCCs (gpu_metrics v1.6+ xcp_stats). num_partitions = 0
    // when the firmware reports none (then whole-GPU busy_pct is the only
    // utilization series).
    int num_partitions = 0;
    double xcp_busy_pct[kMaxXcp] = {-1, -1, -1, -1, -1, -1, -1, -1};
};

struct GpuInfo {
    int index = 0;
    std::string name;            // marketing or gfx name
    std::string uuid;            // unique id (hex) if available
    std::string pci_bdf;         // 0000:0a:00.0
    std::string drm_render;      // renderD128 style device name (attribution key)
    // partitioning (MI3xx XCP/NPS): on a partitioned node each partition
    // enumerates as its own device; partition_id says which slice of the
    // physical GPU this device is. -1 = not reported.
    std::string compute_partition;  // SPX/DPX/TPX/QPX/CPX, "" unknown
    std::string memory_partition;   // NPS1/NPS2/NPS4/NPS8, "" unknown
    int partition_id = -1;
};

// One probed counter family: available, or a reason why not (SURVEY.md §7:
// per-counter availability must be probed with graceful degradation, not
// assumed). Rendered as amd_counter_unavailable{counter,reason}.
struct CounterProbe {
    std::string counter;
    bool available = false;
    std::string reason;
};

class Backend {
  public:
    virtual ~Backend() = default;
    virtual int device_count() = 0;
    virtual GpuInfo info(int dev) = 0;
    virtual GpuSample sample(int dev) = 0;
    virtual std::string name() const = 0;
    // Probe results for counters this backend cannot serve (and for ones
    // it confirmed). Called once at startup; rendered as meta-metrics.
    virtual std::vector<CounterProbe> probes() { return {}; }
};

// rsmi backend: returns nullptr (with err set) when librocm_smi64 is
// unavailable or rsmi_init fails (no GPU).
std::unique_ptr<Backend> make_rsmi_backend(std::string* err);

// amd_smi backend: the successor library (rocm_smi_lib is in maintenance
// mode upstream); nullptr + err when libamd_smi is unavailable or finds
// no GPU. Same per-metric degradation contract as the rsmi backend.
std::unique_ptr<Backend> make_amdsmi_backend(std::string* err);

// mock backend: n synthetic devices; busy% follows a deterministic waveform
// or the value in env MI355X_MOCK_BUSY / file `busy_file` when given.
// Env MI355X_MOCK_PARTITIONS (1..8) makes each device report that many
// compute partitions with per-XCP busy around the device busy%.
std::unique_ptr<Backend> make_mock_backend(int n_devices, const std::string& busy_file);

} // namespace mi355x
