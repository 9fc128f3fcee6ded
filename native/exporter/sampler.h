// sampler.h — periodic GPU counter sampling + derived-rate computation.
//
// The analog of the reference exporter's `-c <ms>` collect loop
// (dcgm-exporter.yaml:37, `-c 10000`): samples every device on a fixed
// tick, keeps the latest sample per device, and derives windowed rates
// (xGMI bytes/s, busy% over the window from activity accumulators) from
// consecutive samples. Thread-safe snapshot for the HTTP renderer.

#pragma once

#include "backend.h"

#include <atomic>
#include <condition_variable>
#include <mutex>
#include <thread>
#include <vector>

namespace mi355x {

struct DeviceMetrics {
    GpuInfo info;
    GpuSample sample;              // latest raw sample
    // derived over the last window:
    double xgmi_read_bps[kMaxXgmiLinks] = {0};
    double xgmi_write_bps[kMaxXgmiLinks] = {0};
    double xgmi_total_bps = 0;
    double busy_windowed_pct = -1; // from gfx_activity_acc delta when available
    double mem_busy_windowed_pct = -1;
    double sample_pass_ms = -1;    // wall time of the last sampling pass
    double pviol_pct = -1;         // power-throttle residency % over the window
    double tviol_pct = -1;         // thermal-throttle residency % over the window
    long long samples_taken = 0;
};

class Sampler {
  public:
    Sampler(Backend* backend, double interval_ms);
    ~Sampler();

    void start();
    void stop();
    // One synchronous sampling pass over all devices (also what the loop
    // calls); exposed so tests can step deterministically.
    void sample_once();

    std::vector<DeviceMetrics> snapshot() const;
    bool ready() const { return ready_.load(); } // first successful sample
    double interval_ms() const { return interval_ms_; }

  private:
    void loop();

    Backend* backend_;
    double interval_ms_;
    mutable std::mutex mu_;
    std::vector<DeviceMetrics> state_;
    std::atomic<bool> ready_{false};
    std::atomic<bool> stop_{false};
    std::thread thread_;
    std::condition_variable cv_;
    std::mutex cv_mu_;
};

} // namespace mi355x
