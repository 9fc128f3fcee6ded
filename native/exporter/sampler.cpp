#include "sampler.h"

#include <chrono>

namespace mi355x {

Sampler::Sampler(Backend* backend, double interval_ms)
    : backend_(backend), interval_ms_(interval_ms)
{
    int n = backend_->device_count();
    state_.resize(n);
    for (int i = 0; i < n; ++i) state_[i].info = backend_->info(i);
}

Sampler::~Sampler() { stop(); }

void Sampler::start()
{
    stop_ = false;
    thread_ = std::thread([this] { loop(); });
}

void Sampler::stop()
{
    stop_ = true;
    cv_.notify_all();
    if (thread_.joinable()) thread_.join();
}

void Sampler::sample_once()
{
    int n = backend_->device_count();
    std::vector<GpuSample> fresh(n);
    auto t0 = std::chrono::steady_clock::now();
    for (int i = 0; i < n; ++i) fresh[i] = backend_->sample(i);
    double pass_ms = std::chrono::duration<double, std::milli>(
                         std::chrono::steady_clock::now() - t0)
                         .count() / (n > 0 ? n : 1);

    std::lock_guard<std::mutex> lk(mu_);
    bool any_ok = false;
    for (int i = 0; i < n && i < (int)state_.size(); ++i) {
        DeviceMetrics& d = state_[i];
        const GpuSample& prev = d.sample;
        const GpuSample& cur = fresh[i];
        if (!cur.ok) continue;
        any_ok = true;
        double dt_ms = cur.ts_ms - prev.ts_ms;
        if (d.samples_taken > 0 && dt_ms > 1.0) {
            double dt_s = dt_ms / 1e3;
            d.xgmi_total_bps = 0;
            for (int l = 0; l < cur.xgmi_num_links; ++l) {
                double dr = (cur.xgmi_read_acc_kb[l] - prev.xgmi_read_acc_kb[l]);
                double dw = (cur.xgmi_write_acc_kb[l] - prev.xgmi_write_acc_kb[l]);
                if (dr < 0) dr = 0; // counter reset
                if (dw < 0) dw = 0;
                d.xgmi_read_bps[l] = dr * 1024.0 / dt_s;
                d.xgmi_write_bps[l] = dw * 1024.0 / dt_s;
                d.xgmi_total_bps += d.xgmi_read_bps[l] + d.xgmi_write_bps[l];
            }
            // gfx_activity_acc counts busy-time; windowed busy% = delta busy
            // time / delta wall time. Units are firmware-defined but cancel
            // as long as the accumulator advances at 1 unit per busy-ms
            // (observed on gfx9xx+); clamp to [0,100] to be robust.
            if (cur.gfx_activity_acc >= 0 && prev.gfx_activity_acc >= 0 &&
                cur.gfx_activity_acc >= prev.gfx_activity_acc) {
                double pct =
                    (cur.gfx_activity_acc - prev.gfx_activity_acc) / dt_ms * 100.0;
                if (pct <= 100.0) d.busy_windowed_pct = pct;
            }
            // PVIOL/TVIOL per the gpu_metrics v1.6 formula
            if (cur.accumulation_counter > 0 && prev.accumulation_counter > 0 &&
                cur.accumulation_counter > prev.accumulation_counter) {
                double dacc = cur.accumulation_counter - prev.accumulation_counter;
                if (cur.ppt_residency_acc >= prev.ppt_residency_acc &&
                    prev.ppt_residency_acc >= 0)
                    d.pviol_pct =
                        (cur.ppt_residency_acc - prev.ppt_residency_acc) * 100.0 /
                        dacc;
                if (cur.thm_residency_acc >= prev.thm_residency_acc &&
                    prev.thm_residency_acc >= 0)
                    d.tviol_pct =
                        (cur.thm_residency_acc - prev.thm_residency_acc) * 100.0 /
                        dacc;
            }
            if (cur.mem_activity_acc >= 0 && prev.mem_activity_acc >= 0 &&
                cur.mem_activity_acc >= prev.mem_activity_acc) {
                double pct =
                    (cur.mem_activity_acc - prev.mem_activity_acc) / dt_ms * 100.0;
                if (pct <= 100.0) d.mem_busy_windowed_pct = pct;
            }
        }
        d.sample = cur;
        d.sample_pass_ms = pass_ms;
        d.samples_taken++;
    }
    if (any_ok) ready_ = true;
}

std::vector<DeviceMetrics> Sampler::snapshot() const
{
    std::lock_guard<std::mutex> lk(mu_);
    return state_;
}

void Sampler::loop()
{
    while (!stop_) {
        sample_once();
        std::unique_lock<std::mutex> lk(cv_mu_);
        cv_.wait_for(lk, std::chrono::duration<double, std::milli>(interval_ms_),
                     [this] { return stop_.load(); });
    }
}

} // namespace mi355x
