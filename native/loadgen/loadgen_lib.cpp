// loadgen_lib.cpp — host-side C API around the CDNA4 load kernels.
//
// MI355X-native replacement for the reference's load workload machinery
// (reference: cuda-test-deployment.yaml:18-19 runs 5000 sequential CUDA
// `vectorAdd` *processes*; README.md:115 doubles the load with a second
// loop). Here the same load shapes are produced from one resident process:
//   * lg_vector_add_loop(): N launches of the vectorAdd kernel with a host
//     sync between launches — reproduces the reference's partial-utilization
//     shape (launch/driver overhead dominates) without 5000 process forks.
//   * lg_gemm_burn(): duty-cycled MFMA bf16 GEMM bursts targeting a given
//     GPU-busy percentage — the high/tunable load the scale-up experiments
//     need (reference has no equivalent).
//   * lg_*_verify(): run one kernel on caller-provided host data so tests
//     can check numerics against a plain fp32 reference.
//
// Exposed as a plain C ABI: consumed by loadgen_main.cpp (CLI) and by
// mi355x_gpu_hpa/loadgen (ctypes).

#include <hip/hip_runtime.h>

#include <chrono>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

extern "C" __global__ void vector_add_f32(const float*, const float*, float*, int);
extern "C" __global__ void vector_add_f32x4(const float4*, const float4*, float4*, int);
extern "C" __global__ void triad_f32x4(const float4*, const float4*, float4*, float,
                                       long);
extern "C" __global__ void gemm_bf16_tn(const unsigned short*, const unsigned short*,
                                        float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_linear(const unsigned short*,
                                               const unsigned short*, float*, int,
                                               int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d21(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d20(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d19(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d18(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d9e(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d9w(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d6(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d14(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d9(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d9nr(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d2(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d7(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d8(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_w32(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_soft(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d5(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d4(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256_d1(const unsigned short*, const unsigned short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_tn_256(const unsigned short*,
                                            const unsigned short*, float*, int,
                                            int, int, int);
extern "C" __global__ void gemm_fp8_tn_256(const unsigned char*,
                                           const unsigned char*, float*, int,
                                           int, int, int);
extern "C" __global__ void gemm_fp8_tn_256_db(const unsigned char*,
                                              const unsigned char*, float*, int,
                                              int, int, int);
extern "C" __global__ void gemm_fp8_tn_256_nr(const unsigned char*,
                                              const unsigned char*, float*, int,
                                              int, int, int);

#define LG_CHECK(expr)                                                        \
    do {                                                                      \
        hipError_t _e = (expr);                                               \
        if (_e != hipSuccess) {                                               \
            std::snprintf(g_last_error, sizeof(g_last_error), "%s:%d %s: %s", \
                          __FILE__, __LINE__, #expr, hipGetErrorString(_e));  \
            return -1;                                                        \
        }                                                                     \
    } while (0)

static char g_last_error[512] = "";

static inline double now_ms()
{
    return std::chrono::duration<double, std::milli>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

static inline unsigned short f32_to_bf16(float f)
{
    uint32_t u;
    std::memcpy(&u, &f, 4);
    // round-to-nearest-even
    uint32_t lsb = (u >> 16) & 1;
    u += 0x7fffu + lsb;
    return (unsigned short)(u >> 16);
}

extern "C" {

const char* lg_last_error() { return g_last_error; }

int lg_device_count()
{
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

// ---------------------------------------------------------------------------
// vectorAdd
// ---------------------------------------------------------------------------

// One allocation + `iters` kernel launches (sync per launch, like one
// process-per-launch in the reference but without exec overhead).
// Returns 0 on success; *ms_out = total wall ms over the launches.
int lg_vector_add_loop(int device, int n, int iters, double* ms_out)
{
    LG_CHECK(hipSetDevice(device));
    float *a, *b, *c;
    size_t bytes = (size_t)n * 4;
    LG_CHECK(hipMalloc(&a, bytes));
    LG_CHECK(hipMalloc(&b, bytes));
    LG_CHECK(hipMalloc(&c, bytes));
    std::vector<float> h(n);
    for (int i = 0; i < n; ++i) h[i] = (float)((i * 2654435761u) % 1000) * 1e-3f;
    LG_CHECK(hipMemcpy(a, h.data(), bytes, hipMemcpyHostToDevice));
    LG_CHECK(hipMemcpy(b, h.data(), bytes, hipMemcpyHostToDevice));

    int threads = 256;
    int blocks = (n + threads - 1) / threads;
    if (blocks > 4096) blocks = 4096; // grid-stride covers the rest
    double t0 = now_ms();
    for (int i = 0; i < iters; ++i) {
        hipLaunchKernelGGL(vector_add_f32, dim3(blocks), dim3(threads), 0, 0, a, b, c, n);
        LG_CHECK(hipDeviceSynchronize());
    }
    double t1 = now_ms();
    if (ms_out) *ms_out = t1 - t0;
    LG_CHECK(hipFree(a));
    LG_CHECK(hipFree(b));
    LG_CHECK(hipFree(c));
    return 0;
}

// Numerics entry: c_out[i] = a[i] + b[i] computed on the GPU.
int lg_vector_add_verify(int device, const float* a_h, const float* b_h, float* c_out, int n)
{
    LG_CHECK(hipSetDevice(device));
    float *a, *b, *c;
    size_t bytes = (size_t)n * 4;
    LG_CHECK(hipMalloc(&a, bytes));
    LG_CHECK(hipMalloc(&b, bytes));
    LG_CHECK(hipMalloc(&c, bytes));
    LG_CHECK(hipMemcpy(a, a_h, bytes, hipMemcpyHostToDevice));
    LG_CHECK(hipMemcpy(b, b_h, bytes, hipMemcpyHostToDevice));
    int threads = 256;
    if (n % 4 == 0) {
        int n4 = n / 4;
        int blocks = (n4 + threads - 1) / threads;
        if (blocks > 4096) blocks = 4096;
        hipLaunchKernelGGL(vector_add_f32x4, dim3(blocks), dim3(threads), 0, 0,
                           (const float4*)a, (const float4*)b, (float4*)c, n4);
    } else {
        int blocks = (n + threads - 1) / threads;
        if (blocks > 4096) blocks = 4096;
        hipLaunchKernelGGL(vector_add_f32, dim3(blocks), dim3(threads), 0, 0, a, b, c, n);
    }
    LG_CHECK(hipDeviceSynchronize());
    LG_CHECK(hipMemcpy(c_out, c, bytes, hipMemcpyDeviceToHost));
    LG_CHECK(hipFree(a));
    LG_CHECK(hipFree(b));
    LG_CHECK(hipFree(c));
    return 0;
}

// ---------------------------------------------------------------------------
// MFMA bf16 GEMM
// ---------------------------------------------------------------------------

struct GemmBufs {
    unsigned short* a = nullptr;
    unsigned short* bt = nullptr;
    float* c = nullptr;
    int m = 0, n = 0, k = 0;
};

static int gemm_alloc(GemmBufs& g, int m, int n, int k, bool fill_random)
{
    LG_CHECK(hipMalloc(&g.a, (size_t)m * k * 2));
    LG_CHECK(hipMalloc(&g.bt, (size_t)n * k * 2));
    LG_CHECK(hipMalloc(&g.c, (size_t)m * n * 4));
    g.m = m; g.n = n; g.k = k;
    if (fill_random) {
        // Random-ish bf16 in [-1, 1): DVFS-honest load (zero-filled operands
        // clock higher and overstate TF/s — playbook §5.4 rule 25).
        size_t na = (size_t)m * k, nb = (size_t)n * k;
        std::vector<unsigned short> h(na > nb ? na : nb);
        uint32_t s = 0x12345678u;
        for (size_t i = 0; i < h.size(); ++i) {
            s = s * 1664525u + 1013904223u;
            float f = ((s >> 8) & 0xffff) / 32768.0f - 1.0f;
            h[i] = f32_to_bf16(f);
        }
        LG_CHECK(hipMemcpy(g.a, h.data(), na * 2, hipMemcpyHostToDevice));
        LG_CHECK(hipMemcpy(g.bt, h.data(), nb * 2, hipMemcpyHostToDevice));
    }
    return 0;
}

static void gemm_free(GemmBufs& g)
{
    (void)hipFree(g.a); (void)hipFree(g.bt); (void)hipFree(g.c);
    g = GemmBufs{};
}

// variant: 0 = 128^2 linear LDS, 1 = 128^2 swizzled, 2 = 256^2 8-phase.
static int gemm_launch(const GemmBufs& g, hipStream_t stream, int variant = 1)
{
    if (variant == 13) {  // d14: 1024-thread CTA, 4 waves/SIMD
        int n_tiles = (g.m / 256) * (g.n / 256);
        int blocks = n_tiles < 2048 ? n_tiles : 2048;
        int tiles_per_cta = (n_tiles + blocks - 1) / blocks;
        hipLaunchKernelGGL(gemm_bf16_tn_256_d14, dim3(blocks), dim3(1024), 0,
                           stream, g.a, g.bt, g.c, g.m, g.n, g.k, tiles_per_cta);
        return 0;
    }
    if (variant >= 2) {
        int n_tiles = (g.m / 256) * (g.n / 256);
        int blocks = n_tiles < 2048 ? n_tiles : 2048;
        int tiles_per_cta = (n_tiles + blocks - 1) / blocks;
        // product auto-select: at huge grids (>2048 tiles, 16k-class
        // shapes) the 3-deep-B d18 wins by ~16% (DMA latency under heavy
        // memory-system load); at <=8192-class shapes d9 ties or wins.
        if (variant == 2 && n_tiles > 2048) variant = 17;
        if (variant == 19 || variant == 20) {  // d20/d21: 256-thread CTA,
            hipLaunchKernelGGL(                    // 1 wave/SIMD, AGPR acc
                variant == 19 ? gemm_bf16_tn_256_d20 : gemm_bf16_tn_256_d21,
                dim3(blocks), dim3(256), 0, stream, g.a, g.bt, g.c, g.m,
                g.n, g.k, tiles_per_cta);
            return 0;
        }
        hipLaunchKernelGGL(variant == 3 ? gemm_bf16_tn_256_d1 : variant == 4 ? gemm_bf16_tn_256_d4 : variant == 5 ? gemm_bf16_tn_256_d5 : variant == 6 ? gemm_bf16_tn_256_d2 : variant == 7 ? gemm_bf16_tn_256_d7 : variant == 8 ? gemm_bf16_tn_256_d8 : variant == 9 ? gemm_bf16_tn_256_w32 : variant == 10 ? gemm_bf16_tn_256_soft : variant == 11 ? gemm_bf16_tn_256_d9 : variant == 12 ? gemm_bf16_tn_256_d9nr : variant == 14 ? gemm_bf16_tn_256_d6 : variant == 15 ? gemm_bf16_tn_256_d9w : variant == 16 ? gemm_bf16_tn_256_d9e : variant == 17 ? gemm_bf16_tn_256_d18 : variant == 18 ? gemm_bf16_tn_256_d19 : gemm_bf16_tn_256, dim3(blocks), dim3(512), 0, stream,
                           g.a, g.bt, g.c, g.m, g.n, g.k, tiles_per_cta);
        return 0;
    }
    int n_tiles = (g.m / 128) * (g.n / 128);
    int blocks = n_tiles < 2048 ? n_tiles : 2048;
    int tiles_per_cta = (n_tiles + blocks - 1) / blocks;
    if (variant == 0)
        hipLaunchKernelGGL(gemm_bf16_tn_linear, dim3(blocks), dim3(256), 0, stream,
                           g.a, g.bt, g.c, g.m, g.n, g.k, tiles_per_cta);
    else
        hipLaunchKernelGGL(gemm_bf16_tn, dim3(blocks), dim3(256), 0, stream,
                           g.a, g.bt, g.c, g.m, g.n, g.k, tiles_per_cta);
    return 0;
}

static bool gemm_dims_ok(int m, int n, int k, int variant)
{
    int mt = variant >= 2 ? 256 : 128;
    if (m % mt || n % mt || k % 64) {
        std::snprintf(g_last_error, sizeof(g_last_error),
                      "gemm dims must be multiples of %d/%d/64", mt, mt);
        return false;
    }
    return true;
}

// Timed GEMM: `iters` back-to-back launches after `warmup` untimed ones.
// *ms_out = mean ms per GEMM, *tflops_out = 2*M*N*K / time.
// variant: 1 = st_16x32-swizzled LDS (default), 0 = linear LDS (A/B ref).
// 256-tile kernels need >=128 CTAs to fill the 256-CU chip; below that the
// 128-tile kernel wins (4x the CTAs). Perf entry points auto-select; the
// verify entry points do NOT (tests must exercise the exact kernel asked).
static int gemm_effective_variant(int m, int n, int variant)
{
    if (variant >= 2 && (m / 256) * (n / 256) < 128 && m % 128 == 0 &&
        n % 128 == 0)
        return 1;
    return variant;
}

int lg_gemm_bf16_bench_variant(int device, int m, int n, int k, int warmup,
                               int iters, int variant, double* ms_out,
                               double* tflops_out)
{
    if (!gemm_dims_ok(m, n, k, variant)) return -1;
    variant = gemm_effective_variant(m, n, variant);
    LG_CHECK(hipSetDevice(device));
    GemmBufs g;
    if (gemm_alloc(g, m, n, k, true)) return -1;
    for (int i = 0; i < warmup; ++i) gemm_launch(g, 0, variant);
    LG_CHECK(hipDeviceSynchronize());
    double t0 = now_ms();
    for (int i = 0; i < iters; ++i) gemm_launch(g, 0, variant);
    LG_CHECK(hipDeviceSynchronize());
    double t1 = now_ms();
    double ms = (t1 - t0) / iters;
    if (ms_out) *ms_out = ms;
    if (tflops_out) *tflops_out = 2.0 * m * n * k / (ms * 1e-3) / 1e12;
    gemm_free(g);
    return 0;
}

int lg_gemm_bf16_bench(int device, int m, int n, int k, int warmup, int iters,
                       double* ms_out, double* tflops_out)
{
    return lg_gemm_bf16_bench_variant(device, m, n, k, warmup, iters, 1, ms_out,
                                      tflops_out);
}

// Numerics entry: C f32 = A bf16 @ B^T bf16 on caller data (A row-major
// [m][k] as f32 -> converted; Bt row-major [n][k]).
// variant: 1 = swizzled (product kernel), 0 = linear LDS.
int lg_gemm_bf16_verify_variant(int device, const float* a_h, const float* bt_h,
                                float* c_out, int m, int n, int k, int variant)
{
    if (!gemm_dims_ok(m, n, k, variant)) return -1;
    LG_CHECK(hipSetDevice(device));
    GemmBufs g;
    if (gemm_alloc(g, m, n, k, false)) return -1;
    std::vector<unsigned short> tmp((size_t)m * k);
    for (size_t i = 0; i < tmp.size(); ++i) tmp[i] = f32_to_bf16(a_h[i]);
    LG_CHECK(hipMemcpy(g.a, tmp.data(), tmp.size() * 2, hipMemcpyHostToDevice));
    tmp.resize((size_t)n * k);
    for (size_t i = 0; i < tmp.size(); ++i) tmp[i] = f32_to_bf16(bt_h[i]);
    LG_CHECK(hipMemcpy(g.bt, tmp.data(), tmp.size() * 2, hipMemcpyHostToDevice));
    gemm_launch(g, 0, variant);
    LG_CHECK(hipDeviceSynchronize());
    LG_CHECK(hipGetLastError());
    LG_CHECK(hipMemcpy(c_out, g.c, (size_t)m * n * 4, hipMemcpyDeviceToHost));
    gemm_free(g);
    return 0;
}

int lg_gemm_bf16_verify(int device, const float* a_h, const float* bt_h,
                        float* c_out, int m, int n, int k)
{
    return lg_gemm_bf16_verify_variant(device, a_h, bt_h, c_out, m, n, k, 1);
}

// Streaming-triad bandwidth burn: duty-cycled HBM traffic at
// `target_util_pct` of wall time; `gb` working set (3 buffers summing to
// ~gb GiB). Returns achieved GB/s over the busy bursts in *gbps_out.
// The bandwidth-axis load for the multi-metric HPA (config 5).
int lg_bw_burn(int device, double target_util_pct, double seconds, double gb,
               double period_ms, volatile int* stop_flag, double* gbps_out)
{
    if (gb <= 0) gb = 6.0;
    if (period_ms <= 0) period_ms = 100.0;
    if (target_util_pct < 0) target_util_pct = 0;
    if (target_util_pct > 100) target_util_pct = 100;
    LG_CHECK(hipSetDevice(device));
    long n4 = (long)(gb * (1ull << 30) / 3.0 / 16.0);
    float4 *a, *b, *c;
    LG_CHECK(hipMalloc(&a, n4 * 16));
    LG_CHECK(hipMalloc(&b, n4 * 16));
    LG_CHECK(hipMalloc(&c, n4 * 16));
    LG_CHECK(hipMemset(a, 0x3f, n4 * 16));
    LG_CHECK(hipMemset(b, 0x3e, n4 * 16));
    int blocks = 8192;
    double busy_ms_total = 0, bytes_total = 0;
    double t_end = now_ms() + seconds * 1e3;
    while (now_ms() < t_end) {
        if (stop_flag && *stop_flag) break;
        double period_start = now_ms();
        double busy_until = period_start + period_ms * target_util_pct / 100.0;
        while (now_ms() < busy_until) {
            double t0 = now_ms();
            hipLaunchKernelGGL(triad_f32x4, dim3(blocks), dim3(256), 0, 0, a, b,
                               c, 1.5f, n4);
            LG_CHECK(hipDeviceSynchronize());
            busy_ms_total += now_ms() - t0;
            bytes_total += 3.0 * n4 * 16;
        }
        double rest = period_start + period_ms - now_ms();
        if (rest > 0)
            std::this_thread::sleep_for(
                std::chrono::duration<double, std::milli>(rest));
    }
    if (gbps_out)
        *gbps_out = busy_ms_total > 0 ? bytes_total / (busy_ms_total * 1e-3) / 1e9
                                      : 0;
    (void)hipFree(a); (void)hipFree(b); (void)hipFree(c);
    return 0;
}

} // extern "C"

// Shared closed-loop duty engine: runs `launch4()` (4 queued kernel
// launches + sync, hipEvent-measured) in busy bursts of duty*period and
// trims the duty fraction by the measured GPU-active fraction per actual
// period. Used by the bf16 and fp8 burn entries.
template <typename LaunchFn>
static int duty_burn_loop(double target_util_pct, double seconds,
                          double period_ms, volatile int* stop_flag,
                          LaunchFn&& launch4)
{
    hipEvent_t ev0, ev1;
    LG_CHECK(hipEventCreate(&ev0));
    LG_CHECK(hipEventCreate(&ev1));
    double duty = target_util_pct / 100.0;
    const double duty_lo = duty > 0.25 ? duty - 0.25 : 0.0;
    const double duty_hi = duty + 0.25 < 1.0 ? duty + 0.25 : 1.0;
    const double kI = 0.004;
    double ema = -1;
    double t_end = now_ms() + seconds * 1e3;
    while (now_ms() < t_end) {
        if (stop_flag && *stop_flag) break;
        double period_start = now_ms();
        double busy_until = period_start + period_ms * duty;
        float active_ms = 0;
        while (now_ms() < busy_until) {
            LG_CHECK(hipEventRecord(ev0, 0));
            launch4();
            LG_CHECK(hipEventRecord(ev1, 0));
            LG_CHECK(hipDeviceSynchronize());
            float dt = 0;
            LG_CHECK(hipEventElapsedTime(&dt, ev0, ev1));
            active_ms += dt;
        }
        if (target_util_pct > 0 && target_util_pct < 100) {
            double actual_ms = now_ms() - period_start;
            if (actual_ms < period_ms) actual_ms = period_ms;
            double active_pct = active_ms / actual_ms * 100.0;
            ema = ema < 0 ? active_pct : 0.7 * ema + 0.3 * active_pct;
            duty += kI * (target_util_pct - ema);
            if (duty < duty_lo) duty = duty_lo;
            if (duty > duty_hi) duty = duty_hi;
        }
        double rest = period_start + period_ms - now_ms();
        if (rest > 0)
            std::this_thread::sleep_for(
                std::chrono::duration<double, std::milli>(rest));
    }
    (void)hipEventDestroy(ev0);
    (void)hipEventDestroy(ev1);
    return 0;
}

extern "C" {

// Duty-cycled GEMM burn: aim at `target_util_pct` GPU-busy for `seconds`.
// Duty cycle over a `period_ms` window: run GEMM launches for duty*period,
// sleep the rest. Two mechanisms close the gap between wall-clock duty and
// the GRBM busy% that rocm-smi/the exporter report (round 1 measured an
// open-loop undershoot of ~0.72x: launch+sync gaps inside a "busy" burst
// are GPU-idle wall time):
//   * launches are BATCHED (4 queued back-to-back per sync) so intra-burst
//     gaps mostly vanish;
//   * the duty fraction is CLOSED-LOOP on the measured GPU-active time per
//     period (hipEvent elapsed time — the kernel-resident time GRBM
//     counts) over the period's ACTUAL length, trimmed with bounded
//     integral action. A sysfs gpu_busy_percent blend was tried and
//     removed: the controller reads right after a burst, so that sample
//     is busy-biased and settled ~8pp low (profiles/duty_closed_loop.md).
// Round-1 verdict item 8: the open-loop burn needed a +/-25pp test band;
// closed-loop targets +/-10pp.
// stop_flag: optional; polled between periods (set non-zero to stop early).
int lg_gemm_burn(int device, double target_util_pct, double seconds,
                 int m, int n, int k, double period_ms, volatile int* stop_flag)
{
    if (m <= 0) m = 4096;
    if (n <= 0) n = 4096;
    if (k <= 0) k = 4096;
    if (period_ms <= 0) period_ms = 100.0;
    const int burn_variant = gemm_effective_variant(m, n, 2);
    if (target_util_pct < 0) target_util_pct = 0;
    if (target_util_pct > 100) target_util_pct = 100;
    LG_CHECK(hipSetDevice(device));
    GemmBufs g;
    if (gemm_alloc(g, m, n, k, true)) return -1;
    // one calibration launch so the first period isn't all compile/warmup
    gemm_launch(g, 0, burn_variant);
    LG_CHECK(hipDeviceSynchronize());
    int rc = duty_burn_loop(target_util_pct, seconds, period_ms, stop_flag,
                            [&] {
                                for (int b = 0; b < 4; ++b)
                                    gemm_launch(g, 0, burn_variant);
                            });
    gemm_free(g);
    return rc;
}

} // extern "C"

// ---------------------------------------------------------------------------
// FP8 (E4M3) MFMA GEMM — the CDNA4 2x-peak datatype (gemm_fp8_256.hip)
// ---------------------------------------------------------------------------

// f32 -> OCP E4M3 (saturating, round-to-nearest-even via the FPU's default
// rounding). Max finite 448; subnormal step 2^-9.
static unsigned char f32_to_fp8_e4m3(float f)
{
    if (f != f) return 0x7f;  // NaN
    unsigned char sign = f < 0 ? 0x80 : 0;
    float af = f < 0 ? -f : f;
    if (af > 448.f) return sign | 0x7e;  // saturate to max finite
    int ef;
    float m = std::frexp(af, &ef);  // af = m * 2^ef, m in [0.5, 1)
    int k = ef - 1;                 // af = (2m) * 2^k, 2m in [1, 2)
    if (af == 0.f || k < -20) return sign;  // below half the min subnormal
    if (k < -6) {  // subnormal band: units of 2^-9, codes 1..8
        int q = (int)std::nearbyint(af * 512.f);
        if (q == 0) return sign;
        return (unsigned char)(sign | q);  // q==8 lands on min normal 0x08
    }
    int q = (int)std::nearbyint(std::ldexp(af, 3 - k));  // in [8, 16]
    if (q == 16) {
        ++k;
        q = 8;
        if (k > 8) return sign | 0x7e;
    }
    return (unsigned char)(sign | ((k + 7) << 3) | (q - 8));
}

static float fp8_e4m3_to_f32(unsigned char v)
{
    int sign = v >> 7;
    int exp = (v >> 3) & 0xf;
    int man = v & 7;
    float r;
    if (exp == 0)
        r = man * (1.f / 512.f);
    else if (exp == 15 && man == 7)
        r = __builtin_nanf("");
    else
        r = (1.f + man / 8.f) * std::ldexp(1.f, exp - 7);
    return sign ? -r : r;
}

struct Fp8GemmBufs {
    unsigned char* a = nullptr;
    unsigned char* bt = nullptr;
    float* c = nullptr;
    int m = 0, n = 0, k = 0;
};

static int fp8_gemm_alloc(Fp8GemmBufs& g, int m, int n, int k, bool fill_random)
{
    LG_CHECK(hipMalloc(&g.a, (size_t)m * k));
    LG_CHECK(hipMalloc(&g.bt, (size_t)n * k));
    LG_CHECK(hipMalloc(&g.c, (size_t)m * n * 4));
    g.m = m; g.n = n; g.k = k;
    if (fill_random) {
        // random fp8 in [-1, 1): DVFS-honest load (playbook rule 25)
        size_t na = (size_t)m * k, nb = (size_t)n * k;
        std::vector<unsigned char> h(na > nb ? na : nb);
        uint32_t st = 0x2468ace1u;
        for (size_t i = 0; i < h.size(); ++i) {
            st = st * 1664525u + 1013904223u;
            float f = ((st >> 8) & 0xffff) / 32768.0f - 1.0f;
            h[i] = f32_to_fp8_e4m3(f);
        }
        LG_CHECK(hipMemcpy(g.a, h.data(), na, hipMemcpyHostToDevice));
        LG_CHECK(hipMemcpy(g.bt, h.data(), nb, hipMemcpyHostToDevice));
    }
    return 0;
}

static void fp8_gemm_free(Fp8GemmBufs& g)
{
    (void)hipFree(g.a); (void)hipFree(g.bt); (void)hipFree(g.c);
    g = Fp8GemmBufs{};
}

static int fp8_gemm_launch(const Fp8GemmBufs& g, hipStream_t stream, int raster)
{
    int n_tiles = (g.m / 256) * (g.n / 256);
    int blocks = n_tiles < 2048 ? n_tiles : 2048;
    int tiles_per_cta = (n_tiles + blocks - 1) / blocks;
    // raster: 1 = product (4x4 super-tile), 0 = linear, 2 = deep-B rotation
    hipLaunchKernelGGL(raster == 2   ? gemm_fp8_tn_256_db
                       : raster == 1 ? gemm_fp8_tn_256
                                     : gemm_fp8_tn_256_nr,
                       dim3(blocks), dim3(512), 0, stream, g.a, g.bt, g.c,
                       g.m, g.n, g.k, tiles_per_cta);
    return 0;
}

extern "C" {

// Timed fp8 GEMM: mean ms + TF/s over `iters` launches. raster: 4x4
// super-tile rasterization on (1, the product config) or off (0).
int lg_gemm_fp8_bench(int device, int m, int n, int k, int warmup, int iters,
                      int raster, double* ms_out, double* tflops_out)
{
    if (m % 256 || n % 256 || k % 128) {
        std::snprintf(g_last_error, sizeof(g_last_error),
                      "fp8 gemm dims must be multiples of 256/256/128");
        return -1;
    }
    LG_CHECK(hipSetDevice(device));
    Fp8GemmBufs g;
    if (fp8_gemm_alloc(g, m, n, k, true)) return -1;
    for (int i = 0; i < warmup; ++i) fp8_gemm_launch(g, 0, raster);
    LG_CHECK(hipDeviceSynchronize());
    double t0 = now_ms();
    for (int i = 0; i < iters; ++i) fp8_gemm_launch(g, 0, raster);
    LG_CHECK(hipDeviceSynchronize());
    double t1 = now_ms();
    double ms = (t1 - t0) / iters;
    if (ms_out) *ms_out = ms;
    if (tflops_out) *tflops_out = 2.0 * m * n * k / (ms * 1e-3) / 1e12;
    fp8_gemm_free(g);
    return 0;
}

// Numerics entry: quantizes the f32 inputs to E4M3, runs the kernel, and
// writes C plus the DEQUANTIZED inputs (aq/btq, f32) back so the caller
// can compute the exact reference matmul on what the GPU actually saw.
int lg_gemm_fp8_verify(int device, const float* a_h, const float* bt_h,
                       float* c_out, float* aq_out, float* btq_out,
                       int m, int n, int k)
{
    if (m % 256 || n % 256 || k % 128) {
        std::snprintf(g_last_error, sizeof(g_last_error),
                      "fp8 gemm dims must be multiples of 256/256/128");
        return -1;
    }
    LG_CHECK(hipSetDevice(device));
    Fp8GemmBufs g;
    if (fp8_gemm_alloc(g, m, n, k, false)) return -1;
    std::vector<unsigned char> q((size_t)m * k);
    for (size_t i = 0; i < q.size(); ++i) {
        q[i] = f32_to_fp8_e4m3(a_h[i]);
        if (aq_out) aq_out[i] = fp8_e4m3_to_f32(q[i]);
    }
    LG_CHECK(hipMemcpy(g.a, q.data(), q.size(), hipMemcpyHostToDevice));
    q.resize((size_t)n * k);
    for (size_t i = 0; i < q.size(); ++i) {
        q[i] = f32_to_fp8_e4m3(bt_h[i]);
        if (btq_out) btq_out[i] = fp8_e4m3_to_f32(q[i]);
    }
    LG_CHECK(hipMemcpy(g.bt, q.data(), q.size(), hipMemcpyHostToDevice));
    fp8_gemm_launch(g, 0, 1);
    LG_CHECK(hipDeviceSynchronize());
    LG_CHECK(hipMemcpy(c_out, g.c, (size_t)m * n * 4, hipMemcpyDeviceToHost));
    fp8_gemm_free(g);
    return 0;
}

// variant form: raster 0/1/2 selects the kernel (2 = deep-B rotation).
int lg_gemm_fp8_verify_variant(int device, const float* a_h, const float* bt_h,
                               float* c_out, float* aq_out, float* btq_out,
                               int m, int n, int k, int raster)
{
    if (m % 256 || n % 256 || k % 128) {
        std::snprintf(g_last_error, sizeof(g_last_error),
                      "fp8 gemm dims must be multiples of 256/256/128");
        return -1;
    }
    LG_CHECK(hipSetDevice(device));
    Fp8GemmBufs g;
    if (fp8_gemm_alloc(g, m, n, k, false)) return -1;
    std::vector<unsigned char> q((size_t)m * k);
    for (size_t i = 0; i < q.size(); ++i) {
        q[i] = f32_to_fp8_e4m3(a_h[i]);
        if (aq_out) aq_out[i] = fp8_e4m3_to_f32(q[i]);
    }
    LG_CHECK(hipMemcpy(g.a, q.data(), q.size(), hipMemcpyHostToDevice));
    q.resize((size_t)n * k);
    for (size_t i = 0; i < q.size(); ++i) {
        q[i] = f32_to_fp8_e4m3(bt_h[i]);
        if (btq_out) btq_out[i] = fp8_e4m3_to_f32(q[i]);
    }
    LG_CHECK(hipMemcpy(g.bt, q.data(), q.size(), hipMemcpyHostToDevice));
    fp8_gemm_launch(g, 0, raster);
    LG_CHECK(hipDeviceSynchronize());
    LG_CHECK(hipMemcpy(c_out, g.c, (size_t)m * n * 4, hipMemcpyDeviceToHost));
    fp8_gemm_free(g);
    return 0;
}

// fp8 (E4M3) variant of the burn: same closed-loop duty engine over the
// fp8 MFMA kernel — the 2x-peak datatype's power/clock profile as a load.
int lg_gemm_fp8_burn(int device, double target_util_pct, double seconds,
                     int m, int n, int k, double period_ms,
                     volatile int* stop_flag)
{
    if (m <= 0) m = 4096;
    if (n <= 0) n = 4096;
    if (k <= 0) k = 4096;
    if (period_ms <= 0) period_ms = 100.0;
    if (m % 256 || n % 256 || k % 128) {
        std::snprintf(g_last_error, sizeof(g_last_error),
                      "fp8 gemm dims must be multiples of 256/256/128");
        return -1;
    }
    if (target_util_pct < 0) target_util_pct = 0;
    if (target_util_pct > 100) target_util_pct = 100;
    LG_CHECK(hipSetDevice(device));
    Fp8GemmBufs g;
    if (fp8_gemm_alloc(g, m, n, k, true)) return -1;
    fp8_gemm_launch(g, 0, 1);
    LG_CHECK(hipDeviceSynchronize());
    int rc = duty_burn_loop(target_util_pct, seconds, period_ms, stop_flag,
                            [&] {
                                for (int b = 0; b < 4; ++b)
                                    fp8_gemm_launch(g, 0, 1);
                            });
    fp8_gemm_free(g);
    return rc;
}

} // extern "C"

