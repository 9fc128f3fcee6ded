// loadgen_main.cpp — `mi355x-loadgen` CLI.
//
// The container entrypoint for the autoscaled workload pods (replacement for
// the reference's `for (( c=1; c<=5000; c++ )); do ./vectorAdd; done`,
// cuda-test-deployment.yaml:19). Modes:
//
//   mi355x-loadgen vectoradd [--n 50000] [--iters 5000] [--device 0]
//       the reference's load shape: tiny kernel, launch-bound, a few % util
//   mi355x-loadgen gemm [--m/--n-dim/--k 4096] [--iters 50] [--device 0]
//       one timed MFMA bf16 GEMM burst; prints ms + TFLOP/s
//   mi355x-loadgen burn --util 80 [--seconds 60] [--device 0]
//       duty-cycled GEMM load at a target busy%% (tunable HPA-test load)
//
// Exit code 0 on success, 1 on usage error, 2 on HIP error.

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>

extern "C" {
const char* lg_last_error();
int lg_device_count();
int lg_vector_add_loop(int device, int n, int iters, double* ms_out);
int lg_gemm_bf16_bench(int device, int m, int n, int k, int warmup, int iters,
                       double* ms_out, double* tflops_out);
int lg_gemm_bf16_bench_variant(int device, int m, int n, int k, int warmup,
                               int iters, int variant, double* ms_out,
                               double* tflops_out);
int lg_gemm_fp8_bench(int device, int m, int n, int k, int warmup, int iters,
                      int raster, double* ms_out, double* tflops_out);
int lg_gemm_burn(int device, double target_util_pct, double seconds,
                 int m, int n, int k, double period_ms, volatile int* stop_flag);
int lg_gemm_fp8_burn(int device, double target_util_pct, double seconds,
                     int m, int n, int k, double period_ms,
                     volatile int* stop_flag);
int lg_bw_burn(int device, double target_util_pct, double seconds, double gb,
               double period_ms, volatile int* stop_flag, double* gbps_out);
}

static double argd(int argc, char** argv, const char* flag, double dflt)
{
    for (int i = 1; i + 1 < argc; ++i)
        if (!std::strcmp(argv[i], flag)) return std::atof(argv[i + 1]);
    return dflt;
}

int main(int argc, char** argv)
{
    if (argc < 2) {
        std::fprintf(stderr,
                     "usage: mi355x-loadgen {vectoradd|gemm|fp8gemm|burn|bwburn|devices} [flags]\n");
        return 1;
    }
    std::string mode = argv[1];
    int device = (int)argd(argc, argv, "--device", 0);

    if (mode == "devices") {
        std::printf("%d\n", lg_device_count());
        return 0;
    }
    if (mode == "vectoradd") {
        int n = (int)argd(argc, argv, "--n", 50000);
        int iters = (int)argd(argc, argv, "--iters", 5000);
        double ms = 0;
        if (lg_vector_add_loop(device, n, iters, &ms)) {
            std::fprintf(stderr, "error: %s\n", lg_last_error());
            return 2;
        }
        std::printf("vectoradd n=%d iters=%d total_ms=%.1f ms_per_iter=%.3f\n",
                    n, iters, ms, ms / iters);
        return 0;
    }
    if (mode == "gemm") {
        int m = (int)argd(argc, argv, "--m", 4096);
        int n = (int)argd(argc, argv, "--n-dim", 4096);
        int k = (int)argd(argc, argv, "--k", 4096);
        int iters = (int)argd(argc, argv, "--iters", 50);
        int warmup = (int)argd(argc, argv, "--warmup", 5);
        int variant = 1;
        for (int i = 1; i < argc; ++i) {
            if (!std::strcmp(argv[i], "--linear")) variant = 0;
            if (!std::strcmp(argv[i], "--v256")) variant = 2;
        }
        int vflag = (int)argd(argc, argv, "--variant", -1);
        if (vflag >= 0) variant = vflag;
        double ms = 0, tf = 0;
        if (lg_gemm_bf16_bench_variant(device, m, n, k, warmup, iters, variant,
                                       &ms, &tf)) {
            std::fprintf(stderr, "error: %s\n", lg_last_error());
            return 2;
        }
        char vname[16] = "";
        if (variant == 0) std::snprintf(vname, sizeof(vname), "_linear");
        else if (variant >= 2) std::snprintf(vname, sizeof(vname), "_v%d", variant);
        std::printf("gemm_bf16%s %dx%dx%d ms=%.3f tflops=%.1f\n",
                    vname, m, n, k, ms, tf);
        return 0;
    }
    if (mode == "fp8gemm") {
        int m = (int)argd(argc, argv, "--m", 8192);
        int n = (int)argd(argc, argv, "--n-dim", 8192);
        int k = (int)argd(argc, argv, "--k", 8192);
        int iters = (int)argd(argc, argv, "--iters", 10);
        int warmup = (int)argd(argc, argv, "--warmup", 2);
        int raster = 1;
        for (int i = 1; i < argc; ++i)
            if (!std::strcmp(argv[i], "--no-raster")) raster = 0;
        double ms = 0, tf = 0;
        if (lg_gemm_fp8_bench(device, m, n, k, warmup, iters, raster, &ms,
                              &tf)) {
            std::fprintf(stderr, "error: %s\n", lg_last_error());
            return 2;
        }
        std::printf("gemm_fp8 %dx%dx%d ms=%.3f tflops=%.1f\n", m, n, k, ms, tf);
        return 0;
    }
    if (mode == "burn") {
        double util = argd(argc, argv, "--util", 80.0);
        double seconds = argd(argc, argv, "--seconds", 60.0);
        int m = (int)argd(argc, argv, "--m", 4096);
        int n = (int)argd(argc, argv, "--n-dim", 4096);
        int k = (int)argd(argc, argv, "--k", 4096);
        double period = argd(argc, argv, "--period-ms", 100.0);
        bool fp8 = false;
        for (int i = 1; i < argc; ++i)
            if (!std::strcmp(argv[i], "--fp8")) fp8 = true;
        int rc = fp8 ? lg_gemm_fp8_burn(device, util, seconds, m, n, k,
                                        period, nullptr)
                     : lg_gemm_burn(device, util, seconds, m, n, k, period,
                                    nullptr);
        if (rc) {
            std::fprintf(stderr, "error: %s\n", lg_last_error());
            return 2;
        }
        return 0;
    }
    if (mode == "bwburn") {
        double util = argd(argc, argv, "--util", 80.0);
        double seconds = argd(argc, argv, "--seconds", 60.0);
        double gb = argd(argc, argv, "--gb", 6.0);
        double period = argd(argc, argv, "--period-ms", 100.0);
        double gbps = 0;
        if (lg_bw_burn(device, util, seconds, gb, period, nullptr, &gbps)) {
            std::fprintf(stderr, "error: %s\n", lg_last_error());
            return 2;
        }
        std::printf("bwburn util=%.0f%% achieved=%.0f GB/s during bursts\n",
                    util, gbps);
        return 0;
    }
    std::fprintf(stderr, "unknown mode '%s'\n", mode.c_str());
    return 1;
}
