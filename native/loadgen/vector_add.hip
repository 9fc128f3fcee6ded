// vector_add.hip — CDNA4 (gfx950) elementwise vector-add load kernel.
//
// MI355X-native replacement for the reference workload's CUDA `vectorAdd`
// binary (reference: cuda-test-deployment.yaml:18-19 — image
// k8s.gcr.io/cuda-vector-add:v0.1 running `C[i]=A[i]+B[i]` over 50'000 floats,
// one tiny kernel launch per process invocation). Re-designed for CDNA4:
//   * 256-thread workgroups = 4 wavefronts of 64 lanes (wave64, not warp32);
//   * grid-stride loop so one launch geometry covers any N and, for large N,
//     produces >256 workgroups to fill all 8 XCDs (256 CUs);
//   * float4 vectorized path for the HBM3E-bound regime (16 B/lane/instr).
//
// The "partial utilization" load shape of the reference (launch overhead
// dominating a ~200 us kernel) is reproduced at the host level by looping
// kernel launches from one process (see loadgen_lib.cpp), not by forking
// 5000 processes.

#include <hip/hip_runtime.h>

extern "C" __global__ void __launch_bounds__(256) vector_add_f32(
    const float* __restrict__ a,
    const float* __restrict__ b,
    float* __restrict__ c,
    int n)
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    int stride = gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        c[i] = a[i] + b[i];
    }
}

// Vectorized variant: 4 floats per lane per iteration. n must be a multiple
// of 4 (the scalar kernel handles tails; the library dispatches).
extern "C" __global__ void __launch_bounds__(256) vector_add_f32x4(
    const float4* __restrict__ a,
    const float4* __restrict__ b,
    float4* __restrict__ c,
    int n4)
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    int stride = gridDim.x * blockDim.x;
    for (; i < n4; i += stride) {
        float4 va = a[i];
        float4 vb = b[i];
        float4 vc;
        vc.x = va.x + vb.x;
        vc.y = va.y + vb.y;
        vc.z = va.z + vb.z;
        vc.w = va.w + vb.w;
        c[i] = vc;
    }
}

// Streaming triad: c[i] = a[i] + s*b[i], float4-vectorized, grid-stride.
// The BANDWIDTH-bound load (3 x 16 B per lane-iteration, ~6 TB/s at full
// duty): drives the HBM3E/UMC-activity metric family independently of
// busy% so the multi-metric HPA (BASELINE config 5) can be exercised on
// its bandwidth axis — the GEMM load is compute-heavy, this one is pure
// memory traffic.
extern "C" __global__ void __launch_bounds__(256) triad_f32x4(
    const float4* __restrict__ a,
    const float4* __restrict__ b,
    float4* __restrict__ c,
    float s, long n4)
{
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; i < n4; i += stride) {
        float4 va = a[i];
        float4 vb = b[i];
        float4 vc;
        vc.x = va.x + s * vb.x;
        vc.y = va.y + s * vb.y;
        vc.z = va.z + s * vb.z;
        vc.w = va.w + s * vb.w;
        c[i] = vc;
    }
}
