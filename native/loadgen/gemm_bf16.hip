// gemm_bf16.hip — MFMA/LDS-tiled bf16 GEMM load-generator kernel for
// MI355X (gfx950, CDNA4).
//
// This is the high-utilization load generator the reference lacks (its only
// kernel is the tiny CUDA vectorAdd, cuda-test-deployment.yaml:19, which caps
// GPU utilization at a few percent). For the 1->8 replica scale-up experiments
// we need a load whose utilization is HIGH and TUNABLE; a matrix-core GEMM is
// the natural MI355X-native choice.
//
// Design (CDNA4-first, per the MI355X kernel playbook):
//   * v_mfma_f32_16x16x32_bf16 matrix-core tiles (gfx950 2xK form).
//     Fragment layout (verified on hardware, tests/test_loadgen_gpu.py):
//       A: lane l holds A[i = l%16][k = 8*(l/16) + j], j = 0..7  (16 B/lane)
//       B: lane l holds B[k = 8*(l/16) + j][n = l%16]
//       C: lane l, reg r -> C[row = (l/16)*4 + r][col = l%16]
//   * 128x128 output tile per 256-thread workgroup (4 waves in a 2x2 grid;
//     each wave owns a 64x64 sub-tile = 4x4 MFMA fragments, 64 f32 acc VGPRs).
//   * BK=64 K-step, double-buffered LDS (2 x (A+B) x 128x64 bf16 = 64 KiB of
//     the CU's 160 KiB), staged with __builtin_amdgcn_global_load_lds
//     (16 B/lane direct HBM->LDS DMA; no VGPR round-trip, no ds_write pass),
//     next tile's DMA left in flight across the barrier (counted vmcnt, raw
//     s_barrier — __syncthreads() would drain the DMA queue).
//   * st_16x32 LDS XOR swizzle (byte ^= ((byte>>9)&1)<<5 within each 1024-B
//     subtile): the linear [128][64] bf16 image puts each ds_read_b128 lane
//     group 8-way on one bank pair (32 LDS cycles); the swizzle spreads it
//     4-way. glds writes lane-linear, so the swizzle is applied by
//     PRE-SWIZZLING the per-lane GLOBAL source address and reading LDS with
//     the same XOR'd offsets.
//   * B is consumed as B^T (column-major B = row-major [N][K]) so both
//     operands' fragments are 8 contiguous k-elements = one ds_read_b128.
//   * Grid-stride over output tiles so one launch fills all 256 CUs / 8 XCDs
//     regardless of problem size, with an XCD-aware bijective tile remap for
//     L2 locality (block b runs on XCD b%8).
//
// The kernel computes C[M][N] f32 = A[M][K] bf16 @ B^T[N][K] bf16.
// M, N, K must be multiples of the tile sizes (load generator: we pick them).

#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(8))) unsigned short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 64
#define TILE_HW (BM * BK)       // halfwords per operand tile (16 KiB)

// One glds instruction moves 64 lanes x 16 B = 1 KiB into LDS, wave-uniform
// LDS base + lane*16. A 16 KiB operand tile is 16 glds across the block's 4
// waves = 4 per wave.
#define GLDS_PER_TILE_PER_WAVE 4

// st_16x32 swizzle on a byte offset within the operand tile image
// (subtile = 1024 B = 8 rows x 128 B).
template <int SWZ>
__device__ __forceinline__ int swz(int byte_off)
{
    if (SWZ) byte_off ^= ((byte_off >> 9) & 1) << 5;
    return byte_off;
}

template <int SWZ>
__device__ __forceinline__ void gemm_bf16_tn_impl(
    const unsigned short* __restrict__ A,  // [M][K] bf16 row-major
    const unsigned short* __restrict__ Bt, // [N][K] bf16 row-major (= B col-major)
    float* __restrict__ C,                 // [M][N] f32 row-major
    int M, int N, int K, int tiles_per_cta)
{
    // Single __shared__ object (a second one forces vmcnt(0) before every
    // ds_read of a glds pipeline on ROCm 7.2).
    __shared__ unsigned short lds[2 * 2 * TILE_HW]; // [buf][A/B][128][64]

    const int tid = threadIdx.x;
    const int wid = tid >> 6;      // wave 0..3
    const int lane = tid & 63;
    const int wr = wid >> 1;       // wave row 0..1
    const int wc = wid & 1;        // wave col 0..1

    const int n_tiles_m = M / BM;
    const int n_tiles_n = N / BN;
    const int n_tiles = n_tiles_m * n_tiles_n;

    // XCD-aware bijective remap: consecutive tiles land on one XCD so an
    // XCD's L2 sees contiguous A-rows. q/r form handles n_tiles % 8 != 0.
    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    // Per-lane glds source for this lane's 16 B of each 1-KiB piece.
    // Piece p (wave wid, iter it: p = wid*4+it) covers LDS bytes
    // [p*1024, +1024) of the tile image = tile rows p*8 .. p*8+7.
    // Lane l's LDS destination byte is p*1024 + l*16 (lane-linear, fixed by
    // the hardware); the element that must LAND there is the one at linear
    // image offset swz(p*1024 + l*16) — so the source address is derived
    // from the swizzled offset while the LDS write stays linear.
    const int in_piece = swz<SWZ>(lane * 16);     // bit9 of l*16 < 512 for l<32
    // NOTE: swz must be evaluated on the offset within the 1024-B SUBTILE;
    // l*16 spans exactly one subtile per piece, so bit9 = (l>=32).
    const int src_row = in_piece >> 7;            // row within the 8-row piece
    const int src_kk = (in_piece & 127) >> 1;     // halfword within the row

    f32x4 acc[4][4];

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        const int tm = tile / n_tiles_n;
        const int tn = tile % n_tiles_n;
        const long row0 = (long)tm * BM;
        const long col0 = (long)tn * BN;

#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        const int kTiles = K / BK;
        const int piece_row0 = wid * (GLDS_PER_TILE_PER_WAVE * 8);

        // Prologue: stage k-tile 0 into buffer 0.
        {
            const unsigned short* ga = A + (row0 + piece_row0 + src_row) * (long)K + src_kk;
            const unsigned short* gb = Bt + (col0 + piece_row0 + src_row) * (long)K + src_kk;
            unsigned short* la = &lds[0];
            unsigned short* lb = &lds[TILE_HW];
#pragma unroll
            for (int it = 0; it < GLDS_PER_TILE_PER_WAVE; ++it) {
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)(ga + (long)it * 8 * K),
                    (__attribute__((address_space(3))) unsigned int*)(la + wid * 4 * 512 + it * 512),
                    16, 0, 0);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)(gb + (long)it * 8 * K),
                    (__attribute__((address_space(3))) unsigned int*)(lb + wid * 4 * 512 + it * 512),
                    16, 0, 0);
            }
        }

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[buf * 2 * TILE_HW];
            const unsigned short* lb = &lds[buf * 2 * TILE_HW + TILE_HW];

            // Issue next tile's glds into the other buffer before consuming
            // this one; the counted vmcnt below drains only the CURRENT
            // buffer's 8 DMAs and leaves these 8 in flight across the
            // barrier (raw s_barrier: __syncthreads would emit vmcnt(0)).
            if (kt + 1 < kTiles) {
                const long k0 = (long)(kt + 1) * BK;
                const unsigned short* ga =
                    A + (row0 + piece_row0 + src_row) * (long)K + k0 + src_kk;
                const unsigned short* gb =
                    Bt + (col0 + piece_row0 + src_row) * (long)K + k0 + src_kk;
                unsigned short* na = &lds[(buf ^ 1) * 2 * TILE_HW];
                unsigned short* nb = &lds[(buf ^ 1) * 2 * TILE_HW + TILE_HW];
#pragma unroll
                for (int it = 0; it < GLDS_PER_TILE_PER_WAVE; ++it) {
                    __builtin_amdgcn_global_load_lds(
                        (const __attribute__((address_space(1))) unsigned int*)(ga + (long)it * 8 * K),
                        (__attribute__((address_space(3))) unsigned int*)(na + wid * 4 * 512 + it * 512),
                        16, 0, 0);
                    __builtin_amdgcn_global_load_lds(
                        (const __attribute__((address_space(1))) unsigned int*)(gb + (long)it * 8 * K),
                        (__attribute__((address_space(3))) unsigned int*)(nb + wid * 4 * 512 + it * 512),
                        16, 0, 0);
                }
                asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
            } else {
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            }
            __builtin_amdgcn_s_barrier();

            // 2 MFMA k-steps of 32 over this 64-deep tile.
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                bf16x8 afrag[4], bfrag[4];
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int arow = wr * 64 + i * 16 + (lane & 15);
                    const int off = swz<SWZ>(arow * 128 + ks * 64 + (lane >> 4) * 16);
                    afrag[i] = *(const bf16x8*)((const char*)la + off);
                }
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bcol = wc * 64 + j * 16 + (lane & 15);
                    const int off = swz<SWZ>(bcol * 128 + ks * 64 + (lane >> 4) * 16);
                    bfrag[j] = *(const bf16x8*)((const char*)lb + off);
                }
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int i = 0; i < 4; ++i)
#pragma unroll
                    for (int j = 0; j < 4; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            // All waves done reading this buffer before it is refilled next
            // iteration. lgkmcnt only; glds-in-flight must survive.
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }

        // Epilogue: f32 store, coalesced by 16-lane row groups.
#pragma unroll
        for (int i = 0; i < 4; ++i) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 64 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        // Re-converge before the next grid-stride tile reuses LDS.
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(256, 1) gemm_bf16_tn(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_impl<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// Linear-LDS variant kept for A/B perf comparison (see profiles/).
extern "C" __global__ void __launch_bounds__(256, 1) gemm_bf16_tn_linear(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_impl<0>(A, Bt, C, M, N, K, tiles_per_cta);
}
