// gemm_bf16_256.hip — 256x256-tile, 8-phase MFMA bf16 GEMM for MI355X
// (gfx950, CDNA4). The deep-pipelined big-tile schedule: where the 128^2
// 2-barrier kernel (gemm_bf16.hip) tops out near the structural ~900 TF
// ceiling (its barrier drains the LDS-DMA queue once per K-step), this one
// keeps half-tile DMAs in flight across every barrier with counted vmcnt
// waits and splits each K-step into 4 MFMA phases so staging, LDS reads and
// matrix math overlap continuously.
//
// Geometry (per 512-thread workgroup = 8 waves in a 2(M) x 4(N) grid):
//   output tile     256 x 256 (wave: 128 x 64 = 8 x 4 fragments of 16x16)
//   K-step (BK)     64  (2 x mfma_f32_16x16x32_bf16 per fragment)
//   LDS             128 KiB = 2 buffers x 4 half-tiles x 16 KiB
//                   half-tiles: A rows 0-127 (A0) / 128-255 (A1),
//                               B^T cols 0-127 (B0) / 128-255 (B1),
//                   each a [128][64] bf16 image, st_16x32-swizzled
//   staging         __builtin_amdgcn_global_load_lds 16 B/lane; one
//                   half-tile = 2 glds per wave (16 KiB / (8 waves x 1 KiB))
//
// PRODUCT (round 2): the d9/d18 single-barrier schedules defined further
// down (impl9/impl18; auto-selected by grid size in loadgen_lib.cpp —
// ~1.0 PF/s @8192^3, 1.25 PF/s @16384^3, profiles/gemm_bf16_256_ladder.md).
// The remainder of this header documents the round-1 d6 schedule, kept as
// the A/B baseline (variant 14) together with the full measured ablation
// family (d1..d21).
//
// d6 phase schedule: phase q computes m-frags {2q, 2q+1} x ALL FOUR
// n-frags (16 MFMA):
//   q0 reads: A frags 0-1 (4x ds_read_b128) + all B frags (8x), B held to q3
//   q1-q3 reads: 4x A each  -> 16 LDS reads per K-tile (vs 40 for the
//   quadrant schedules), and every B slot is dead after q0, which lets B
//   halves stage TWO K-tiles ahead:
//   q0 stages (kt+1, A0) + (kt+1, A1) into the other buffer
//   q1 stages (kt+2, B0) into the CURRENT buffer (B0 dead after q0)
//   q2 stages (kt+2, B1) into the CURRENT buffer
//   one per-wave drain at q3: vmcnt(4) -> all of kt+1 landed, kt+2's B
//   halves stay in flight across the boundary. Every half-tile gets >=4
//   phases of DMA-latency budget.
//
// Measured ablations kept as _d* entry points (same template, different
// DEPTH): d1 full-drain depth-1; d2 half-per-phase with one B0 in flight
// (846-880 TF); d4 quadrant phases + all-B-held (register spills, 666 TF);
// d5 latency-balanced d2 (==d2). Src K-tiles are clamped at the tail (a
// clamped restage writes byte-identical data, so the overlap is benign).
//
// C[M][N] f32 = A[M][K] bf16 @ B^T[N][K] bf16; M,N % 256 == 0, K % 64 == 0.

#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(8))) unsigned short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define HALF_HW 8192          // halfwords per half-tile image (128x64)
#define HALF_BYTES 16384

static __device__ __forceinline__ int swz256(int byte_off)
{
    return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

// DEPTH selects the measured schedule variants (see the header comment and
// profiles/gemm_bf16_256_ladder.md): 1 = depth-1 full drain; 2 = one B0
// half in flight; 4 = quadrant+all-B-held (spills); 5 = latency-balanced 2;
// 6 = PRODUCT (m-quarter phases, all-B-held, B two tiles ahead).
// RASTER = 1 adds the 4x4 super-tile rasterization.
template <int DEPTH, int RASTER = 0, int SOFT_LGKM = 0>
__device__ __forceinline__ void gemm_bf16_tn_256_impl(
    const unsigned short* __restrict__ A,  // [M][K] bf16
    const unsigned short* __restrict__ Bt, // [N][K] bf16
    float* __restrict__ C,                 // [M][N] f32
    int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW]; // [buf][half][128][64]

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2; // 0..1: A-half this wave consumes
    const int wc = w & 3;  // 0..3: B cols wc*64..+64 (B-half = wc>>1)

    const int n_tiles_n = N / 256;
    const int n_tiles = (M / 256) * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    // glds source mapping for this lane's 16 B of each 1-KiB piece
    // (piece p = w*2+it covers half-image rows p*8..p*8+7).
    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    // LDS read byte offsets (within a half image) for fragment loads.
    // A-frag mf (0..7), k-step ks: row_in_half = mf*16 + (lane&15)
    // B-frag nf (0..3): col_in_half = (wc&1)*64 + nf*16 + (lane&15)
    auto frag_off = [&](int row_in_half, int ks) {
        return swz256(row_in_half * 128 + ks * 64 + ((lane >> 4) * 16));
    };

    const int n_tiles_m = M / 256;
    // RASTER: 4x4 super-tile rasterization — an XCD's consecutive tiles
    // cover a 4x4 block of the tile grid, so 4 A-panels x 4 B-panels are
    // re-read from L2/L3 instead of HBM (fetch ~4x lower per block).
    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][4];
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        // stage half `h` (0=A0,1=A1,2=B0,3=B1) of K-tile `kt` into buffer
        // `buf`; 2 glds per wave.
        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1; // tail clamp (benign restage)
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        // ---- prologue: kt0's 4 halves + (kt1,B0) [B0 is staged two tiles
        // ahead by the q3 slot, so kt1's B0 belongs to "kt=-1 q3"] ----
        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        if (DEPTH == 4 || DEPTH == 6) {
            // B halves are staged two K-tiles ahead (B frags all read at
            // q0, so B slots die a whole tile early)
            stage(1, 2, 1);
            stage(1, 3, 1);
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // kt0 landed
        } else if (DEPTH == 2 || DEPTH == 5) {
            stage(1, 2, 1);
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); // kt0 landed
        } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[4][2]; // current m-half-range fragments
        bf16x8 bfrag[4][2]; // n fragments (DEPTH==4: all four held from q0;
                            // otherwise the current pair, held 2 phases)

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[(buf * 4 + wr) * HALF_HW];
            const unsigned short* lb = &lds[(buf * 4 + 2 + (wc >> 1)) * HALF_HW];
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                // DEPTH 6: phase q = m-frags {2q, 2q+1} x all n (2 A reads
                // held per phase, all 4 B frags held from q0 — 16 LDS reads
                // per K-tile instead of 40). Others: m-half-major quadrants.
                const int mbase = DEPTH == 6 ? q * 2 : (q & 1) * 4;
                const int npair = (q >> 1);    // q0,q1: n0-1; q2,q3: n2-3
                const int n_m = DEPTH == 6 ? 2 : 4;

                // fragment ds_reads for this phase
#pragma unroll
                for (int m = 0; m < 4; ++m) {
                    if (m >= n_m) break;
                    const int row = (mbase + m) * 16 + (lane & 15);
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        afrag[m][ks] =
                            *(const bf16x8*)((const char*)la + frag_off(row, ks));
                }
                if (DEPTH == 4 || DEPTH == 6) { // all four B frags at q0
                    if (q == 0) {
#pragma unroll
                        for (int n = 0; n < 4; ++n) {
                            const int col = bcol0 + n * 16 + (lane & 15);
#pragma unroll
                            for (int ks = 0; ks < 2; ++ks)
                                bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                                frag_off(col, ks));
                        }
                    }
                } else if ((q & 1) == 0) { // q0/q2: refresh B pair
#pragma unroll
                    for (int n = 0; n < 2; ++n) {
                        const int col = bcol0 + (npair * 2 + n) * 16 + (lane & 15);
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            bfrag[n][ks] =
                                *(const bf16x8*)((const char*)lb + frag_off(col, ks));
                    }
                }

                // staging for this phase (see schedule above)
                if (DEPTH == 6) {
                    // A halves one tile ahead at q0 (slots dead since kt-1
                    // end); B halves TWO tiles ahead at q1/q2 (B slots dead
                    // after q0's reads)
                    if (q == 0) {
                        stage(kt + 1, 0, buf ^ 1);
                        stage(kt + 1, 1, buf ^ 1);
                    } else if (q == 1) {
                        stage(kt + 2, 2, buf);
                    } else if (q == 2) {
                        stage(kt + 2, 3, buf);
                    }
                } else if (DEPTH == 4) {
                    // A halves one tile ahead at q0; B halves TWO tiles
                    // ahead at q1/q2 (their current-buffer slots are dead
                    // after q0's reads)
                    if (q == 0) {
                        stage(kt + 1, 0, buf ^ 1);
                        stage(kt + 1, 1, buf ^ 1);
                    } else if (q == 1) {
                        stage(kt + 2, 2, buf);
                    } else if (q == 2) {
                        stage(kt + 2, 3, buf);
                    }
                } else if (DEPTH == 5) {
                    // latency-balanced: both A halves at q0, B1 at q1 — the
                    // last-staged kt+1 half gets 3 phases to land
                    if (q == 0) {
                        stage(kt + 1, 0, buf ^ 1);
                        stage(kt + 1, 1, buf ^ 1);
                    } else if (q == 1) {
                        stage(kt + 1, 3, buf ^ 1);
                    } else if (q == 3) {
                        stage(kt + 2, 2, buf);
                    }
                } else if (q == 0) stage(kt + 1, 0, buf ^ 1);
                else if (q == 1) stage(kt + 1, 1, buf ^ 1);
                else if (q == 2) stage(kt + 1, 3, buf ^ 1);
                else if (DEPTH == 2) stage(kt + 2, 2, buf);
                else stage(kt + 1, 2, buf ^ 1);

                __builtin_amdgcn_s_barrier();
                // SOFT_LGKM: rely on hipcc's counted lgkm waits between each
                // ds_read and its consuming MFMA instead of a full drain —
                // the first MFMA can issue ~50 cyc earlier per phase. All
                // reads still retire before the phase-closing barrier (their
                // consumers issue before it), so the staging-deadness proofs
                // hold unchanged.
                if (!SOFT_LGKM)
                    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

                __builtin_amdgcn_s_setprio(1);
                if (DEPTH == 6) {
#pragma unroll
                    for (int m = 0; m < 2; ++m)
#pragma unroll
                        for (int n = 0; n < 4; ++n)
#pragma unroll
                            for (int ks = 0; ks < 2; ++ks)
                                acc[mbase + m][n] =
                                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                        afrag[m][ks], bfrag[n][ks],
                                        acc[mbase + m][n], 0, 0, 0);
                } else {
#pragma unroll
                    for (int m = 0; m < 4; ++m)
#pragma unroll
                        for (int n = 0; n < 2; ++n)
#pragma unroll
                            for (int ks = 0; ks < 2; ++ks)
                                acc[mbase + m][npair * 2 + n] =
                                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                        afrag[m][ks],
                                        bfrag[DEPTH == 4 ? npair * 2 + n : n][ks],
                                        acc[mbase + m][npair * 2 + n], 0, 0, 0);
                }
                __builtin_amdgcn_s_setprio(0);

                // per-wave DMA drain, before the barrier that publishes it
                if (q == 3) {
                    if (DEPTH == 4 || DEPTH == 6)
                        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
                    else if (DEPTH == 2 || DEPTH == 5)
                        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
                    else
                        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                }
                __builtin_amdgcn_s_barrier();
            }
        }

        // epilogue: per-wave 128x64 f32 store
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

// the round-1 product (d6 schedule: 8 barriers/K-tile, m-quarter phases,
// all-B-held, B staged two tiles ahead) — kept as the A/B baseline.
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d6(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<6, 1, 1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// the earlier half-per-phase schedule (one B0 half in flight) — ablation.
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d2(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<2>(A, Bt, C, M, N, K, tiles_per_cta);
}

// depth-1 pipeline (full DMA drain per K-tile) — correctness bisect + A/B.
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d1(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// deep variant: B halves staged two K-tiles ahead, all B frags held from q0
// (2 half-tiles in flight across the boundary; >=4 phases of DMA latency
// budget per half).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d4(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<4>(A, Bt, C, M, N, K, tiles_per_cta);
}

// latency-balanced staging order (A0+A1 at q0, B1 at q1, B0 two ahead).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d5(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<5>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d7: the d6 schedule with 5 barriers per K-tile instead of 8.
//
// PMC on d6 shows 34% of wave time parked at barriers/waitcnts. Mid-tile,
// every fragment read targets data published at the TILE boundary (the
// per-phase publishes only matter for the next tile), so the closing
// barrier of each phase is unnecessary if each phase drains its LDS reads
// (lgkmcnt(0)) BEFORE its single barrier — barrier passage then proves all
// waves' reads retired, which is exactly the deadness proof the next
// phase's glds staging needs. Phase body becomes
//     reads -> lgkmcnt(0) -> s_barrier -> MFMA -> stage
// with one extra boundary barrier per K-tile after the q3 vmcnt drain
// (the only point where freshly landed halves must be published to reads).
// A fast wave's next-phase reads now overlap a slow wave's MFMA segment.
// ---------------------------------------------------------------------------
template <int RASTER = 0>
__device__ __forceinline__ void gemm_bf16_tn_256_impl7(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW];

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2;
    const int wc = w & 3;

    const int n_tiles_n = N / 256;
    const int n_tiles = (M / 256) * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    auto frag_off = [&](int row_in_half, int ks) {
        return swz256(row_in_half * 128 + ks * 64 + ((lane >> 4) * 16));
    };

    const int n_tiles_m = M / 256;
    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][4];
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        stage(1, 2, 1);
        stage(1, 3, 1);
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[2][2];
        bf16x8 bfrag[4][2];

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[(buf * 4 + wr) * HALF_HW];
            const unsigned short* lb = &lds[(buf * 4 + 2 + (wc >> 1)) * HALF_HW];
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int mbase = q * 2;

                // reads, drained BEFORE the phase barrier
#pragma unroll
                for (int m = 0; m < 2; ++m) {
                    const int row = (mbase + m) * 16 + (lane & 15);
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        afrag[m][ks] =
                            *(const bf16x8*)((const char*)la + frag_off(row, ks));
                }
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 4; ++n) {
                        const int col = bcol0 + n * 16 + (lane & 15);
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                            frag_off(col, ks));
                    }
                }
                asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
                __builtin_amdgcn_s_barrier();

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int m = 0; m < 2; ++m)
#pragma unroll
                    for (int n = 0; n < 4; ++n)
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            acc[mbase + m][n] =
                                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                    afrag[m][ks], bfrag[n][ks],
                                    acc[mbase + m][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);

                if (q == 0) {
                    stage(kt + 1, 0, buf ^ 1);
                    stage(kt + 1, 1, buf ^ 1);
                } else if (q == 1) {
                    stage(kt + 2, 2, buf);
                } else if (q == 2) {
                    stage(kt + 2, 3, buf);
                }
            }
            // tile boundary: land kt+1, publish to the next tile's reads
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }

        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d7(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl7<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// d8 = the d6 schedule WITHOUT super-tile rasterization (raster ablation;
// the product entry above includes it: +6%% @8192^3, fetch 4.4->? GB).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d8(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<6, 0>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// w32: the d6 schedule on v_mfma_f32_32x32x16_bf16 tiles.
//
// The 32x32 shape's measured ceiling is ~15% above 16x16's (2382 vs 2075
// TF µbench) and each phase issues 8 long MFMAs instead of 16 short ones —
// half the issue slots for the same FLOPs. Per-wave tile stays 128x64:
// 4 m-frags x 2 n-frags of 32x32, acc 8 x 16 f32 = 128 regs (unchanged).
// Fragment maps (CK/ISA): A: lane l -> A[i=l%32][k=8*(l/32)+j];
// C/D: reg r -> row (r&3)+8*(r>>2)+4*(lane>>5), col lane&31.
// Staging, liveness proofs, drains, swizzle and raster are d6's verbatim
// (B is still read only at q0; A-halves live through q3).
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16;

extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_w32(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW];

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2;
    const int wc = w & 3;

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    // 32x32 fragment read: row_in_half = frag_row + (lane&31),
    // k = ks16*16 + (lane>>5)*8  -> byte off swizzled
    auto frag_off32 = [&](int row_in_half, int ks16) {
        return swz256(row_in_half * 128 + ks16 * 32 + ((lane >> 5) * 16));
    };

    const bool super4 = (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x16 acc[4][2];
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
#pragma unroll
                for (int e = 0; e < 16; ++e) acc[i][j][e] = 0.f;

        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        stage(1, 2, 1);
        stage(1, 3, 1);
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[4]; // current m-frag, 4 k-steps of 16
        bf16x8 bfrag[2][4];

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[(buf * 4 + wr) * HALF_HW];
            const unsigned short* lb = &lds[(buf * 4 + 2 + (wc >> 1)) * HALF_HW];
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int arow = q * 32 + (lane & 31);
#pragma unroll
                for (int ks = 0; ks < 4; ++ks)
                    afrag[ks] = *(const bf16x8*)((const char*)la +
                                                 frag_off32(arow, ks));
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 2; ++n) {
                        const int bcol = bcol0 + n * 32 + (lane & 31);
#pragma unroll
                        for (int ks = 0; ks < 4; ++ks)
                            bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                            frag_off32(bcol, ks));
                    }
                }

                if (q == 0) {
                    stage(kt + 1, 0, buf ^ 1);
                    stage(kt + 1, 1, buf ^ 1);
                } else if (q == 1) {
                    stage(kt + 2, 2, buf);
                } else if (q == 2) {
                    stage(kt + 2, 3, buf);
                }

                __builtin_amdgcn_s_barrier();
                asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int n = 0; n < 2; ++n)
#pragma unroll
                    for (int ks = 0; ks < 4; ++ks)
                        acc[q][n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            afrag[ks], bfrag[n][ks], acc[q][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);

                if (q == 3)
                    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
                __builtin_amdgcn_s_barrier();
            }
        }

        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 4; ++i) {
#pragma unroll
            for (int j = 0; j < 2; ++j) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const long row = row0 + wr * 128 + i * 32 + (r & 3) +
                                     8 * (r >> 2) + 4 * (lane >> 5);
                    const long col = col0 + wc * 64 + j * 32 + (lane & 31);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

// product schedule with the explicit post-barrier full drain (ablation;
// the product entry uses compiler-counted waits, measured +0.5-1%).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_soft(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl<6, 1, 0>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d9: ONE barrier per K-tile — the whole tile is a single scheduling window.
//
// The round-1 PMC bound: 34% of wave time parked at the 8 per-K-tile
// barriers/waits, MFMA pipe ~50% busy (profiles/gemm_bf16_256_ladder.md),
// and the d7 experiment showed shaving barriers while keeping per-phase
// read drains is not the lever. The d9 observation is a liveness fact: if
// EVERY half (A0,A1,B0,B1) of K-tile kt+1 stages into buffer buf^1, then
// every staged slot has been dead since the kt-1 -> kt boundary barrier
// (buf^1 was fully consumed during tile kt-1; B is read only at q0, A by
// q3, and all reads retire before their consuming MFMAs which precede
// that barrier). So NO mid-tile publish points are needed at all:
//
//     boundary barrier
//       q0: read B(all 4)+A(2) | stage kt+1 A0,A1 -> buf^1 | MFMA q0
//       q1: read A(2)          | stage kt+1 B0    -> buf^1 | MFMA q1
//       q2: read A(2)          | stage kt+1 B1    -> buf^1 | MFMA q2
//       q3: read A(2)          |                           | MFMA q3
//     s_waitcnt vmcnt(0)   (kt+1 landed; DMAs had ~3 phases >= 1.5 us)
//     boundary barrier
//
// with NO fences between the phases: the compiler schedules 64 MFMAs,
// 24 ds_reads and 6 glds of a tile as one block, so each phase's LDS
// reads issue under the previous phase's MFMA segment (counted lgkm
// waits), and co-resident waves drift freely within the tile — the
// partner-overlap the barrier-locked d6 could never reach. The staging
// depth shrinks from d6's two-tiles-ahead to one, which is why the
// drain can be a full vmcnt(0): by q3 the q0-issued DMAs are ~1.5 us
// old vs ~0.4 us HBM latency.
// ---------------------------------------------------------------------------
// WIDE_EPI: per-wave LDS transpose of the accumulators so C stores are
// global_store_dwordx4 (4 rows x 64 B -> 16 rows x 64 B per instruction,
// 4x fewer store instructions). The 128 KiB tile LDS is dead after the
// K-loop's final barrier; each wave uses a private padded 16x68-f32
// region (bank-conflict-free for both the scatter writes and the b128
// row reads; pad 68 makes bank = 4*row + col mod 64 a permutation).
// STAGE_Q0: issue all four kt+1 stages at q0 (instead of spreading over
// q0-q2) — maximizes DMA landing margin before the boundary vmcnt(0).
template <int RASTER = 1, int WIDE_EPI = 0, int STAGE_Q0 = 0>
__device__ __forceinline__ void gemm_bf16_tn_256_impl9(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW];

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2;
    const int wc = w & 3;

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    auto frag_off = [&](int row_in_half, int ks) {
        return swz256(row_in_half * 128 + ks * 64 + ((lane >> 4) * 16));
    };

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][4];
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[2][2];
        bf16x8 bfrag[4][2];

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[(buf * 4 + wr) * HALF_HW];
            const unsigned short* lb = &lds[(buf * 4 + 2 + (wc >> 1)) * HALF_HW];
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int mbase = q * 2;
#pragma unroll
                for (int m = 0; m < 2; ++m) {
                    const int row = (mbase + m) * 16 + (lane & 15);
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        afrag[m][ks] =
                            *(const bf16x8*)((const char*)la + frag_off(row, ks));
                }
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 4; ++n) {
                        const int col = bcol0 + n * 16 + (lane & 15);
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                            frag_off(col, ks));
                    }
                }

                // next tile's halves -> buf^1 (all slots dead since the
                // previous boundary barrier; no publish needed mid-tile)
                if (q == 0) {
                    stage(kt + 1, 0, buf ^ 1);
                    stage(kt + 1, 1, buf ^ 1);
                    if (STAGE_Q0) {
                        stage(kt + 1, 2, buf ^ 1);
                        stage(kt + 1, 3, buf ^ 1);
                    }
                } else if (!STAGE_Q0 && q == 1) {
                    stage(kt + 1, 2, buf ^ 1);
                } else if (!STAGE_Q0 && q == 2) {
                    stage(kt + 1, 3, buf ^ 1);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int m = 0; m < 2; ++m)
#pragma unroll
                    for (int n = 0; n < 4; ++n)
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            acc[mbase + m][n] =
                                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                    afrag[m][ks], bfrag[n][ks],
                                    acc[mbase + m][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            // single boundary: land kt+1's four halves, then publish
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }

        if (WIDE_EPI) {
            float* scr = (float*)lds + w * 1088; // private 16 x 68 f32
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                if (i)  // WAR: chunk i-1's reads must retire first
                    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
                for (int j = 0; j < 4; ++j)
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        scr[((lane >> 4) * 4 + r) * 68 + j * 16 + (lane & 15)] =
                            acc[i][j][r];
                // RAW: DS ops are per-wave in-order; drain before reads
                asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
                const long row = row0 + wr * 128 + i * 16 + (lane & 15);
                float* dst = C + row * (long)N + col0 + wc * 64;
#pragma unroll
                for (int rq = 0; rq < 4; ++rq) {
                    const int cq = (lane >> 4) + rq * 4;
                    f32x4 v = *(const f32x4*)&scr[(lane & 15) * 68 + cq * 4];
                    *(f32x4*)(dst + cq * 4) = v;
                }
            }
        } else {
#pragma unroll
            for (int i = 0; i < 8; ++i) {
#pragma unroll
                for (int j = 0; j < 4; ++j) {
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const long row =
                            row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                        const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                        C[row * (long)N + col] = acc[i][j][r];
                    }
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d9(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl9<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// d9 + wide epilogue (LDS-transposed dwordx4 stores) — A/B candidate.
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d9w(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl9<1, 1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d18: d9 + 3-deep B rotation in the spare 32 KiB of LDS (160 KiB total).
//
// d9's residual wait is its boundary vmcnt(0): the B halves staged at
// q1/q2 are only ~0.5-1 us old when the drain hits. MI355X has 160 KiB of
// LDS per CU and the d9 layout uses 128: spending the spare 32 KiB on a
// THIRD pair of B slots lets B stage TWO tiles ahead (d6's depth) while
// keeping d9's zero mid-tile barriers:
//   A: 2 buffers x 2 halves (64 KiB), abuf = kt & 1, staged one ahead at q0
//   B: 3 buffers x 2 halves (96 KiB), bbuf = kt % 3, staged two ahead at
//      q1/q2 into (kt+2) % 3 — that slot was consumed at tile kt-1 and its
//      reads retired before the kt-1 -> kt boundary barrier, so it is dead
//      with NO mid-tile publish point.
// Boundary drain becomes vmcnt(4): retires A(kt+1) (staged q0, ~1.5 us
// old) and B(kt+2) (staged LAST tile, ~3.5 us old) — both long landed —
// while this tile's B(kt+3) DMAs stay in flight across the barrier.
// ---------------------------------------------------------------------------
template <int RASTER = 1>
__device__ __forceinline__ void gemm_bf16_tn_256_impl18(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[(4 + 6) * HALF_HW];  // 160 KiB

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2;
    const int wc = w & 3;

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    auto frag_off = [&](int row_in_half, int ks) {
        return swz256(row_in_half * 128 + ks * 64 + ((lane >> 4) * 16));
    };

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][4];
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        // A slots: lds[(abuf*2 + h) * HALF_HW], h 0/1
        // B slots: lds[(4 + bbuf*2 + hb) * HALF_HW], hb 0/1
        auto stage_a = [&](int kt, int h, int abuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = A + (row0 + h * 128) * (long)K + k0;
            unsigned short* dst = &lds[(abuf * 2 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };
        auto stage_b = [&](int kt, int hb, int bbuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = Bt + (col0 + hb * 128) * (long)K + k0;
            unsigned short* dst = &lds[(4 + bbuf * 2 + hb) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage_a(0, 0, 0);
        stage_a(0, 1, 0);
        stage_b(0, 0, 0);
        stage_b(0, 1, 0);
        stage_b(1, 0, 1);
        stage_b(1, 1, 1);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[2][2];
        bf16x8 bfrag[4][2];

        int bbuf = 0;       // kt % 3 without a divide
        for (int kt = 0; kt < kTiles; ++kt) {
            const int abuf = kt & 1;
            const int bnext2 = bbuf + 2 >= 3 ? bbuf - 1 : bbuf + 2;  // (kt+2)%3
            const unsigned short* la = &lds[(abuf * 2 + wr) * HALF_HW];
            const unsigned short* lb =
                &lds[(4 + bbuf * 2 + (wc >> 1)) * HALF_HW];
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int mbase = q * 2;
#pragma unroll
                for (int m = 0; m < 2; ++m) {
                    const int row = (mbase + m) * 16 + (lane & 15);
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        afrag[m][ks] =
                            *(const bf16x8*)((const char*)la + frag_off(row, ks));
                }
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 4; ++n) {
                        const int col = bcol0 + n * 16 + (lane & 15);
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                            frag_off(col, ks));
                    }
                }

                if (q == 0) {
                    stage_a(kt + 1, 0, abuf ^ 1);
                    stage_a(kt + 1, 1, abuf ^ 1);
                } else if (q == 1) {
                    stage_b(kt + 2, 0, bnext2);
                } else if (q == 2) {
                    stage_b(kt + 2, 1, bnext2);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int m = 0; m < 2; ++m)
#pragma unroll
                    for (int n = 0; n < 4; ++n)
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            acc[mbase + m][n] =
                                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                    afrag[m][ks], bfrag[n][ks],
                                    acc[mbase + m][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            // boundary: A(kt+1) + B(kt+2) long landed; B(kt+3) stays in
            // flight across the barrier (FIFO: 4 newest glds)
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            bbuf = bbuf + 1 >= 3 ? 0 : bbuf + 1;
        }

        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d18(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl18<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d19: d18 + linearized LDS addressing + software-pipelined A reads.
//
// ISA inspection of d9/d18 shows each phase's MFMA block opens with a full
// s_waitcnt lgkmcnt(0): the phase's fragment reads issue immediately
// before their consumers, so every phase eats the whole LDS latency. Two
// changes remove that:
//  1. The st_16x32 swizzle XOR depends only on bits 0-10 of the byte
//     offset, i.e. NOT on the fragment index (mf*2048 or nf*2048) — so
//     swz(x) = frag*2048 + swz(base(lane, ks)). All 24 fragment reads
//     collapse onto TWO per-lane base offsets with compile-time immediate
//     deltas (the compiler previously held ~24 hoisted address VGPRs).
//  2. The freed registers fund a double-buffered afrag: phase q issues
//     phase q+1's A reads BEFORE its own MFMA segment, so the reads
//     retire under the matrix math (counted lgkm waits, ~0 exposed
//     latency for q1-q3).
// B stays read-at-q0 (its 8 reads overlap q1's prefetched As poorly only
// at q0; with the deep-B rotation the data has been resident for a full
// tile, so only LDS latency — not DMA — is exposed there).
// ---------------------------------------------------------------------------
template <int RASTER = 1>
__device__ __forceinline__ void gemm_bf16_tn_256_impl19(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[(4 + 6) * HALF_HW];  // 160 KiB (d18 layout)

    const int tid = threadIdx.x;
    const int w = tid >> 6;
    const int lane = tid & 63;
    const int wr = w >> 2;
    const int wc = w & 3;

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    // linearized read addressing: swz(frag_row*128 + ks*64 + lg*16) =
    // frag*2048 + swz_base[ks] for frag_row = frag*16 + (lane&15)
    const int swz_base0 = swz256((lane & 15) * 128 + ((lane >> 4) * 16));
    const int swz_base1 = swz256((lane & 15) * 128 + 64 + ((lane >> 4) * 16));

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][4];
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        auto stage_a = [&](int kt, int h, int abuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = A + (row0 + h * 128) * (long)K + k0;
            unsigned short* dst = &lds[(abuf * 2 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };
        auto stage_b = [&](int kt, int hb, int bbuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = Bt + (col0 + hb * 128) * (long)K + k0;
            unsigned short* dst = &lds[(4 + bbuf * 2 + hb) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 2; ++it) {
                const int p = w * 2 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage_a(0, 0, 0);
        stage_a(0, 1, 0);
        stage_b(0, 0, 0);
        stage_b(0, 1, 0);
        stage_b(1, 0, 1);
        stage_b(1, 1, 1);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 af[2][2][2];  // [pipe][m][ks] — double-buffered A fragments
        bf16x8 bfrag[4][2];

        int bbuf = 0;
        for (int kt = 0; kt < kTiles; ++kt) {
            const int abuf = kt & 1;
            const int bnext2 = bbuf + 2 >= 3 ? bbuf - 1 : bbuf + 2;
            const char* la0 = (const char*)&lds[(abuf * 2 + wr) * HALF_HW] +
                              swz_base0;
            const char* la1 = (const char*)&lds[(abuf * 2 + wr) * HALF_HW] +
                              swz_base1;
            const char* lb0 =
                (const char*)&lds[(4 + bbuf * 2 + (wc >> 1)) * HALF_HW] +
                (wc & 1) * 8192 + swz_base0;
            const char* lb1 =
                (const char*)&lds[(4 + bbuf * 2 + (wc >> 1)) * HALF_HW] +
                (wc & 1) * 8192 + swz_base1;

            // preload phase 0's A fragments (m-frags 0, 1)
#pragma unroll
            for (int m = 0; m < 2; ++m) {
                af[0][m][0] = *(const bf16x8*)(la0 + m * 2048);
                af[0][m][1] = *(const bf16x8*)(la1 + m * 2048);
            }
#pragma unroll
            for (int n = 0; n < 4; ++n) {
                bfrag[n][0] = *(const bf16x8*)(lb0 + n * 2048);
                bfrag[n][1] = *(const bf16x8*)(lb1 + n * 2048);
            }

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int mbase = q * 2;
                // pipeline: issue phase q+1's A reads before q's MFMAs
                if (q < 3) {
#pragma unroll
                    for (int m = 0; m < 2; ++m) {
                        const int fr = (q + 1) * 2 + m;
                        af[(q + 1) & 1][m][0] =
                            *(const bf16x8*)(la0 + fr * 2048);
                        af[(q + 1) & 1][m][1] =
                            *(const bf16x8*)(la1 + fr * 2048);
                    }
                }

                if (q == 0) {
                    stage_a(kt + 1, 0, abuf ^ 1);
                    stage_a(kt + 1, 1, abuf ^ 1);
                } else if (q == 1) {
                    stage_b(kt + 2, 0, bnext2);
                } else if (q == 2) {
                    stage_b(kt + 2, 1, bnext2);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int m = 0; m < 2; ++m)
#pragma unroll
                    for (int n = 0; n < 4; ++n)
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            acc[mbase + m][n] =
                                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                    af[q & 1][m][ks], bfrag[n][ks],
                                    acc[mbase + m][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            bbuf = bbuf + 1 >= 3 ? 0 : bbuf + 1;
        }

        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 64 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d19(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl19<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d20: 4 waves x 128x128 wave tiles — the hipBLASLt/Tensile design point.
//
// Kernel-trace of torch.matmul at 8192³ shows the library's winner is
// MT256x256x64_MI16x16 — OUR macro-tile and MFMA shape — at ~0.80 ms vs
// our 1.10 (gpurun_out/prof_blaslt). Its structural difference: ONE wave
// per SIMD with the full 512-register budget (256 arch VGPRs + 256
// accumulation AGPRs), i.e. 4 waves of 128x128 output each. Effects:
//   * per-FLOP LDS read traffic HALVES (each A row read by 2 wave-cols
//     and each B col by 2 wave-rows instead of 4/2): 32 KB per wave per
//     K-tile x 4 waves = 128 KB vs the 8-wave schedules' 192 KB;
//   * the MFMA pipe is fed by a single wave with 64 independent acc
//     chains — no co-resident-wave arbitration, issue slack for reads;
//   * barriers synchronize 4 waves instead of 8 (less skew).
// acc[8][8] f32x4 = 256 registers lands in AGPRs (launch_bounds(256,1)
// gives the 512-reg budget); B frags all held from q0 (64 VGPRs), A
// fragments read per phase with the d19 linearized addressing. Staging,
// liveness and the single boundary barrier are d9's, with 4 glds per
// wave per half (16 KiB / 4 waves).
// ---------------------------------------------------------------------------
template <int RASTER = 1>
__device__ __forceinline__ void gemm_bf16_tn_256_impl20(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW];

    const int tid = threadIdx.x;
    const int w = tid >> 6;       // 0..3
    const int lane = tid & 63;
    const int wr = w >> 1;        // 0..1: A half (rows wr*128..+128)
    const int wc = w & 1;         // 0..1: B half (cols wc*128..+128)

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    const int swz_base0 = swz256((lane & 15) * 128 + ((lane >> 4) * 16));
    const int swz_base1 = swz256((lane & 15) * 128 + 64 + ((lane >> 4) * 16));

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][8];  // 256 registers -> AGPR file
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 8; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        // 4 glds per wave per half (pieces w*4 .. w*4+3)
        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 4; ++it) {
                const int p = w * 4 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[2];
        bf16x8 bfrag[8][2];  // all 8 B fragments held from q0 (64 VGPRs)

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const char* la0 = (const char*)&lds[(buf * 4 + wr) * HALF_HW] +
                              swz_base0;
            const char* la1 = (const char*)&lds[(buf * 4 + wr) * HALF_HW] +
                              swz_base1;
            const char* lb0 = (const char*)&lds[(buf * 4 + 2 + wc) * HALF_HW] +
                              swz_base0;
            const char* lb1 = (const char*)&lds[(buf * 4 + 2 + wc) * HALF_HW] +
                              swz_base1;

#pragma unroll
            for (int q = 0; q < 8; ++q) {
                afrag[0] = *(const bf16x8*)(la0 + q * 2048);
                afrag[1] = *(const bf16x8*)(la1 + q * 2048);
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 8; ++n) {
                        bfrag[n][0] = *(const bf16x8*)(lb0 + n * 2048);
                        bfrag[n][1] = *(const bf16x8*)(lb1 + n * 2048);
                    }
                }

                if (q == 0) {
                    stage(kt + 1, 0, buf ^ 1);
                } else if (q == 1) {
                    stage(kt + 1, 1, buf ^ 1);
                } else if (q == 2) {
                    stage(kt + 1, 2, buf ^ 1);
                } else if (q == 3) {
                    stage(kt + 1, 3, buf ^ 1);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int n = 0; n < 8; ++n)
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        acc[q][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag[ks], bfrag[n][ks], acc[q][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }

#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 128 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(256, 1) gemm_bf16_tn_256_d20(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl20<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d21: d20 (4 waves, 128x128 tiles, AGPR acc) + the latency-hiding a
// single wave per SIMD actually needs:
//   * 3-deep B rotation in 160 KiB LDS (d18): B(kt+1) is PUBLISHED a full
//     tile early, so the next tile's 16 B-fragment reads prefetch at q7
//     of the current tile into a double-buffered bfrag set (128 VGPRs);
//   * A fragments double-buffered and prefetched one phase ahead (d19);
//     only q0's A reads after the boundary barrier expose LDS latency.
// With no co-resident wave to cover stalls (d20's regression), every
// latency must hide inside this wave's own MFMA stream — this is what
// the Tensile MT256x256x64 kernels do that the plain d20 port did not.
// ---------------------------------------------------------------------------
template <int RASTER = 1>
__device__ __forceinline__ void gemm_bf16_tn_256_impl21(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[(4 + 6) * HALF_HW];  // A 2x2, B 3x2 slots

    const int tid = threadIdx.x;
    const int w = tid >> 6;       // 0..3
    const int lane = tid & 63;
    const int wr = w >> 1;
    const int wc = w & 1;

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    const int swz_base0 = swz256((lane & 15) * 128 + ((lane >> 4) * 16));
    const int swz_base1 = swz256((lane & 15) * 128 + 64 + ((lane >> 4) * 16));

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[8][8];  // 256 registers -> AGPR file
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
            for (int j = 0; j < 8; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        auto stage_a = [&](int kt, int h, int abuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = A + (row0 + h * 128) * (long)K + k0;
            unsigned short* dst = &lds[(abuf * 2 + h) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 4; ++it) {
                const int p = w * 4 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };
        auto stage_b = [&](int kt, int hb, int bbuf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src = Bt + (col0 + hb * 128) * (long)K + k0;
            unsigned short* dst = &lds[(4 + bbuf * 2 + hb) * HALF_HW];
#pragma unroll
            for (int it = 0; it < 4; ++it) {
                const int p = w * 4 + it;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) unsigned int*)
                        (src + (long)(p * 8 + src_row) * K),
                    (__attribute__((address_space(3))) unsigned int*)
                        (dst + p * 512),
                    16, 0, 0);
            }
        };

        stage_a(0, 0, 0);
        stage_a(0, 1, 0);
        stage_b(0, 0, 0);
        stage_b(0, 1, 0);
        stage_b(1, 0, 1);
        stage_b(1, 1, 1);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 af[2][2];      // [pipe][ks]
        bf16x8 bf[8][2];      // B fragments, read once per tile at q0
                              // (double-buffering spilled: MFMA operands
                              // must be arch VGPRs and 128 extra regs do
                              // not fit beside af + addressing)

        int bbuf = 0;
        for (int kt = 0; kt < kTiles; ++kt) {
            const int abuf = kt & 1;
            const int bnext2 = bbuf + 2 >= 3 ? bbuf - 1 : bbuf + 2;
            const int bbuf1 = bbuf + 1 >= 3 ? 0 : bbuf + 1;  // (kt+1)%3
            const char* la0 = (const char*)&lds[(abuf * 2 + wr) * HALF_HW] +
                              swz_base0;
            const char* la1 = (const char*)&lds[(abuf * 2 + wr) * HALF_HW] +
                              swz_base1;
            const char* lb0 =
                (const char*)&lds[(4 + bbuf * 2 + wc) * HALF_HW] + swz_base0;
            const char* lb1 =
                (const char*)&lds[(4 + bbuf * 2 + wc) * HALF_HW] + swz_base1;

            // q0's fragments (the reads the barrier exposes; B is resident
            // since a tile ago — deep-B — so this is LDS latency only)
            af[0][0] = *(const bf16x8*)(la0);
            af[0][1] = *(const bf16x8*)(la1);
#pragma unroll
            for (int n = 0; n < 8; ++n) {
                bf[n][0] = *(const bf16x8*)(lb0 + n * 2048);
                bf[n][1] = *(const bf16x8*)(lb1 + n * 2048);
            }

#pragma unroll
            for (int q = 0; q < 8; ++q) {
                // prefetch next phase's A under this phase's MFMAs
                if (q < 7) {
                    af[(q + 1) & 1][0] =
                        *(const bf16x8*)(la0 + (q + 1) * 2048);
                    af[(q + 1) & 1][1] =
                        *(const bf16x8*)(la1 + (q + 1) * 2048);
                }

                if (q == 0) {
                    stage_a(kt + 1, 0, abuf ^ 1);
                } else if (q == 1) {
                    stage_a(kt + 1, 1, abuf ^ 1);
                } else if (q == 2) {
                    stage_b(kt + 2, 0, bnext2);
                } else if (q == 3) {
                    stage_b(kt + 2, 1, bnext2);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int n = 0; n < 8; ++n)
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        acc[q][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            af[q & 1][ks], bf[n][ks], acc[q][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            bbuf = bbuf1;
        }

#pragma unroll
        for (int i = 0; i < 8; ++i) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const long row = row0 + wr * 128 + i * 16 + (lane >> 4) * 4 + r;
                    const long col = col0 + wc * 128 + j * 16 + (lane & 15);
                    C[row * (long)N + col] = acc[i][j][r];
                }
            }
        }
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(256, 1) gemm_bf16_tn_256_d21(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl21<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// d9 with all four stages issued at q0 — A/B candidate.
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d9e(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl9<1, 0, 1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// d9 without the super-tile raster (isolates the schedule effect).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256_d9nr(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl9<0>(A, Bt, C, M, N, K, tiles_per_cta);
}

// ---------------------------------------------------------------------------
// d14: 16-wave CTA (1024 threads), 64x64 wave tiles, 4 waves/SIMD.
//
// The occupancy lever d9 cannot reach: with 128 acc VGPRs per wave the
// 128x64-tile schedules cap at 2 waves/SIMD (512-VGPR file), and the
// measured MFMA-pipe busy ~40-50% is then bounded by per-wave stall
// fraction. Shrinking the wave tile to 64x64 (4x4 wave grid over the same
// 256x256 CTA tile) drops the budget to acc 64 + B-frags 32 + A-frags 8
// ~= 104+addressing VGPRs — under the 128 needed for FOUR waves per SIMD,
// so the SIMD holds 2x the MFMA issue sources at every point. Same LDS
// image/swizzle/glds machinery (16 waves -> 1 KiB piece per wave), same
// single-barrier-per-K-tile liveness as d9. LDS read traffic doubles per
// FLOP vs d9 (B re-read by 4 waves per column block instead of 2) to
// ~78% of LDS bandwidth at 2 PF/s — the conflict-free swizzle is what
// makes that budget real.
// __launch_bounds__(1024, 4): the 4-waves/SIMD guarantee (<=128 VGPRs,
// enforced at compile time — check the .sdata VGPR count, do not assume).
// ---------------------------------------------------------------------------
template <int RASTER = 1>
__device__ __forceinline__ void gemm_bf16_tn_256_impl14(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int tiles_per_cta)
{
    __shared__ unsigned short lds[2 * 4 * HALF_HW];

    const int tid = threadIdx.x;
    const int w = tid >> 6;       // 0..15
    const int lane = tid & 63;
    const int wr = w >> 2;        // 0..3: A rows wr*64..+64
    const int wc = w & 3;         // 0..3: B cols wc*64..+64

    const int n_tiles_n = N / 256;
    const int n_tiles_m = M / 256;
    const int n_tiles = n_tiles_m * n_tiles_n;
    const int kTiles = K / 64;

    const int nwg = gridDim.x;
    int wgid = blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = wgid & 7, pos = wgid >> 3;
        wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    }

    const int in_piece = swz256(lane * 16) & 1023;
    const int src_row = in_piece >> 7;
    const int src_kk = (in_piece & 127) >> 1;

    auto frag_off = [&](int row_in_half, int ks) {
        return swz256(row_in_half * 128 + ks * 64 + ((lane >> 4) * 16));
    };

    const bool super4 = RASTER && (n_tiles_n % 4 == 0) && (n_tiles_m % 4 == 0);

    for (int t = 0; t < tiles_per_cta; ++t) {
        const int tile = wgid + t * nwg;
        if (tile >= n_tiles) return;
        int tm, tn;
        if (super4) {
            const int sb = tile >> 4, wi = tile & 15;
            const int sbn = n_tiles_n >> 2;
            tm = (sb / sbn) * 4 + (wi >> 2);
            tn = (sb % sbn) * 4 + (wi & 3);
        } else {
            tm = tile / n_tiles_n;
            tn = tile % n_tiles_n;
        }
        const long row0 = (long)tm * 256;
        const long col0 = (long)tn * 256;

        f32x4 acc[4][4];
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

        // one 1-KiB glds piece per wave (16 waves x 1 KiB = one half-tile)
        auto stage = [&](int kt, int h, int buf) {
            if (kt >= kTiles) kt = kTiles - 1;
            const long k0 = (long)kt * 64 + src_kk;
            const unsigned short* src =
                (h < 2) ? A + (row0 + h * 128) * (long)K + k0
                        : Bt + (col0 + (h - 2) * 128) * (long)K + k0;
            unsigned short* dst = &lds[(buf * 4 + h) * HALF_HW];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)
                    (src + (long)(w * 8 + src_row) * K),
                (__attribute__((address_space(3))) unsigned int*)(dst + w * 512),
                16, 0, 0);
        };

        stage(0, 0, 0);
        stage(0, 1, 0);
        stage(0, 2, 0);
        stage(0, 3, 0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        bf16x8 afrag[2];
        bf16x8 bfrag[4][2];

        for (int kt = 0; kt < kTiles; ++kt) {
            const int buf = kt & 1;
            const unsigned short* la = &lds[(buf * 4 + (wr >> 1)) * HALF_HW];
            const unsigned short* lb = &lds[(buf * 4 + 2 + (wc >> 1)) * HALF_HW];
            const int arow0 = (wr & 1) * 64;
            const int bcol0 = (wc & 1) * 64;

#pragma unroll
            for (int q = 0; q < 4; ++q) {
                const int row = arow0 + q * 16 + (lane & 15);
#pragma unroll
                for (int ks = 0; ks < 2; ++ks)
                    afrag[ks] =
                        *(const bf16x8*)((const char*)la + frag_off(row, ks));
                if (q == 0) {
#pragma unroll
                    for (int n = 0; n < 4; ++n) {
                        const int col = bcol0 + n * 16 + (lane & 15);
#pragma unroll
                        for (int ks = 0; ks < 2; ++ks)
                            bfrag[n][ks] = *(const bf16x8*)((const char*)lb +
                                                            frag_off(col, ks));
                    }
                }

                if (q == 0) {
                    stage(kt + 1, 0, buf ^ 1);
                    stage(kt + 1, 1, buf ^ 1);
                } else if (q == 1) {
                    stage(kt + 1, 2, buf ^ 1);
                } else if (q == 2) {
                    stage(kt + 1, 3, buf ^ 1);
                }

                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int n = 0; n < 4; ++n)
#pragma unroll
                    for (int ks = 0; ks < 2; ++ks)
                        acc[q][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag[ks], bfrag[n][ks], acc[q][n], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }

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
        __syncthreads();
    }
}

extern "C" __global__ void __launch_bounds__(1024, 4) gemm_bf16_tn_256_d14(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl14<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

// PRODUCT kernel: the d9 schedule (single barrier per K-tile, all four
// halves staged one tile ahead into the dead buffer) — measured 990-998
// TF/s @8192^3, 1124-1139 TF/s @16k x 16k x 8k on random bf16, +6% over
// the round-1 d6 product (see profiles/gemm_bf16_256_ladder.md).
extern "C" __global__ void __launch_bounds__(512, 2) gemm_bf16_tn_256(
    const unsigned short* A, const unsigned short* Bt, float* C,
    int M, int N, int K, int tiles_per_cta)
{
    gemm_bf16_tn_256_impl9<1>(A, Bt, C, M, N, K, tiles_per_cta);
}

