"""GPU numerics tests for the CDNA4 HIP load kernels.

Each HIP kernel is checked against a plain PyTorch fp32 reference of the
same op (vectorAdd: exact; bf16 MFMA GEMM: against a bf16-quantized fp32
matmul with a bf16-appropriate tolerance). Asymmetric inputs are used so a
transposed C-write cannot pass (playbook: symmetric-input tests miss it).
"""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _loadgen():
    from mi355x_gpu_hpa import loadgen

    assert loadgen.available(), (
        "libmi355x_loadgen.so missing on a GPU box — native path must load"
    )
    return loadgen


def test_device_visible(gpu):
    lg = _loadgen()
    assert lg.device_count() >= 1


def test_vector_add_numerics(gpu):
    lg = _loadgen()
    rng = np.random.default_rng(0)
    n = 50_000  # the reference's vectorAdd N (cuda-test-deployment.yaml:19)
    a = rng.standard_normal(n, dtype=np.float32)
    b = rng.standard_normal(n, dtype=np.float32)
    out = lg.vector_add(a, b)
    np.testing.assert_array_equal(out, a + b)


def test_vector_add_odd_n(gpu):
    lg = _loadgen()
    rng = np.random.default_rng(1)
    n = 50_001  # scalar-kernel tail path
    a = rng.standard_normal(n, dtype=np.float32)
    b = rng.standard_normal(n, dtype=np.float32)
    np.testing.assert_array_equal(lg.vector_add(a, b), a + b)


@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 384, 128), (512, 512, 512)])
def test_gemm_bf16_numerics(gpu, m, n, k):
    import torch

    lg = _loadgen()
    rng = np.random.default_rng(2)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    # Make inputs asymmetric/structured enough to catch layout transposes
    a[:, 0] += np.arange(m) * 0.01
    bt[:, 0] -= np.arange(n) * 0.01

    got = lg.gemm_bf16(a, bt)

    ta = torch.from_numpy(a).bfloat16().float()
    tb = torch.from_numpy(bt).bfloat16().float()
    ref = (ta @ tb.T).numpy()

    # fp32 accumulation over bf16 products on both sides; small tolerance
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2 * np.sqrt(k) / 8)


@pytest.mark.parametrize("m,n,k", [(256, 256, 64), (256, 256, 128),
                                   (512, 256, 256), (512, 512, 1024)])
def test_gemm_bf16_256_numerics(gpu, m, n, k):
    """The 256^2 8-phase kernel vs torch fp32 reference (multi-K-tile shapes
    exercise the deep staging pipeline; k=64 the clamped tail path)."""
    import torch

    lg = _loadgen()
    rng = np.random.default_rng(7)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    a[:, 0] += np.arange(m) * 0.01
    bt[:, 0] -= np.arange(n) * 0.01
    got = lg.gemm_bf16(a, bt, variant=2)
    ref = (torch.from_numpy(a).bfloat16().float()
           @ torch.from_numpy(bt).bfloat16().float().T).numpy()
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2 * np.sqrt(k) / 8)


@pytest.mark.parametrize("variant", [11, 15, 17, 18, 19, 20])  # d9..d21
@pytest.mark.parametrize("m,n,k", [(256, 256, 64), (256, 256, 128),
                                   (512, 256, 256), (512, 512, 1024),
                                   (1024, 1024, 2048)])
def test_gemm_bf16_256_d9_numerics(gpu, m, n, k, variant):
    """The single-barrier-per-K-tile d9 schedule vs torch fp32 (multi-K-tile
    shapes exercise the all-four-halves-ahead staging; k=64 the clamped
    tail). Same accumulation order as the product kernel, so also bitwise-
    comparable to it."""
    import torch

    lg = _loadgen()
    rng = np.random.default_rng(11)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    a[:, 0] += np.arange(m) * 0.01
    bt[:, 0] -= np.arange(n) * 0.01
    got = lg.gemm_bf16(a, bt, variant=variant)
    ref = (torch.from_numpy(a).bfloat16().float()
           @ torch.from_numpy(bt).bfloat16().float().T).numpy()
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2 * np.sqrt(k) / 8)
    same = lg.gemm_bf16(a, bt, variant=2)
    np.testing.assert_allclose(got, same, rtol=1e-6, atol=1e-5)


@pytest.mark.parametrize("m,n,k", [(256, 256, 64), (256, 256, 128),
                                   (512, 512, 1024), (1024, 1024, 2048)])
def test_gemm_bf16_256_d14_numerics(gpu, m, n, k):
    """The 16-wave 4-waves/SIMD kernel vs torch fp32 + bitwise vs the
    product kernel (same accumulation order)."""
    import torch

    lg = _loadgen()
    rng = np.random.default_rng(14)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    a[:, 0] += np.arange(m) * 0.01
    bt[:, 0] -= np.arange(n) * 0.01
    got = lg.gemm_bf16(a, bt, variant=13)
    ref = (torch.from_numpy(a).bfloat16().float()
           @ torch.from_numpy(bt).bfloat16().float().T).numpy()
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2 * np.sqrt(k) / 8)
    same = lg.gemm_bf16(a, bt, variant=14)   # d6: independent schedule,
    np.testing.assert_allclose(got, same, rtol=1e-6, atol=1e-5)  # same sums


def test_gemm_bf16_256_matches_128(gpu):
    """Cross-check: both kernels compute identical bf16 sums (same
    accumulation order over K) — results should agree to fp32 rounding."""
    lg = _loadgen()
    rng = np.random.default_rng(8)
    m = n = 512
    k = 512
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    c1 = lg.gemm_bf16(a, bt, variant=1)
    c2 = lg.gemm_bf16(a, bt, variant=2)
    np.testing.assert_allclose(c1, c2, rtol=1e-6, atol=1e-5)


def test_gemm_bf16_large_shape(gpu):
    """Grid-stride path: more tiles than CTAs."""
    import torch

    lg = _loadgen()
    rng = np.random.default_rng(3)
    m = n = 1024
    k = 256
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    got = lg.gemm_bf16(a, bt)
    ref = (torch.from_numpy(a).bfloat16().float()
           @ torch.from_numpy(bt).bfloat16().float().T).numpy()
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2 * np.sqrt(k) / 8)


def test_vector_add_loop_runs(gpu):
    lg = _loadgen()
    ms = lg.vector_add_loop(n=50_000, iters=50)
    assert ms > 0


def _floor_check(bench_fn, floor_tf, what):
    """Fallback detector, not a perf test: an implausibly low reading is
    re-measured once after a settle (observed one 2.5 TF/s transient when
    this ran right after a 20 s HBM burn on one box; isolated re-runs
    measured 450 TF/s — gpurun_out/gputest_final3.log)."""
    import time as _t

    ms, tf = bench_fn()
    if tf <= floor_tf:
        _t.sleep(2.0)
        ms, tf = bench_fn()
    assert ms > 0
    assert tf > floor_tf, f"{what} at {tf:.0f} TF/s — MFMA path not engaged?"


def test_gemm_bench_sane(gpu):
    lg = _loadgen()
    # MFMA path must be in play: even an untuned MFMA GEMM clears 100 TF/s;
    # a VALU/eager fallback cannot.
    _floor_check(lambda: lg.gemm_bench(m=2048, n=2048, k=2048, warmup=2,
                                       iters=10), 100, "bf16 GEMM")


def test_bw_burn_hits_hbm(gpu):
    """The streaming-triad load must push real HBM bandwidth (config 5's
    bandwidth axis): >3 TB/s during bursts on MI355X (8 TB/s peak)."""
    lg = _loadgen()
    gbps = lg.bw_burn(100.0, 4.0, gb=6.0)
    assert gbps > 3000, f"triad only {gbps:.0f} GB/s"


@pytest.mark.parametrize("raster", [1, 2])  # product, deep-B rotation
@pytest.mark.parametrize("m,n,k", [(256, 256, 128), (256, 256, 256),
                                   (512, 512, 1024), (1024, 1024, 2048)])
def test_gemm_fp8_numerics(gpu, m, n, k, raster):
    """FP8 (E4M3) MFMA kernel vs an exact fp32 reference over the
    dequantized operands (fp8 products are exact in f32; only the f32
    accumulation rounds, so the tolerance is tight)."""
    lg = _loadgen()
    rng = np.random.default_rng(88)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    a[:, 0] += np.arange(m) * 0.01
    bt[:, 0] -= np.arange(n) * 0.01
    c, aq, btq = lg.gemm_fp8(a, bt, raster=raster)
    # quantization sanity: E4M3 RNE error <= half a step = |x|/32 for
    # normals (3 mantissa bits), 2^-10 floor in the subnormal band
    assert (np.abs(aq - a) <= np.maximum(np.abs(a) / 16 + 1e-6,
                                         1.0 / 1024)).all()
    assert np.abs(aq).max() > 0.5
    ref = (aq.astype(np.float64) @ btq.astype(np.float64).T).astype(np.float32)
    # fp8 products are exact in f32; the error is K-long f32 accumulation
    # vs the float64 reference (measured max ~1e-3 relative at k=2048:
    # gpurun_out/fp8test4.log) — bound scales with sqrt(k)
    np.testing.assert_allclose(c, ref, rtol=2e-3, atol=2e-3 * np.sqrt(k))


def test_gemm_fp8_bench_sane(gpu):
    """The fp8 path must engage the 2x-peak matrix pipe: even untuned,
    >200 TF/s at a small shape (measured 450); a non-MFMA path cannot."""
    lg = _loadgen()
    _floor_check(lambda: lg.gemm_fp8_bench(m=2048, n=2048, k=2048, warmup=2,
                                           iters=10), 200, "fp8 GEMM")


def test_gemm_fp8_burn_tracks_target(gpu):
    """The fp8 burn shares the closed-loop duty engine: 50% target must
    land within the same ±10pp band as the bf16 burn."""
    import ctypes
    import threading
    import time

    from mi355x_gpu_hpa import loadgen
    from mi355x_gpu_hpa.control import parse_prometheus_text
    from mi355x_gpu_hpa.exporter import ExporterProcess

    stop = ctypes.c_int(0)

    def burn():
        loadgen._load().lg_gemm_fp8_burn(
            0, ctypes.c_double(50.0), ctypes.c_double(25.0),
            4096, 4096, 4096, ctypes.c_double(100.0), ctypes.byref(stop))

    t = threading.Thread(target=burn, daemon=True)
    t.start()
    vals = []
    try:
        time.sleep(3.0)
        with ExporterProcess(interval_ms=250) as exp:
            for _ in range(8):
                time.sleep(0.5)
                for s in parse_prometheus_text(exp.scrape()):
                    if (s.name == "dcgm_gpu_utilization"
                            and s.labels["gpu"] == "0"):
                        vals.append(s.value)
    finally:
        stop.value = 1
        t.join(timeout=15)
    mean = sum(vals) / len(vals)
    assert 40 <= mean <= 60, f"fp8 burn mean busy {mean}% for 50% target"
