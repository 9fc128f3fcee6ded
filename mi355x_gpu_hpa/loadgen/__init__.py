"""ctypes bindings for the CDNA4 HIP load-generator library.

Native side: native/loadgen/{vector_add,gemm_bf16}.hip + loadgen_lib.cpp,
built by ``make -C native loadgen`` into native/build/libmi355x_loadgen.so.

These are the MI355X-native replacements for the reference's CUDA workload
(`k8s.gcr.io/cuda-vector-add:v0.1`, cuda-test-deployment.yaml:18-19).

On a GPU box the HIP library MUST load and the kernels MUST run — there is
no CPU fallback here by design (a silent eager fallback would fake the GPU
tests); calls raise LoadgenError with the native error string on failure.
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

from .. import NATIVE_BUILD

_LIB_PATH = os.environ.get(
    "MI355X_LOADGEN_LIB", str(NATIVE_BUILD / "libmi355x_loadgen.so")
)


class LoadgenError(RuntimeError):
    pass


_lib = None


def _load():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise LoadgenError(
                f"libmi355x_loadgen.so not found at {_LIB_PATH}; "
                "run `make -C native loadgen`"
            )
        lib = ctypes.CDLL(_LIB_PATH)
        lib.lg_last_error.restype = ctypes.c_char_p
        lib.lg_device_count.restype = ctypes.c_int
        lib.lg_vector_add_loop.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.POINTER(ctypes.c_double)
        ]
        lib.lg_vector_add_verify.argtypes = [
            ctypes.c_int,
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            ctypes.c_int,
        ]
        lib.lg_gemm_bf16_bench.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, ctypes.c_int,
            ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double),
        ]
        lib.lg_gemm_bf16_verify.argtypes = [
            ctypes.c_int,
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ]
        lib.lg_gemm_bf16_verify_variant.argtypes = (
            lib.lg_gemm_bf16_verify.argtypes + [ctypes.c_int]
        )
        lib.lg_gemm_bf16_bench_variant.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double),
        ]
        lib.lg_gemm_burn.argtypes = [
            ctypes.c_int, ctypes.c_double, ctypes.c_double,
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_double, ctypes.c_void_p,
        ]
        lib.lg_gemm_fp8_burn.argtypes = lib.lg_gemm_burn.argtypes
        lib.lg_gemm_fp8_bench.argtypes = [
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double),
        ]
        lib.lg_gemm_fp8_verify_variant.argtypes = [
            ctypes.c_int,
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ]
        lib.lg_gemm_fp8_verify.argtypes = [
            ctypes.c_int,
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            np.ctypeslib.ndpointer(np.float32, flags="C_CONTIGUOUS"),
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ]
        lib.lg_bw_burn.argtypes = [
            ctypes.c_int, ctypes.c_double, ctypes.c_double, ctypes.c_double,
            ctypes.c_double, ctypes.c_void_p, ctypes.POINTER(ctypes.c_double),
        ]
        _lib = lib
    return _lib


def _check(rc: int):
    if rc != 0:
        raise LoadgenError(_load().lg_last_error().decode())


def available() -> bool:
    return os.path.exists(_LIB_PATH)


def device_count() -> int:
    return _load().lg_device_count()


def vector_add_loop(device: int = 0, n: int = 50_000, iters: int = 100) -> float:
    """Run the reference load shape (launch loop); returns total wall ms."""
    ms = ctypes.c_double()
    _check(_load().lg_vector_add_loop(device, n, iters, ctypes.byref(ms)))
    return ms.value


def vector_add(a: np.ndarray, b: np.ndarray, device: int = 0) -> np.ndarray:
    a = np.ascontiguousarray(a, np.float32)
    b = np.ascontiguousarray(b, np.float32)
    assert a.shape == b.shape and a.ndim == 1
    out = np.empty_like(a)
    _check(_load().lg_vector_add_verify(device, a, b, out, a.size))
    return out


def gemm_bf16(a: np.ndarray, bt: np.ndarray, device: int = 0,
              variant: int = 1) -> np.ndarray:
    """C f32 [m,n] = bf16(a) [m,k] @ bf16(bt).T [k,n] on the GPU.

    variant: 1 = 128^2 swizzled tile (default), 0 = 128^2 linear LDS,
    2 = 256^2 8-phase schedule (m,n must be multiples of 256)."""
    a = np.ascontiguousarray(a, np.float32)
    bt = np.ascontiguousarray(bt, np.float32)
    m, k = a.shape
    n, k2 = bt.shape
    assert k == k2
    out = np.empty((m, n), np.float32)
    _check(_load().lg_gemm_bf16_verify_variant(device, a, bt, out, m, n, k,
                                               variant))
    return out


def gemm_bench(m=4096, n=4096, k=4096, warmup=5, iters=50, device=0,
               variant=1):
    """Returns (ms_per_gemm, tflops)."""
    ms = ctypes.c_double()
    tf = ctypes.c_double()
    _check(_load().lg_gemm_bf16_bench_variant(device, m, n, k, warmup, iters,
                                              variant,
                                              ctypes.byref(ms), ctypes.byref(tf)))
    return ms.value, tf.value


def gemm_fp8(a: np.ndarray, bt: np.ndarray, device: int = 0,
             raster: int = 1):
    """FP8 (E4M3) MFMA GEMM numerics entry: quantizes the f32 inputs to
    E4M3, computes C = Aq @ Btq^T on the GPU, and returns
    (c, aq, btq) where aq/btq are the dequantized (f32) operands the GPU
    actually multiplied — the caller computes the exact reference from
    them. M,N %% 256 == 0, K %% 128 == 0."""
    m, k = a.shape
    n, k2 = bt.shape
    assert k == k2
    a = np.ascontiguousarray(a, np.float32)
    bt = np.ascontiguousarray(bt, np.float32)
    c = np.empty((m, n), np.float32)
    aq = np.empty_like(a)
    btq = np.empty_like(bt)
    _check(_load().lg_gemm_fp8_verify_variant(device, a, bt, c, aq, btq,
                                              m, n, k, raster))
    return c, aq, btq


def gemm_fp8_bench(m=8192, n=8192, k=8192, warmup=2, iters=10, raster=1,
                   device=0):
    """Returns (ms_per_gemm, tflops) for the fp8 kernel."""
    ms = ctypes.c_double()
    tf = ctypes.c_double()
    _check(_load().lg_gemm_fp8_bench(device, m, n, k, warmup, iters, raster,
                                     ctypes.byref(ms), ctypes.byref(tf)))
    return ms.value, tf.value


def gemm_burn(target_util_pct: float, seconds: float, device: int = 0,
              m: int = 4096, n: int = 4096, k: int = 4096,
              period_ms: float = 100.0, fp8: bool = False):
    """Duty-cycled GEMM load at a target GPU-busy percentage (closed-loop
    on measured GPU-active time); fp8=True burns with the E4M3 kernel."""
    fn = _load().lg_gemm_fp8_burn if fp8 else _load().lg_gemm_burn
    _check(fn(device, target_util_pct, seconds, m, n, k, period_ms, None))


def bw_burn(target_util_pct: float, seconds: float, device: int = 0,
            gb: float = 6.0, period_ms: float = 100.0) -> float:
    """Duty-cycled streaming-triad HBM load (the bandwidth axis of the
    multi-metric HPA). Returns achieved GB/s during the busy bursts."""
    gbps = ctypes.c_double()
    _check(_load().lg_bw_burn(device, target_util_pct, seconds, gb,
                              period_ms, None, ctypes.byref(gbps)))
    return gbps.value
