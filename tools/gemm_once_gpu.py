#!/usr/bin/env python3
"""Run a few dispatches of one gemm variant (for rocprofv3 wrapping).
Usage: python tools/gemm_once_gpu.py [variant] [m n k] [iters]"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from mi355x_gpu_hpa import loadgen  # noqa: E402

v = int(sys.argv[1]) if len(sys.argv) > 1 else 11
m, n, k = (int(x) for x in sys.argv[2:5]) if len(sys.argv) > 4 else (8192,) * 3
iters = int(sys.argv[5]) if len(sys.argv) > 5 else 4
if v >= 100:  # 100/101: fp8 kernel, raster on/off
    ms, tf = loadgen.gemm_fp8_bench(m, n, k, warmup=1, iters=iters,
                                    raster=(v == 100))
else:
    ms, tf = loadgen.gemm_bench(m, n, k, warmup=1, iters=iters, variant=v)
print(f"variant {v} {m}x{n}x{k}: {tf:.0f} TF/s ({ms:.2f} ms)")
