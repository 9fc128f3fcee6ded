#!/usr/bin/env python3
"""gemm_ab_gpu.py — interleaved A/B of gemm_bf16_256 variants on MI355X.

Usage: python tools/gemm_ab_gpu.py [variants...] [--shapes 8192 16k8k]
Runs each variant's bench entry several times interleaved (A,B,A,B,...) so
DVFS drift hits all variants equally; prints per-run TF/s and the max.
Variant map: 2=PRODUCT (auto d9/d18 by grid size), 7=d7 5-barrier,
8=d6 no-raster, 10=soft-lgkm d6, 11=d9 single-barrier+raster,
12=d9 no-raster, 13=d14 16-wave 4-waves/SIMD, 14=d6 round-1 product,
15=d9 wide-epilogue, 16=d9e burst-staging, 17=d18 3-deep-B (160 KiB LDS),
18=d19 pipelined reads, 19=d20 4-wave AGPR-acc, 20=d21 d20+deep-B+prefetch,
100/101/102=fp8 E4M3 raster/no-raster/deep-B.
"""

import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402

SHAPES = {
    "4096": (4096, 4096, 4096),
    "8192": (8192, 8192, 8192),
    "16k8k": (16384, 16384, 8192),
    "16k": (16384, 16384, 16384),
}


def main():
    args = [a for a in sys.argv[1:]]
    shapes = ["8192", "16k8k"]
    if "--shapes" in args:
        i = args.index("--shapes")
        shapes = args[i + 1:]
        args = args[:i]
    variants = [int(a) for a in args] or [2, 11]
    rounds = 3
    out = {}
    for sh in shapes:
        m, n, k = SHAPES[sh]
        res = {v: [] for v in variants}
        for r in range(rounds):
            for v in variants:
                if v >= 100:  # 100/101/102: fp8 raster / no-raster / deep-B
                    ms, tf = loadgen.gemm_fp8_bench(
                        m, n, k, warmup=2, iters=4,
                        raster={100: 1, 101: 0, 102: 2}[v])
                else:
                    ms, tf = loadgen.gemm_bench(m, n, k, warmup=2, iters=4,
                                                variant=v)
                res[v].append(round(tf, 1))
                print(f"{sh} v{v} round{r}: {tf:.0f} TF/s", file=sys.stderr)
        out[sh] = {str(v): {"tf_runs": res[v], "tf_max": max(res[v])}
                   for v in variants}
    print(json.dumps(out))


if __name__ == "__main__":
    main()
