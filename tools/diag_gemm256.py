#!/usr/bin/env python3
"""Diagnose gemm_bf16_tn_256 numeric mismatches: localize wrong elements by
(m,n) region (-> which wave/quadrant) and test whether the error equals a
stale/missing K-tile contribution."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402


def check_shapes(m, n, k):
    rng = np.random.default_rng(7)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)
    import torch
    abf = torch.from_numpy(a).bfloat16().float().numpy()
    btbf = torch.from_numpy(bt).bfloat16().float().numpy()
    ref = abf @ btbf.T
    for variant in (2, 3, 4):
        got = loadgen.gemm_bf16(a, bt, variant=variant)
        bad = np.abs(got - ref) > 0.5
        print(f"variant {variant} ({m}x{n}x{k}): mismatch {bad.sum()}/{bad.size}"
              f" = {bad.mean()*100:.2f}%")


def main():
    for shape in ((256, 256, 128), (512, 256, 256), (512, 512, 1024),
                  (1024, 1024, 2048)):
        check_shapes(*shape)


if __name__ == "__main__":
    main()
