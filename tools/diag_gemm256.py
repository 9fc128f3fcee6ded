#!/usr/bin/env python3
"""Diagnose gemm_bf16_tn_256 numeric mismatches: localize wrong elements by
(m,n) region (-> which wave/quadrant) and test whether the error equals a
stale/missing K-tile contribution."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402


def main():
    m, n, k = 512, 512, 1024
    rng = np.random.default_rng(7)
    a = rng.uniform(-1, 1, (m, k)).astype(np.float32)
    bt = rng.uniform(-1, 1, (n, k)).astype(np.float32)

    import torch

    abf = torch.from_numpy(a).bfloat16().float().numpy()
    btbf = torch.from_numpy(bt).bfloat16().float().numpy()
    ref = abf @ btbf.T

    got = loadgen.gemm_bf16(a, bt, variant=2)
    err = got - ref
    bad = np.abs(err) > 0.5
    print(f"mismatch: {bad.sum()}/{bad.size} = {bad.mean()*100:.2f}%")
    if not bad.any():
        print("clean")
        return

    rows = np.where(bad.any(axis=1))[0]
    cols = np.where(bad.any(axis=0))[0]
    print(f"bad rows: {rows.min()}..{rows.max()} (n={len(rows)})")
    print(f"bad cols: {cols.min()}..{cols.max()} (n={len(cols)})")
    # per-128 block histogram (tile = 256; waves: wr in 0..1 over 128-row
    # halves, wc over 64-col quarters)
    h = np.zeros((m // 64, n // 64))
    for i in range(m // 64):
        for j in range(n // 64):
            h[i, j] = bad[i * 64:(i + 1) * 64, j * 64:(j + 1) * 64].mean()
    print("bad fraction per 64x64 block:")
    for r in h:
        print(" ".join(f"{x:4.2f}" for x in r))

    # hypothesis tests: error equals +/- contribution of K-tile kt
    kt_contrib = [abf[:, kt * 64:(kt + 1) * 64] @ btbf[:, kt * 64:(kt + 1) * 64].T
                  for kt in range(k // 64)]
    i, j = rows[0], cols[0]
    e = err[i, j]
    print(f"first bad [{i},{j}]: got {got[i,j]:.4f} ref {ref[i,j]:.4f} err {e:.4f}")
    for kt, c in enumerate(kt_contrib):
        if abs(e - c[i, j]) < 0.05:
            print(f"  err ~= +contrib of K-tile {kt} (double-counted)")
        if abs(e + c[i, j]) < 0.05:
            print(f"  err ~= -contrib of K-tile {kt} (missing)")
        for kt2, c2 in enumerate(kt_contrib):
            if kt2 != kt and abs(e - (c2[i, j] - c[i, j])) < 0.05:
                print(f"  err ~= contrib({kt2}) - contrib({kt}) (stale swap)")


if __name__ == "__main__":
    main()
