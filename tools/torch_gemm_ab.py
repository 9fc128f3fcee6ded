#!/usr/bin/env python3
"""Library-GEMM comparison: torch.matmul (hipBLASLt) vs the hand-written
kernels at the bench shapes, interleaved. Context for profiles/: the burn
kernel exists for in-kernel duty control and fusion freedom; this records
honestly where the library stands on plain GEMMs."""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from mi355x_gpu_hpa import loadgen  # noqa: E402

SHAPES = {"4096": (4096,)*3, "8192": (8192,)*3, "16k": (16384,)*3}


def torch_tf(m, n, k, iters=4):
    a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
    # bf16 out (the library's fast path; ours writes f32 C — noted in the
    # profile when comparing)
    out = torch.empty(m, n, dtype=torch.bfloat16, device="cuda")
    for _ in range(2):
        torch.matmul(a, b.T, out=out)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        torch.matmul(a, b.T, out=out)
    torch.cuda.synchronize()
    dt = (time.monotonic() - t0) / iters
    return 2.0 * m * n * k / dt / 1e12


def main():
    out = {}
    for name, (m, n, k) in SHAPES.items():
        res = {"torch_bf16": [], "ours_bf16": []}
        for _ in range(3):
            res["torch_bf16"].append(round(torch_tf(m, n, k), 1))
            _, tf = loadgen.gemm_bench(m, n, k, warmup=2, iters=4, variant=2)
            res["ours_bf16"].append(round(tf, 1))
            print(name, res, file=sys.stderr)
        out[name] = res
    print(json.dumps(out))


if __name__ == "__main__":
    main()
