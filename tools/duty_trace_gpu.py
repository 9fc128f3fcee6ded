#!/usr/bin/env python3
"""duty_trace_gpu.py — characterize lg_gemm_burn's closed-loop duty control.

Runs the GEMM burn at several duty targets, samples the exporter's
dcgm_gpu_utilization each tick, and prints per-target mean/std/min/max plus
the settling behavior. Evidence base for the duty-band test tightening
(round-1 verdict item 8). Usage: python tools/duty_trace_gpu.py [secs]
"""

import ctypes
import json
import statistics
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen                      # noqa: E402
from mi355x_gpu_hpa.control import parse_prometheus_text  # noqa: E402
from mi355x_gpu_hpa.exporter import ExporterProcess     # noqa: E402


def trace(target_pct: float, secs: float):
    stop = ctypes.c_int(0)

    def burn():
        loadgen._load().lg_gemm_burn(
            0, ctypes.c_double(target_pct), ctypes.c_double(secs + 10),
            4096, 4096, 4096, ctypes.c_double(100.0), ctypes.byref(stop))

    t = threading.Thread(target=burn, daemon=True)
    t.start()
    vals = []
    try:
        time.sleep(3.0)  # settle: integral trim converges over ~2 s
        with ExporterProcess(interval_ms=250) as exp:
            t_end = time.time() + secs
            while time.time() < t_end:
                time.sleep(0.5)
                for s in parse_prometheus_text(exp.scrape()):
                    if (s.name == "dcgm_gpu_utilization"
                            and s.labels["gpu"] == "0"):
                        vals.append(s.value)
    finally:
        stop.value = 1
        t.join(timeout=15)
    time.sleep(1.0)
    return vals


def main():
    secs = float(sys.argv[1]) if len(sys.argv) > 1 else 15.0
    out = {}
    for target in (20.0, 50.0, 80.0):
        vals = trace(target, secs)
        out[target] = {
            "target_pct": target,
            "n": len(vals),
            "mean": round(statistics.mean(vals), 2) if vals else None,
            "stdev": round(statistics.stdev(vals), 2) if len(vals) > 1 else None,
            "min": min(vals) if vals else None,
            "max": max(vals) if vals else None,
            "err_of_mean": (round(abs(statistics.mean(vals) - target), 2)
                            if vals else None),
            "samples": vals,
        }
        print(f"target {target}%: mean {out[target]['mean']}% "
              f"stdev {out[target]['stdev']} n {len(vals)}", file=sys.stderr)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
