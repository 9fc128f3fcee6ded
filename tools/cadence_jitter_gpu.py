#!/usr/bin/env python3
"""cadence_jitter_gpu.py — exporter tick-cadence stability on a real GPU.

Scrapes the exporter at high frequency for `--seconds`, detecting sample
updates by the gfx activity accumulator changing, and reports the
inter-update interval distribution (the realized collect cadence + jitter)
plus scrape-latency percentiles — all under GEMM load. Also cross-checks
busy% against the `amd-smi` CLI (a second oracle besides rocm-smi)."""

import argparse
import ctypes
import json
import subprocess
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402
from mi355x_gpu_hpa.control import parse_prometheus_text  # noqa: E402
from mi355x_gpu_hpa.exporter import ExporterProcess  # noqa: E402


def amd_smi_busy():
    for args in (["amd-smi", "metric", "-g", "0", "--usage", "--json"],
                 ["amd-smi", "metric", "--usage", "--json"]):
        try:
            out = subprocess.run(args, capture_output=True, timeout=15)
            data = json.loads(out.stdout.decode())
            # {"gpu_data": [{"gpu":0,"usage":{"gfx_activity":{"value":N}}}]}
            if isinstance(data, dict):
                data = data.get("gpu_data", [])
            if isinstance(data, list) and data:
                usage = data[0].get("usage", {})
                g = usage.get("gfx_activity")
                if isinstance(g, dict):
                    return float(g.get("value"))
                if g is not None:
                    return float(g)
        except Exception:
            continue
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=30.0)
    ap.add_argument("--interval-ms", type=float, default=100.0)
    ap.add_argument("--backend", default=None)
    args = ap.parse_args()

    stop = ctypes.c_int(0)
    t = threading.Thread(
        target=lambda: loadgen._load().lg_gemm_burn(
            0, ctypes.c_double(70.0), ctypes.c_double(args.seconds + 30),
            4096, 4096, 4096, ctypes.c_double(50.0), ctypes.byref(stop)),
        daemon=True)
    t.start()
    time.sleep(2)

    intervals = []
    scrape_ms = []
    try:
        with ExporterProcess(interval_ms=args.interval_ms,
                         backend=args.backend) as exp:
            last_acc = None
            last_change = None
            t_end = time.monotonic() + args.seconds
            while time.monotonic() < t_end:
                t0 = time.monotonic()
                text = exp.scrape()
                scrape_ms.append((time.monotonic() - t0) * 1e3)
                acc = None
                for s in parse_prometheus_text(text):
                    if s.name == "amd_gfx_activity_accumulated":
                        acc = s.value
                        break
                now = time.monotonic()
                if acc is not None and acc != last_acc:
                    if last_change is not None:
                        intervals.append((now - last_change) * 1e3)
                    last_change = now
                    last_acc = acc
                time.sleep(0.02)
            ours = None
            for s in parse_prometheus_text(exp.scrape()):
                if s.name == "dcgm_gpu_utilization" and s.labels["gpu"] == "0":
                    ours = s.value
            oracle = amd_smi_busy()
    finally:
        stop.value = 1
        t.join(timeout=20)

    intervals.sort()
    scrape_ms.sort()

    def pct(v, p):
        return round(v[min(len(v) - 1, int(len(v) * p))], 2) if v else None

    print(json.dumps({
        "configured_interval_ms": args.interval_ms,
        "n_updates": len(intervals),
        "update_interval_ms": {
            "p10": pct(intervals, 0.10), "p50": pct(intervals, 0.50),
            "p90": pct(intervals, 0.90), "p99": pct(intervals, 0.99),
            "max": pct(intervals, 1.0),
        },
        "scrape_latency_ms": {
            "p50": pct(scrape_ms, 0.50), "p99": pct(scrape_ms, 0.99),
            "max": pct(scrape_ms, 1.0),
        },
        "busy_ours_pct": ours,
        "busy_amd_smi_pct": oracle,
    }, indent=1))


if __name__ == "__main__":
    main()
