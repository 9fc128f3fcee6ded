#!/usr/bin/env python3
"""longevity_gpu.py — sustained-operation soak on a real MI355X.

Runs the exporter (real rsmi backend, 100 ms tick) plus a duty-cycled GEMM
burn for `--minutes`, scraping once a second, and reports:
  * exporter RSS at start/middle/end (memory-leak watch)
  * tick count observed vs expected (missed-tick watch)
  * scrape failures
  * busy% drift vs target
"""

import argparse
import ctypes
import json
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402
from mi355x_gpu_hpa.control import parse_prometheus_text  # noqa: E402
from mi355x_gpu_hpa.exporter import ExporterProcess  # noqa: E402


def rss_kb(pid: int) -> int:
    with open(f"/proc/{pid}/status") as f:
        for line in f:
            if line.startswith("VmRSS:"):
                return int(line.split()[1])
    return -1


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=8.0)
    ap.add_argument("--util", type=float, default=60.0)
    args = ap.parse_args()
    seconds = args.minutes * 60

    stop = ctypes.c_int(0)
    t = threading.Thread(
        target=lambda: loadgen._load().lg_gemm_burn(
            0, ctypes.c_double(args.util), ctypes.c_double(seconds + 60),
            4096, 4096, 4096, ctypes.c_double(50.0), ctypes.byref(stop)),
        daemon=True)
    t.start()

    rss = []
    busy = []
    fails = 0
    last_acc = None
    ticks = 0
    try:
        with ExporterProcess(interval_ms=100) as exp:
            pid = exp.proc.pid
            t_end = time.monotonic() + seconds
            next_rss = time.monotonic()
            while time.monotonic() < t_end:
                try:
                    samples = parse_prometheus_text(exp.scrape())
                except Exception:
                    fails += 1
                    time.sleep(1.0)
                    continue
                for s in samples:
                    if s.name == "amd_gfx_activity_accumulated":
                        if s.value != last_acc:
                            ticks += 1
                            last_acc = s.value
                    if (s.name == "dcgm_gpu_utilization"
                            and s.labels.get("gpu") == "0"):
                        busy.append(s.value)
                if time.monotonic() >= next_rss:
                    rss.append(rss_kb(pid))
                    next_rss += 30.0
                time.sleep(1.0)
            rss.append(rss_kb(pid))
    finally:
        stop.value = 1
        t.join(timeout=30)

    busy_tail = busy[10:]
    print(json.dumps({
        "minutes": args.minutes,
        "target_util_pct": args.util,
        "scrape_failures": fails,
        "ticks_observed_per_s_sampling": ticks,
        "rss_kb_timeline": rss,
        "rss_growth_kb": rss[-1] - rss[0] if len(rss) > 1 else None,
        "busy_mean_pct": round(sum(busy_tail) / max(1, len(busy_tail)), 1),
        "busy_min_pct": min(busy_tail) if busy_tail else None,
        "busy_max_pct": max(busy_tail) if busy_tail else None,
    }, indent=1))


if __name__ == "__main__":
    main()
