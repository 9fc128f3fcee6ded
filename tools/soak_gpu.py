#!/usr/bin/env python3
"""soak_gpu.py — closed-loop soak on a real MI355X.

Drives the MFMA GEMM burn through a schedule of utilization targets while
the native exporter (real rsmi backend, 100 ms tick) feeds the reference
recording rule and HPA. Records, per load step-change, the END-TO-END
detection latency: wall time from changing the burn target to the first
control-loop step whose metric lands in the new level's band. This is the
real-hardware version of the reference's manual probe (README.md:112-122)
with numbers attached — its loop took tens of seconds by construction
(10 s tick + 30 s rule); ours is bounded by the 100 ms exporter tick.

Run (on a GPU box):  python tools/soak_gpu.py [--seconds-per-level 20]
Writes a JSON report to stdout.
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
from mi355x_gpu_hpa.control import (  # noqa: E402
    ControlLoop,
    HpaSpec,
    Scraper,
    ScrapeTarget,
    synth_pod_labels,
)
from mi355x_gpu_hpa.exporter import ExporterProcess  # noqa: E402


class Burner:
    """Restartable duty-cycled GEMM load."""

    def __init__(self, device=0):
        self.device = device
        self.thread = None
        self.stop_flag = None

    def set_target(self, util_pct: float):
        self.stop()
        if util_pct <= 0:
            return
        self.stop_flag = ctypes.c_int(0)
        flag = self.stop_flag

        def run():
            loadgen._load().lg_gemm_burn(
                self.device, ctypes.c_double(util_pct), ctypes.c_double(3600.0),
                4096, 4096, 4096, ctypes.c_double(50.0), ctypes.byref(flag))

        self.thread = threading.Thread(target=run, daemon=True)
        self.thread.start()

    def stop(self):
        if self.stop_flag is not None:
            self.stop_flag.value = 1
        if self.thread is not None:
            self.thread.join(timeout=15)
        self.thread = None
        self.stop_flag = None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds-per-level", type=float, default=20.0)
    ap.add_argument("--levels", type=str, default="0,40,80,20,0")
    ap.add_argument("--interval-ms", type=float, default=100.0)
    ap.add_argument("--loop-period-s", type=float, default=0.5)
    args = ap.parse_args()
    levels = [float(x) for x in args.levels.split(",")]

    burner = Burner()
    report = {"levels": [], "interval_ms": args.interval_ms,
              "loop_period_s": args.loop_period_s}
    with ExporterProcess(interval_ms=args.interval_ms) as exp:
        scraper = Scraper([ScrapeTarget(exp.url, node="node0")])
        orig = scraper.scrape_once

        def with_pods():
            samples = orig()
            for s in samples:
                g = s.labels.get("gpu")
                if g is not None and "pod" not in s.labels:
                    s.labels["pod"] = f"cuda-test-{g}"
                    s.labels.setdefault("namespace", "default")
            return samples

        scraper.scrape_once = with_pods
        loop = ControlLoop(
            scraper,
            hpa_spec=HpaSpec(min_replicas=1, max_replicas=8, target_value=5.0,
                             downscale_stabilization_s=30.0),
            extra_samples=lambda: synth_pod_labels(
                ["cuda-test-0"]),
            use_adapter=True,  # full L4 hop (adapter discovery + GET)
        )

        t_start = time.monotonic()
        for li, level in enumerate(levels):
            t_change = time.monotonic()
            burner.set_target(level)
            detected_at = None
            timeline = []
            while time.monotonic() - t_change < args.seconds_per_level:
                r = loop.step()
                now = time.monotonic()
                timeline.append({
                    "t": round(now - t_start, 2),
                    "metric": r.metric_value,
                    "replicas": r.replicas,
                    "loop_ms": round(r.total_s * 1e3, 3),
                })
                if (detected_at is None and r.metric_value is not None
                        and abs(r.metric_value - level) <= max(5.0, 0.25 * level)):
                    detected_at = now - t_change
                time.sleep(args.loop_period_s)
            report["levels"].append({
                "target_util_pct": level,
                "detection_latency_s": (round(detected_at, 3)
                                        if detected_at is not None else None),
                "final_metric": timeline[-1]["metric"] if timeline else None,
                "final_replicas": timeline[-1]["replicas"] if timeline else None,
                "samples": timeline[:: max(1, len(timeline) // 10)],
            })
        burner.stop()

    lats = [x["detection_latency_s"] for x in report["levels"][1:]
            if x["detection_latency_s"] is not None]
    report["median_detection_latency_s"] = sorted(lats)[len(lats) // 2] if lats else None
    report["reference_equivalent_s"] = "10-40 (10 s tick + <=30 s rule eval)"
    print(json.dumps(report, indent=1))


if __name__ == "__main__":
    main()
