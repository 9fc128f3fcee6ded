#!/usr/bin/env python3
"""endurance_gpu.py — everything-at-once sustained run on a real MI355X.

One exporter daemon (auto backend, DaemonSet-default 1 s cadence, FULL
metric surface, kubelet attribution against a real-protocol fake kubelet
serving the device's actual renderD id) + the adapter-routed control loop,
while the load alternates through every burn the stack ships:

    bf16 MFMA burn -> fp8 MFMA burn -> streaming-triad HBM burn -> idle

Per phase it records the loop's metric trajectory, detection latency of
the phase change, scrape failures, attribution presence and exporter RSS.
Usage: python tools/endurance_gpu.py [--minutes-per-phase 3] [--cycles 1]
"""

import argparse
import ctypes
import json
import sys
import threading
import time
from concurrent import futures
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa import loadgen  # noqa: E402
from mi355x_gpu_hpa.control import (  # noqa: E402
    ControlLoop,
    HpaSpec,
    Scraper,
    ScrapeTarget,
    parse_prometheus_text,
    synth_pod_labels,
)
from mi355x_gpu_hpa.exporter import ExporterProcess  # noqa: E402

from mi355x_gpu_hpa.podresources_wire import (  # noqa: E402
    container,
    container_devices,
    list_response,
    pod,
)


def rss_kb(pid):
    try:
        with open(f"/proc/{pid}/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    return int(line.split()[1])
    except OSError:
        return -1
    return -1


class Burner:
    def __init__(self):
        self.thread = None
        self.flag = None

    def start(self, kind, util):
        self.stop()
        if kind == "idle":
            return
        self.flag = ctypes.c_int(0)
        flag = self.flag
        lib = loadgen._load()

        def run():
            if kind == "bf16":
                lib.lg_gemm_burn(0, ctypes.c_double(util),
                                 ctypes.c_double(3600.0), 4096, 4096, 4096,
                                 ctypes.c_double(50.0), ctypes.byref(flag))
            elif kind == "fp8":
                lib.lg_gemm_fp8_burn(0, ctypes.c_double(util),
                                     ctypes.c_double(3600.0), 4096, 4096,
                                     4096, ctypes.c_double(50.0),
                                     ctypes.byref(flag))
            elif kind == "triad":
                g = ctypes.c_double()
                lib.lg_bw_burn(0, ctypes.c_double(util),
                               ctypes.c_double(3600.0), ctypes.c_double(6.0),
                               ctypes.c_double(100.0), ctypes.byref(flag),
                               ctypes.byref(g))

        self.thread = threading.Thread(target=run, daemon=True)
        self.thread.start()

    def stop(self):
        if self.flag is not None:
            self.flag.value = 1
        if self.thread is not None:
            self.thread.join(timeout=20)
        self.thread = None
        self.flag = None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes-per-phase", type=float, default=3.0)
    ap.add_argument("--cycles", type=int, default=1)
    ap.add_argument("--interval-ms", type=float, default=1000.0)
    ap.add_argument("--util", type=float, default=60.0)
    args = ap.parse_args()

    grpc = __import__("grpc")

    # learn the real device identity, then serve it from the fake kubelet
    with ExporterProcess(interval_ms=200) as probe:
        time.sleep(0.6)
        samples = parse_prometheus_text(probe.scrape())
    render = next(s.labels["device"] for s in samples
                  if s.name == "dcgm_gpu_utilization"
                  and s.labels["gpu"] == "0")

    response = list_response([
        pod("cuda-test-0", "default", [
            container("main", [container_devices("amd.com/gpu", [render])]),
        ]),
    ])

    class Handler(grpc.GenericRpcHandler):
        def service(self, hcd):
            if hcd.method == "/v1.PodResourcesLister/List":
                return grpc.unary_unary_rpc_method_handler(
                    lambda req, ctx: response,
                    request_deserializer=None, response_serializer=None)
            return None

    sockdir = Path("/tmp/endurance-kubelet")
    sockdir.mkdir(exist_ok=True)
    sock = sockdir / "kubelet.sock"
    if sock.exists():
        sock.unlink()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    server.add_generic_rpc_handlers((Handler(),))
    server.add_insecure_port(f"unix:{sock}")
    server.start()

    phases = ["bf16", "fp8", "triad", "idle"] * args.cycles
    report = {"interval_ms": args.interval_ms, "util": args.util,
              "render": render, "phases": []}
    burner = Burner()
    try:
        with ExporterProcess(interval_ms=args.interval_ms, kubernetes=True,
                             pod_resources_socket=str(sock)) as exp:
            scraper = Scraper([ScrapeTarget(exp.url, node="node0")])
            loop = ControlLoop(
                scraper,
                hpa_spec=HpaSpec(min_replicas=1, max_replicas=8,
                                 target_value=5.0,
                                 downscale_stabilization_s=30.0),
                extra_samples=lambda: synth_pod_labels(["cuda-test-0"]),
                use_adapter=True,
            )
            scrape_failures = 0
            for phase in phases:
                burner.start(phase, args.util)
                t0 = time.monotonic()
                vals, attributed, detected = [], 0, None
                steps = 0
                while time.monotonic() - t0 < args.minutes_per_phase * 60:
                    r = loop.step()
                    steps += 1
                    if r.metric_value is None:
                        scrape_failures += 1
                    else:
                        vals.append(r.metric_value)
                        target = 0.0 if phase == "idle" else args.util
                        if (detected is None
                                and abs(r.metric_value - target)
                                <= max(8.0, 0.25 * target)):
                            detected = time.monotonic() - t0
                    if any(s.labels.get("pod") == "cuda-test-0"
                           for s in scraper.last.get(exp.url, [])):
                        attributed += 1
                    time.sleep(1.0)
                settled = vals[len(vals) // 2:]
                report["phases"].append({
                    "phase": phase,
                    "steps": steps,
                    "detected_s": round(detected, 2) if detected else None,
                    "metric_mean_settled": (round(sum(settled) / len(settled), 2)
                                            if settled else None),
                    "attributed_steps": attributed,
                    "replicas": loop.hpa_state.current_replicas,
                    "exporter_rss_kb": rss_kb(exp.proc.pid),
                })
                print(f"[endurance] {phase}: {report['phases'][-1]}",
                      file=sys.stderr)
            report["scrape_failures"] = scrape_failures
    finally:
        burner.stop()
        server.stop(0)
    print(json.dumps(report, indent=1))


if __name__ == "__main__":
    main()
