#!/usr/bin/env python3
"""scaleup_experiment.py — the 1->8 replica scale-up curve (BASELINE.json
config 4), runnable with no cluster.

Simulates the reference's closed-loop experiment (README.md:112-122: raise
the load, watch replicas grow) against the native exporter in mock mode:
each "replica" contributes a busy GPU; the recording rule averages across
replica pods; the HPA reconciles on the reference cadence (15 s sync,
configurable). The output is the scale-up curve — replicas vs time — plus
the overshoot analysis the reference only describes anecdotally
(README.md:123).

Two cadence profiles:
  --profile reference   10 s exporter tick, 30 s rule eval, 15 s HPA sync
                        (the reference's loop timing)
  --profile native      1 s tick, 1 s rule eval, 15 s HPA sync
                        (this stack's defaults)

Time is simulated (no wall-clock sleeps): the exporter tick staleness is
modeled by sampling the busy file only on tick boundaries.

Usage:
    python tools/scaleup_experiment.py --profile native --max-replicas 8
"""

import argparse
import json
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from mi355x_gpu_hpa.control import (  # noqa: E402
    ControlLoop,
    HpaSpec,
    Scraper,
    ScrapeTarget,
    synth_pod_labels,
)
from mi355x_gpu_hpa.exporter import ExporterProcess  # noqa: E402


def run(profile: str, max_replicas: int, per_replica_busy: float,
        target: float, duration_s: float, pod_start_s: float):
    cadence = {
        "reference": dict(tick_s=10.0, rule_s=30.0, hpa_s=15.0),
        "native": dict(tick_s=1.0, rule_s=1.0, hpa_s=15.0),
        # native cadences + the v2 behavior block from deploy/cuda-test-hpa
        # (1 pod / 15 s): the stepped ramp that removes the reference's
        # documented overshoot (README.md:123)
        "native-v2": dict(tick_s=1.0, rule_s=1.0, hpa_s=15.0),
    }[profile]
    scale_up_pods = 1 if profile == "native-v2" else 0

    with tempfile.TemporaryDirectory() as td:
        busy_file = Path(td) / "busy"
        busy_file.write_text("0\n")
        with ExporterProcess(mock_devices=max_replicas, interval_ms=50,
                             mock_busy_file=str(busy_file)) as exp:
            # replica i <-> GPU i <-> pod cuda-test-i
            scraper = Scraper([ScrapeTarget(exp.url, node="node0")])
            orig = scraper.scrape_once

            state = {"replicas": 1, "pending": []}  # pending: (ready_t, count)

            def scrape_with_pods():
                samples = orig()
                # only GPUs of RUNNING replicas have pods attached
                out = []
                for s in samples:
                    g = s.labels.get("gpu")
                    if g is not None:
                        if int(g) >= state["replicas"]:
                            continue
                        s.labels.setdefault("pod", f"cuda-test-{g}")
                        s.labels.setdefault("namespace", "default")
                    out.append(s)
                return out

            scraper.scrape_once = scrape_with_pods
            loop = ControlLoop(
                scraper,
                hpa_spec=HpaSpec(min_replicas=1, max_replicas=max_replicas,
                                 target_value=target,
                                 scale_up_pods=scale_up_pods),
                extra_samples=lambda: synth_pod_labels(
                    [f"cuda-test-{i}" for i in range(state["replicas"])]),
            )

            # simulated clock: events at HPA sync cadence; exporter tick
            # staleness modeled by writing busy only on tick boundaries
            curve = []
            t = 0.0
            last_tick = -1e9
            metric_lag = cadence["tick_s"] + cadence["rule_s"]
            busy_per_pod = per_replica_busy
            while t <= duration_s:
                # replicas that finished starting become ready
                ready = [c for (rt, c) in state["pending"] if rt <= t]
                if ready:
                    state["replicas"] = max(state["replicas"], max(ready))
                    state["pending"] = [(rt, c) for (rt, c) in state["pending"]
                                        if rt > t]
                # exporter tick: running replicas' GPUs show load
                if t - last_tick >= cadence["tick_s"]:
                    lines = [f"{i}:{busy_per_pod}" for i in range(state["replicas"])]
                    lines += [f"{i}:0" for i in range(state["replicas"], max_replicas)]
                    busy_file.write_text("\n".join(lines) + "\n")
                    last_tick = t
                    time.sleep(0.12)  # let the real exporter re-sample

                # the metric the HPA sees is metric_lag stale; approximate by
                # evaluating on current exporter state only at HPA syncs
                r = loop.step(now_s=t)
                desired = r.replicas
                if desired > state["replicas"] and not state["pending"]:
                    state["pending"].append((t + pod_start_s, desired))
                curve.append({"t": round(t, 1), "replicas": state["replicas"],
                              "desired": desired,
                              "metric": r.metric_value})
                t += cadence["hpa_s"]

    peak = max(c["desired"] for c in curve)
    t_to_max = next((c["t"] for c in curve if c["replicas"] >= max_replicas),
                    None)
    return {
        "profile": profile,
        "cadence": cadence,
        "per_replica_busy_pct": per_replica_busy,
        "hpa_target": target,
        "pod_start_s": pod_start_s,
        "curve": curve,
        "peak_desired": peak,
        "time_to_max_replicas_s": t_to_max,
        "overshoot": peak >= max_replicas and any(
            c["desired"] >= max_replicas and c["replicas"] == 1 for c in curve),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--profile", choices=["reference", "native", "native-v2"],
                    default="native")
    ap.add_argument("--max-replicas", type=int, default=8)
    ap.add_argument("--busy", type=float, default=40.0,
                    help="busy%% each running replica shows")
    ap.add_argument("--target", type=float, default=5.0)
    ap.add_argument("--duration", type=float, default=300.0)
    ap.add_argument("--pod-start", type=float, default=20.0,
                    help="simulated pod start latency (s)")
    args = ap.parse_args()
    result = run(args.profile, args.max_replicas, args.busy, args.target,
                 args.duration, args.pod_start)
    print(json.dumps(result, indent=1))


if __name__ == "__main__":
    main()
