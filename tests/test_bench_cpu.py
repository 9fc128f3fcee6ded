"""CPU coverage of bench.py's contract, including the multi-process
(torch.distributed, gloo, world_size 2) launch path the driver uses."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _run(cmd, timeout=240):
    return subprocess.run(
        cmd, cwd=REPO, capture_output=True, timeout=timeout,
        env={**os.environ, "PYTHONUNBUFFERED": "1"},
    )


def _last_json_line(out: bytes):
    for line in reversed(out.decode().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out.decode()[-2000:]}")


def check_contract(r, n_gpus):
    assert r["metric"] == "p50_scrape_to_hpa_scale_latency"
    assert r["unit"] == "ms"
    assert r["n_gpus"] == n_gpus
    assert r["higher_is_better"] is False
    assert r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert r["value"] > 0
    assert r["ms_per_step"] > 0
    assert r["vs_baseline"] is None  # reference publishes no latency number


def test_bench_single():
    p = _run([sys.executable, "bench.py", "--steps", "10", "--warmup", "2"])
    assert p.returncode == 0, p.stderr.decode()[-2000:]
    r = _last_json_line(p.stdout)
    check_contract(r, 1)
    # with the mock exporter the whole loop must be comfortably sub-10ms
    assert r["value"] < 100
    # self-evidencing fields (round-1 verdict item 3)
    cfg = r["config"]
    assert "observed_busy_pct" in cfg
    assert "load_step_detection_s" in cfg
    # tick alignment: every timed step must have seen a fresh exporter
    # sample (the mock exporter ticks reliably on CPU)
    assert cfg["tick_aligned_steps"] == 10
    assert cfg["stale_steps"] == 0
    # ms_per_step is now tick-bound: ~the exporter interval (100 ms), far
    # above the per-cycle latency — the freshness-honest shape
    assert r["ms_per_step"] >= 50


def test_bench_world2_gloo():
    p = _run([
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", "29517", "bench.py",
        "--gpus", "2", "--steps", "5", "--warmup", "2",
    ])
    assert p.returncode == 0, p.stderr.decode()[-2000:]
    r = _last_json_line(p.stdout)
    check_contract(r, 2)


def test_bench_world8_gloo():
    """The exact N=8 launch shape the driver uses for SCALE (round-1
    verdict item 9: keep multi-GPU readiness warm — world-8 rendezvous,
    8-device mock exporter, per-rank loop roles must not bit-rot)."""
    p = _run([
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
        "--master-port", "29519", "bench.py",
        "--gpus", "8", "--steps", "3", "--warmup", "1",
    ], timeout=420)
    assert p.returncode == 0, p.stderr.decode()[-2000:]
    r = _last_json_line(p.stdout)
    check_contract(r, 8)
    assert r["config"]["gpus_exported"] in (0, 8)  # 8 once on-GPU validated
