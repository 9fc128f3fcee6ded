"""stub.py — pure-Python stub exporter (BASELINE.json config 1).

A dependency-free /metrics endpoint emitting the exact dcgm_* 1.x-compat
label schema of the native exporter, for the CPU-only kind-cluster
integration harness (deploy/kind/): it lets kube-prometheus-stack +
prometheus-adapter + the HPA be exercised with zero GPUs and zero native
code in the pod image (`python3 -m mi355x_gpu_hpa.exporter.stub`).

Busy% control: env STUB_BUSY (static), or POST /busy with a float body
(scriptable step-changes for scale-up tests — the integration test drives
the HPA through its trigger threshold this way).
"""

from __future__ import annotations

import os
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class _State:
    def __init__(self, n_gpus: int, busy: float):
        self.n_gpus = n_gpus
        self.busy = busy
        self.lock = threading.Lock()

    def render(self) -> str:
        pod = os.environ.get("POD_NAME", "")
        ns = os.environ.get("POD_NAMESPACE", "")
        k8s_labels = ""
        if pod:
            k8s_labels = f',container="main",namespace="{ns or "default"}",pod="{pod}"'
        out = [
            "# HELP dcgm_gpu_utilization GPU utilization (%).",
            "# TYPE dcgm_gpu_utilization gauge",
        ]
        with self.lock:
            busy, n = self.busy, self.n_gpus
        for i in range(n):
            base = (
                f'gpu="{i}",uuid="stub-{i:016x}",device="renderD{128+i}",'
                f'modelName="AMD Instinct MI355X (stub)"{k8s_labels}'
            )
            out.append(f"dcgm_gpu_utilization{{{base}}} {busy:g}")
        out += ["# HELP dcgm_gpu_temp GPU temperature (in C).",
                "# TYPE dcgm_gpu_temp gauge"]
        for i in range(n):
            base = (
                f'gpu="{i}",uuid="stub-{i:016x}",device="renderD{128+i}",'
                f'modelName="AMD Instinct MI355X (stub)"{k8s_labels}'
            )
            out.append(f"dcgm_gpu_temp{{{base}}} {40+i}")
        return "\n".join(out) + "\n"


def serve(port: int = 9400, n_gpus: int = 1, busy: float = 0.0):
    state = _State(n_gpus, busy)

    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):  # quiet
            pass

        def do_GET(self):
            if self.path.startswith("/metrics"):
                body = state.render().encode()
                self.send_response(200)
                self.send_header("Content-Type",
                                 "text/plain; version=0.0.4; charset=utf-8")
            elif self.path.startswith(("/healthz", "/readyz")):
                body = b"ok\n"
                self.send_response(200)
                self.send_header("Content-Type", "text/plain")
            else:
                body = b"see /metrics\n"
                self.send_response(404)
                self.send_header("Content-Type", "text/plain")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_POST(self):
            if self.path.startswith("/busy"):
                n = int(self.headers.get("Content-Length", 0))
                try:
                    with state.lock:
                        state.busy = float(self.rfile.read(n))
                    self.send_response(200)
                except ValueError:
                    self.send_response(400)
            else:
                self.send_response(404)
            self.send_header("Content-Length", "0")
            self.end_headers()

    srv = ThreadingHTTPServer(("0.0.0.0", port), Handler)
    return srv, state


def main():
    import argparse

    ap = argparse.ArgumentParser(description="stub dcgm_*-schema exporter")
    ap.add_argument("--port", type=int, default=int(os.environ.get("STUB_PORT", 9400)))
    ap.add_argument("--gpus", type=int, default=int(os.environ.get("STUB_GPUS", 1)))
    ap.add_argument("--busy", type=float,
                    default=float(os.environ.get("STUB_BUSY", 0.0)))
    args = ap.parse_args()
    srv, _ = serve(args.port, args.gpus, args.busy)
    print(f"stub exporter on :{args.port} ({args.gpus} GPUs, busy={args.busy}%)",
          flush=True)
    srv.serve_forever()


if __name__ == "__main__":
    main()
