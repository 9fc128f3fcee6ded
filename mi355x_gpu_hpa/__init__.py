"""mi355x_gpu_hpa — MI355X-native Kubernetes GPU-metric HPA stack.

Built from scratch with the capabilities of ``ashrafgt/k8s-gpu-hpa``
(reference layout in SURVEY.md): a native gfx950 exporter daemon over
rocm_smi (native/exporter/), hand-written CDNA4 HIP load kernels
(native/loadgen/), the drop-in manifest surface (deploy/), and this Python
package: bindings to the native pieces, the stub exporter for GPU-less
integration testing, and the control-plane harness (scraper, recording-rule
evaluator, HPA controller algorithm) used by tests and bench.py.
"""

__version__ = "0.1.0"

from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent
NATIVE_BUILD = REPO_ROOT / "native" / "build"
