"""The native paths must fail LOUDLY when their components are missing —
a silent fallback would let GPU tests pass without the native code
(exactly what the round-end native-code-loaded check exists to catch)."""

import os
import subprocess

import pytest

from mi355x_gpu_hpa.exporter import EXPORTER_BIN


def test_loadgen_raises_without_library(monkeypatch):
    monkeypatch.setenv("MI355X_LOADGEN_LIB", "/nonexistent/lib.so")
    import importlib

    import mi355x_gpu_hpa.loadgen as lg

    importlib.reload(lg)
    try:
        assert not lg.available()
        with pytest.raises(lg.LoadgenError, match="not found"):
            lg.device_count()
    finally:
        monkeypatch.delenv("MI355X_LOADGEN_LIB")
        importlib.reload(lg)


@pytest.mark.skipif(not os.path.exists(EXPORTER_BIN), reason="not built")
def test_exporter_refuses_to_fake_gpus_without_backend():
    """Without --mock and without a GPU, the exporter exits rc=3 with a
    clear error instead of serving fabricated data."""
    import torch

    if torch.cuda.is_available():
        pytest.skip("real GPU present; rsmi backend would start")
    r = subprocess.run([EXPORTER_BIN, "-c", "100", "-l", "127.0.0.1:0"],
                       capture_output=True, timeout=30)
    assert r.returncode == 3
    assert b"no GPU backend available" in r.stderr
