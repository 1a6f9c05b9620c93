"""Shared pytest config for the MI355X-native gpu-pruner.

Markers:
  gpu — needs a real MI355X (run via gpurun / the driver); everything else
        runs hermetically on CPU.
"""

import os
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (gfx950)")


@pytest.fixture
def fake_prom():
    from gpu_pruner_amd.fixtures import FakePrometheus

    with FakePrometheus() as p:
        yield p


@pytest.fixture
def fake_api(monkeypatch):
    """Fake kube-apiserver with the env override pointed at it."""
    from gpu_pruner_amd.fixtures import FakeApiServer

    with FakeApiServer() as a:
        monkeypatch.setenv("GPU_PRUNER_K8S_URL", a.url)
        monkeypatch.delenv("GPU_PRUNER_K8S_TOKEN", raising=False)
        monkeypatch.delenv("GPU_PRUNER_K8S_TOKEN_FILE", raising=False)
        monkeypatch.delenv("GPU_PRUNER_K8S_CA", raising=False)
        yield a


@pytest.fixture
def core():
    from gpu_pruner_amd import _pruner_core

    return _pruner_core


@pytest.fixture
def pruner_bin():
    path = REPO_ROOT / "bin" / "gpu-pruner"
    if not path.exists():
        pytest.skip("bin/gpu-pruner not built (make -C native bins)")
    return str(path)
