"""gpu_pruner_amd — MI355X-native Kubernetes idle-GPU culler.

A brand-new AMD-first implementation of the capabilities of
``wseaton/gpu-pruner``: a native C++ culler daemon
(``bin/gpu-pruner``), a first-party ROCm/gfx950 metrics exporter
(``bin/mi355-exporter``), a gfx950 HIP busy-loop self-test probe, and this
Python package providing bindings (`_pruner_core`, `_gpumon`), test fixtures
(fake Prometheus / fake kube-apiserver), and the benchmark harness.
"""

from pathlib import Path

__version__ = "0.1.0"

PKG_DIR = Path(__file__).resolve().parent
REPO_ROOT = PKG_DIR.parent
BIN_DIR = REPO_ROOT / "bin"


def core():
    """Import and return the native _pruner_core module (raises if unbuilt)."""
    from gpu_pruner_amd import _pruner_core

    return _pruner_core


def gpumon():
    """Import and return the native _gpumon module (raises if unbuilt)."""
    from gpu_pruner_amd import _gpumon

    return _gpumon
