"""HIP busy-probe loader (gfx950 self-test fixture).

ctypes wrapper over ``bin/libbusy_probe.so`` (native/probe/busy_probe.hip).
Fails loudly if the shared object is missing on a GPU host — the probe is the
fixture proving the sampler's counter semantics (util > 0 under load, == 0
idle), so silently skipping it would hide a broken native path.
"""

from __future__ import annotations

import ctypes
from pathlib import Path

from gpu_pruner_amd import BIN_DIR

_LIB = None


def _lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is None:
        path = BIN_DIR / "libbusy_probe.so"
        if not path.exists():
            raise FileNotFoundError(
                f"{path} not built — run `make -C native probe` (hipcc --offload-arch=gfx950)"
            )
        lib = ctypes.CDLL(str(path))
        lib.busy_probe_start.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_double]
        lib.busy_probe_start.restype = ctypes.c_int
        lib.busy_probe_stop.restype = ctypes.c_int
        lib.busy_probe_run_for_ms.argtypes = [ctypes.c_int, ctypes.c_int]
        lib.busy_probe_run_for_ms.restype = ctypes.c_int
        lib.busy_probe_device_count.restype = ctypes.c_int
        lib.busy_probe_last_error.restype = ctypes.c_char_p
        _LIB = lib
    return _LIB


def device_count() -> int:
    return _lib().busy_probe_device_count()


def _check(rc: int):
    if rc != 0:
        raise RuntimeError(f"busy probe error: {_lib().busy_probe_last_error().decode()}")


def start(device: int = 0, blocks: int = 0, max_seconds: float = 60.0):
    """Launch the persistent wave64 FMA spin kernel (self-bounded)."""
    _check(_lib().busy_probe_start(device, blocks, max_seconds))


def stop():
    _check(_lib().busy_probe_stop())


def run_for_ms(device: int = 0, ms: int = 1000):
    """Blocking full-load burst."""
    _check(_lib().busy_probe_run_for_ms(device, ms))


class busy_load:
    """Context manager: GPU under full load inside the block."""

    def __init__(self, device: int = 0, max_seconds: float = 60.0):
        self.device = device
        self.max_seconds = max_seconds

    def __enter__(self):
        start(self.device, 0, self.max_seconds)
        return self

    def __exit__(self, *exc):
        stop()
