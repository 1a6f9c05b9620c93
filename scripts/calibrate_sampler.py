#!/usr/bin/env python3
"""Sampler calibration on a real MI355X: prints idle and under-load counter
traces and derives the gfx_activity_acc accumulation rate, validating the
windowed-ratio math in native/exporter/sampler.cpp (Δacc / (Δfw_ts·100)).

Run on a GPU box:  python scripts/calibrate_sampler.py
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from gpu_pruner_amd import _gpumon, probe  # noqa: E402


def line(d):
    return (f"busy={d['busy_percent']:5.1f} ratio={d['gr_engine_active']:.4f} "
            f"acc={d['gfx_activity_acc']} fw_ts={d['firmware_timestamp']} "
            f"power={d['power_w']:.0f}W clk={d['gfx_clock_mhz']}MHz")


def main():
    s = _gpumon.Sampler(poll_interval_ms=100)
    s.init()
    print("devices:", s.device_count)
    for d in s.snapshot():
        print({k: d[k] for k in ("index", "model_name", "unique_id", "pci_bdf",
                                 "drm_render_minor", "kfd_gpu_id", "vram_total_b")})
    print("--- idle 2s ---")
    for _ in range(4):
        time.sleep(0.5)
        s.poll_once()
        print(line(s.snapshot()[0]))

    print("--- probe load 3s ---")
    probe.start(0, 0, 30.0)
    acc0 = None
    try:
        t0 = time.time()
        while time.time() - t0 < 3:
            time.sleep(0.25)
            s.poll_once()
            d = s.snapshot()[0]
            if acc0 is None and d["busy_percent"] > 90:
                acc0 = (d["gfx_activity_acc"], d["firmware_timestamp"])
            print(line(d))
    finally:
        probe.stop()
    d = s.snapshot()[0]
    if acc0:
        dacc = d["gfx_activity_acc"] - acc0[0]
        dfw = d["firmware_timestamp"] - acc0[1]
        # gfx950 firmware_timestamp ticks in ns (header says 10 ns; verified
        # against wall time in profiles/raw/calibration.log)
        dfw_s = dfw * 1e-9
        rate = dacc / max(dfw_s, 1e-9)
        print(f"calib: dacc={dacc} dfw_ns={dfw} rate={rate:.1f}/s at 100% busy "
              f"(sampler model uses 100000/s)")
    print("--- settle ---")
    for _ in range(10):
        time.sleep(0.5)
        s.poll_once()
        d = s.snapshot()[0]
        print(f"busy={d['busy_percent']:5.1f} ratio={d['gr_engine_active']:.4f}")
    s.stop()


if __name__ == "__main__":
    main()
