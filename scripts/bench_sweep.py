#!/usr/bin/env python3
"""Decision-throughput sweep: concurrency × injected apiserver latency.

Quantifies the headroom over the reference's implied envelope (BASELINE.md:
10 in-flight pod evaluations hard-cap, ~1-3 API round-trips per pod): with
realistic apiserver RTTs the configurable worker pool should scale decision
throughput roughly linearly until RTT×concurrency saturates.

Writes one JSON line per cell to stdout; run on any box
(`python scripts/bench_sweep.py [--pods N]`).
"""

import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
os.environ.setdefault("GPU_PRUNER_LOG", "error")

from gpu_pruner_amd import _pruner_core as core  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pods", type=int, default=1000)
    ap.add_argument("--trials", type=int, default=3)
    ap.add_argument("--latencies-us", type=int, nargs="*",
                    default=[0, 500, 2000])  # 0 / 0.5ms / 2ms RTT
    ap.add_argument("--concurrencies", type=int, nargs="*",
                    default=[1, 10, 32, 64, 128])
    args = ap.parse_args()

    os.environ["PROMETHEUS_TOKEN"] = "sweep"
    results = []
    for lat in args.latencies_us:
        backend = core.SyntheticBackend(n_pods=args.pods, latency_us=lat)
        backend.start()
        os.environ["GPU_PRUNER_K8S_URL"] = backend.k8s_url
        for conc in args.concurrencies:
            cfg = json.dumps({
                "duration": 30, "grace_period": 300, "run_mode": "scale-down",
                "prometheus_url": backend.prom_url, "max_concurrency": conc,
            })
            core.run_tick(cfg)  # warmup
            best = float("inf")
            for _ in range(args.trials):
                t0 = time.perf_counter()
                out = core.run_tick(cfg)
                best = min(best, time.perf_counter() - t0)
            row = {
                "latency_us": lat, "concurrency": conc, "n_pods": args.pods,
                "tick_ms": round(best * 1000, 2),
                "pods_per_sec": round(out["num_unique_pods"] / best, 1),
            }
            results.append(row)
            print(json.dumps(row), flush=True)
        backend.stop()

    # summary: speedup of best concurrency vs the reference's cap of 10
    print("\n# speedup vs concurrency=10 (the reference's hard cap):",
          file=sys.stderr)
    for lat in args.latencies_us:
        rows = [r for r in results if r["latency_us"] == lat]
        base = next(r for r in rows if r["concurrency"] == 10)
        best = max(rows, key=lambda r: r["pods_per_sec"])
        print(f"#   latency={lat}us: cap10={base['pods_per_sec']:.0f} pods/s, "
              f"best(c={best['concurrency']})={best['pods_per_sec']:.0f} pods/s "
              f"({best['pods_per_sec']/base['pods_per_sec']:.2f}x)", file=sys.stderr)


if __name__ == "__main__":
    main()
