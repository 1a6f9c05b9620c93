"""Distributed bench-path test (CPU, gloo, world_size 2).

The driver runs bench.py under torch.distributed.run with one rank per GPU;
this pins the multi-rank path (init, barriers, MAX/SUM reductions, single
rank-0 JSON line) on CPU so it is correct by construction before it ever
reaches an 8-GPU node.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent


def test_bench_two_ranks_gloo():
    env = dict(os.environ)
    env.setdefault("GPU_PRUNER_LOG", "error")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", str(REPO_ROOT / "bench.py"),
         "--gpus", "2", "--steps", "3", "--warmup", "1", "--pods", "100"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO_ROOT), env=env)
    assert r.returncode == 0, r.stderr[-3000:]
    all_lines = r.stdout.strip().splitlines()
    assert len(all_lines) == 1, f"stdout must be exactly one line, got: {r.stdout!r}"
    lines = [l for l in all_lines if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected, got: {r.stdout!r}"
    result = json.loads(lines[0])
    assert result["n_gpus"] == 2
    assert result["scaling"] == "strong"
    # strong scaling: 100 pods split across 2 ranks
    assert result["config"]["pods_per_rank"] == 50
    assert result["config"]["n_pods"] == 100
    assert result["value"] > 0


def test_bench_single_rank_json_contract():
    r = subprocess.run(
        [sys.executable, str(REPO_ROOT / "bench.py"), "--steps", "2",
         "--warmup", "1", "--pods", "50"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO_ROOT))
    assert r.returncode == 0, r.stderr[-2000:]
    result = json.loads(r.stdout.strip().splitlines()[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in result, f"missing contract key {key}"
    assert result["data"] == "synthetic"
    assert result["config"]["p50_tick_latency_ms"] > 0
    assert result["config"]["per_pod_decision_latency_us"] > 0
    # BASELINE config 5 shape by default: OTLP export on + an RTT-bound point
    assert result["config"]["otlp"].startswith("enabled")
    assert result["config"]["rtt_bound"]["apiserver_latency_us"] == 2000
    assert result["config"]["rtt_bound"]["pods_per_sec"] > 0


def test_bench_latency_injection_flag():
    """--latency-us feeds through to the synthetic apiserver."""
    r = subprocess.run(
        [sys.executable, str(REPO_ROOT / "bench.py"), "--steps", "2",
         "--warmup", "1", "--pods", "40", "--latency-us", "2000"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO_ROOT))
    assert r.returncode == 0, r.stderr[-2000:]
    result = json.loads(r.stdout.strip().splitlines()[-1])
    assert result["config"]["apiserver_latency_us"] == 2000
    # with 2ms per request the tick cannot be instant
    assert result["ms_per_step"] > 4
