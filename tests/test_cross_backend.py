"""Cross-implementation check: the engine must reach identical decisions
against the Python fixture apiserver (tests) and the native C++ synthetic
backend (bench harness) for the same cluster shape — guarding against either
fake drifting from real apiserver semantics.
"""

import json
import os

import pytest


def test_engine_outcomes_match_across_backends(core, monkeypatch):
    from gpu_pruner_amd.fixtures import FakeApiServer, FakePrometheus, build_synthetic_cluster

    n_pods, per_parent = 30, 2
    monkeypatch.setenv("PROMETHEUS_TOKEN", "t")

    # --- python fixtures ---
    with FakeApiServer() as api, FakePrometheus() as prom:
        info = build_synthetic_cluster(api, prom, n_pods=n_pods,
                                       pods_per_parent=per_parent)
        monkeypatch.setenv("GPU_PRUNER_K8S_URL", api.url)
        cfg_py = json.dumps({"duration": 30, "grace_period": 300,
                             "run_mode": "scale-down",
                             "prometheus_url": prom.url})
        out_py = core.run_tick(cfg_py)

    # --- native synthetic backend (same generator parameters) ---
    b = core.SyntheticBackend(n_pods=n_pods, pods_per_parent=per_parent)
    b.start()
    try:
        monkeypatch.setenv("GPU_PRUNER_K8S_URL", b.k8s_url)
        cfg_cc = json.dumps({"duration": 30, "grace_period": 300,
                             "run_mode": "scale-down",
                             "prometheus_url": b.prom_url})
        out_cc = core.run_tick(cfg_cc)
    finally:
        b.stop()

    assert out_py["num_unique_pods"] == out_cc["num_unique_pods"] == n_pods
    assert (out_py["shutdown_events"] == out_cc["shutdown_events"]
            == info["expected_shutdown_events"])
    assert out_py["scaled"] == out_cc["scaled"] == info["expected_shutdown_events"]
