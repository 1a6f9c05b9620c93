"""Lookback-window semantics, end to end.

MiniProm *executes* the culler's wire query over raw time-series, so these
tests pin the one semantic no canned fixture can: ``max_over_time`` —
being idle right now is not enough; the pod must have been idle for the
WHOLE window. This is the product's core safety property (never cull a
recently-active workload).
"""

import os
import subprocess

import pytest

from gpu_pruner_amd.fixtures.miniprom import MiniProm


@pytest.fixture
def miniprom():
    with MiniProm() as p:
        yield p


def run_pruner(pruner_bin, fake_api, prom_url, *args, timeout=30):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    return subprocess.run([pruner_bin, "--prometheus-url", prom_url, *args],
                          capture_output=True, text=True, timeout=timeout, env=env)


def add_deployment_pod(fake_api, name, ns="ml"):
    dep = fake_api.add_deployment(name, ns)
    rs = fake_api.add_replicaset(f"{name}-rs", ns, owner=dep)
    fake_api.add_pod(f"{name}-0", ns, owner_kind="ReplicaSet", owner_name=f"{name}-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    return dep


def test_fully_idle_pod_is_culled(pruner_bin, fake_api, miniprom):
    add_deployment_pod(fake_api, "idle")
    # samples across the whole 30m window, all zero
    for age in (1700, 1200, 600, 10):
        miniprom.ingest_activity("idle-0", "ml", 0.0, age_s=age)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "idle")["spec"]["replicas"] == 0


def test_recently_busy_pod_is_not_culled(pruner_bin, fake_api, miniprom):
    """Idle NOW but busy 10 minutes ago: peak over the window > 0 → safe."""
    add_deployment_pod(fake_api, "recent")
    miniprom.ingest_activity("recent-0", "ml", 0.0, age_s=1700)
    miniprom.ingest_activity("recent-0", "ml", 0.85, age_s=600)  # busy burst
    miniprom.ingest_activity("recent-0", "ml", 0.0, age_s=5)     # idle now
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "recent")["spec"]["replicas"] == 1
    assert fake_api.events == []


def test_activity_outside_window_does_not_protect(pruner_bin, fake_api, miniprom):
    """Busy 45 minutes ago with --duration 30: outside the window → culled."""
    add_deployment_pod(fake_api, "old-burst")
    miniprom.ingest_activity("old-burst-0", "ml", 0.9, age_s=45 * 60)
    miniprom.ingest_activity("old-burst-0", "ml", 0.0, age_s=600)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "old-burst")["spec"]["replicas"] == 0


def test_shorter_duration_flag_shrinks_window(pruner_bin, fake_api, miniprom):
    """-t 5: a burst 10 minutes ago no longer protects the pod."""
    add_deployment_pod(fake_api, "shortwin")
    miniprom.ingest_activity("shortwin-0", "ml", 0.85, age_s=600)
    miniprom.ingest_activity("shortwin-0", "ml", 0.0, age_s=60)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down",
                   "-t", "5", "--grace-period", "0")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "shortwin")["spec"]["replicas"] == 0


def test_gpu_util_fallback_metric(pruner_bin, fake_api, miniprom):
    """Series only in DCGM_FI_DEV_GPU_UTIL (0-100) still drive decisions."""
    add_deployment_pod(fake_api, "fallback")
    miniprom.ingest("DCGM_FI_DEV_GPU_UTIL", {
        "Hostname": "node-0", "exported_pod": "fallback-0",
        "exported_namespace": "ml", "exported_container": "main",
        "gpu": "0", "modelName": "AMD Instinct MI355X"}, 0.0, age_s=60)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "fallback")["spec"]["replicas"] == 0


def test_power_threshold_vetoes_zero_compute(pruner_bin, fake_api, miniprom):
    """Compute idle all window, but peak power 420 W: --power-threshold 300
    excludes the pod (corroborating-signal semantics)."""
    add_deployment_pod(fake_api, "hotidle")
    miniprom.ingest_activity("hotidle-0", "ml", 0.0, age_s=600)
    miniprom.ingest_power("hotidle-0", "ml", 420.0, age_s=500)
    add_deployment_pod(fake_api, "coldidle")
    miniprom.ingest_activity("coldidle-0", "ml", 0.0, age_s=600)
    miniprom.ingest_power("coldidle-0", "ml", 250.0, age_s=500)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down",
                   "--power-threshold", "300")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "hotidle")["spec"]["replicas"] == 1
    assert fake_api.get("Deployment", "ml", "coldidle")["spec"]["replicas"] == 0


def test_namespace_filter_applies_to_series(pruner_bin, fake_api, miniprom):
    add_deployment_pod(fake_api, "inns", ns="ml-team-a")
    add_deployment_pod(fake_api, "outns", ns="web")
    miniprom.ingest_activity("inns-0", "ml-team-a", 0.0, age_s=60)
    miniprom.ingest_activity("outns-0", "web", 0.0, age_s=60)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down",
                   "--namespace", "ml-team-.*")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml-team-a", "inns")["spec"]["replicas"] == 0
    assert fake_api.get("Deployment", "web", "outns")["spec"]["replicas"] == 1


def test_honor_labels_end_to_end(pruner_bin, fake_api, miniprom):
    add_deployment_pod(fake_api, "native")
    miniprom.ingest_activity("native-0", "ml", 0.0, age_s=60, honor_labels=True)
    r = run_pruner(pruner_bin, fake_api, miniprom.url, "--run-mode", "scale-down",
                   "--honor-labels")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("Deployment", "ml", "native")["spec"]["replicas"] == 0
