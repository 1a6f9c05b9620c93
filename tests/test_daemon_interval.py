"""Daemon-loop cadence + scale-consumer tests."""

import os
import subprocess
import time

import pytest


def test_daemon_mode_ticks_on_interval(pruner_bin, fake_api, fake_prom):
    """--check-interval=1: at least 2 and at most ~5 queries in ~3.5s."""
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    p = subprocess.Popen(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--daemon-mode",
         "--check-interval", "1"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        # wait until at least 2 ticks landed (loaded CI boxes can stretch
        # startup); then verify the cadence upper bound over the window
        deadline = time.monotonic() + 15
        t0 = time.monotonic()
        while time.monotonic() < deadline and len(fake_prom.queries) < 2:
            time.sleep(0.1)
        elapsed = time.monotonic() - t0
        assert p.poll() is None, "daemon should keep running"
        n = len(fake_prom.queries)
        assert n >= 2, f"expected >=2 ticks at 1s interval within 15s, saw {n}"
        assert n <= elapsed + 3, f"{n} ticks in {elapsed:.1f}s — interval ignored?"
    finally:
        p.kill()
        p.wait()


def test_one_shot_mode_runs_once_and_exits(pruner_bin, fake_api, fake_prom):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run([pruner_bin, "--prometheus-url", fake_prom.url],
                       capture_output=True, text=True, timeout=30, env=env)
    assert r.returncode == 0
    assert len(fake_prom.queries) == 1


@pytest.mark.parametrize("n_pods", [200])
def test_large_cluster_scale_down_complete(pruner_bin, fake_api, fake_prom, n_pods):
    """Stress shape (BASELINE config 5, scaled down for CI): every parent of
    an idle pod is scaled exactly once and announced exactly once."""
    from gpu_pruner_amd.fixtures import build_synthetic_cluster

    info = build_synthetic_cluster(fake_api, fake_prom, n_pods=n_pods,
                                   pods_per_parent=4)
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--run-mode", "scale-down",
         "--max-concurrency", "64"],
        capture_output=True, text=True, timeout=120, env=env)
    assert r.returncode == 0, r.stderr[-1000:]
    assert len(fake_api.events) == info["expected_shutdown_events"]
    scaled = sum(
        1 for (kind, _, _), obj in fake_api.objects.items()
        if (kind in ("Deployment", "StatefulSet") and obj["spec"].get("replicas") == 0)
        or (kind == "Notebook" and "kubeflow-resource-stopped" in obj["metadata"].get("annotations", {}))
        or (kind == "InferenceService" and obj["spec"]["predictor"].get("minReplicas") == 0))
    assert scaled == info["expected_shutdown_events"]


def test_sigterm_graceful_shutdown(pruner_bin, fake_api, fake_prom):
    """K8s pod termination: SIGTERM interrupts the tick wait, drains, exits 0."""
    import signal

    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    p = subprocess.Popen(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--daemon-mode",
         "--check-interval", "180"],  # long interval: the wait must be interruptible
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        time.sleep(1.0)
        assert p.poll() is None
        t0 = time.monotonic()
        p.send_signal(signal.SIGTERM)
        rc = p.wait(timeout=10)
        assert time.monotonic() - t0 < 5, "shutdown should not wait out the interval"
        assert rc == 0
        assert b"draining" in p.stderr.read()
    finally:
        if p.poll() is None:
            p.kill()
            p.wait()


def test_self_metrics_endpoint(pruner_bin, fake_api, fake_prom):
    """--metrics-port serves the counters + /healthz (MI355X-native add)."""
    import socket
    import urllib.request

    # ephemeral port: fixed ports collide when suites run in parallel
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml")
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    p = subprocess.Popen(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--daemon-mode",
         "--check-interval", "1", "--run-mode", "scale-down",
         "--metrics-port", str(port)],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        text = None
        for _ in range(40):
            time.sleep(0.2)
            try:
                text = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
                if "gpu_pruner_query_successes_total" in text:
                    break
            except OSError:
                continue
        assert text and "gpu_pruner_query_successes_total" in text
        assert "gpu_pruner_scale_successes_total" in text
        health = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/healthz", timeout=2).read()
        assert health == b"ok\n"
    finally:
        p.terminate()
        p.wait(timeout=10)


def test_five_thousand_pod_tick_correctness(core, monkeypatch):
    """10x-BASELINE stress (native backend): one tick evaluates 5000 pods and
    culls every parent exactly once — guards against O(n^2) regressions and
    dedup breakage at scale. (Timing is profiled separately; this asserts
    correctness only.)"""
    import json as _json

    monkeypatch.setenv("PROMETHEUS_TOKEN", "t")
    b = core.SyntheticBackend(n_pods=5000)
    b.start()
    try:
        monkeypatch.setenv("GPU_PRUNER_K8S_URL", b.k8s_url)
        cfg = _json.dumps({"duration": 30, "grace_period": 300,
                           "run_mode": "scale-down",
                           "prometheus_url": b.prom_url, "max_concurrency": 32})
        out = core.run_tick(cfg)
        assert out["num_unique_pods"] == 5000
        assert out["shutdown_events"] == 2500
        assert out["scaled"] == 2500
        assert b.events_posted == 2500
    finally:
        b.stop()


def test_latency_injected_tick_uses_concurrency(core, monkeypatch):
    """Catastrophic-concurrency-regression guard: with 2 ms injected RTT and
    200 pods, a serial engine would need >= 200 x 3 x 2 ms ~ 1.2 s per tick;
    the concurrent engine must finish far faster (very loose bound to stay
    flake-free in CI)."""
    import json as _json
    import time as _time

    monkeypatch.setenv("PROMETHEUS_TOKEN", "t")
    b = core.SyntheticBackend(n_pods=200, latency_us=2000)
    b.start()
    try:
        monkeypatch.setenv("GPU_PRUNER_K8S_URL", b.k8s_url)
        cfg = _json.dumps({"duration": 30, "grace_period": 300,
                           "run_mode": "scale-down",
                           "prometheus_url": b.prom_url, "max_concurrency": 32,
                           "eval_strategy": "get"})
        core.run_tick(cfg)  # warmup
        # best of 3 to shrug off CI load spikes; a serial engine needs
        # >= 1.2 s regardless
        dt = 10.0
        for _ in range(3):
            t0 = _time.perf_counter()
            out = core.run_tick(cfg)
            dt = min(dt, _time.perf_counter() - t0)
        assert out["num_unique_pods"] == 200
        assert dt < 1.0, f"tick took {dt:.2f}s — concurrency regression?"
    finally:
        b.stop()
