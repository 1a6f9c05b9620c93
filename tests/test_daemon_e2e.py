"""End-to-end tests of the native gpu-pruner binary against fake services.

Equivalent of the reference's kind-cluster e2e suite
(gpu-pruner/tests/e2e.rs) but hermetic: the full daemon — CLI → query build →
Prometheus round-trip → pod eligibility → owner walk → scale patch → Event —
runs as a subprocess against the in-process fakes. Covers BASELINE.json
config 1 (dry-run, mocked Prometheus + apiserver, Deployment pod) and the
scale-down semantics of every kind.
"""

import json
import os
import subprocess

import pytest


def run_pruner(pruner_bin, fake_api, fake_prom, *args, timeout=30, env_extra=None):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env.pop("KUBERNETES_SERVICE_HOST", None)
    env["PROMETHEUS_TOKEN"] = "test-token"
    if env_extra:
        env.update(env_extra)
    cmd = [pruner_bin, "--prometheus-url", fake_prom.url, *args]
    return subprocess.run(cmd, capture_output=True, text=True, timeout=timeout, env=env)


@pytest.fixture
def cluster(fake_api, fake_prom):
    """One idle Deployment pod, ready to be culled."""
    dep = fake_api.add_deployment("model-server", "ml", replicas=1)
    rs = fake_api.add_replicaset("model-server-rs", "ml", owner=dep)
    fake_api.add_pod("model-server-rs-0", "ml", owner_kind="ReplicaSet",
                     owner_name="model-server-rs", owner_uid=rs["metadata"]["uid"],
                     age_s=3 * 3600)
    fake_prom.add_idle_series("model-server-rs-0", "ml")
    return fake_api


def test_dry_run_does_not_scale(pruner_bin, cluster, fake_prom):
    """BASELINE.json config 1: dry-run against mocked Prom + apiserver."""
    r = run_pruner(pruner_bin, cluster, fake_prom)  # default run-mode = dry-run
    assert r.returncode == 0, r.stderr
    assert "Would have sent" in r.stderr
    assert cluster.get("Deployment", "ml", "model-server")["spec"]["replicas"] == 1
    assert cluster.events == []
    # exactly one Prometheus query in one-shot mode
    assert len(fake_prom.queries) == 1
    assert "DCGM_FI_PROF_GR_ENGINE_ACTIVE" in fake_prom.queries[0]


def test_scale_down_deployment_to_zero(pruner_bin, cluster, fake_prom):
    r = run_pruner(pruner_bin, cluster, fake_prom, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert cluster.get("Deployment", "ml", "model-server")["spec"]["replicas"] == 0
    # one gpuscaler- Event announcing the action
    assert len(cluster.events) == 1
    ev = cluster.events[0]
    assert ev["metadata"]["name"].startswith("gpuscaler-")
    assert ev["involvedObject"]["kind"] == "Deployment"
    assert ev["involvedObject"]["name"] == "model-server"
    assert ev["reason"] == "Pod ml::model-server was not using GPU"


def test_scale_down_statefulset(pruner_bin, fake_api, fake_prom):
    ss = fake_api.add_statefulset("db", "ml", replicas=1)
    fake_api.add_pod("db-0", "ml", owner_kind="StatefulSet", owner_name="db",
                     owner_uid=ss["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("db-0", "ml")
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert fake_api.get("StatefulSet", "ml", "db")["spec"]["replicas"] == 0


def test_scale_down_notebook_sets_stop_annotation(pruner_bin, fake_api, fake_prom):
    nb = fake_api.add_notebook("workbench", "ml")
    fake_api.add_statefulset("workbench-ss", "ml", notebook_owner=nb)
    fake_api.add_pod("workbench-ss-0", "ml", owner_kind="StatefulSet",
                     owner_name="workbench-ss", age_s=3 * 3600)
    fake_prom.add_idle_series("workbench-ss-0", "ml")
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    nb_after = fake_api.get("Notebook", "ml", "workbench")
    anno = nb_after["metadata"].get("annotations", {})
    assert "kubeflow-resource-stopped" in anno
    assert anno["kubeflow-resource-stopped"].endswith("Z")


def test_scale_down_inferenceservice_min_replicas(pruner_bin, fake_api, fake_prom):
    fake_api.add_inferenceservice("llm", "serving", min_replicas=1)
    fake_api.add_pod("llm-predictor-0", "serving", age_s=3 * 3600,
                     labels={"serving.kserve.io/inferenceservice": "llm"})
    fake_prom.add_idle_series("llm-predictor-0", "serving")
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    isvc = fake_api.get("InferenceService", "serving", "llm")
    assert isvc["spec"]["predictor"]["minReplicas"] == 0


def test_disabled_resource_not_scaled(pruner_bin, cluster, fake_prom):
    """Deployment resolved but filtered out by --enabled-resources."""
    r = run_pruner(pruner_bin, cluster, fake_prom, "--run-mode", "scale-down",
                   "--enabled-resources", "n")
    assert r.returncode == 0, r.stderr
    assert "not enabled" in r.stderr
    assert cluster.get("Deployment", "ml", "model-server")["spec"]["replicas"] == 1


def test_apiserver_429_retried_tick_completes(pruner_bin, cluster, fake_prom):
    """Apiserver priority-and-fairness throttling: 429 + Retry-After responses
    are retried (bounded) and the tick still makes the right decisions
    (VERDICT r1 #8)."""
    cluster.throttle_next = 3  # first 3 requests bounce with 429
    cluster.retry_after_s = 0
    r = run_pruner(pruner_bin, cluster, fake_prom, "--run-mode", "scale-down")
    assert r.returncode == 0, r.stderr
    assert cluster.throttled == 3
    assert cluster.get("Deployment", "ml", "model-server")["spec"]["replicas"] == 0
    assert len(cluster.events) == 1


def test_apiserver_429_storm_exhausts_retries_gracefully(pruner_bin, cluster,
                                                         fake_prom):
    """When every request is throttled past the retry budget the pod is
    skipped (skip-and-continue) — the process does not crash."""
    cluster.throttle_next = 10_000
    r = run_pruner(pruner_bin, cluster, fake_prom, "--run-mode", "scale-down",
                   timeout=120)
    assert r.returncode == 0, r.stderr
    assert cluster.get("Deployment", "ml", "model-server")["spec"]["replicas"] == 1
    assert cluster.throttled >= 4  # 1 try + 3 retries on the first GET at least


def test_failure_breaker_exits_nonzero(pruner_bin, fake_api, fake_prom):
    """Daemon mode aborts after more than --max-failures consecutive failures."""
    fake_prom.fail_next = 100
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--daemon-mode",
                   "--check-interval", "0", "--max-failures", "2", timeout=60)
    assert r.returncode != 0
    assert "Too many consecutive failures" in r.stderr
    # Reference parity (main.rs:310-321 compares the PRE-increment counter):
    # with max 2 the breaker trips on the 4th consecutive failure — failures
    # with pre-increment values 0, 1, 2 are tolerated, 3 > 2 exits.
    assert len(fake_prom.queries) == 4


def test_failure_then_recovery_does_not_trip(pruner_bin, fake_api, cluster, fake_prom):
    """A success between failures resets the breaker (reference main.rs:299)."""
    import subprocess as sp, time
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    fake_prom.fail_next = 1
    p = sp.Popen([pruner_bin, "--prometheus-url", fake_prom.url, "--daemon-mode",
                  "--check-interval", "0", "--max-failures", "1"],
                 env=env, stdout=sp.PIPE, stderr=sp.PIPE)
    try:
        time.sleep(2.0)
        assert p.poll() is None, "daemon exited although breaker should have reset"
        assert len(fake_prom.queries) > 3
    finally:
        p.kill()
        p.wait()


def test_honor_labels_series(pruner_bin, fake_api, fake_prom):
    """honor_labels=true: query + parsing use native label names end-to-end."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml", honor_labels=True)
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down",
                   "--honor-labels")
    assert r.returncode == 0, r.stderr
    assert "exported_pod" not in fake_prom.queries[0]
    assert fake_api.get("Deployment", "ml", "d")["spec"]["replicas"] == 0


def test_filters_appear_in_wire_query(pruner_bin, fake_api, fake_prom):
    r = run_pruner(pruner_bin, fake_api, fake_prom,
                   "--namespace", "ml-.*", "--model-name", "AMD Instinct MI355X",
                   "--power-threshold", "120")
    assert r.returncode == 0, r.stderr
    q = fake_prom.queries[0]
    assert 'exported_namespace =~ "ml-.*"' in q
    assert 'modelName =~ "AMD Instinct MI355X"' in q
    assert "DCGM_FI_DEV_POWER_USAGE" in q and ">= 120" in q


def test_bearer_token_sent_to_prometheus(pruner_bin, fake_api, fake_prom):
    run_pruner(pruner_bin, fake_api, fake_prom, "--prometheus-token", "flag-token")
    assert fake_prom.bearer_tokens[-1] == "flag-token"


def test_env_token_fallback(pruner_bin, fake_api, fake_prom):
    run_pruner(pruner_bin, fake_api, fake_prom,
               env_extra={"PROMETHEUS_TOKEN": "env-token"})
    assert fake_prom.bearer_tokens[-1] == "env-token"


def test_multi_pod_mixed_cluster(pruner_bin, fake_api, fake_prom):
    """BASELINE config 3 shape: 50 mixed Deploy/SS+Notebook/InferenceService
    pods — every parent culled exactly once, announced exactly once."""
    from gpu_pruner_amd.fixtures import build_synthetic_cluster

    info = build_synthetic_cluster(fake_api, fake_prom, n_pods=50, pods_per_parent=2)
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down",
                   "--max-concurrency", "16")
    assert r.returncode == 0, r.stderr
    scaled = 0
    for (kind, ns, name), obj in fake_api.objects.items():
        if kind in ("Deployment", "StatefulSet") and obj.get("spec", {}).get("replicas") == 0:
            scaled += 1
        if kind == "Notebook" and "kubeflow-resource-stopped" in obj["metadata"].get(
                "annotations", {}):
            scaled += 1
        if kind == "InferenceService" and obj["spec"]["predictor"].get("minReplicas") == 0:
            scaled += 1
    assert scaled == info["expected_shutdown_events"]
    # every action announced
    assert len(fake_api.events) == info["expected_shutdown_events"]


def test_scale_down_with_list_strategy(pruner_bin, fake_api, fake_prom):
    """The daemon binary with --eval-strategy list reaches identical outcomes."""
    from gpu_pruner_amd.fixtures import build_synthetic_cluster

    info = build_synthetic_cluster(fake_api, fake_prom, n_pods=30, pods_per_parent=3)
    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down",
                   "--eval-strategy", "list")
    assert r.returncode == 0, r.stderr
    assert len(fake_api.events) == info["expected_shutdown_events"]
    # LIST path: far fewer GETs than pods
    gets = [p for (m, p) in fake_api.requests if m == "GET" and "/pods/" in p]
    assert len(gets) == 0, f"list strategy should not GET individual pods: {gets[:3]}"


def test_config4_filters_grace_and_partially_busy(pruner_bin, fake_api, fake_prom):
    """BASELINE config 4: namespace + model-name filters with grace period on
    a partially-busy cluster. Busy pods never appear in the idle query result
    (Prometheus's == 0 predicate), young pods are age-filtered, other-model
    and other-namespace pods are excluded by the pushed-down regex filters —
    only the old idle matching pod's parent is culled."""
    def deployment_with_pod(name, ns, age_s=3 * 3600):
        dep = fake_api.add_deployment(name, ns)
        rs = fake_api.add_replicaset(f"{name}-rs", ns, owner=dep)
        fake_api.add_pod(f"{name}-0", ns, owner_kind="ReplicaSet",
                         owner_name=f"{name}-rs", owner_uid=rs["metadata"]["uid"],
                         age_s=age_s)
        return dep

    deployment_with_pod("idle-old", "ml-team-a")          # culled
    deployment_with_pod("idle-young", "ml-team-a", age_s=60)  # grace-filtered
    deployment_with_pod("busy", "ml-team-a")              # busy → no series
    deployment_with_pod("other-ns", "web")                # namespace filter
    deployment_with_pod("other-gpu", "ml-team-a")         # model filter

    # Prometheus only returns series matching the filters and == 0:
    fake_prom.add_idle_series("idle-old-0", "ml-team-a",
                              model_name="AMD Instinct MI355X")
    fake_prom.add_idle_series("idle-young-0", "ml-team-a",
                              model_name="AMD Instinct MI355X")
    # busy pod: excluded by == 0; other-ns/other-gpu: excluded by the regex
    # filters — the real Prometheus applies those, so they emit no series.

    r = run_pruner(pruner_bin, fake_api, fake_prom, "--run-mode", "scale-down",
                   "--namespace", "ml-team-.*",
                   "--model-name", "AMD Instinct MI355.*",
                   "--grace-period", "300")
    assert r.returncode == 0, r.stderr
    # filters were pushed into the wire query
    q = fake_prom.queries[0]
    assert 'exported_namespace =~ "ml-team-.*"' in q
    assert 'modelName =~ "AMD Instinct MI355.*"' in q
    # only the old idle matching deployment was culled
    assert fake_api.get("Deployment", "ml-team-a", "idle-old")["spec"]["replicas"] == 0
    for name, ns in [("idle-young", "ml-team-a"), ("busy", "ml-team-a"),
                     ("other-gpu", "ml-team-a")]:
        assert fake_api.get("Deployment", ns, name)["spec"]["replicas"] == 1, name
    assert fake_api.get("Deployment", "web", "other-ns")["spec"]["replicas"] == 1
    assert len(fake_api.events) == 1


def test_non_vector_response_counts_as_failure(pruner_bin, fake_api, fake_prom):
    """A matrix (range) response is a query failure, not a crash — the
    breaker handles it (reference expects a vector, main.rs:405-409)."""
    fake_prom.data_override = {"resultType": "matrix", "result": []}
    r = run_pruner(pruner_bin, fake_api, fake_prom)
    # one-shot: failure logged, exit 0 (breaker only trips in daemon mode)
    assert r.returncode == 0
    assert "expected vector" in r.stderr
