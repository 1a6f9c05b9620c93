"""Decision-engine tests against the fake kube-apiserver.

Covers the owner-reference walk (reference lib.rs:437-513, e2e.rs:168-252),
eligibility filters (missing / Pending / young / no-creation-timestamp pods —
reference main.rs:411-532), (pod, namespace) series dedup, and shared-parent
dedup — all paths the reference only exercises against a live kind cluster
(SURVEY.md §4).
"""

import json

import pytest


CFG = {"duration": 30, "grace_period": 300, "run_mode": "dry-run"}


def series(pod, ns, gpu="0"):
    return {
        "metric": {
            "Hostname": "node-0",
            "exported_pod": pod,
            "exported_namespace": ns,
            "exported_container": "main",
            "gpu": gpu,
            "modelName": "AMD Instinct MI355X",
        },
        "value": [1700000000.0, "0"],
    }


def evaluate(core, result, cfg=None):
    return core.evaluate_candidates(json.dumps(result), json.dumps(cfg or CFG))


# ---- owner walk -------------------------------------------------------------


def test_walk_resolves_deployment_not_replicaset(core, fake_api):
    dep = fake_api.add_deployment("model-server", "ml")
    rs = fake_api.add_replicaset("model-server-abc", "ml", owner=dep)
    fake_api.add_pod("model-server-abc-xyz", "ml", owner_kind="ReplicaSet",
                     owner_name="model-server-abc", owner_uid=rs["metadata"]["uid"])
    out = evaluate(core, [series("model-server-abc-xyz", "ml")])
    assert out["shutdown_events"] == 1
    root = out["roots"][0]
    assert root.kind == "Deployment"
    assert root.name == "model-server"


def test_walk_orphan_replicaset_scales_directly(core, fake_api):
    rs = fake_api.add_replicaset("solo-rs", "ml")
    fake_api.add_pod("solo-rs-pod", "ml", owner_kind="ReplicaSet",
                     owner_name="solo-rs", owner_uid=rs["metadata"]["uid"])
    out = evaluate(core, [series("solo-rs-pod", "ml")])
    assert [r.kind for r in out["roots"]] == ["ReplicaSet"]


def test_walk_statefulset_without_notebook(core, fake_api):
    ss = fake_api.add_statefulset("db", "ml")
    fake_api.add_pod("db-0", "ml", owner_kind="StatefulSet",
                     owner_name="db", owner_uid=ss["metadata"]["uid"])
    out = evaluate(core, [series("db-0", "ml")])
    assert [r.kind for r in out["roots"]] == ["StatefulSet"]


def test_walk_statefulset_with_notebook_owner(core, fake_api):
    nb = fake_api.add_notebook("workbench", "ml")
    fake_api.add_statefulset("workbench-ss", "ml", notebook_owner=nb)
    fake_api.add_pod("workbench-ss-0", "ml", owner_kind="StatefulSet",
                     owner_name="workbench-ss")
    out = evaluate(core, [series("workbench-ss-0", "ml")])
    root = out["roots"][0]
    assert root.kind == "Notebook"
    assert root.name == "workbench"
    assert root.api_version == "v1"


def test_walk_kserve_label_shortcut(core, fake_api):
    fake_api.add_inferenceservice("llm", "serving")
    fake_api.add_pod("llm-predictor-0", "serving",
                     labels={"serving.kserve.io/inferenceservice": "llm"})
    out = evaluate(core, [series("llm-predictor-0", "serving")])
    root = out["roots"][0]
    assert root.kind == "InferenceService"
    assert root.name == "llm"


def test_walk_orphan_pod_yields_nothing(core, fake_api):
    fake_api.add_pod("orphan", "ml")
    out = evaluate(core, [series("orphan", "ml")])
    assert out["shutdown_events"] == 0


def test_walk_unknown_owner_kind_ignored(core, fake_api):
    fake_api.add_pod("job-pod", "ml", owner_kind="Job", owner_name="some-job")
    out = evaluate(core, [series("job-pod", "ml")])
    assert out["shutdown_events"] == 0


# ---- eligibility filters ----------------------------------------------------


def test_missing_pod_skipped(core, fake_api):
    out = evaluate(core, [series("ghost", "ml")])
    assert out["num_unique_pods"] == 1
    assert out["shutdown_events"] == 0


def test_pending_pod_skipped(core, fake_api):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("pending-pod", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], phase="Pending")
    out = evaluate(core, [series("pending-pod", "ml")])
    assert out["shutdown_events"] == 0


def test_pod_without_creation_timestamp_skipped(core, fake_api):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("no-ts", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=None)
    out = evaluate(core, [series("no-ts", "ml")])
    assert out["shutdown_events"] == 0


def test_young_pod_skipped(core, fake_api):
    """A pod younger than duration+grace cannot have a trustworthy idle window."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    # lookback = 30min + 300s = 2100s; pod is 60s old
    fake_api.add_pod("young", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=60.0)
    out = evaluate(core, [series("young", "ml")])
    assert out["shutdown_events"] == 0


def test_old_pod_eligible(core, fake_api):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("old", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600.0)
    out = evaluate(core, [series("old", "ml")])
    assert out["shutdown_events"] == 1


# ---- dedup ------------------------------------------------------------------


def test_multi_gpu_pod_series_deduped(core, fake_api):
    """Multi-GPU pods emit one series per GPU; the owner chain resolves once."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p8", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"])
    result = [series("p8", "ml", gpu=str(g)) for g in range(8)]
    out = evaluate(core, result)
    assert out["num_series"] == 8
    assert out["num_unique_pods"] == 1
    assert out["shutdown_events"] == 1


def test_pods_sharing_parent_collapse_to_one_event(core, fake_api):
    dep = fake_api.add_deployment("shared", "ml")
    rs = fake_api.add_replicaset("shared-rs", "ml", owner=dep)
    for i in range(3):
        fake_api.add_pod(f"shared-{i}", "ml", owner_kind="ReplicaSet",
                         owner_name="shared-rs", owner_uid=rs["metadata"]["uid"])
    out = evaluate(core, [series(f"shared-{i}", "ml") for i in range(3)])
    assert out["num_unique_pods"] == 3
    assert out["shutdown_events"] == 1


def test_same_pod_name_different_namespaces_not_deduped(core, fake_api):
    for ns in ("a", "b"):
        dep = fake_api.add_deployment("d", ns)
        rs = fake_api.add_replicaset("d-rs", ns, owner=dep)
        fake_api.add_pod("p", ns, owner_kind="ReplicaSet", owner_name="d-rs",
                         owner_uid=rs["metadata"]["uid"])
    out = evaluate(core, [series("p", "a"), series("p", "b")])
    assert out["num_unique_pods"] == 2
    assert out["shutdown_events"] == 2


def test_malformed_series_skipped_not_fatal(core, fake_api):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("good", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"])
    bad = {"metric": {"exported_pod": "x"}, "value": [0, "0"]}  # missing labels
    out = evaluate(core, [bad, series("good", "ml")])
    assert out["num_unique_pods"] == 1
    assert out["shutdown_events"] == 1


# ---- concurrency ------------------------------------------------------------


@pytest.mark.parametrize("concurrency", [1, 8, 64])
def test_concurrent_evaluation_matches_serial(core, fake_api, concurrency):
    n = 40
    for i in range(n):
        dep = fake_api.add_deployment(f"d{i}", "ml")
        rs = fake_api.add_replicaset(f"d{i}-rs", "ml", owner=dep)
        fake_api.add_pod(f"p{i}", "ml", owner_kind="ReplicaSet", owner_name=f"d{i}-rs",
                         owner_uid=rs["metadata"]["uid"])
    cfg = dict(CFG, max_concurrency=concurrency)
    out = evaluate(core, [series(f"p{i}", "ml") for i in range(n)], cfg)
    assert out["num_unique_pods"] == n
    assert out["shutdown_events"] == n
    assert sorted(r.name for r in out["roots"]) == sorted(f"d{i}" for i in range(n))


def test_owner_precedence_first_scalable_wins(core, fake_api):
    """Mixed owner refs: unknown kinds are skipped, first scalable resolves
    (reference lib.rs:458-506 iterates in order)."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    pod = fake_api.add_pod("multi-owner", "ml")
    pod["metadata"]["ownerReferences"] = [
        {"apiVersion": "batch/v1", "kind": "Job", "name": "some-job", "uid": "j1"},
        {"apiVersion": "apps/v1", "kind": "ReplicaSet", "name": "d-rs",
         "uid": rs["metadata"]["uid"]},
    ]
    out = evaluate(core, [series("multi-owner", "ml")])
    assert [r.kind for r in out["roots"]] == ["Deployment"]


def test_kserve_label_takes_precedence_over_owners(core, fake_api):
    """The KServe label shortcut is checked before owner refs (lib.rs:448-456)."""
    fake_api.add_inferenceservice("llm", "ml")
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("pred-0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"],
                     labels={"serving.kserve.io/inferenceservice": "llm"})
    out = evaluate(core, [series("pred-0", "ml")])
    assert [r.kind for r in out["roots"]] == ["InferenceService"]


def test_repeat_tick_idempotent(core, fake_api):
    """Re-running the decision over an already-culled cluster is a no-op
    patch (reference §5.4: stateless between ticks, idempotent)."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"])
    result = [series("p", "ml")]
    for _ in range(3):
        out = evaluate(core, result)
        assert out["shutdown_events"] == 1


@pytest.mark.parametrize("strategy", ["get", "list", "auto"])
def test_eval_strategies_identical_outcomes(core, fake_api, strategy):
    """LIST-based evaluation must reach exactly the per-GET decisions
    (native/pruner/objcache.hpp)."""
    # mixed cluster incl. a gone pod, a pending pod, and a KServe shortcut
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    for i in range(12):
        fake_api.add_pod(f"p{i}", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                         owner_uid=rs["metadata"]["uid"])
    nb = fake_api.add_notebook("wb", "ml")
    fake_api.add_statefulset("wb-ss", "ml", notebook_owner=nb)
    fake_api.add_pod("wb-ss-0", "ml", owner_kind="StatefulSet", owner_name="wb-ss")
    fake_api.add_inferenceservice("llm", "ml")
    fake_api.add_pod("pred-0", "ml",
                     labels={"serving.kserve.io/inferenceservice": "llm"})
    fake_api.add_pod("pending", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], phase="Pending")
    result = ([series(f"p{i}", "ml") for i in range(12)] +
              [series("wb-ss-0", "ml"), series("pred-0", "ml"),
               series("pending", "ml"), series("ghost", "ml")])
    cfg = dict(CFG, eval_strategy=strategy)
    out = evaluate(core, result, cfg)
    assert out["num_unique_pods"] == 16
    assert out["shutdown_events"] == 3  # Deployment + Notebook + InferenceService
    assert sorted(r.kind for r in out["roots"]) == [
        "Deployment", "InferenceService", "Notebook"]


def test_list_strategy_uses_few_requests(core, fake_api):
    """At 40 candidates in one namespace the LIST path costs O(1) requests."""
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    for i in range(40):
        fake_api.add_pod(f"p{i}", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                         owner_uid=rs["metadata"]["uid"])
    result = [series(f"p{i}", "ml") for i in range(40)]
    fake_api.requests.clear()
    evaluate(core, result, dict(CFG, eval_strategy="list"))
    n_list = len(fake_api.requests)
    fake_api.requests.clear()
    evaluate(core, result, dict(CFG, eval_strategy="get"))
    n_get = len(fake_api.requests)
    assert n_list <= 8, f"LIST strategy made {n_list} requests"
    assert n_get >= 40, f"GET strategy made {n_get} requests"


def test_kserve_label_missing_isvc_skips(core, fake_api):
    """KServe label pointing at a deleted InferenceService: skip, no crash
    (reference propagates the error → pod skipped)."""
    fake_api.add_pod("stale-pred", "ml",
                     labels={"serving.kserve.io/inferenceservice": "gone"})
    out = evaluate(core, [series("stale-pred", "ml")])
    assert out["shutdown_events"] == 0


def test_garbage_creation_timestamp_skipped(core, fake_api):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    pod = fake_api.add_pod("badts", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                           owner_uid=rs["metadata"]["uid"])
    pod["metadata"]["creationTimestamp"] = "not-a-date"
    out = evaluate(core, [series("badts", "ml")])
    assert out["shutdown_events"] == 0


def test_owner_fetch_error_falls_through_to_next_owner(core, fake_api):
    """A failing (500) owner GET tries the next ownerReference (reference
    lib.rs:464 `if let Ok(rs)` semantics)."""
    ss = fake_api.add_statefulset("db", "ml")
    pod = fake_api.add_pod("multi", "ml")
    pod["metadata"]["ownerReferences"] = [
        # first owner: a ReplicaSet whose name breaks the fake's routing →
        # the GET raises; the walk must continue to the StatefulSet
        {"apiVersion": "apps/v1", "kind": "ReplicaSet", "name": "x/y", "uid": "u1"},
        {"apiVersion": "apps/v1", "kind": "StatefulSet", "name": "db",
         "uid": ss["metadata"]["uid"]},
    ]
    out = evaluate(core, [series("multi", "ml")])
    assert [r.kind for r in out["roots"]] == ["StatefulSet"]
