"""Scaling-model tests.

Ports the reference's 33 lib unit tests (reference gpu-pruner/src/lib.rs:578-998):
resource-flag parsing, ScaleKind→ResourceKind conversion, uid-based
equality/hashing and set dedup, the Meta surface per variant, and Event
generation.
"""

import json

import pytest


def make(core, kind, name, ns, uid=None, extra=None):
    obj = {"metadata": {"name": name, "namespace": ns}}
    if uid is not None:
        obj["metadata"]["uid"] = uid
    if extra:
        obj.update(extra)
    return core.ScaleKind(kind, json.dumps(obj))


def make_deployment(core, name, ns, uid=None):
    return make(core, "Deployment", name, ns, uid)


def make_replica_set(core, name, ns, uid=None):
    return make(core, "ReplicaSet", name, ns, uid)


def make_stateful_set(core, name, ns, uid=None):
    return make(core, "StatefulSet", name, ns, uid)


def make_notebook(core, name, ns, uid=None):
    return make(core, "Notebook", name, ns, uid, {"spec": {"template": None}})


def make_inference_service(core, name, ns, uid=None):
    return make(core, "InferenceService", name, ns, uid, {"spec": {"predictor": {}}})


# ---- get_enabled_resources --------------------------------------------------


def test_enabled_resources_all_flags(core):
    rk = core.get_enabled_resources("drsin")
    for flag in (core.RK_DEPLOYMENT, core.RK_REPLICA_SET, core.RK_STATEFUL_SET,
                 core.RK_INFERENCE_SERVICE, core.RK_NOTEBOOK):
        assert rk & flag


def test_enabled_resources_single_flag(core):
    rk = core.get_enabled_resources("n")
    assert rk & core.RK_NOTEBOOK
    for flag in (core.RK_DEPLOYMENT, core.RK_REPLICA_SET, core.RK_STATEFUL_SET,
                 core.RK_INFERENCE_SERVICE):
        assert not rk & flag


def test_enabled_resources_subset(core):
    rk = core.get_enabled_resources("di")
    assert rk & core.RK_DEPLOYMENT and rk & core.RK_INFERENCE_SERVICE
    assert not rk & core.RK_NOTEBOOK
    assert not rk & core.RK_REPLICA_SET
    assert not rk & core.RK_STATEFUL_SET


def test_enabled_resources_empty_string(core):
    assert core.get_enabled_resources("") == 0


def test_enabled_resources_ignores_unknown_chars(core):
    rk = core.get_enabled_resources("xdqz")
    assert rk & core.RK_DEPLOYMENT
    assert not rk & core.RK_NOTEBOOK


def test_enabled_resources_duplicate_chars_are_idempotent(core):
    assert core.get_enabled_resources("dddd") == core.get_enabled_resources("d")


# ---- ResourceKind bitflags --------------------------------------------------


def test_resource_kind_union(core):
    combined = core.RK_DEPLOYMENT | core.RK_NOTEBOOK
    assert combined & core.RK_DEPLOYMENT
    assert combined & core.RK_NOTEBOOK
    assert not combined & core.RK_STATEFUL_SET


def test_resource_kind_empty_contains_nothing(core):
    for flag in (core.RK_DEPLOYMENT, core.RK_REPLICA_SET, core.RK_STATEFUL_SET,
                 core.RK_INFERENCE_SERVICE, core.RK_NOTEBOOK):
        assert not 0 & flag


# ---- ScaleKind → ResourceKind ----------------------------------------------


@pytest.mark.parametrize(
    "maker,flag_name",
    [
        (make_deployment, "RK_DEPLOYMENT"),
        (make_replica_set, "RK_REPLICA_SET"),
        (make_stateful_set, "RK_STATEFUL_SET"),
        (make_inference_service, "RK_INFERENCE_SERVICE"),
        (make_notebook, "RK_NOTEBOOK"),
    ],
)
def test_scale_kind_to_resource_kind(core, maker, flag_name):
    sk = maker(core, "x", "ns")
    assert sk.resource_kind == getattr(core, flag_name)


# ---- equality ---------------------------------------------------------------


def test_same_deployment_is_equal(core):
    assert make_deployment(core, "d", "ns", "uid-1") == make_deployment(core, "d", "ns", "uid-1")


def test_different_uid_deployments_not_equal(core):
    assert make_deployment(core, "d", "ns", "uid-1") != make_deployment(core, "d", "ns", "uid-2")


def test_different_variants_not_equal(core):
    assert make_deployment(core, "x", "ns", "uid-1") != make_replica_set(core, "x", "ns", "uid-1")


def test_notebook_equality_uses_uid(core):
    # CRDs compare by uid alone: different names, same uid → equal
    assert make_notebook(core, "nb-a", "ns", "same-uid") == make_notebook(core, "nb-b", "ns", "same-uid")


def test_inference_service_equality_uses_uid(core):
    assert make_inference_service(core, "is-a", "ns", "uid-x") == make_inference_service(
        core, "is-b", "ns", "uid-x")


# ---- hashing / set dedup ----------------------------------------------------


def test_set_deduplicates_same_deployment(core):
    s = {make_deployment(core, "d", "ns", "uid-1"), make_deployment(core, "d", "ns", "uid-1")}
    assert len(s) == 1


def test_set_keeps_different_uid_deployments(core):
    s = {make_deployment(core, "d", "ns", "uid-1"), make_deployment(core, "d", "ns", "uid-2")}
    assert len(s) == 2


def test_set_keeps_different_variants_same_uid(core):
    s = {make_deployment(core, "x", "ns", "uid-1"), make_replica_set(core, "x", "ns", "uid-1")}
    assert len(s) == 2


def test_set_deduplicates_notebooks_by_uid(core):
    s = {make_notebook(core, "nb-a", "ns", "uid-nb"), make_notebook(core, "nb-b", "ns", "uid-nb")}
    assert len(s) == 1


def test_set_mixed_resources(core):
    s = {
        make_deployment(core, "d1", "ns", "uid-d"),
        make_replica_set(core, "r1", "ns", "uid-r"),
        make_stateful_set(core, "s1", "ns", "uid-s"),
        make_inference_service(core, "i1", "ns", "uid-i"),
        make_notebook(core, "n1", "ns", "uid-n"),
        make_deployment(core, "d1", "ns", "uid-d"),  # duplicate
    }
    assert len(s) == 5


# ---- Meta surface -----------------------------------------------------------


def test_meta_deployment(core):
    sk = make_deployment(core, "my-dep", "prod", "dep-uid")
    assert sk.name == "my-dep"
    assert sk.namespace == "prod"
    assert sk.kind == "Deployment"
    assert sk.uid == "dep-uid"
    assert sk.api_version == "apps/v1"


def test_meta_replica_set(core):
    sk = make_replica_set(core, "my-rs", "staging", "rs-uid")
    assert (sk.name, sk.namespace, sk.kind, sk.uid, sk.api_version) == (
        "my-rs", "staging", "ReplicaSet", "rs-uid", "apps/v1")


def test_meta_stateful_set(core):
    sk = make_stateful_set(core, "my-ss", "dev", "ss-uid")
    assert (sk.name, sk.namespace, sk.kind, sk.uid, sk.api_version) == (
        "my-ss", "dev", "StatefulSet", "ss-uid", "apps/v1")


def test_meta_notebook(core):
    sk = make_notebook(core, "my-nb", "ml", "nb-uid")
    assert (sk.name, sk.namespace, sk.kind, sk.uid, sk.api_version) == (
        "my-nb", "ml", "Notebook", "nb-uid", "v1")


def test_meta_inference_service(core):
    sk = make_inference_service(core, "my-is", "serving", "is-uid")
    assert (sk.name, sk.namespace, sk.kind, sk.uid, sk.api_version) == (
        "my-is", "serving", "InferenceService", "is-uid", "v1beta1")


# ---- Event generation -------------------------------------------------------


def test_event_for_notebook(core):
    sk = make_notebook(core, "gpu-test", "rhoai--weaton", "nb-uid-1")
    ev = json.loads(core.generate_scale_event(sk))
    io = ev["involvedObject"]
    assert io["name"] == "gpu-test"
    assert io["namespace"] == "rhoai--weaton"
    assert io["kind"] == "Notebook"
    assert io["uid"] == "nb-uid-1"
    assert io["apiVersion"] == "v1"
    assert ev["action"] == "scale_down"
    assert ev["type"] == "Normal"
    assert ev["reason"] == "Pod rhoai--weaton::gpu-test was not using GPU"
    assert ev["reportingComponent"] == "gpu-pruner"
    assert ev["metadata"]["name"].startswith("gpuscaler-")
    assert ev["metadata"]["namespace"] == "rhoai--weaton"
    assert ev["firstTimestamp"] and ev["lastTimestamp"] and ev["eventTime"]


def test_event_for_deployment(core):
    sk = make_deployment(core, "my-dep", "prod", "dep-uid")
    ev = json.loads(core.generate_scale_event(sk))
    assert ev["involvedObject"]["kind"] == "Deployment"
    assert ev["involvedObject"]["apiVersion"] == "apps/v1"
    assert ev["reason"] == "Pod prod::my-dep was not using GPU"


def test_event_for_replica_set(core):
    sk = make_replica_set(core, "my-rs", "staging", None)
    ev = json.loads(core.generate_scale_event(sk))
    assert ev["involvedObject"]["kind"] == "ReplicaSet"
    assert "uid" not in ev["involvedObject"]


def test_event_for_stateful_set(core):
    sk = make_stateful_set(core, "my-ss", "dev", "ss-uid")
    ev = json.loads(core.generate_scale_event(sk))
    assert ev["involvedObject"]["kind"] == "StatefulSet"
    assert ev["involvedObject"]["apiVersion"] == "apps/v1"


def test_event_for_inference_service(core):
    sk = make_inference_service(core, "my-is", "serving", "is-uid")
    ev = json.loads(core.generate_scale_event(sk))
    assert ev["involvedObject"]["kind"] == "InferenceService"
    assert ev["involvedObject"]["apiVersion"] == "v1beta1"


def test_event_names_are_unique(core):
    sk = make_notebook(core, "nb", "ns", None)
    e1 = json.loads(core.generate_scale_event(sk))
    e2 = json.loads(core.generate_scale_event(sk))
    assert e1["metadata"]["name"] != e2["metadata"]["name"]


def test_event_with_no_namespace(core):
    sk = core.ScaleKind("Deployment", json.dumps({"metadata": {"name": "orphan"}}))
    ev = json.loads(core.generate_scale_event(sk))
    assert "namespace" not in ev["involvedObject"]
    assert ev["reason"] == "Pod ::orphan was not using GPU"


def test_event_reporting_instance_from_pod_name(core, monkeypatch):
    monkeypatch.setenv("POD_NAME", "gpu-pruner-abc123")
    sk = make_deployment(core, "d", "ns", "u")
    ev = json.loads(core.generate_scale_event(sk))
    assert ev["reportingInstance"] == "gpu-pruner-abc123"
    monkeypatch.delenv("POD_NAME")
    ev2 = json.loads(core.generate_scale_event(sk))
    assert ev2["reportingInstance"] == "gpu_pruner"


# ---- resource filtering integration -----------------------------------------


def test_enabled_resources_filter_accepts_matching_scale_kind(core):
    enabled = core.get_enabled_resources("dn")
    dep = make_deployment(core, "d", "ns").resource_kind
    nb = make_notebook(core, "n", "ns").resource_kind
    ss = make_stateful_set(core, "s", "ns").resource_kind
    assert enabled & dep
    assert enabled & nb
    assert not enabled & ss
