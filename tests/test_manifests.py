"""Deploy-manifest sanity tests (parity: reference hack/*.yaml bundle).

Validates the kustomize bundle parses, references only files that exist, and
that the RBAC surface covers every API group + resource (incl. /scale
subresources) the daemon actually touches.
"""

from pathlib import Path

import yaml

DEPLOY = Path(__file__).resolve().parent.parent / "deploy"


def load_all(name):
    return list(yaml.safe_load_all((DEPLOY / name).read_text()))


def test_kustomization_resources_exist():
    k = load_all("kustomization.yaml")[0]
    for res in k["resources"]:
        assert (DEPLOY / res).exists(), f"kustomization references missing {res}"


def test_all_manifests_parse():
    for f in DEPLOY.glob("*.yaml"):
        docs = list(yaml.safe_load_all(f.read_text()))
        assert docs, f"{f.name} is empty"
        for d in docs:
            assert "kind" in d, f"{f.name}: doc without kind"


def test_clusterrole_covers_daemon_surface():
    """Every (group, resource) the engine touches must be grantable."""
    docs = load_all("clusterrole.yaml")
    pruner_cr = next(d for d in docs if d["metadata"]["name"] == "gpu-pruner-cr")
    granted = set()
    for rule in pruner_cr["rules"]:
        for g in rule["apiGroups"]:
            for r in rule["resources"]:
                granted.add((g, r))
    needed = [
        ("", "pods"), ("", "events"),
        ("apps", "deployments"), ("apps", "deployments/scale"),
        ("apps", "replicasets"), ("apps", "replicasets/scale"),
        ("apps", "statefulsets"), ("apps", "statefulsets/scale"),
        ("kubeflow.org", "notebooks"),
        ("serving.kserve.io", "inferenceservices"),
        ("coordination.k8s.io", "leases"),  # --leader-elect
    ]
    for pair in needed:
        assert pair in granted, f"ClusterRole missing {pair}"


def test_clusterrole_lease_verbs():
    """--leader-elect needs exactly get/create/update on Leases: get to read
    the current holder, create for first acquisition, update (rv-fenced PUT)
    for renew/takeover. list/watch/delete are NOT used by the elector and
    must not be granted (least privilege)."""
    docs = load_all("clusterrole.yaml")
    pruner_cr = next(d for d in docs if d["metadata"]["name"] == "gpu-pruner-cr")
    lease_rules = [r for r in pruner_cr["rules"]
                   if "coordination.k8s.io" in r["apiGroups"]
                   and "leases" in r["resources"]]
    assert lease_rules, "no Lease rule in gpu-pruner-cr"
    verbs = {v for r in lease_rules for v in r["verbs"]}
    assert verbs == {"get", "create", "update"}, verbs


def test_crd_manifests_parse_as_v1_crds():
    """deploy/crds/ ships minimal Notebook + InferenceService CRDs so the
    envtest/e2e tiers (and air-gapped installs without Kubeflow/KServe) can
    register the custom kinds the walk resolves. Both must be well-formed
    apiextensions.k8s.io/v1 with the group/kind the engine expects."""
    crd_dir = DEPLOY / "crds"
    expected = {
        ("kubeflow.org", "Notebook"),
        ("serving.kserve.io", "InferenceService"),
    }
    seen = set()
    for f in sorted(crd_dir.glob("*.yaml")):
        for d in yaml.safe_load_all(f.read_text()):
            assert d["apiVersion"] == "apiextensions.k8s.io/v1", f.name
            assert d["kind"] == "CustomResourceDefinition", f.name
            spec = d["spec"]
            names = spec["names"]
            seen.add((spec["group"], names["kind"]))
            # name must be <plural>.<group> per apiserver validation
            assert d["metadata"]["name"] == f"{names['plural']}.{spec['group']}"
            assert any(v.get("served") for v in spec["versions"]), f.name
    assert seen == expected, seen


def test_deployment_resource_budget():
    """Reference budget: the daemon fits in 250m-500m CPU / 64-128Mi."""
    docs = load_all("deployment.yaml")
    dep = docs[0]
    res = dep["spec"]["template"]["spec"]["containers"][0]["resources"]
    assert res["limits"]["cpu"] == "500m"
    assert res["limits"]["memory"] == "128Mi"


def test_exporter_daemonset_mounts():
    docs = load_all("exporter-daemonset.yaml")
    ds = next(d for d in docs if d["kind"] == "DaemonSet")
    spec = ds["spec"]["template"]["spec"]
    mounts = {m["mountPath"] for c in spec["containers"] for m in c["volumeMounts"]}
    assert "/dev/kfd" in mounts and "/dev/dri" in mounts and "/sys" in mounts
    env_names = {e["name"] for c in spec["containers"] for e in c.get("env", [])}
    assert "NODE_NAME" in env_names


def test_serviceaccounts_and_bindings_align():
    sas = {d["metadata"]["name"] for d in load_all("serviceaccount.yaml")}
    for crb in load_all("crs.yaml"):
        for subj in crb["subjects"]:
            assert subj["name"] in sas, f"CRB {crb['metadata']['name']} binds unknown SA"
