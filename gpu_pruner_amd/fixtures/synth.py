"""Synthetic cluster generator for benchmark configs.

Builds the BASELINE.json stress fixtures: N pods spread over mixed
Deployment / StatefulSet(+Notebook) / InferenceService parents in several
namespaces, with one idle series per (pod, GPU) in the fake Prometheus —
matching the decision-throughput configs (50 mixed pods; 1000-pod stress).
"""

from __future__ import annotations

from .fake_apiserver import FakeApiServer
from .fake_prom import FakePrometheus


def build_synthetic_cluster(
    api: FakeApiServer,
    prom: FakePrometheus,
    n_pods: int,
    pods_per_parent: int = 2,
    gpus_per_pod: int = 1,
    n_namespaces: int = 4,
    model_name: str = "AMD Instinct MI355X",
    n_nodes: int = 1,
    age_s: float = 7200.0,
) -> dict:
    """Populate fixtures with n_pods idle pods; returns expected-outcome info.

    Parents rotate Deployment → StatefulSet+Notebook → InferenceService, so
    all five ScaleKind variants appear. Expected shutdown events =
    number of distinct parents (pods_per_parent pods share each parent).
    """
    parents = []
    n_parents = (n_pods + pods_per_parent - 1) // pods_per_parent
    for p in range(n_parents):
        ns = f"ml-team-{p % n_namespaces}"
        flavor = p % 3
        if flavor == 0:
            dep = api.add_deployment(f"dep-{p}", ns)
            rs = api.add_replicaset(f"dep-{p}-rs", ns, owner=dep)
            parents.append(("ReplicaSet", rs, ns, "Deployment"))
        elif flavor == 1:
            nb = api.add_notebook(f"nb-{p}", ns)
            ss = api.add_statefulset(f"nb-{p}-ss", ns, notebook_owner=nb)
            parents.append(("StatefulSet", ss, ns, "Notebook"))
        else:
            isvc = api.add_inferenceservice(f"isvc-{p}", ns)
            parents.append(("InferenceService", isvc, ns, "InferenceService"))

    for i in range(n_pods):
        p = i // pods_per_parent
        owner_kind, owner, ns, _root = parents[p]
        pod_name = f"pod-{i}"
        if owner_kind == "InferenceService":
            api.add_pod(
                pod_name, ns, age_s=age_s,
                labels={"serving.kserve.io/inferenceservice": owner["metadata"]["name"]},
            )
        else:
            api.add_pod(
                pod_name, ns, owner_kind=owner_kind,
                owner_name=owner["metadata"]["name"],
                owner_uid=owner["metadata"]["uid"], age_s=age_s,
            )
        for g in range(gpus_per_pod):
            prom.add_idle_series(
                pod_name, ns, gpu=str(g), model_name=model_name,
                hostname=f"mi355-node-{i % max(1, n_nodes)}",
            )

    return {
        "n_pods": n_pods,
        "n_parents": n_parents,
        "n_series": n_pods * gpus_per_pod,
        "expected_shutdown_events": n_parents,
    }
