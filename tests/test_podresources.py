"""kubelet PodResources client tests.

The native client (native/exporter/podresources.cpp) is a from-scratch
HTTP/2-cleartext + protobuf implementation; these tests validate it against
grpcio + the real protobuf runtime serving v1.PodResourcesLister/List on a
unix socket — the independent-implementation cross-check for the
"Pod↔GPU attribution" hard part.
"""

import json

import pytest

pytest.importorskip("grpc")


@pytest.fixture
def gpumon():
    from gpu_pruner_amd import _gpumon

    return _gpumon


ENTRIES = [
    {
        "pod": "train-0", "namespace": "ml",
        "containers": [
            {"name": "worker",
             "devices": [{"resource_name": "amd.com/gpu", "device_ids": ["56525"]}]},
        ],
    },
    {
        "pod": "notebook-1", "namespace": "workbench",
        "containers": [
            {"name": "jupyter",
             "devices": [{"resource_name": "amd.com/gpu",
                          "device_ids": ["abcdef0123456789", "77777"]}]},
            {"name": "sidecar", "devices": []},  # no GPU
        ],
    },
    {
        "pod": "cpu-only", "namespace": "misc",
        "containers": [{"name": "app",
                        "devices": [{"resource_name": "cpu", "device_ids": ["0"]}]}],
    },
]


@pytest.fixture
def podres_server(tmp_path):
    from gpu_pruner_amd.fixtures.fake_podresources import FakePodResources

    sock = str(tmp_path / "kubelet.sock")
    with FakePodResources(sock, ENTRIES) as srv:
        yield srv


def test_list_pod_resources_roundtrip(gpumon, podres_server):
    entries = gpumon.list_pod_resources(podres_server.socket_path)
    assert podres_server.calls == 1
    by_container = {(e["pod"], e["container"]): e for e in entries}
    assert ("train-0", "worker") in by_container
    worker = by_container[("train-0", "worker")]
    assert worker["namespace"] == "ml"
    assert worker["devices"] == [{"resource_name": "amd.com/gpu", "device_ids": ["56525"]}]
    jup = by_container[("notebook-1", "jupyter")]
    assert jup["devices"][0]["device_ids"] == ["abcdef0123456789", "77777"]
    assert ("notebook-1", "sidecar") in by_container
    assert ("cpu-only", "app") in by_container


def test_list_missing_socket_raises(gpumon, tmp_path):
    with pytest.raises(gpumon.PodResourcesError):
        gpumon.list_pod_resources(str(tmp_path / "nope.sock"), 1000)


def test_decode_empty_response(gpumon):
    assert gpumon.decode_list_response(b"") == []


def test_decode_skips_unknown_fields(gpumon):
    """Future kubelet fields (e.g. cpu_ids, memory) must be skipped cleanly."""
    from gpu_pruner_amd.fixtures.fake_podresources import _MSGS

    resp = _MSGS["Response"]()
    pr = resp.pod_resources.add()
    pr.name = "p"
    setattr(pr, "namespace", "ns")
    cr = pr.containers.add()
    cr.name = "c"
    payload = resp.SerializeToString()
    # append an unknown field (field 9, varint) to the top-level message
    payload += bytes([0x48, 0x2A])
    entries = gpumon.decode_list_response(payload)
    assert entries[0]["pod"] == "p"


# ---- allocation-first attribution chain -------------------------------------


SAMPLES = [
    {"index": 0, "kfd_gpu_id": 56525, "unique_id": "32da0b77724e0fe2",
     "drm_render_minor": 152, "pci_bdf": "0000:23:00.0"},
    {"index": 1, "kfd_gpu_id": 77777, "unique_id": "abcdef0123456789",
     "drm_render_minor": 160, "pci_bdf": "0000:26:00.0"},
]


def test_resolve_full_prefers_podresources(gpumon, podres_server, tmp_path, monkeypatch):
    monkeypatch.setenv("GPU_EXPORTER_PODRESOURCES_SOCKET", podres_server.socket_path)
    monkeypatch.setenv("GPU_EXPORTER_SYSFS_ROOT", str(tmp_path))   # empty KFD tree
    monkeypatch.setenv("GPU_EXPORTER_PROCFS_ROOT", str(tmp_path))
    monkeypatch.delenv("GPU_EXPORTER_POD_MAP_FILE", raising=False)
    monkeypatch.delenv("GPU_PRUNER_K8S_URL", raising=False)
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    a = gpumon.Attributor()
    out = a.resolve_full(json.dumps(SAMPLES))
    # device 0 matched by KFD gpu_id, device 1 by unique_id — both from
    # *allocations*, no process needs the GPU open (idle-pod case)
    assert out[0] == {"pod": "train-0", "namespace": "ml", "container": "worker"}
    assert out[1] == {"pod": "notebook-1", "namespace": "workbench",
                      "container": "jupyter"}


def test_resolve_full_kfd_fallback_without_socket(gpumon, tmp_path, monkeypatch):
    """No kubelet socket → the KFD usage path still attributes."""
    uid = "8f7e1a2b-3c4d-5e6f-7a8b-9c0d1e2f3a4b"
    kfd = tmp_path / "sys/class/kfd/kfd/proc/4242"
    kfd.mkdir(parents=True)
    (kfd / "vram_56525").write_text("1\n")
    proc = tmp_path / "proc/4242"
    proc.mkdir(parents=True)
    (proc / "cgroup").write_text(
        f"0::/kubepods.slice/kubepods-pod{uid.replace('-', '_')}.slice/x.scope\n")
    map_file = tmp_path / "map.json"
    map_file.write_text(json.dumps({uid: {"pod": "kfd-pod", "namespace": "ml",
                                          "container": "c"}}))
    monkeypatch.setenv("GPU_EXPORTER_PODRESOURCES_SOCKET", str(tmp_path / "no.sock"))
    monkeypatch.setenv("GPU_EXPORTER_SYSFS_ROOT", str(tmp_path))
    monkeypatch.setenv("GPU_EXPORTER_PROCFS_ROOT", str(tmp_path))
    monkeypatch.setenv("GPU_EXPORTER_POD_MAP_FILE", str(map_file))
    a = gpumon.Attributor()
    out = a.resolve_full(json.dumps(SAMPLES))
    assert out[0]["pod"] == "kfd-pod"
    assert 1 not in out


# ---- AMD device-plugin ID-form fixtures ---------------------
#
# ROCm/k8s-device-plugin keys its pluginapi.Device IDs by the PCI address it
# discovers under /sys/module/amdgpu/drivers/pci:amdgpu; other stacks (GPU
# operator builds, forks) have shipped KFD gpu_ids, 64-bit unique ids, DRM
# node names, or bare indices. Each known form must attribute the same GPU.

DEVICE0 = {"index": 0, "kfd_gpu_id": 56525, "unique_id": "0012345678abcdef",
           "drm_render_minor": 152, "pci_bdf": "0000:23:00.0"}


@pytest.mark.parametrize("device_id", [
    "0000:23:00.0",          # PCI BDF (ROCm k8s-device-plugin's native form)
    "0000:23:00",            # BDF without function
    "23:00.0",               # BDF without PCI domain
    "0000:23:00.0".upper(),  # case-insensitive
    "56525",                 # KFD topology gpu_id
    "0012345678abcdef",      # unique id, padded
    "0x0012345678abcdef",    # unique id, 0x-prefixed
    "0x12345678abcdef",      # unique id, unpadded
    "12345678abcdef",        # unique id, unpadded, no prefix
    "renderD152",            # DRM render node name
    "/dev/dri/renderD152",   # DRM render node path
    "card24",                # card index (renderD 128+N ↔ cardN)
    "0",                     # device index
])
def test_device_id_forms_attribute(gpumon, tmp_path, monkeypatch, device_id):
    from gpu_pruner_amd.fixtures.fake_podresources import FakePodResources

    entries = [{
        "pod": "train-0", "namespace": "ml",
        "containers": [{"name": "worker",
                        "devices": [{"resource_name": "amd.com/gpu",
                                     "device_ids": [device_id]}]}],
    }]
    sock = str(tmp_path / "kubelet.sock")
    monkeypatch.setenv("GPU_EXPORTER_PODRESOURCES_SOCKET", sock)
    monkeypatch.setenv("GPU_EXPORTER_SYSFS_ROOT", str(tmp_path))
    monkeypatch.setenv("GPU_EXPORTER_PROCFS_ROOT", str(tmp_path))
    monkeypatch.delenv("GPU_EXPORTER_POD_MAP_FILE", raising=False)
    monkeypatch.delenv("GPU_PRUNER_K8S_URL", raising=False)
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    with FakePodResources(sock, entries):
        a = gpumon.Attributor()
        out = a.resolve_full(json.dumps([DEVICE0]))
    assert out.get(0) == {"pod": "train-0", "namespace": "ml",
                          "container": "worker"}, device_id


@pytest.mark.parametrize("bogus_id", [
    "0000:99:00.0",     # different BDF
    "99999",            # unknown gpu_id
    "renderD200",       # other render node
    "gpu-not-a-form",   # garbage
])
def test_non_matching_ids_do_not_attribute(gpumon, tmp_path, monkeypatch, bogus_id):
    from gpu_pruner_amd.fixtures.fake_podresources import FakePodResources

    entries = [{
        "pod": "train-0", "namespace": "ml",
        "containers": [{"name": "worker",
                        "devices": [{"resource_name": "amd.com/gpu",
                                     "device_ids": [bogus_id]}]}],
    }]
    sock = str(tmp_path / "kubelet.sock")
    monkeypatch.setenv("GPU_EXPORTER_PODRESOURCES_SOCKET", sock)
    monkeypatch.setenv("GPU_EXPORTER_SYSFS_ROOT", str(tmp_path))
    monkeypatch.setenv("GPU_EXPORTER_PROCFS_ROOT", str(tmp_path))
    monkeypatch.delenv("GPU_EXPORTER_POD_MAP_FILE", raising=False)
    monkeypatch.delenv("GPU_PRUNER_K8S_URL", raising=False)
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    with FakePodResources(sock, entries):
        a = gpumon.Attributor()
        out = a.resolve_full(json.dumps([DEVICE0]))
    assert 0 not in out, (bogus_id, out)


def test_kubelet_grpc_error_raises_loudly(gpumon, tmp_path):
    """A kubelet returning a gRPC error (trailers-only: grpc-status != 0, no
    DATA) must raise PodResourcesError carrying the status — before response
    headers were decoded, it looked like an EMPTY pod list, silently
    disabling GPU→pod attribution."""
    import grpc
    from concurrent import futures

    class Failing(grpc.GenericRpcHandler):
        def service(self, hcd):
            def unary_unary(request, context):
                context.abort(grpc.StatusCode.UNAVAILABLE, "kubelet shutting down")
            return grpc.unary_unary_rpc_method_handler(
                unary_unary, request_deserializer=None,
                response_serializer=None)

    sock = str(tmp_path / "kubelet.sock")
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=2),
                         handlers=(Failing(),))
    server.add_insecure_port(f"unix:{sock}")
    server.start()
    try:
        with pytest.raises(gpumon.PodResourcesError) as ei:
            gpumon.list_pod_resources(sock)
        assert "grpc-status 14" in str(ei.value)  # UNAVAILABLE
        assert "kubelet shutting down" in str(ei.value)
    finally:
        server.stop(grace=None)
