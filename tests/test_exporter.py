"""mi355-exporter CPU-side tests: exposition format + pod attribution chain.

The sampler itself needs gfx950 silicon (tests/test_gpu.py); everything
around it — the Prometheus text rendering the pruner's PromQL consumes and
the KFD/cgroup → pod UID → name attribution — is pinned here with fixture
sysfs/procfs trees (native/exporter/{registry,attrib}.cpp).
"""

import json

import pytest


@pytest.fixture
def gpumon():
    from gpu_pruner_amd import _gpumon

    return _gpumon


SAMPLE = {
    "index": 0,
    "model_name": "AMD Instinct MI355X",
    "unique_id": "abcdef0123456789",
    "drm_render_minor": 128,
    "busy_percent": 42.0,
    "gr_engine_active": 0.415,
    "mem_busy_percent": 7.0,
    "power_w": 612.5,
    "vram_used_b": 2147483648.0,
    "vram_total_b": 309237645312.0,  # 288 GiB HBM3E
    "temp_edge_c": 55.0,
    "gfx_clock_mhz": 2400.0,
}


def test_render_families_and_values(gpumon):
    text = gpumon.render_metrics(json.dumps([SAMPLE]))
    for fam in ("DCGM_FI_PROF_GR_ENGINE_ACTIVE", "DCGM_FI_DEV_GPU_UTIL",
                "DCGM_FI_DEV_POWER_USAGE", "DCGM_FI_DEV_FB_USED",
                "DCGM_FI_DEV_FB_FREE", "DCGM_FI_DEV_GPU_TEMP",
                "DCGM_FI_DEV_SM_CLOCK", "DCGM_FI_DEV_MEM_COPY_UTIL"):
        assert f"# HELP {fam} " in text
        assert f"# TYPE {fam} " in text
    assert 'modelName="AMD Instinct MI355X"' in text
    assert 'gpu="0"' in text
    assert 'device="renderD128"' in text
    assert "} 0.415" in text  # GR_ENGINE_ACTIVE value
    assert "} 612.5" in text  # power
    # FB_USED = 2 GiB in MiB
    assert "} 2048\n" in text


def test_render_idle_is_exact_zero(gpumon):
    """The == 0 PromQL predicate needs a literal 0, not 1e-9."""
    idle = dict(SAMPLE, busy_percent=0.0, gr_engine_active=0.0)
    text = gpumon.render_metrics(json.dumps([idle]))
    line = [l for l in text.splitlines()
            if l.startswith("DCGM_FI_PROF_GR_ENGINE_ACTIVE{")][0]
    assert line.endswith("} 0")


def test_render_attribution_labels(gpumon):
    attribs = {"0": {"pod": "train-abc", "namespace": "ml", "container": "worker"}}
    text = gpumon.render_metrics(json.dumps([SAMPLE]), json.dumps(attribs))
    assert 'pod="train-abc"' in text
    assert 'namespace="ml"' in text
    assert 'container="worker"' in text


def test_render_node_type_const_label(gpumon):
    text = gpumon.render_metrics(json.dumps([SAMPLE]), "", "node-7", "amd-mi355x")
    assert 'Hostname="node-7"' in text
    assert 'node_type="amd-mi355x"' in text


def test_render_multiple_gpus(gpumon):
    samples = [dict(SAMPLE, index=i, drm_render_minor=128 + i) for i in range(8)]
    text = gpumon.render_metrics(json.dumps(samples))
    for i in range(8):
        assert f'gpu="{i}"' in text


def test_render_label_escaping(gpumon):
    weird = dict(SAMPLE, model_name='AMD "MI355X"\\test')
    text = gpumon.render_metrics(json.dumps([weird]))
    assert 'modelName="AMD \\"MI355X\\"\\\\test"' in text


# ---- sampler read-health -----------------------------------


def test_render_unhealthy_device_withholds_activity_series(gpumon):
    """A device whose SMU reads are failing must NOT publish (stale) DCGM
    series — a withheld series can never satisfy the pruner's == 0 idle
    predicate, so broken telemetry fails safe instead of culling."""
    sick = dict(SAMPLE, healthy=False, staleness_s=42.5)
    text = gpumon.render_metrics(json.dumps([sick]))
    assert "DCGM_FI_PROF_GR_ENGINE_ACTIVE{" not in text
    assert "DCGM_FI_DEV_GPU_UTIL{" not in text
    # ...but the health families still report it for alerting
    assert 'mi355_sampler_healthy{gpu="0"' in text
    line = [l for l in text.splitlines() if l.startswith("mi355_sampler_healthy{")][0]
    assert line.endswith("} 0")
    age = [l for l in text.splitlines()
           if l.startswith("mi355_sampler_last_good_read_age_seconds{")][0]
    assert age.endswith("} 42.5")


def test_render_healthy_device_has_health_series(gpumon):
    text = gpumon.render_metrics(json.dumps([SAMPLE]))
    line = [l for l in text.splitlines() if l.startswith("mi355_sampler_healthy{")][0]
    assert line.endswith("} 1")
    assert "DCGM_FI_PROF_GR_ENGINE_ACTIVE{" in text


def test_render_mixed_health_per_device(gpumon):
    samples = [dict(SAMPLE, index=0),
               dict(SAMPLE, index=1, healthy=False, staleness_s=10.0)]
    text = gpumon.render_metrics(json.dumps(samples))
    util = [l for l in text.splitlines() if l.startswith("DCGM_FI_DEV_GPU_UTIL{")]
    assert len(util) == 1 and 'gpu="0"' in util[0]
    health = [l for l in text.splitlines() if l.startswith("mi355_sampler_healthy{")]
    assert len(health) == 2


# ---- ActivityWindow: scrape-idempotent sliding window -----------------------


def test_activity_window_basic_ratio(gpumon):
    w = gpumon.ActivityWindow()
    for i in range(11):  # 10 x 1s segments, half busy
        w.add(float(i), 1.0 if i % 2 == 0 else 0.0)
    r, known = w.ratio(10.0, 10.0)
    assert known == pytest.approx(10.0)
    assert 0.3 < r < 0.7


def test_activity_window_is_idempotent(gpumon):
    w = gpumon.ActivityWindow()
    for i in range(6):
        w.add(float(i), 0.8)
    first = w.ratio(5.0, 5.0)
    second = w.ratio(5.0, 5.0)  # a second "scraper" sees the same value
    assert first == second == (pytest.approx(0.8), pytest.approx(5.0))


def test_activity_window_burst_ages_out(gpumon):
    w = gpumon.ActivityWindow()
    w.add(0.0, 0.0)
    w.add(1.0, 1.0)   # busy during [0,1)
    for t in range(2, 40):
        w.add(float(t), 0.0)
    r, _ = w.ratio(39.0, 5.0)  # window [34,39] is all idle
    assert r == 0.0


def test_activity_window_unknown_segments_never_count_as_idle(gpumon):
    """Failed reads (known=False) contribute to neither busy nor known time:
    a dead read path cannot decay the ratio toward a false 0."""
    w = gpumon.ActivityWindow()
    w.add(0.0, 0.0)
    w.add(1.0, 1.0)  # known busy second
    for t in range(2, 8):
        w.add(float(t), 0.0, False)  # reads failing
    r, known = w.ratio(7.0, 7.0)
    assert known == pytest.approx(1.0)
    assert r == pytest.approx(1.0)  # the only known time was busy


def test_activity_window_no_data(gpumon):
    w = gpumon.ActivityWindow()
    r, known = w.ratio(100.0, 30.0)
    assert r == 0.0 and known == 0.0


def test_activity_window_retention_bounds_memory(gpumon):
    w = gpumon.ActivityWindow()
    w.set_retention(10.0)
    for i in range(1000):
        w.add(float(i), 0.5)
    assert w.size < 20


def test_activity_window_partial_segment_weighting(gpumon):
    w = gpumon.ActivityWindow()
    w.add(0.0, 0.0)
    w.add(10.0, 1.0)  # one long busy segment [0,10)
    r, known = w.ratio(10.0, 5.0)  # window covers only [5,10)
    assert known == pytest.approx(5.0)
    assert r == pytest.approx(1.0)


# ---- cgroup → pod UID -------------------------------------------------------


def test_pod_uid_cgroup_v2(gpumon):
    text = ("0::/kubepods.slice/kubepods-burstable.slice/"
            "kubepods-burstable-pod8f7e1a2b_3c4d_5e6f_7a8b_9c0d1e2f3a4b.slice/"
            "cri-containerd-deadbeef.scope\n")
    assert gpumon.pod_uid_from_cgroup(text) == "8f7e1a2b-3c4d-5e6f-7a8b-9c0d1e2f3a4b"


def test_pod_uid_cgroup_v1(gpumon):
    text = ("11:cpu,cpuacct:/kubepods/besteffort/"
            "pod12345678-abcd-ef01-2345-6789abcdef01/deadbeef\n")
    assert gpumon.pod_uid_from_cgroup(text) == "12345678-abcd-ef01-2345-6789abcdef01"


def test_pod_uid_absent(gpumon):
    assert gpumon.pod_uid_from_cgroup("0::/user.slice/user-0.slice/session-1.scope\n") is None
    assert gpumon.pod_uid_from_cgroup("0::/system.slice/podman.service\n") is None


# ---- KFD proc scan + full attribution chain --------------------------------


@pytest.fixture
def fake_node(tmp_path, monkeypatch):
    """Fixture sysfs/procfs: pid 4242 holds KFD gpu_id 777, owned by pod uid."""
    uid = "8f7e1a2b-3c4d-5e6f-7a8b-9c0d1e2f3a4b"
    kfd = tmp_path / "sys/class/kfd/kfd/proc/4242"
    kfd.mkdir(parents=True)
    (kfd / "vram_777").write_text("1048576\n")
    (kfd / "pasid").write_text("32768\n")
    proc = tmp_path / "proc/4242"
    proc.mkdir(parents=True)
    (proc / "cgroup").write_text(
        "0::/kubepods.slice/kubepods-burstable.slice/"
        f"kubepods-burstable-pod{uid.replace('-', '_')}.slice/cri-x.scope\n")
    monkeypatch.setenv("GPU_EXPORTER_SYSFS_ROOT", str(tmp_path))
    monkeypatch.setenv("GPU_EXPORTER_PROCFS_ROOT", str(tmp_path))
    return {"uid": uid, "pid": 4242, "kfd_gpu_id": 777, "root": tmp_path}


def test_kfd_gpu_pids(gpumon, fake_node):
    pids = gpumon.kfd_gpu_pids()
    assert pids == {777: [4242]}


def test_attribution_via_map_file(gpumon, fake_node, tmp_path, monkeypatch):
    map_file = tmp_path / "podmap.json"
    map_file.write_text(json.dumps({
        fake_node["uid"]: {"pod": "train-0", "namespace": "ml", "container": "worker"},
    }))
    monkeypatch.setenv("GPU_EXPORTER_POD_MAP_FILE", str(map_file))
    monkeypatch.delenv("GPU_PRUNER_K8S_URL", raising=False)
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    a = gpumon.Attributor()
    out = a.resolve([(0, 777)])
    assert out == {0: {"pod": "train-0", "namespace": "ml", "container": "worker"}}


def test_attribution_via_apiserver(gpumon, fake_node, fake_api, monkeypatch):
    monkeypatch.delenv("GPU_EXPORTER_POD_MAP_FILE", raising=False)
    pod = fake_api.add_pod("train-1", "ml")
    pod["metadata"]["uid"] = fake_node["uid"]
    pod["spec"]["containers"] = [{"name": "worker"}]
    a = gpumon.Attributor()
    out = a.resolve([(0, 777)])
    assert out[0]["pod"] == "train-1"
    assert out[0]["namespace"] == "ml"
    assert out[0]["container"] == "worker"


def test_attribution_unowned_gpu(gpumon, fake_node, monkeypatch):
    monkeypatch.delenv("GPU_EXPORTER_POD_MAP_FILE", raising=False)
    monkeypatch.delenv("GPU_PRUNER_K8S_URL", raising=False)
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    a = gpumon.Attributor()
    # kfd id 888 has no pids → no attribution
    assert a.resolve([(0, 888)]) == {}


def test_exporter_binary_fails_loudly_without_gpu():
    """On a host without amdgpu the exporter must exit nonzero with a clear
    error — never serve fake zeros (the == 0 predicate would cull everything)."""
    import subprocess
    from pathlib import Path

    binary = Path(__file__).resolve().parent.parent / "bin" / "mi355-exporter"
    if not binary.exists():
        pytest.skip("exporter not built")
    import torch

    if torch.cuda.is_available():
        pytest.skip("host has a GPU; negative test is for CPU CI")
    r = subprocess.run([str(binary), "-p", "19499"], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode == 1
    assert "sampler init failed" in r.stderr
