"""GPU tests — require a real MI355X (gfx950); run via gpurun / the driver.

Prove the counter-semantics contract on silicon (SURVEY.md §7 "Hard parts"):
the first-party sampler must read exactly 0 on an idle GPU and rise under the
gfx950 HIP busy probe, in both the instantaneous busy percent and the
windowed GR_ENGINE_ACTIVE ratio — then the whole culler stack must make the
right decision from that real signal.
"""

import json
import os
import subprocess
import time
import urllib.request
from pathlib import Path

import pytest

pytestmark = pytest.mark.gpu

REPO_ROOT = Path(__file__).resolve().parent.parent


def _require_gpu():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU visible")


@pytest.fixture(scope="module")
def sampler():
    _require_gpu()
    from gpu_pruner_amd import _gpumon

    # 2 s sliding window: short enough that idle/busy transitions age out
    # within a test, long enough to span many 100 ms polls
    s = _gpumon.Sampler(poll_interval_ms=100, window_s=2.0)
    s.init()  # must not silently fall back — raises SamplerError if broken
    yield s
    s.stop()


def _settle_idle(sampler, seconds=5.0, target=0.0):
    """Poll until busy_percent settles at `target` (returns last value)."""
    deadline = time.monotonic() + seconds
    busy = None
    while time.monotonic() < deadline:
        sampler.poll_once()
        busy = sampler.snapshot()[0]["busy_percent"]
        if busy == target:
            return busy
        time.sleep(0.1)
    return busy


def test_sampler_enumerates_mi355x(sampler):
    assert sampler.device_count >= 1
    snap = sampler.snapshot()[0]
    assert snap["model_name"], "model name must be non-empty"
    # 288 GB HBM3E per MI355X
    assert snap["vram_total_b"] > 200 * 2**30, snap["vram_total_b"]
    assert snap["pci_bdf"], "PCI BDF must resolve"


def test_idle_utilization_is_exactly_zero(sampler):
    """The PromQL `== 0` predicate depends on a literal zero when idle."""
    busy = _settle_idle(sampler, seconds=10.0)
    assert busy == 0.0, f"idle GPU reports busy_percent={busy}"
    # keep polling so the 2 s sliding window is entirely idle time
    for _ in range(25):
        time.sleep(0.1)
        sampler.poll_once()
    ratio = sampler.snapshot()[0]["gr_engine_active"]
    assert ratio < 0.01, f"idle GR_ENGINE_ACTIVE={ratio}"
    assert sampler.snapshot()[0]["healthy"] is True


def test_busy_probe_raises_utilization(sampler):
    from gpu_pruner_amd import probe

    _settle_idle(sampler, seconds=5.0)
    probe.start(device=0, max_seconds=30.0)
    try:
        busy = 0.0
        for _ in range(40):
            time.sleep(0.1)
            sampler.poll_once()
            busy = sampler.snapshot()[0]["busy_percent"]
            if busy >= 90.0:
                break
        assert busy >= 90.0, f"probe should saturate the GPU (busy={busy})"
        # hold the load for a full 2 s sliding window before reading the ratio
        for _ in range(22):
            time.sleep(0.1)
            sampler.poll_once()
        ratio = sampler.snapshot()[0]["gr_engine_active"]
    finally:
        probe.stop()
    assert ratio > 0.3, f"windowed ratio under load = {ratio}"
    # and utilization must fall back to zero afterwards
    busy = _settle_idle(sampler, seconds=10.0)
    assert busy == 0.0, f"busy stuck at {busy} after probe stop"


def test_power_and_clock_sane(sampler):
    sampler.poll_once()
    snap = sampler.snapshot()[0]
    assert 20.0 < snap["power_w"] < 1600.0, snap["power_w"]  # idle..TDP range
    assert snap["metrics_table_ok"], "gpu_metrics table must be readable"


def test_activity_acc_counter_advances_under_load(sampler):
    """The firmware gfx_activity_acc accumulator must advance under load —
    it is the GRBM_GUI_ACTIVE-derived signal for the windowed ratio."""
    from gpu_pruner_amd import probe

    sampler.poll_once()
    before = sampler.snapshot()[0]["gfx_activity_acc"]
    with probe.busy_load(device=0, max_seconds=15.0):
        time.sleep(1.5)
        sampler.poll_once()
    after = sampler.snapshot()[0]["gfx_activity_acc"]
    assert after > before, f"gfx_activity_acc did not advance ({before} -> {after})"


def test_exporter_binary_serves_real_metrics():
    _require_gpu()
    port = 19400
    proc = subprocess.Popen(
        [str(REPO_ROOT / "bin" / "mi355-exporter"), "-p", str(port),
         "-b", "127.0.0.1", "-i", "200", "--node-type", "amd-mi355x"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        text = None
        for _ in range(50):
            time.sleep(0.2)
            if proc.poll() is not None:
                raise AssertionError(
                    f"exporter died: {proc.stderr.read().decode()[:500]}")
            try:
                text = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
                break
            except OSError:
                continue
        assert text is not None, "exporter never served /metrics"
        assert "DCGM_FI_PROF_GR_ENGINE_ACTIVE{" in text
        assert "DCGM_FI_DEV_GPU_UTIL{" in text
        assert "DCGM_FI_DEV_POWER_USAGE{" in text
        assert 'node_type="amd-mi355x"' in text
        assert 'modelName=' in text
        # full family surface incl. the AMD-native xGMI extras
        for fam in ("DCGM_FI_DEV_FB_USED", "DCGM_FI_DEV_FB_FREE",
                    "DCGM_FI_DEV_GPU_TEMP", "DCGM_FI_DEV_SM_CLOCK",
                    "DCGM_FI_DEV_TOTAL_ENERGY_CONSUMPTION",
                    "mi355_xgmi_link_width", "mi355_xgmi_read_kb_total",
                    "mi355_sampler_healthy",
                    "mi355_sampler_last_good_read_age_seconds"):
            assert f"# TYPE {fam} " in text, f"missing family {fam}"
        healthy = _parse_prom_text(text, "mi355_sampler_healthy")
        assert healthy and all(v == 1.0 for _, v in healthy), healthy
        # scrapes are idempotent: two back-to-back scrapes agree on the
        # windowed activity value (reset-on-scrape would zero the second)
        text2 = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
        a1 = _parse_prom_text(text, "DCGM_FI_PROF_GR_ENGINE_ACTIVE")
        a2 = _parse_prom_text(text2, "DCGM_FI_PROF_GR_ENGINE_ACTIVE")
        for (l1, v1), (l2, v2) in zip(a1, a2):
            assert abs(v1 - v2) < 0.05, (v1, v2)
        health = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/healthz", timeout=2).read()
        assert health == b"ok\n"
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def _parse_prom_text(text, family):
    """family{labels} value → list of (labels_dict, float)."""
    out = []
    for line in text.splitlines():
        if not line.startswith(family + "{"):
            continue
        labels_part, value = line.rsplit("} ", 1)
        labels = {}
        for item in labels_part[len(family) + 1:].split('",'):
            if "=" in item:
                k, v = item.split("=", 1)
                labels[k.strip(",")] = v.strip('"')
        out.append((labels, float(value)))
    return out


def test_full_stack_idle_cull_from_real_gpu_signal(fake_api, fake_prom, pruner_bin):
    """The headline e2e (BASELINE config 2 analog): a real idle MI355X, its
    activity scraped from the live mi355-exporter, drives the culler to scale
    an idle Notebook to zero."""
    _require_gpu()
    port = 19401
    exporter = subprocess.Popen(
        [str(REPO_ROOT / "bin" / "mi355-exporter"), "-p", str(port),
         "-b", "127.0.0.1", "-i", "200"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        text = None
        for _ in range(50):
            time.sleep(0.2)
            try:
                text = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
                break
            except OSError:
                continue
        assert text, "exporter unreachable"
        time.sleep(1.0)  # let an idle window accumulate
        text = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
        series = _parse_prom_text(text, "DCGM_FI_PROF_GR_ENGINE_ACTIVE")
        assert series, "no activity series"
        gpu0 = [s for s in series if s[0].get("gpu") == "0"][0]
        activity = gpu0[1]
        assert activity == 0.0, f"idle GPU activity={activity}"

        # stand in for Prometheus: the scraped value becomes the query result
        nb = fake_api.add_notebook("workbench", "ml")
        fake_api.add_statefulset("workbench-ss", "ml", notebook_owner=nb)
        fake_api.add_pod("workbench-ss-0", "ml", owner_kind="StatefulSet",
                         owner_name="workbench-ss", age_s=3 * 3600)
        fake_prom.add_idle_series("workbench-ss-0", "ml", value=activity,
                                  model_name=gpu0[0].get("modelName", "AMD"))

        env = dict(os.environ)
        env["GPU_PRUNER_K8S_URL"] = fake_api.url
        env["PROMETHEUS_TOKEN"] = "t"
        r = subprocess.run(
            [pruner_bin, "--prometheus-url", fake_prom.url, "--run-mode", "scale-down"],
            capture_output=True, text=True, timeout=60, env=env)
        assert r.returncode == 0, r.stderr
        nb_after = fake_api.get("Notebook", "ml", "workbench")
        assert "kubeflow-resource-stopped" in nb_after["metadata"].get("annotations", {})
    finally:
        exporter.terminate()
        exporter.wait(timeout=10)


def test_bench_one_gpu_quick():
    """bench.py runs with the real sampler and prints the contract line."""
    _require_gpu()
    r = subprocess.run(
        [os.environ.get("PYTHON", "python3"), str(REPO_ROOT / "bench.py"),
         "--steps", "5", "--warmup", "1", "--pods", "200"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO_ROOT))
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    result = json.loads(line)
    assert result["metric"] == "pods_evaluated_per_sec"
    assert result["value"] > 0
    assert "real GPU" in result["config"]["utilization_source"]


def test_busy_gpu_is_not_culled_idle_gpu_is():
    """Closed-loop semantic test on silicon: while the busy probe runs, the
    idle query yields no candidates (the synthetic Prometheus honors the
    == 0 predicate on the real activity value); once the GPU settles idle,
    the same cluster is culled."""
    _require_gpu()
    import json
    from gpu_pruner_amd import _gpumon, _pruner_core as core, probe

    sampler = _gpumon.Sampler(poll_interval_ms=100, window_s=1.0)
    sampler.init()
    backend = core.SyntheticBackend(n_pods=20)
    backend.start()
    os.environ["GPU_PRUNER_K8S_URL"] = backend.k8s_url
    os.environ["PROMETHEUS_TOKEN"] = "t"
    cfg = json.dumps({"duration": 30, "grace_period": 300,
                      "run_mode": "scale-down",
                      "prometheus_url": backend.prom_url})
    try:
        # phase 1: GPU busy → no candidates, nothing scaled
        probe.start(device=0, max_seconds=30.0)
        try:
            busy = 0.0
            for _ in range(40):
                time.sleep(0.1)
                sampler.poll_once()
                busy = sampler.snapshot()[0]["busy_percent"]
                if busy >= 90.0:
                    break
            # poll through a full 1 s window of busy time
            for _ in range(12):
                time.sleep(0.1)
                sampler.poll_once()
            ratio = sampler.snapshot()[0]["gr_engine_active"]
            backend.set_series_value(ratio)
            assert ratio > 0.0
            out = core.run_tick(cfg)
            assert out["num_unique_pods"] == 0, "busy GPU must yield no candidates"
            assert backend.scale_patches == 0
        finally:
            probe.stop()

        # phase 2: GPU idle → all pods candidates, parents culled
        for _ in range(100):
            time.sleep(0.1)
            sampler.poll_once()
            if sampler.snapshot()[0]["busy_percent"] == 0.0:
                break
        # poll through a full idle window so the busy burst ages out
        for _ in range(15):
            time.sleep(0.1)
            sampler.poll_once()
        ratio = sampler.snapshot()[0]["gr_engine_active"]
        backend.set_series_value(ratio)
        assert ratio == 0.0, f"idle ratio={ratio}"
        out = core.run_tick(cfg)
        assert out["num_unique_pods"] == 20
        assert out["scaled"] == out["shutdown_events"] == 10
    finally:
        backend.stop()
        sampler.stop()


def test_window_semantics_with_real_gpu_signal(fake_api, pruner_bin):
    """Three-tier pipeline on silicon: the exporter's real activity series
    are scraped into MiniProm (which executes max_over_time like Prometheus)
    and the daemon decides from them. A real probe burst inside the 1-minute
    window protects the pod; once the burst ages out, the pod is culled."""
    _require_gpu()
    from gpu_pruner_amd.fixtures import MiniProm
    from gpu_pruner_amd import probe

    port = 19402
    # short sliding window so the burst ages out of the *exporter's* window
    # quickly; the lookback semantics under test live in MiniProm/PromQL
    exporter = subprocess.Popen(
        [str(REPO_ROOT / "bin" / "mi355-exporter"), "-p", str(port),
         "-b", "127.0.0.1", "-i", "200", "--activity-window", "3"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)

    def scrape_into(prom):
        text = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=3).read().decode()
        series = _parse_prom_text(text, "DCGM_FI_PROF_GR_ENGINE_ACTIVE")
        gpu0 = [s for s in series if s[0].get("gpu") == "0"][0]
        prom.ingest_activity("train-0", "ml", gpu0[1])
        return gpu0[1]

    dep = fake_api.add_deployment("train", "ml")
    rs = fake_api.add_replicaset("train-rs", "ml", owner=dep)
    fake_api.add_pod("train-0", "ml", owner_kind="ReplicaSet", owner_name="train-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)

    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"

    with MiniProm() as prom:
        try:
            for _ in range(50):  # wait for the exporter
                time.sleep(0.2)
                try:
                    urllib.request.urlopen(f"http://127.0.0.1:{port}/healthz",
                                           timeout=2).read()
                    break
                except OSError:
                    continue
            # real burst: probe load, scraped into the store
            with probe.busy_load(device=0, max_seconds=20.0):
                time.sleep(1.5)
                busy_val = scrape_into(prom)
            burst_t = time.monotonic()
            assert busy_val > 0.0
            # settle + scrape an idle sample
            deadline = time.monotonic() + 15
            idle_val = 1.0
            while time.monotonic() < deadline:
                time.sleep(1.0)
                idle_val = scrape_into(prom)
                if idle_val == 0.0:
                    break
            assert idle_val == 0.0

            args = [pruner_bin, "--prometheus-url", prom.url, "--run-mode",
                    "scale-down", "-t", "1", "--grace-period", "0"]
            # burst still inside the 1-minute window → protected
            r = subprocess.run(args, capture_output=True, text=True, timeout=30, env=env)
            assert r.returncode == 0, r.stderr
            assert fake_api.get("Deployment", "ml", "train")["spec"]["replicas"] == 1

            # keep scraping idle samples until the burst has aged out of the
            # 1-minute lookback (plus margin for the scrape-timestamp grid)
            while time.monotonic() - burst_t < 75.0:
                time.sleep(1.0)
                scrape_into(prom)
            r = subprocess.run(args, capture_output=True, text=True, timeout=30, env=env)
            assert r.returncode == 0, r.stderr
            assert fake_api.get("Deployment", "ml", "train")["spec"]["replicas"] == 0
        finally:
            exporter.terminate()
            exporter.wait(timeout=10)
