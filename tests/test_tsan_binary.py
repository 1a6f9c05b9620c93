"""Full-daemon ThreadSanitizer e2e (sanitizers stand in for
the reference's Rust guarantees).

The `make -C native tsan` unit tier covers single-threaded components; THIS
runs the whole daemon binary compiled with -fsanitize=thread — informer
watch threads, the consumer pool, parallel_for fan-outs, OTLP span buffers —
through real scale-down ticks and asserts TSan stayed silent. Built on
demand (skipped when bin/gpu-pruner-tsan is absent and make is unavailable).
"""

import os
import subprocess
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
TSAN_BIN = REPO_ROOT / "bin" / "gpu-pruner-tsan"


@pytest.fixture(scope="module")
def tsan_bin():
    if not TSAN_BIN.exists():
        r = subprocess.run(["make", "-C", str(REPO_ROOT / "native"), "tsan-bin"],
                           capture_output=True, text=True, timeout=600)
        if r.returncode != 0 or not TSAN_BIN.exists():
            pytest.skip("could not build the TSan daemon binary")
    return str(TSAN_BIN)


def test_tsan_daemon_watch_scaledown(tsan_bin, fake_api, fake_prom):
    """Several daemon ticks with watch informers + OTLP + leader election
    under TSan (elector thread included in the race surface)."""
    from gpu_pruner_amd.fixtures import FakeOtlpCollector, build_synthetic_cluster

    build_synthetic_cluster(fake_api, fake_prom, n_pods=40, pods_per_parent=2)
    with FakeOtlpCollector() as col:
        env = dict(os.environ)
        env["GPU_PRUNER_K8S_URL"] = fake_api.url
        env["PROMETHEUS_TOKEN"] = "t"
        env["GPU_PRUNER_LOG"] = "error"
        env["OTEL_EXPORTER_OTLP_ENDPOINT"] = col.url
        env["OTEL_METRIC_EXPORT_INTERVAL"] = "500"
        env["TSAN_OPTIONS"] = "halt_on_error=0 exitcode=66"
        env["POD_NAME"] = "tsan-replica"
        env["POD_NAMESPACE"] = "gpu-pruner-system"
        p = subprocess.Popen(
            [tsan_bin, "--prometheus-url", fake_prom.url, "--daemon-mode",
             "--run-mode", "scale-down", "--check-interval", "1",
             "--eval-strategy", "watch", "--leader-elect",
             "--leader-elect-renew-period", "1"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        import time
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and len(fake_prom.queries) < 5:
            time.sleep(0.2)
        p.terminate()
        out, err = p.communicate(timeout=30)
        text = err.decode(errors="replace")
    assert len(fake_prom.queries) >= 5, text[-2000:]
    assert "WARNING: ThreadSanitizer" not in text, text[-4000:]
    assert p.returncode != 66, "TSan reported races"
    # the daemon actually worked: parents scaled + events posted
    assert fake_api.get("Deployment", "ml-team-0", "dep-0")["spec"]["replicas"] == 0
    assert any(e["metadata"]["name"].startswith("gpuscaler-") for e in fake_api.events)
