"""TLS-mode tests for the Prometheus client + querytest debug-tool tests.

Covers the reference's TLS surface (skip | verify | custom PEM bundle —
reference lib.rs:232-282) against a self-signed HTTPS fixture, and the
querytest binary's table/CSV output (reference src/bin/querytest.rs).
"""

import csv
import os
import subprocess
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent


@pytest.fixture(scope="module")
def self_signed(tmp_path_factory):
    """Self-signed cert for 127.0.0.1 (SAN IP) + key."""
    d = tmp_path_factory.mktemp("certs")
    cert, key = d / "prom.crt", d / "prom.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True)
    return {"cert": str(cert), "key": str(key)}


@pytest.fixture
def tls_prom(self_signed):
    from gpu_pruner_amd.fixtures import FakePrometheus

    with FakePrometheus(certfile=self_signed["cert"], keyfile=self_signed["key"]) as p:
        yield p


def run_pruner(pruner_bin, fake_api, prom_url, *args, timeout=60):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    return subprocess.run(
        [pruner_bin, "--prometheus-url", prom_url, *args],
        capture_output=True, text=True, timeout=timeout, env=env)


def test_tls_verify_rejects_self_signed(pruner_bin, fake_api, tls_prom):
    r = run_pruner(pruner_bin, fake_api, tls_prom.url)  # default: verify
    # one-shot: the query fails (logged), nothing evaluated
    assert "Failed to run query" in r.stderr
    assert tls_prom.queries == []


def test_tls_skip_accepts_self_signed(pruner_bin, fake_api, tls_prom):
    r = run_pruner(pruner_bin, fake_api, tls_prom.url, "--prometheus-tls-mode", "skip")
    assert r.returncode == 0, r.stderr
    assert len(tls_prom.queries) == 1


def test_tls_custom_ca_bundle(pruner_bin, fake_api, tls_prom, self_signed):
    """verify mode + --prometheus-tls-cert pointing at the self-signed cert."""
    r = run_pruner(pruner_bin, fake_api, tls_prom.url,
                   "--prometheus-tls-cert", self_signed["cert"])
    assert r.returncode == 0, r.stderr
    assert len(tls_prom.queries) == 1


def test_tls_bad_ca_file_fails(pruner_bin, fake_api, tls_prom, tmp_path):
    bad = tmp_path / "bad.pem"
    bad.write_text("not a pem")
    r = run_pruner(pruner_bin, fake_api, tls_prom.url,
                   "--prometheus-tls-cert", str(bad))
    assert "Failed to run query" in r.stderr or r.returncode != 0


# ---- querytest --------------------------------------------------------------


@pytest.fixture
def querytest_bin():
    path = REPO_ROOT / "bin" / "querytest"
    if not path.exists():
        pytest.skip("bin/querytest not built")
    return str(path)


def test_querytest_vector_table_and_csv(querytest_bin, fake_prom, tmp_path):
    fake_prom.add_idle_series("pod-a", "ml", value=0.25)
    fake_prom.add_idle_series("pod-b", "ml", value=0.5)
    env = dict(os.environ)
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run(
        [querytest_bin, "up{job='x'}", fake_prom.url],
        capture_output=True, text=True, timeout=30, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr
    assert "pod-a" in r.stdout and "pod-b" in r.stdout
    assert "0.25" in r.stdout
    # raw query forwarded verbatim
    assert fake_prom.queries == ["up{job='x'}"]
    # CSV written next to cwd
    out_csv = tmp_path / "output.csv"
    assert out_csv.exists()
    rows = list(csv.reader(out_csv.open()))
    assert len(rows) == 2
    assert any("pod-a" in cell for cell in rows[0] + rows[1])


def test_querytest_requires_args(querytest_bin):
    r = subprocess.run([querytest_bin], capture_output=True, text=True, timeout=10)
    assert r.returncode == 2
    assert "usage" in r.stderr


def test_querytest_matrix_result(querytest_bin, fake_prom, tmp_path):
    """Range-vector (matrix) results print one row per sample."""
    fake_prom.data_override = {
        "resultType": "matrix",
        "result": [{
            "metric": {"pod": "pod-m", "namespace": "ml"},
            "values": [[1700000000, "0.1"], [1700000060, "0.2"]],
        }],
    }
    env = dict(os.environ)
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run(
        [querytest_bin, "rate(x[5m])", fake_prom.url],
        capture_output=True, text=True, timeout=30, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr
    assert r.stdout.count("pod-m") == 2
    assert "0.1" in r.stdout and "0.2" in r.stdout
    rows = list(csv.reader((tmp_path / "output.csv").open()))
    assert len(rows) == 2


def test_querytest_scalar_unsupported(querytest_bin, fake_prom):
    """Scalar results are rejected (parity with reference querytest.rs:54-56)."""
    fake_prom.data_override = {"resultType": "scalar", "result": [1700000000, "1"]}
    env = dict(os.environ)
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run([querytest_bin, "1", fake_prom.url],
                       capture_output=True, text=True, timeout=30, env=env)
    assert r.returncode == 1
    assert "Scalar data not supported" in r.stderr
