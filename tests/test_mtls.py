"""mTLS client-certificate auth against the apiserver.

Kubernetes clusters commonly authenticate out-of-cluster clients with
client certificates (kubeconfig `client-certificate`/`client-key`); the
native HTTP client supports this via GPU_PRUNER_K8S_CLIENT_CERT/KEY
(native/common/http.cpp SSL_CTX_use_certificate_chain_file path).
"""

import os
import subprocess

import pytest


@pytest.fixture(scope="module")
def pki(tmp_path_factory):
    """CA + server cert (SAN 127.0.0.1) + client cert signed by the CA."""
    d = tmp_path_factory.mktemp("mtls")

    def run(*args):
        subprocess.run(["openssl", *args], check=True, capture_output=True)

    ca_key, ca_crt = d / "ca.key", d / "ca.crt"
    run("req", "-x509", "-newkey", "rsa:2048", "-nodes", "-keyout", str(ca_key),
        "-out", str(ca_crt), "-days", "2", "-subj", "/CN=test-ca")

    def issue(name, cn, san=None):
        key, csr, crt = d / f"{name}.key", d / f"{name}.csr", d / f"{name}.crt"
        run("req", "-newkey", "rsa:2048", "-nodes", "-keyout", str(key),
            "-out", str(csr), "-subj", f"/CN={cn}")
        ext = []
        if san:
            extfile = d / f"{name}.ext"
            extfile.write_text(f"subjectAltName={san}\n")
            ext = ["-extfile", str(extfile)]
        run("x509", "-req", "-in", str(csr), "-CA", str(ca_crt), "-CAkey", str(ca_key),
            "-CAcreateserial", "-out", str(crt), "-days", "2", *ext)
        return key, crt

    s_key, s_crt = issue("server", "127.0.0.1", san="IP:127.0.0.1")
    # Realistic apiserver cert shape: CN is NOT the IP, identity lives only in
    # the IP SAN (in-cluster apiserver certs look like this; a client that
    # runs only a DNS-name check fails the handshake against it).
    ip_key, ip_crt = issue("server-ipsan", "kube-apiserver", san="IP:127.0.0.1")
    c_key, c_crt = issue("client", "gpu-pruner-user")
    return {"ca": str(ca_crt), "server_key": str(s_key), "server_crt": str(s_crt),
            "ipsan_key": str(ip_key), "ipsan_crt": str(ip_crt),
            "client_key": str(c_key), "client_crt": str(c_crt)}


@pytest.fixture
def mtls_api(pki):
    from gpu_pruner_amd.fixtures import FakeApiServer

    with FakeApiServer(certfile=pki["server_crt"], keyfile=pki["server_key"],
                       client_ca=pki["ca"]) as a:
        yield a


def run_pruner(pruner_bin, api_url, prom_url, pki, with_client_cert=True):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = api_url
    env["GPU_PRUNER_K8S_CA"] = pki["ca"]
    env.pop("GPU_PRUNER_K8S_SKIP_TLS", None)
    if with_client_cert:
        env["GPU_PRUNER_K8S_CLIENT_CERT"] = pki["client_crt"]
        env["GPU_PRUNER_K8S_CLIENT_KEY"] = pki["client_key"]
    else:
        env.pop("GPU_PRUNER_K8S_CLIENT_CERT", None)
        env.pop("GPU_PRUNER_K8S_CLIENT_KEY", None)
    env["PROMETHEUS_TOKEN"] = "t"
    return subprocess.run(
        [pruner_bin, "--prometheus-url", prom_url, "--run-mode", "scale-down"],
        capture_output=True, text=True, timeout=30, env=env)


def test_mtls_apiserver_full_cull(pruner_bin, mtls_api, fake_prom, pki):
    dep = mtls_api.add_deployment("d", "ml")
    rs = mtls_api.add_replicaset("d-rs", "ml", owner=dep)
    mtls_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml")
    r = run_pruner(pruner_bin, mtls_api.url, fake_prom.url, pki)
    assert r.returncode == 0, r.stderr
    assert mtls_api.get("Deployment", "ml", "d")["spec"]["replicas"] == 0


def test_mtls_without_client_cert_rejected(pruner_bin, mtls_api, fake_prom, pki):
    """The server requires a client cert: handshake fails, pods are skipped."""
    dep = mtls_api.add_deployment("d", "ml")
    rs = mtls_api.add_replicaset("d-rs", "ml", owner=dep)
    mtls_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml")
    r = run_pruner(pruner_bin, mtls_api.url, fake_prom.url, pki, with_client_cert=False)
    assert mtls_api.get("Deployment", "ml", "d")["spec"]["replicas"] == 1
    assert "TLS" in r.stderr or "error" in r.stderr.lower()


def test_ip_literal_host_verified_against_ip_san(pruner_bin, fake_prom, pki):
    """In-cluster operation: the apiserver URL is an IP literal (ClusterIP) and
    the serving cert carries the identity only in an IP SAN (CN is the
    component name, not the address). Verification must match the IP SAN —
    a DNS-name-only check fails this handshake fail-closed
    (native/common/http.cpp start_tls)."""
    from gpu_pruner_amd.fixtures import FakeApiServer

    with FakeApiServer(certfile=pki["ipsan_crt"], keyfile=pki["ipsan_key"]) as api:
        assert api.url.startswith("https://127.0.0.1:")
        dep = api.add_deployment("d", "ml")
        rs = api.add_replicaset("d-rs", "ml", owner=dep)
        api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                    owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
        fake_prom.add_idle_series("p0", "ml")
        r = run_pruner(pruner_bin, api.url, fake_prom.url, pki, with_client_cert=False)
        assert r.returncode == 0, r.stderr
        assert api.get("Deployment", "ml", "d")["spec"]["replicas"] == 0


def test_kubeconfig_with_inline_mtls_material(pruner_bin, mtls_api, fake_prom, pki,
                                              tmp_path, monkeypatch):
    """A kubectl-style kubeconfig with base64 -data CA + client cert/key
    drives the full cull against an mTLS apiserver — no env overrides."""
    import base64

    b64 = lambda p: base64.b64encode(open(p, "rb").read()).decode()
    kc = tmp_path / "kubeconfig"
    kc.write_text(f"""\
apiVersion: v1
kind: Config
current-context: test
clusters:
- name: test-cluster
  cluster:
    server: {mtls_api.url}
    certificate-authority-data: {b64(pki["ca"])}
contexts:
- name: test
  context:
    cluster: test-cluster
    user: test-user
users:
- name: test-user
  user:
    client-certificate-data: {b64(pki["client_crt"])}
    client-key-data: {b64(pki["client_key"])}
""")
    dep = mtls_api.add_deployment("kc-dep", "ml")
    rs = mtls_api.add_replicaset("kc-dep-rs", "ml", owner=dep)
    mtls_api.add_pod("kc-0", "ml", owner_kind="ReplicaSet", owner_name="kc-dep-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("kc-0", "ml")

    env = dict(os.environ)
    for var in ("GPU_PRUNER_K8S_URL", "GPU_PRUNER_K8S_CA",
                "GPU_PRUNER_K8S_CLIENT_CERT", "GPU_PRUNER_K8S_CLIENT_KEY",
                "KUBERNETES_SERVICE_HOST"):
        env.pop(var, None)
    env["KUBECONFIG"] = str(kc)
    env["PROMETHEUS_TOKEN"] = "t"
    r = subprocess.run(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--run-mode", "scale-down"],
        capture_output=True, text=True, timeout=30, env=env)
    assert r.returncode == 0, r.stderr
    assert mtls_api.get("Deployment", "ml", "kc-dep")["spec"]["replicas"] == 0
