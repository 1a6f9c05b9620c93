"""Kubernetes config + Prometheus token-chain resolution tests.

Pins the in-cluster config path (KUBERNETES_SERVICE_HOST + service-account
dir: token file, ca.crt, namespace — what kube-rs Config::infer does for the
reference) and the Prometheus token chain ($PROMETHEUS_TOKEN → SA token file
→ K8s token env → `oc whoami -t`, reference lib.rs:205-230).
"""

import pytest


@pytest.fixture
def sa_dir(tmp_path):
    d = tmp_path / "serviceaccount"
    d.mkdir()
    (d / "token").write_text("sa-token-123\n")
    (d / "ca.crt").write_text("-----BEGIN CERTIFICATE-----\nfake\n-----END CERTIFICATE-----\n")
    (d / "namespace").write_text("gpu-pruner-system")
    return d


@pytest.fixture
def clean_env(monkeypatch):
    for var in ("GPU_PRUNER_K8S_URL", "GPU_PRUNER_K8S_TOKEN",
                "GPU_PRUNER_K8S_TOKEN_FILE", "GPU_PRUNER_K8S_CA",
                "GPU_PRUNER_K8S_CLIENT_CERT", "GPU_PRUNER_K8S_CLIENT_KEY",
                "GPU_PRUNER_K8S_SKIP_TLS", "KUBERNETES_SERVICE_HOST",
                "KUBERNETES_SERVICE_PORT", "GPU_PRUNER_SA_DIR",
                "PROMETHEUS_TOKEN", "KUBECONFIG"):
        monkeypatch.delenv(var, raising=False)
    # a stray ~/.kube/config must not leak into hermetic tests
    monkeypatch.setenv("HOME", "/nonexistent-home")
    return monkeypatch


def test_in_cluster_config(core, clean_env, sa_dir):
    clean_env.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    clean_env.setenv("KUBERNETES_SERVICE_PORT", "6443")
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(sa_dir))
    cfg = core.resolve_kube_config()
    assert cfg["url"] == "https://10.0.0.1:6443"
    assert cfg["token_file"] == str(sa_dir / "token")
    assert cfg["ca_file"] == str(sa_dir / "ca.crt")
    assert cfg["default_namespace"] == "gpu-pruner-system"
    assert cfg["skip_tls"] is False


def test_in_cluster_ipv6_host(core, clean_env, sa_dir):
    clean_env.setenv("KUBERNETES_SERVICE_HOST", "fd00::1")
    clean_env.setenv("KUBERNETES_SERVICE_PORT", "443")
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(sa_dir))
    cfg = core.resolve_kube_config()
    assert cfg["url"] == "https://[fd00::1]:443"


def test_env_override_takes_precedence(core, clean_env, sa_dir):
    clean_env.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(sa_dir))
    clean_env.setenv("GPU_PRUNER_K8S_URL", "http://127.0.0.1:8080")
    clean_env.setenv("GPU_PRUNER_K8S_TOKEN", "override-token")
    cfg = core.resolve_kube_config()
    assert cfg["url"] == "http://127.0.0.1:8080"
    assert cfg["token"] == "override-token"


def test_no_config_raises(core, clean_env):
    with pytest.raises(RuntimeError, match="no Kubernetes config"):
        core.resolve_kube_config()


# ---- Prometheus token chain -------------------------------------------------


def test_prom_token_env_first(core, clean_env, sa_dir):
    clean_env.setenv("PROMETHEUS_TOKEN", "env-tok")
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(sa_dir))
    assert core.get_prometheus_token() == "env-tok"


def test_prom_token_sa_file_second(core, clean_env, sa_dir):
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(sa_dir))
    assert core.get_prometheus_token() == "sa-token-123"


def test_prom_token_k8s_env_third(core, clean_env, tmp_path):
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(tmp_path / "nonexistent"))
    clean_env.setenv("GPU_PRUNER_K8S_TOKEN", "k8s-tok")
    assert core.get_prometheus_token() == "k8s-tok"


def test_prom_token_oc_whoami_last(core, clean_env, tmp_path, monkeypatch):
    """Last resort shells out to `oc whoami -t`."""
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(tmp_path / "nonexistent"))
    oc = tmp_path / "oc"
    oc.write_text("#!/bin/sh\necho oc-token-456\n")
    oc.chmod(0o755)
    monkeypatch.setenv("PATH", f"{tmp_path}:/usr/bin:/bin")
    assert core.get_prometheus_token() == "oc-token-456"


# ---- kubeconfig ($KUBECONFIG / ~/.kube/config) ------------------------------


KUBECONFIG_TMPL = """\
apiVersion: v1
kind: Config
current-context: prod
clusters:
- name: prod-cluster
  cluster:
    server: https://api.prod.example:6443
    certificate-authority-data: {ca_b64}
- name: other
  cluster:
    server: https://other:6443
contexts:
- name: prod
  context:
    cluster: prod-cluster
    user: admin
    namespace: ml-team
- name: stale
  context:
    cluster: other
    user: admin
users:
- name: admin
  user:
    token: kubeconfig-token-123
"""


def test_kubeconfig_token_auth(core, clean_env, tmp_path):
    import base64
    import glob

    ca_pem = "-----BEGIN CERTIFICATE-----\nabc\n-----END CERTIFICATE-----\n"
    kc = tmp_path / "config"
    kc.write_text(KUBECONFIG_TMPL.format(
        ca_b64=base64.b64encode(ca_pem.encode()).decode()))
    clean_env.setenv("KUBECONFIG", str(kc))
    tmp_before = set(glob.glob("/tmp/gpu-pruner-*"))
    cfg = core.resolve_kube_config()
    assert cfg["url"] == "https://api.prod.example:6443"
    assert cfg["token"] == "kubeconfig-token-123"
    assert cfg["default_namespace"] == "ml-team"
    # -data CA decoded IN MEMORY — never written to disk (key material must
    # not accumulate in /tmp across the daemon's per-tick config re-resolves)
    assert cfg["ca_file"] is None
    assert cfg["ca_data"] == ca_pem
    assert set(glob.glob("/tmp/gpu-pruner-*")) == tmp_before


def test_kubeconfig_insecure_and_cert_files(core, clean_env, tmp_path):
    kc = tmp_path / "config"
    kc.write_text("""\
current-context: c
clusters:
- name: cl
  cluster:
    server: https://x:6443
    insecure-skip-tls-verify: true
contexts:
- name: c
  context:
    cluster: cl
    user: u
users:
- name: u
  user:
    client-certificate: /etc/certs/me.crt
    client-key: /etc/certs/me.key
""")
    clean_env.setenv("KUBECONFIG", str(kc))
    cfg = core.resolve_kube_config()
    assert cfg["skip_tls"] is True
    # client cert/key paths surface through KubeClient → http mTLS options


def test_env_override_beats_kubeconfig(core, clean_env, tmp_path):
    kc = tmp_path / "config"
    kc.write_text("current-context: c\n")
    clean_env.setenv("KUBECONFIG", str(kc))
    clean_env.setenv("GPU_PRUNER_K8S_URL", "http://127.0.0.1:1234")
    assert core.resolve_kube_config()["url"] == "http://127.0.0.1:1234"


def test_prom_token_from_kubeconfig(core, clean_env, tmp_path):
    import base64

    kc = tmp_path / "config"
    kc.write_text(KUBECONFIG_TMPL.format(
        ca_b64=base64.b64encode(b"x").decode()))
    clean_env.setenv("KUBECONFIG", str(kc))
    clean_env.setenv("GPU_PRUNER_SA_DIR", str(tmp_path / "nosa"))
    assert core.get_prometheus_token() == "kubeconfig-token-123"


# ---- exec credential plugins (client.authentication.k8s.io) -----------------


EXEC_KUBECONFIG = """\
current-context: c
clusters:
- name: cl
  cluster:
    server: {server}
contexts:
- name: c
  context:
    cluster: cl
    user: u
users:
- name: u
  user:
    exec:
      apiVersion: client.authentication.k8s.io/v1
      command: {command}
"""


@pytest.fixture
def exec_plugin(tmp_path):
    """Stub credential plugin: emits an ExecCredential and counts its runs."""
    counter = tmp_path / "invocations"
    counter.write_text("0")
    script = tmp_path / "fake-auth-plugin"
    script.write_text(f"""#!/bin/sh
n=$(cat {counter})
echo $((n + 1)) > {counter}
cat <<JSON
{{"apiVersion": "client.authentication.k8s.io/v1", "kind": "ExecCredential",
 "status": {{"token": "exec-token-xyz",
            "expirationTimestamp": "2099-01-01T00:00:00Z"}}}}
JSON
""")
    script.chmod(0o755)
    return {"script": script, "counter": counter}


def test_kubeconfig_exec_plugin_parsed(core, clean_env, tmp_path, exec_plugin):
    kc = tmp_path / "config"
    kc.write_text(EXEC_KUBECONFIG.format(server="https://x:6443",
                                         command=exec_plugin["script"]))
    clean_env.setenv("KUBECONFIG", str(kc))
    cfg = core.resolve_kube_config()
    assert cfg["exec_command"] == str(exec_plugin["script"])
    assert cfg["token"] is None  # minted lazily by the client, not at resolve


def test_exec_plugin_token_authenticates_e2e(pruner_bin, clean_env, tmp_path,
                                             exec_plugin, fake_prom):
    """Full cull against an apiserver that only accepts the plugin-minted
    bearer token, via a kubeconfig exec user."""
    import os
    import subprocess

    from gpu_pruner_amd.fixtures import FakeApiServer

    with FakeApiServer(token="exec-token-xyz") as api:
        dep = api.add_deployment("d", "ml")
        rs = api.add_replicaset("d-rs", "ml", owner=dep)
        api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                    owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
        fake_prom.add_idle_series("p0", "ml")
        kc = tmp_path / "config"
        kc.write_text(EXEC_KUBECONFIG.format(server=api.url,
                                             command=exec_plugin["script"]))
        env = dict(os.environ)
        for var in ("GPU_PRUNER_K8S_URL", "KUBERNETES_SERVICE_HOST"):
            env.pop(var, None)
        env["KUBECONFIG"] = str(kc)
        env["PROMETHEUS_TOKEN"] = "t"
        r = subprocess.run(
            [pruner_bin, "--prometheus-url", fake_prom.url,
             "--run-mode", "scale-down"],
            capture_output=True, text=True, timeout=60, env=env)
        assert r.returncode == 0, r.stderr
        assert api.get("Deployment", "ml", "d")["spec"]["replicas"] == 0


def test_exec_plugin_cached_across_ticks(pruner_bin, clean_env, tmp_path,
                                         exec_plugin, fake_prom):
    """The plugin is forked once, not once per tick: its token is cached
    until expirationTimestamp."""
    import os
    import subprocess
    import time

    from gpu_pruner_amd.fixtures import FakeApiServer

    with FakeApiServer(token="exec-token-xyz") as api:
        dep = api.add_deployment("d", "ml")
        rs = api.add_replicaset("d-rs", "ml", owner=dep)
        api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                    owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
        fake_prom.add_idle_series("p0", "ml")
        kc = tmp_path / "config"
        kc.write_text(EXEC_KUBECONFIG.format(server=api.url,
                                             command=exec_plugin["script"]))
        env = dict(os.environ)
        for var in ("GPU_PRUNER_K8S_URL", "KUBERNETES_SERVICE_HOST"):
            env.pop(var, None)
        env["KUBECONFIG"] = str(kc)
        env["PROMETHEUS_TOKEN"] = "t"
        p = subprocess.Popen(
            [pruner_bin, "--prometheus-url", fake_prom.url,
             "--run-mode", "scale-down", "--daemon-mode", "--check-interval", "0"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        try:
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline:
                if len(fake_prom.queries) >= 5:  # several ticks completed
                    break
                time.sleep(0.1)
            assert len(fake_prom.queries) >= 5
        finally:
            p.terminate()
            p.wait(timeout=10)
        assert api.get("Deployment", "ml", "d")["spec"]["replicas"] == 0
        invocations = int(exec_plugin["counter"].read_text().strip())
        assert invocations == 1, f"plugin forked {invocations} times"
