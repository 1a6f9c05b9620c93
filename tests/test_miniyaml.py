"""YAML-subset reader tests (native/common/miniyaml.hpp).

The reader only needs to cover kubectl-generated kubeconfig shapes, but it
must be total over arbitrary text (config files are user input).
"""

import json

import pytest


def y(core, src):
    return json.loads(core._yaml_to_json(src))


def test_basic_mapping_and_scalars(core):
    doc = y(core, "a: 1\nb: text\nc: true\nd: null\ne: \"quoted: x\"\n")
    assert doc == {"a": 1, "b": "text", "c": True, "d": None, "e": "quoted: x"}


def test_nested_blocks_and_same_indent_lists(core):
    doc = y(core, """\
top:
  sub: v
items:
- name: a
  value: 1
- name: b
deep:
- outer: x
  inner:
    k: v
""")
    assert doc["top"] == {"sub": "v"}
    assert doc["items"] == [{"name": "a", "value": 1}, {"name": "b"}]
    assert doc["deep"] == [{"outer": "x", "inner": {"k": "v"}}]


def test_scalar_lists_and_comments(core):
    doc = y(core, """\
# leading comment
letters:
- a
- b # trailing comment
empty:
""")
    assert doc["letters"] == ["a", "b"]
    assert doc["empty"] is None


def test_malformed_raises(core):
    with pytest.raises(core.YamlError):
        y(core, "just a bare scalar line\n")


def test_base64_roundtrip_through_kubeconfig(core, tmp_path, monkeypatch):
    """-data fields survive base64 + temp-file round trip (already covered in
    config tests; here the multiline-wrapped base64 case)."""
    import base64

    pem = b"-----BEGIN CERTIFICATE-----\n" + b"A" * 60 + b"\n-----END CERTIFICATE-----\n"
    wrapped = base64.encodebytes(pem).decode()  # includes newlines
    # miniyaml scalar is single-line; kubeconfigs emit single-line base64,
    # so strip the wrapping as kubectl does
    single = wrapped.replace("\n", "")
    kc = tmp_path / "config"
    kc.write_text(f"""\
current-context: c
clusters:
- name: cl
  cluster:
    server: https://x:6443
    certificate-authority-data: {single}
contexts:
- name: c
  context:
    cluster: cl
    user: u
users:
- name: u
  user:
    token: t
""")
    for var in ("GPU_PRUNER_K8S_URL", "KUBERNETES_SERVICE_HOST"):
        monkeypatch.delenv(var, raising=False)
    monkeypatch.setenv("KUBECONFIG", str(kc))
    cfg = core.resolve_kube_config()
    assert cfg["ca_data"].encode() == pem  # decoded in memory, not to a file


hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings, strategies as st  # noqa: E402


@settings(max_examples=300, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.text(max_size=300))
def test_yaml_reader_total(core, text):
    """Arbitrary text either parses or raises YamlError — never crashes."""
    try:
        core._yaml_to_json(text)
    except core.YamlError:
        pass
