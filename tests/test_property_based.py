"""Property-based tests (hypothesis) over the native parsers/builders.

The JSON library carries every K8s object in the system; the cgroup parser
sees arbitrary host text; the query builder sees user-controlled flag
values — all three must be total functions over their input domains.
"""

import json as pyjson

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings, strategies as st  # noqa: E402


json_values = st.recursive(
    st.none() | st.booleans()
    | st.integers(min_value=-(2**53), max_value=2**53)
    | st.floats(allow_nan=False, allow_infinity=False, width=64)
    | st.text(max_size=40),
    lambda children: st.lists(children, max_size=5)
    | st.dictionaries(st.text(max_size=20), children, max_size=5),
    max_leaves=25,
)


@settings(max_examples=200, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(json_values)
def test_json_roundtrip_through_native(core, value):
    """python json → native parse → native dump → python json == original.

    Pods/owners flow through exactly this path (apiserver → jsn::parse →
    dump in fixtures), so cross-implementation round-trip fidelity is the
    contract. (ScaleKind equality for built-ins is full-object equality —
    reference lib.rs:47-49 — so byte-level fidelity matters.)
    """
    text = pyjson.dumps(value)
    sk = core.ScaleKind("Deployment", pyjson.dumps({"metadata": {"name": "x"},
                                                    "payload": value}))
    back = pyjson.loads(sk.object_json())["payload"]
    assert back == value


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=200))
def test_cgroup_parser_total(gpumon_mod, text):
    """Arbitrary cgroup text never crashes; any result is a canonical uuid."""
    out = gpumon_mod.pod_uid_from_cgroup(text)
    if out is not None:
        assert len(out) == 36
        assert out == out.lower()
        assert sum(c in "0123456789abcdef" for c in out) == 32


@settings(max_examples=100, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.text(max_size=60), st.text(max_size=60),
       st.integers(min_value=1, max_value=100000))
def test_query_builder_total_and_contained(core, ns, model, duration):
    """Any flag values produce a query that still ends in the idle predicate
    and whose matcher strings never escape their quotes."""
    q = core.render_query(pyjson.dumps(
        {"duration": duration, "namespace": ns, "model_name": model}))
    assert q.rstrip().endswith("== 0")
    assert f"[{duration}m]" in q


@settings(max_examples=150, deadline=None)
@given(st.binary(max_size=120))
def test_podresources_decoder_total(gpumon_mod, payload):
    """Arbitrary bytes: the protobuf decoder either parses or raises
    PodResourcesError — never crashes or loops."""
    try:
        gpumon_mod.decode_list_response(payload)
    except gpumon_mod.PodResourcesError:
        pass


@pytest.fixture(scope="module")
def gpumon_mod():
    from gpu_pruner_amd import _gpumon

    return _gpumon
