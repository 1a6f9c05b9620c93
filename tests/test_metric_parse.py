"""PodMetricData parsing tests.

The reference has NO coverage of its series parser (SURVEY.md §4 "Gaps");
these pin the exported_* → native label fallback, the node_type default,
required-label errors, and sample-value decoding
(native/pruner/prom.cpp::parse_pod_metric; reference lib.rs:136-187).
"""

import json

import pytest


def parse(core, metric, value=(1700000000.0, "0")):
    return core.parse_pod_metric(json.dumps({"metric": metric, "value": list(value)}))


BASE = {
    "exported_pod": "p1",
    "exported_namespace": "ns1",
    "exported_container": "c1",
    "modelName": "AMD Instinct MI355X",
}


def test_parse_exported_labels(core):
    pmd = parse(core, dict(BASE, node_type="amd-mi355x"))
    assert pmd["name"] == "p1"
    assert pmd["namespace"] == "ns1"
    assert pmd["container"] == "c1"
    assert pmd["node_type"] == "amd-mi355x"
    assert pmd["gpu_model"] == "AMD Instinct MI355X"
    assert pmd["value"] == 0.0


def test_parse_native_labels(core):
    pmd = parse(core, {
        "pod": "p2", "namespace": "ns2", "container": "c2",
        "modelName": "AMD Instinct MI355X",
    })
    assert pmd["name"] == "p2"
    assert pmd["namespace"] == "ns2"
    assert pmd["container"] == "c2"


def test_parse_exported_takes_precedence(core):
    pmd = parse(core, dict(BASE, pod="native-name"))
    assert pmd["name"] == "p1"


def test_parse_node_type_defaults_to_unknown(core):
    pmd = parse(core, BASE)
    assert pmd["node_type"] == "unknown"


@pytest.mark.parametrize("missing", ["exported_pod", "exported_namespace",
                                     "exported_container", "modelName"])
def test_parse_missing_required_label_raises(core, missing):
    metric = dict(BASE)
    del metric[missing]
    with pytest.raises(core.PodConvertError):
        parse(core, metric)


def test_parse_value_decoding(core):
    pmd = parse(core, BASE, value=(1700000000.0, "0.25"))
    assert pmd["value"] == 0.25
