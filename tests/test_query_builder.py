"""Query-builder contract tests.

Ports the reference's 11 template unit tests (reference
gpu-pruner/src/main.rs:572-740) against the native QueryBuilder
(native/pruner/promql.cpp): the rendered idle-GPU query must use
max_over_time over both DCGM-shaped metrics with /100 normalization, carry
the filters into every compute selector, and switch label conventions with
honor_labels.
"""

import json

import pytest


def render(core, args: dict) -> str:
    return core.render_query(json.dumps(args))


def test_query_uses_max_over_time(core):
    q = render(core, {"duration": 30})
    assert "max_over_time(" in q, "should use max_over_time, not avg_over_time"
    assert "avg_over_time(" not in q


def test_query_includes_gpu_util_fallback(core):
    q = render(core, {"duration": 30})
    assert "DCGM_FI_PROF_GR_ENGINE_ACTIVE" in q, "primary metric missing"
    assert "DCGM_FI_DEV_GPU_UTIL" in q, "fallback metric missing"
    assert "/ 100" in q, "fallback should normalize 0-100 to 0-1"


def test_query_without_power_threshold_has_no_unless(core):
    q = render(core, {"duration": 30})
    assert "unless" not in q
    assert "DCGM_FI_DEV_POWER_USAGE" not in q


def test_query_with_power_threshold_adds_unless(core):
    q = render(core, {"duration": 30, "power_threshold": 150.0})
    assert "unless on (exported_pod, exported_namespace)" in q
    assert "DCGM_FI_DEV_POWER_USAGE" in q
    assert ">= 150" in q, "should use the configured threshold"


def test_query_with_namespace_filter(core):
    q = render(core, {"duration": 15, "namespace": "ml-team"})
    # idle block appears twice (enriched + bare fallback), 2 metrics each = 4
    assert q.count('exported_namespace =~ "ml-team"') == 4


def test_query_with_namespace_and_power_threshold(core):
    q = render(core, {"duration": 15, "namespace": "ml-team", "power_threshold": 100.0})
    # 4 from compute (2 paths x 2 metrics) + 1 from power = 5
    assert q.count('exported_namespace =~ "ml-team"') == 5


def test_query_with_model_name_filter(core):
    q = render(core, {"duration": 30, "model_name": "AMD Instinct MI355X"})
    assert q.count('modelName =~ "AMD Instinct MI355X"') == 4


def test_query_duration_is_interpolated(core):
    q = render(core, {"duration": 45})
    assert "[45m]" in q


def test_query_default_uses_exported_labels(core):
    q = render(core, {"duration": 30})
    assert "exported_pod" in q
    assert "exported_namespace" in q
    assert "exported_container" in q


def test_query_honor_labels_uses_native_labels(core):
    q = render(core, {"duration": 30, "honor_labels": True})
    assert "exported_pod" not in q
    assert "exported_namespace" not in q
    assert "pod !=" in q
    assert "sum by (Hostname, container, pod, namespace" in q


def test_query_honor_labels_with_power_threshold(core):
    q = render(core, {"duration": 30, "honor_labels": True, "power_threshold": 120.0})
    assert "unless on (pod, namespace)" in q


# ---- gap-closure additions beyond the ported 11 -----------------------------


def test_query_idle_predicate_and_enrichment(core):
    q = render(core, {"duration": 30})
    assert q.rstrip().endswith("== 0"), "idle predicate must terminate the query"
    assert "node_dmi_info" in q and "group_left(node_type)" in q
    assert q.count("label_replace(") == 2


def test_query_power_threshold_fractional(core):
    q = render(core, {"duration": 30, "power_threshold": 72.5})
    assert ">= 72.5" in q


@pytest.mark.parametrize("duration", [1, 15, 2880])
def test_query_duration_range(core, duration):
    q = render(core, {"duration": duration})
    assert f"[{duration}m]" in q


def test_query_escapes_quotes_in_filters(core):
    """A quote in --namespace/--model-name must not break out of the PromQL
    string matcher (the reference interpolates the raw value)."""
    q = render(core, {"duration": 30, "namespace": 'evil"} or up{x="',
                      "model_name": 'A"B\\C'})
    assert 'evil\\"} or up{x=\\"' in q
    assert 'A\\"B\\\\C' in q
    # the query still terminates with the idle predicate
    assert q.rstrip().endswith("== 0")
