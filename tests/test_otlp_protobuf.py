"""Binary-protobuf OTLP export (OTEL_EXPORTER_OTLP_PROTOCOL=http/protobuf).

The hand-rolled encoder (native/common/pb.hpp + otlp.cpp) is validated by
decoding its payloads with the real protobuf runtime against
opentelemetry-proto field numbers — an independent implementation of the
wire format, mirroring the PodResources decoder test in reverse.
"""

import json
import os
import subprocess

import pytest

pytest.importorskip("google.protobuf")

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory  # noqa: E402

from gpu_pruner_amd.fixtures import FakeOtlpCollector, FakeOtlpGrpcCollector  # noqa: E402


def _build_otlp_messages():
    pool = descriptor_pool.DescriptorPool()
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "otlp_fixture.proto"
    f.package = "o"
    f.syntax = "proto3"
    T = descriptor_pb2.FieldDescriptorProto

    def msg(name):
        m = f.message_type.add()
        m.name = name
        return m

    def field(m, name, num, ftype, label=T.LABEL_OPTIONAL, type_name=None):
        fl = m.field.add()
        fl.name, fl.number, fl.type, fl.label = name, num, ftype, label
        if type_name:
            fl.type_name = type_name

    sp = msg("Span")
    field(sp, "trace_id", 1, T.TYPE_BYTES)
    field(sp, "span_id", 2, T.TYPE_BYTES)
    field(sp, "parent_span_id", 4, T.TYPE_BYTES)
    field(sp, "name", 5, T.TYPE_STRING)
    field(sp, "kind", 6, T.TYPE_INT32)
    field(sp, "start_time_unix_nano", 7, T.TYPE_FIXED64)
    field(sp, "end_time_unix_nano", 8, T.TYPE_FIXED64)

    ss = msg("ScopeSpans")
    field(ss, "spans", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.Span")
    rs = msg("ResourceSpans")
    field(rs, "scope_spans", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.ScopeSpans")
    td = msg("TracesData")
    field(td, "resource_spans", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.ResourceSpans")

    dp = msg("NumberDataPoint")
    field(dp, "start_time_unix_nano", 2, T.TYPE_FIXED64)
    field(dp, "time_unix_nano", 3, T.TYPE_FIXED64)
    field(dp, "as_int", 6, T.TYPE_SFIXED64)
    sm = msg("Sum")
    field(sm, "data_points", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.NumberDataPoint")
    field(sm, "aggregation_temporality", 2, T.TYPE_INT32)
    field(sm, "is_monotonic", 3, T.TYPE_BOOL)
    ga = msg("Gauge")
    field(ga, "data_points", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.NumberDataPoint")
    me = msg("Metric")
    field(me, "name", 1, T.TYPE_STRING)
    field(me, "gauge", 5, T.TYPE_MESSAGE, type_name=".o.Gauge")
    field(me, "sum", 7, T.TYPE_MESSAGE, type_name=".o.Sum")
    scm = msg("ScopeMetrics")
    field(scm, "metrics", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.Metric")
    rm = msg("ResourceMetrics")
    field(rm, "scope_metrics", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.ScopeMetrics")
    md = msg("MetricsData")
    field(md, "resource_metrics", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".o.ResourceMetrics")

    pool.Add(f)
    get = lambda n: message_factory.GetMessageClass(pool.FindMessageTypeByName(n))
    return get("o.TracesData"), get("o.MetricsData")


def test_protobuf_otlp_spans_and_counters(pruner_bin, fake_api, fake_prom):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml")
    with FakeOtlpCollector() as collector:
        env = dict(os.environ)
        env["GPU_PRUNER_K8S_URL"] = fake_api.url
        env["PROMETHEUS_TOKEN"] = "t"
        env["OTEL_EXPORTER_OTLP_ENDPOINT"] = collector.url
        env["OTEL_EXPORTER_OTLP_PROTOCOL"] = "http/protobuf"
        env["OTEL_METRIC_EXPORT_INTERVAL"] = "60000"
        r = subprocess.run(
            [pruner_bin, "--prometheus-url", fake_prom.url, "--run-mode", "scale-down"],
            capture_output=True, text=True, timeout=60, env=env)
        assert r.returncode == 0, r.stderr
        assert collector.traces_pb, "no protobuf trace payloads"
        assert collector.metrics_pb, "no protobuf metric payloads"
        assert collector.traces == [] and collector.metrics == []

        TracesData, MetricsData = _build_otlp_messages()
        spans = []
        for raw in collector.traces_pb:
            td = TracesData()
            td.ParseFromString(raw)
            for rsp in td.resource_spans:
                for ss in rsp.scope_spans:
                    spans.extend(ss.spans)
        names = {s.name for s in spans}
        assert "run_query_and_scale" in names
        assert "find_root_object" in names
        for s in spans:
            assert len(s.trace_id) == 16 and len(s.span_id) == 8
            assert s.end_time_unix_nano >= s.start_time_unix_nano
            assert s.kind == 1

        md = MetricsData()
        md.ParseFromString(collector.metrics_pb[-1])
        by_name = {}
        for rmx in md.resource_metrics:
            for sm in rmx.scope_metrics:
                for m in sm.metrics:
                    if m.HasField("sum"):
                        by_name[m.name] = m.sum.data_points[0].as_int
                        assert m.sum.is_monotonic
                        assert m.sum.aggregation_temporality == 2
                    elif m.HasField("gauge"):
                        by_name[m.name] = m.gauge.data_points[0].as_int
        assert by_name.get("query_successes") == 1
        assert by_name.get("scale_successes") == 1
        assert by_name.get("query_returned_candidates") == 1


def _cull_cluster(fake_api, fake_prom):
    dep = fake_api.add_deployment("d", "ml")
    rs = fake_api.add_replicaset("d-rs", "ml", owner=dep)
    fake_api.add_pod("p0", "ml", owner_kind="ReplicaSet", owner_name="d-rs",
                     owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)
    fake_prom.add_idle_series("p0", "ml")


def _run_scaledown(pruner_bin, fake_api, fake_prom, env_extra):
    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = fake_api.url
    env["PROMETHEUS_TOKEN"] = "t"
    env["OTEL_METRIC_EXPORT_INTERVAL"] = "60000"
    env.update(env_extra)
    return subprocess.run(
        [pruner_bin, "--prometheus-url", fake_prom.url, "--run-mode", "scale-down"],
        capture_output=True, text=True, timeout=60, env=env)


def test_grpc_otlp_export(pruner_bin, fake_api, fake_prom):
    """OTLP/gRPC (the reference's tonic transport, main.rs:206-221): the
    hand-rolled h2c client Exports to a real grpcio server; payloads decode
    with the real protobuf runtime."""
    _cull_cluster(fake_api, fake_prom)
    with FakeOtlpGrpcCollector() as collector:
        r = _run_scaledown(pruner_bin, fake_api, fake_prom, {
            "OTEL_EXPORTER_OTLP_ENDPOINT": collector.url,
            "OTEL_EXPORTER_OTLP_PROTOCOL": "grpc",
        })
        assert r.returncode == 0, r.stderr
        assert collector.traces_pb, "no gRPC trace exports"
        assert collector.metrics_pb, "no gRPC metric exports"

        TracesData, MetricsData = _build_otlp_messages()
        spans = []
        for raw in collector.traces_pb:
            td = TracesData()
            td.ParseFromString(raw)
            for rsp in td.resource_spans:
                for ss in rsp.scope_spans:
                    spans.extend(ss.spans)
        names = {s.name for s in spans}
        assert "run_query_and_scale" in names
        assert "scale" in names

        md = MetricsData()
        md.ParseFromString(collector.metrics_pb[-1])
        by_name = {}
        for rmx in md.resource_metrics:
            for sm in rmx.scope_metrics:
                for m in sm.metrics:
                    if m.HasField("sum"):
                        by_name[m.name] = m.sum.data_points[0].as_int
        assert by_name.get("query_successes") == 1
        assert by_name.get("scale_successes") == 1


def test_grpc_span_nesting(pruner_bin, fake_api, fake_prom):
    """parent_span_id linkage over gRPC: scale_to_zero nests under scale,
    find_root_object under run_query_and_scale (the reference's
    #[tracing::instrument] hierarchy, lib.rs 7 sites)."""
    _cull_cluster(fake_api, fake_prom)
    with FakeOtlpGrpcCollector() as collector:
        r = _run_scaledown(pruner_bin, fake_api, fake_prom, {
            "OTEL_EXPORTER_OTLP_ENDPOINT": collector.url,
            "OTEL_EXPORTER_OTLP_PROTOCOL": "grpc",
        })
        assert r.returncode == 0, r.stderr
        TracesData, _ = _build_otlp_messages()
        spans = []
        for raw in collector.traces_pb:
            td = TracesData()
            td.ParseFromString(raw)
            for rsp in td.resource_spans:
                for ss in rsp.scope_spans:
                    spans.extend(ss.spans)
        by_id = {s.span_id: s for s in spans}
        by_name = {}
        for s in spans:
            by_name.setdefault(s.name, []).append(s)

        # generate_scale_event + scale_to_zero are children of scale
        for child_name, parent_name in (("scale_to_zero", "scale"),
                                        ("generate_scale_event", "scale"),
                                        ("find_root_object", "run_query_and_scale")):
            assert child_name in by_name, f"missing span {child_name}"
            for child in by_name[child_name]:
                assert child.parent_span_id, f"{child_name} has no parent"
                parent = by_id[child.parent_span_id]
                assert parent.name == parent_name, (child_name, parent.name)
                assert parent.trace_id == child.trace_id
                # child runs inside the parent's wall-clock window
                assert parent.start_time_unix_nano <= child.start_time_unix_nano
                assert child.end_time_unix_nano <= parent.end_time_unix_nano

        # roots exist and have no parent
        roots = [s for s in spans if not s.parent_span_id]
        assert any(s.name == "run_query_and_scale" for s in roots)


def test_json_transport_also_carries_parent_ids(pruner_bin, fake_api, fake_prom):
    _cull_cluster(fake_api, fake_prom)
    with FakeOtlpCollector() as collector:
        r = _run_scaledown(pruner_bin, fake_api, fake_prom, {
            "OTEL_EXPORTER_OTLP_ENDPOINT": collector.url,
        })
        assert r.returncode == 0, r.stderr
        spans = collector.spans()
        with_parent = [s for s in spans if s.get("parentSpanId")]
        assert with_parent, "no nested spans on the JSON transport"
        by_id = {s["spanId"]: s for s in spans}
        for s in spans:
            if s["name"] == "scale_to_zero":
                assert by_id[s["parentSpanId"]]["name"] == "scale"


def test_grpc_large_batch_respects_flow_control(core, fake_api, monkeypatch):
    """A 1000-pod tick mints thousands of spans; the gRPC exporter must split
    Export requests to fit the peer's default HTTP/2 flow-control window
    (65535 B) — a single oversized write would be a protocol violation the
    collector tears down."""
    from gpu_pruner_amd.fixtures import FakeOtlpGrpcCollector

    with FakeOtlpGrpcCollector() as collector:
        monkeypatch.setenv("OTEL_EXPORTER_OTLP_ENDPOINT", collector.url)
        monkeypatch.setenv("OTEL_EXPORTER_OTLP_PROTOCOL", "grpc")
        monkeypatch.setenv("OTEL_METRIC_EXPORT_INTERVAL", "60000")
        monkeypatch.setenv("PROMETHEUS_TOKEN", "t")
        b = core.SyntheticBackend(n_pods=1000, pods_per_parent=2)
        b.start()
        try:
            monkeypatch.setenv("GPU_PRUNER_K8S_URL", b.k8s_url)
            core.otlp_init("flowcontrol-test")
            cfg = json.dumps({"duration": 30, "grace_period": 300,
                              "run_mode": "scale-down",
                              "prometheus_url": b.prom_url})
            out = core.run_tick(cfg)
            assert out["num_unique_pods"] == 1000
            core.otlp_shutdown()
        finally:
            b.stop()
        assert len(collector.traces_pb) >= 2, "large batch was not split"
        for raw in collector.traces_pb:
            assert len(raw) <= 60000, f"oversized Export request: {len(raw)} B"
        TracesData, _ = _build_otlp_messages()
        total = 0
        for raw in collector.traces_pb:
            td = TracesData()
            td.ParseFromString(raw)
            for rsp in td.resource_spans:
                for ss in rsp.scope_spans:
                    total += len(ss.spans)
        # 1000 find_root_object + 500 x (scale + helper + event) + roots
        assert total >= 2500, total
