// otlp.hpp — OpenTelemetry export over OTLP/gRPC and OTLP/HTTP.
//
// The reference exports spans + counters over OTLP/gRPC behind the `otel`
// cargo feature, configured purely by OTEL_* env vars (SURVEY.md §5.1, §5.5;
// reference main.rs:138-271, tonic transport main.rs:206-221). This build
// speaks all three OTLP transports:
//   * grpc          — unary Export to TraceService/MetricsService on :4317
//                     (hand-rolled h2c, native/common/grpc_client.cpp);
//   * http/protobuf — binary protobuf POST to /v1/traces + /v1/metrics;
//   * http/json     — OTLP/JSON POST (same paths).
// Selected by OTEL_EXPORTER_OTLP_PROTOCOL; when unset, grpc is used for
// endpoints on the conventional gRPC port 4317 and http/json otherwise.
//
// Spans NEST: each SpanGuard records its parent (the innermost live guard on
// the thread) so scale → scale_to_zero → Event trees export with
// parent_span_id linkage, mirroring the reference's #[tracing::instrument]
// hierarchy. A root guard starts a fresh trace. For fan-out work on pool
// threads, capture current_context() and install it with ContextGuard in the
// worker to keep the tree connected across threads.
//
// Enabled when OTEL_EXPORTER_OTLP_ENDPOINT is set (and OTEL_SDK_DISABLED is
// not "true").
#pragma once

#include <string>

namespace otlp {

// Start the exporter if configured; safe to call when not configured (no-op).
void init(const std::string& service_name);
// Flush pending spans/metrics and stop the export thread (OtelGuard::drop).
void shutdown();
bool enabled();

// Identity of the innermost live span on this thread (empty ids when none).
struct SpanContext {
  std::string trace_id;  // 32 hex chars
  std::string span_id;   // 16 hex chars
};
SpanContext current_context();

// RAII span: records wall-clock duration and ships the span on destruction,
// linked to its parent (innermost enclosing guard or installed context).
class SpanGuard {
public:
  explicit SpanGuard(const std::string& name);
  ~SpanGuard();
  SpanGuard(const SpanGuard&) = delete;
  SpanGuard& operator=(const SpanGuard&) = delete;

private:
  std::string name_;
  std::string trace_id_;
  std::string span_id_;
  std::string parent_id_;
  uint64_t start_ns_;
};

// Installs `parent` as this thread's span context for the guard's scope —
// spans created under it become its children (cross-thread propagation for
// the engine's pool fan-outs).
class ContextGuard {
public:
  explicit ContextGuard(const SpanContext& parent);
  ~ContextGuard();
  ContextGuard(const ContextGuard&) = delete;
  ContextGuard& operator=(const ContextGuard&) = delete;

private:
  std::string saved_trace_id_;
  std::string saved_parent_;
  bool installed_ = false;
};

// Test hook: number of successfully delivered export batches.
uint64_t delivered_batches();

}  // namespace otlp
