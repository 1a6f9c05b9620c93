// otlp.hpp — OpenTelemetry export over OTLP/HTTP+JSON.
//
// The reference exports spans + counters over OTLP/gRPC behind the `otel`
// cargo feature, configured purely by OTEL_* env vars (SURVEY.md §5.1, §5.5;
// reference main.rs:138-271). This build speaks the OTLP **HTTP/JSON**
// protocol instead — same wire semantics, no gRPC/protobuf dependency —
// shipping:
//   * spans for the instrumented operations (run_query_and_scale, scale,
//     find_root_object, …) via SpanGuard,
//   * the six counters of SURVEY.md §5.5 from the logx counter registry,
//     exported periodically.
//
// Enabled when OTEL_EXPORTER_OTLP_ENDPOINT is set (and OTEL_SDK_DISABLED is
// not "true"). Endpoint forms: http(s)://host:4318 — /v1/traces and
// /v1/metrics are appended per the OTLP spec.
#pragma once

#include <string>

namespace otlp {

// Start the exporter if configured; safe to call when not configured (no-op).
void init(const std::string& service_name);
// Flush pending spans/metrics and stop the export thread (OtelGuard::drop).
void shutdown();
bool enabled();

// RAII span: records wall-clock duration and ships the span on destruction.
class SpanGuard {
public:
  explicit SpanGuard(const std::string& name);
  ~SpanGuard();
  SpanGuard(const SpanGuard&) = delete;
  SpanGuard& operator=(const SpanGuard&) = delete;

private:
  std::string name_;
  uint64_t start_ns_;
};

// Test hook: number of successfully delivered export batches.
uint64_t delivered_batches();

}  // namespace otlp
