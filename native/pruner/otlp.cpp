#include "otlp.hpp"

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <mutex>
#include <memory>
#include <thread>
#include <vector>

#include "../common/http.hpp"
#include "../common/pb.hpp"
#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/strutil.hpp"

namespace otlp {

namespace {

struct FinishedSpan {
  std::string name;
  uint64_t start_ns;
  uint64_t end_ns;
};

constexpr size_t kMaxBufferedSpans = 50000;  // drop-oldest beyond this
constexpr size_t kMaxSpansPerPost = 5000;    // bound per-request payload

struct State {
  std::atomic<bool> enabled{false};
  std::atomic<uint64_t> dropped_spans{0};
  std::atomic<bool> running{false};
  std::atomic<uint64_t> delivered{0};
  std::string endpoint;  // base, no trailing slash
  std::string service_name;
  std::string trace_id;  // one id per process run; spans are flat (no parent tracking)
  std::mutex mu;
  std::vector<FinishedSpan> spans;
  std::thread exporter;
  std::condition_variable cv;
  bool stop = false;
  int interval_ms = 5000;
  bool use_protobuf = false;  // OTEL_EXPORTER_OTLP_PROTOCOL=http/protobuf
  std::unique_ptr<http::Client> client;  // persistent export connection pool
};

State& state() {
  static State s;
  return s;
}

uint64_t now_unix_ns() {
  return static_cast<uint64_t>(std::chrono::duration_cast<std::chrono::nanoseconds>(
                                   std::chrono::system_clock::now().time_since_epoch())
                                   .count());
}

std::string rand_hex(size_t bytes) {
  std::string id = strutil::uuid4_simple();  // 32 hex chars
  return id.substr(0, bytes * 2);
}

std::string hex_to_bytes(const std::string& hex) {
  std::string out;
  out.reserve(hex.size() / 2);
  for (size_t i = 0; i + 1 < hex.size(); i += 2) {
    auto nib = [](char c) -> int {
      if (c >= '0' && c <= '9') return c - '0';
      if (c >= 'a' && c <= 'f') return c - 'a' + 10;
      if (c >= 'A' && c <= 'F') return c - 'A' + 10;
      return 0;
    };
    out += static_cast<char>((nib(hex[i]) << 4) | nib(hex[i + 1]));
  }
  return out;
}

// ---- binary-protobuf encoders (opentelemetry-proto field numbers) ----------

std::string pb_resource(const std::string& service_name) {
  std::string any;  // AnyValue{string_value=1}
  pb::put_bytes(any, 1, service_name);
  std::string kv;  // KeyValue{key=1, value=2}
  pb::put_bytes(kv, 1, "service.name");
  pb::put_bytes(kv, 2, any);
  std::string res;  // Resource{attributes=1}
  pb::put_bytes(res, 1, kv);
  return res;
}

std::string pb_scope() {
  std::string scope;  // InstrumentationScope{name=1}
  pb::put_bytes(scope, 1, "gpu_pruner::main");
  return scope;
}

// ExportTraceServiceRequest{resource_spans=1{resource=1, scope_spans=2{scope=1, spans=2}}}
std::string encode_spans_pb(const State& s, const std::vector<FinishedSpan>& batch) {
  std::string spans;
  for (const auto& fs : batch) {
    std::string sp;  // Span
    pb::put_bytes(sp, 1, hex_to_bytes(s.trace_id));  // trace_id (16 bytes)
    pb::put_bytes(sp, 2, hex_to_bytes(rand_hex(8))); // span_id (8 bytes)
    pb::put_bytes(sp, 5, fs.name);                   // name
    pb::put_varint(sp, 6, 1);                        // kind = INTERNAL
    pb::put_fixed64(sp, 7, fs.start_ns);
    pb::put_fixed64(sp, 8, fs.end_ns);
    pb::put_bytes(spans, 2, sp);                     // ScopeSpans.spans
  }
  std::string scope_spans;
  pb::put_bytes(scope_spans, 1, pb_scope());
  scope_spans += spans;
  std::string rs;  // ResourceSpans
  pb::put_bytes(rs, 1, pb_resource(s.service_name));
  pb::put_bytes(rs, 2, scope_spans);
  std::string req;
  pb::put_bytes(req, 1, rs);
  return req;
}

// ExportMetricsServiceRequest{resource_metrics=1{resource=1, scope_metrics=2{scope=1, metrics=2}}}
std::string encode_metrics_pb(const State& s,
                              const std::map<std::string, int64_t>& counters,
                              uint64_t t_ns) {
  std::string metrics;
  for (const auto& [name, value] : counters) {
    bool monotonic = strutil::starts_with(name, "monotonic_counter.");
    std::string short_name = name.substr(name.find('.') + 1);
    std::string dp;  // NumberDataPoint{start=2, time=3, as_int=6 (sfixed64)}
    if (monotonic) pb::put_fixed64(dp, 2, t_ns);
    pb::put_fixed64(dp, 3, t_ns);
    pb::put_fixed64(dp, 6, static_cast<uint64_t>(value));
    std::string m;  // Metric{name=1, gauge=5 | sum=7}
    pb::put_bytes(m, 1, short_name);
    if (monotonic) {
      std::string sum;  // Sum{data_points=1, temporality=2, is_monotonic=3}
      pb::put_bytes(sum, 1, dp);
      pb::put_varint(sum, 2, 2);  // CUMULATIVE
      pb::put_varint(sum, 3, 1);
      pb::put_bytes(m, 7, sum);
    } else {
      std::string gauge;  // Gauge{data_points=1}
      pb::put_bytes(gauge, 1, dp);
      pb::put_bytes(m, 5, gauge);
    }
    pb::put_bytes(metrics, 2, m);  // ScopeMetrics.metrics
  }
  std::string scope_metrics;
  pb::put_bytes(scope_metrics, 1, pb_scope());
  scope_metrics += metrics;
  std::string rm;
  pb::put_bytes(rm, 1, pb_resource(s.service_name));
  pb::put_bytes(rm, 2, scope_metrics);
  std::string req;
  pb::put_bytes(req, 1, rm);
  return req;
}

jsn::Value resource_json(const std::string& service_name) {
  jsn::Value attr = jsn::Value::object();
  attr["key"] = "service.name";
  attr["value"] = jsn::Value::object();
  attr["value"]["stringValue"] = service_name;
  jsn::Value res = jsn::Value::object();
  res["attributes"] = jsn::Value(jsn::Array{attr});
  return res;
}

void post_payload(const std::string& url, std::string body, const char* content_type) {
  State& s = state();
  auto parsed = http::Url::parse(url);
  if (!parsed) return;
  if (!s.client) {
    http::ClientOptions opts;
    opts.connect_timeout_ms = 2000;
    opts.io_timeout_ms = 5000;
    s.client = std::make_unique<http::Client>(*parsed, opts);
  }
  http::Request req;
  req.method = "POST";
  req.path = parsed->path;
  req.body = std::move(body);
  req.headers.emplace_back("Content-Type", content_type);
  http::Response resp = s.client->request(req);
  if (resp.status >= 200 && resp.status < 300)
    s.delivered.fetch_add(1, std::memory_order_relaxed);
}

void post_json(const std::string& url, const jsn::Value& body) {
  post_payload(url, body.dump(), "application/json");
}

void export_once() {
  State& s = state();
  // ---- spans (chunked: a slow collector must not grow our heap) ----
  std::vector<FinishedSpan> all;
  {
    std::lock_guard<std::mutex> lock(s.mu);
    all.swap(s.spans);
  }
  for (size_t base = 0; base < all.size(); base += kMaxSpansPerPost) {
    size_t n = std::min(kMaxSpansPerPost, all.size() - base);
    std::vector<FinishedSpan> batch(all.begin() + static_cast<long>(base),
                                    all.begin() + static_cast<long>(base + n));
    if (s.use_protobuf) {
      try {
        post_payload(s.endpoint + "/v1/traces", encode_spans_pb(s, batch),
                     "application/x-protobuf");
      } catch (const std::exception&) { /* collector away; drop batch */ }
      continue;
    }
    jsn::Value spans = jsn::Value::array();
    for (const auto& fs : batch) {
      jsn::Value sp = jsn::Value::object();
      sp["traceId"] = s.trace_id;
      sp["spanId"] = rand_hex(8);
      sp["name"] = fs.name;
      sp["kind"] = 1;  // SPAN_KIND_INTERNAL
      sp["startTimeUnixNano"] = std::to_string(fs.start_ns);
      sp["endTimeUnixNano"] = std::to_string(fs.end_ns);
      spans.push_back(sp);
    }
    jsn::Value scope_spans = jsn::Value::object();
    scope_spans["scope"] = jsn::Value::object();
    scope_spans["scope"]["name"] = "gpu_pruner::main";
    scope_spans["spans"] = spans;
    jsn::Value rs = jsn::Value::object();
    rs["resource"] = resource_json(s.service_name);
    rs["scopeSpans"] = jsn::Value(jsn::Array{scope_spans});
    jsn::Value payload = jsn::Value::object();
    payload["resourceSpans"] = jsn::Value(jsn::Array{rs});
    try {
      post_json(s.endpoint + "/v1/traces", payload);
    } catch (const std::exception&) { /* collector away; drop batch */ }
  }

  // ---- metrics: the counter registry (monotonic counters + gauges) ----
  auto counters = logx::counters_snapshot();
  if (!counters.empty() && s.use_protobuf) {
    try {
      post_payload(s.endpoint + "/v1/metrics",
                   encode_metrics_pb(s, counters, now_unix_ns()),
                   "application/x-protobuf");
    } catch (const std::exception&) { /* collector away */ }
  } else if (!counters.empty()) {
    uint64_t t = now_unix_ns();
    jsn::Value metrics = jsn::Value::array();
    for (const auto& [name, value] : counters) {
      jsn::Value dp = jsn::Value::object();
      dp["asInt"] = std::to_string(value);
      dp["timeUnixNano"] = std::to_string(t);
      jsn::Value m = jsn::Value::object();
      bool monotonic = strutil::starts_with(name, "monotonic_counter.");
      std::string short_name = name.substr(name.find('.') + 1);
      m["name"] = short_name;
      if (monotonic) {
        dp["startTimeUnixNano"] = std::to_string(t);
        m["sum"] = jsn::Value::object();
        m["sum"]["dataPoints"] = jsn::Value(jsn::Array{dp});
        m["sum"]["aggregationTemporality"] = 2;  // CUMULATIVE
        m["sum"]["isMonotonic"] = true;
      } else {
        m["gauge"] = jsn::Value::object();
        m["gauge"]["dataPoints"] = jsn::Value(jsn::Array{dp});
      }
      metrics.push_back(m);
    }
    jsn::Value scope_metrics = jsn::Value::object();
    scope_metrics["scope"] = jsn::Value::object();
    scope_metrics["scope"]["name"] = "gpu_pruner::main";
    scope_metrics["metrics"] = metrics;
    jsn::Value rm = jsn::Value::object();
    rm["resource"] = resource_json(s.service_name);
    rm["scopeMetrics"] = jsn::Value(jsn::Array{scope_metrics});
    jsn::Value payload = jsn::Value::object();
    payload["resourceMetrics"] = jsn::Value(jsn::Array{rm});
    try {
      post_json(s.endpoint + "/v1/metrics", payload);
    } catch (const std::exception&) { /* collector away */ }
  }
}

}  // namespace

void init(const std::string& service_name) {
  State& s = state();
  const char* disabled = std::getenv("OTEL_SDK_DISABLED");
  if (disabled && strutil::lower(disabled) == "true") return;
  const char* ep = std::getenv("OTEL_EXPORTER_OTLP_ENDPOINT");
  if (!ep || !*ep) return;
  s.endpoint = ep;
  while (!s.endpoint.empty() && s.endpoint.back() == '/') s.endpoint.pop_back();
  s.service_name = service_name;
  if (const char* sn = std::getenv("OTEL_SERVICE_NAME"); sn && *sn) s.service_name = sn;
  s.trace_id = rand_hex(16);
  if (const char* iv = std::getenv("OTEL_METRIC_EXPORT_INTERVAL"); iv && *iv)
    s.interval_ms = std::atoi(iv);
  if (const char* proto = std::getenv("OTEL_EXPORTER_OTLP_PROTOCOL"); proto && *proto)
    s.use_protobuf = std::string(proto) == "http/protobuf";
  s.enabled.store(true);
  s.running.store(true);
  s.exporter = std::thread([&s] {
    std::unique_lock<std::mutex> lock(s.mu);
    while (!s.stop) {
      s.cv.wait_for(lock, std::chrono::milliseconds(s.interval_ms));
      if (s.stop) break;
      lock.unlock();
      export_once();
      lock.lock();
    }
  });
}

void shutdown() {
  State& s = state();
  if (!s.running.load()) return;
  {
    std::lock_guard<std::mutex> lock(s.mu);
    s.stop = true;
  }
  s.cv.notify_all();
  s.exporter.join();
  export_once();  // final flush
  s.running.store(false);
  s.enabled.store(false);
}

bool enabled() { return state().enabled.load(std::memory_order_relaxed); }

SpanGuard::SpanGuard(const std::string& name) : name_(name), start_ns_(now_unix_ns()) {}

SpanGuard::~SpanGuard() {
  if (!enabled()) return;
  State& s = state();
  std::lock_guard<std::mutex> lock(s.mu);
  if (s.spans.size() >= kMaxBufferedSpans) {
    // bounded buffer: a stalled exporter drops oldest spans instead of
    // growing the daemon heap without limit
    s.spans.erase(s.spans.begin(),
                  s.spans.begin() + static_cast<long>(kMaxBufferedSpans / 10));
    s.dropped_spans.fetch_add(kMaxBufferedSpans / 10, std::memory_order_relaxed);
  }
  s.spans.push_back({name_, start_ns_, now_unix_ns()});
}

uint64_t delivered_batches() { return state().delivered.load(std::memory_order_relaxed); }

}  // namespace otlp
