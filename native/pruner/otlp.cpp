#include "otlp.hpp"

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <mutex>
#include <memory>
#include <random>
#include <thread>
#include <vector>

#include "../common/grpc_client.hpp"
#include "../common/http.hpp"
#include "../common/pb.hpp"
#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../common/tsan_compat.hpp"

namespace otlp {

namespace {

struct FinishedSpan {
  std::string name;
  std::string trace_id;   // 32 hex
  std::string span_id;    // 16 hex
  std::string parent_id;  // 16 hex or empty (root)
  uint64_t start_ns;
  uint64_t end_ns;
};

constexpr size_t kMaxBufferedSpans = 50000;  // drop-oldest beyond this
constexpr size_t kMaxSpansPerPost = 5000;    // bound per-request payload

enum class Transport { Grpc, HttpProtobuf, HttpJson };

struct State {
  std::atomic<bool> enabled{false};
  std::atomic<uint64_t> dropped_spans{0};
  std::atomic<bool> running{false};
  std::atomic<uint64_t> delivered{0};
  std::string endpoint;  // base, no trailing slash
  std::string grpc_host;
  uint16_t grpc_port = 4317;
  std::string service_name;
  std::mutex mu;
  std::vector<FinishedSpan> spans;
  std::thread exporter;
  std::condition_variable cv;
  bool stop = false;
  int interval_ms = 5000;
  Transport transport = Transport::HttpJson;
  std::unique_ptr<http::Client> client;  // persistent export connection pool
};

// Per-thread span lineage: innermost live SpanGuard = back of the stack.
// ContextGuard seeds `installed_parent` so pool workers attach to the
// caller's span instead of starting disconnected traces.
struct ThreadCtx {
  std::string trace_id;
  std::vector<std::string> stack;
  std::string installed_parent;
};
thread_local ThreadCtx t_ctx;

// Finished spans land in PER-THREAD buffers (a single global mutex was the
// hot path: ~3k spans per 1000-pod tick across the worker pool). Buffers
// register once in a global list the exporter drains; shared_ptr keeps a
// buffer alive past thread exit so its tail still exports.
struct ThreadSpanBuf {
  std::mutex mu;  // uncontended except while the exporter drains
  std::vector<FinishedSpan> spans;
};

struct BufRegistry {
  std::mutex mu;
  std::vector<std::shared_ptr<ThreadSpanBuf>> bufs;
};

BufRegistry& buf_registry() {
  static BufRegistry r;
  return r;
}

ThreadSpanBuf& thread_buf() {
  thread_local std::shared_ptr<ThreadSpanBuf> buf = [] {
    auto b = std::make_shared<ThreadSpanBuf>();
    auto& reg = buf_registry();
    std::lock_guard<std::mutex> lock(reg.mu);
    reg.bufs.push_back(b);
    return b;
  }();
  return *buf;
}

constexpr size_t kMaxPerThreadSpans = 16384;  // drop-oldest beyond this

State& state() {
  static State s;
  return s;
}

uint64_t now_unix_ns() {
  return static_cast<uint64_t>(std::chrono::duration_cast<std::chrono::nanoseconds>(
                                   std::chrono::system_clock::now().time_since_epoch())
                                   .count());
}

// Span/trace ids need uniqueness, not cryptographic strength; the hot path
// mints ~3k ids per 1000-pod tick, so avoid uuid4_simple's per-call
// /dev/urandom open/read/close — a thread_local PRNG seeded once from the
// OS is plenty.
std::string rand_hex(size_t bytes) {
  static thread_local std::mt19937_64 rng{[] {
    std::random_device rd;
    return (static_cast<uint64_t>(rd()) << 32) ^ rd() ^
           std::hash<std::thread::id>{}(std::this_thread::get_id());
  }()};
  static const char* hex = "0123456789abcdef";
  std::string out(bytes * 2, '0');
  for (size_t i = 0; i < bytes; i += 8) {
    uint64_t v = rng();
    for (size_t b = 0; b < 8 && i + b < bytes; b++) {
      out[(i + b) * 2] = hex[(v >> (b * 8 + 4)) & 0xF];
      out[(i + b) * 2 + 1] = hex[(v >> (b * 8)) & 0xF];
    }
  }
  return out;
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
    pb::put_bytes(sp, 1, hex_to_bytes(fs.trace_id));  // trace_id (16 bytes)
    pb::put_bytes(sp, 2, hex_to_bytes(fs.span_id));   // span_id (8 bytes)
    if (!fs.parent_id.empty())
      pb::put_bytes(sp, 4, hex_to_bytes(fs.parent_id));  // parent_span_id
    pb::put_bytes(sp, 5, fs.name);                    // name
    pb::put_varint(sp, 6, 1);                         // kind = INTERNAL
    pb::put_fixed64(sp, 7, fs.start_ns);
    pb::put_fixed64(sp, 8, fs.end_ns);
    pb::put_bytes(spans, 2, sp);                      // ScopeSpans.spans
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

// Unary OTLP/gRPC Export (the reference's tonic transport, main.rs:206-221):
// ExportTrace/MetricsServiceRequest protobuf over h2c to the collector.
void grpc_export(const State& s, const char* service_method, std::string msg) {
  grpcx::Target t;
  t.host = s.grpc_host;
  t.port = s.grpc_port;
  t.authority = s.grpc_host + ":" + std::to_string(s.grpc_port);
  grpcx::unary_call(t, service_method, msg, /*timeout_ms=*/5000);
  state().delivered.fetch_add(1, std::memory_order_relaxed);
}

void export_once() {
  State& s = state();
  // ---- spans (chunked: a slow collector must not grow our heap) ----
  std::vector<FinishedSpan> all;
  {
    std::lock_guard<std::mutex> lock(s.mu);
    all.swap(s.spans);
  }
  {
    // drain every thread's buffer
    std::vector<std::shared_ptr<ThreadSpanBuf>> bufs;
    {
      auto& reg = buf_registry();
      std::lock_guard<std::mutex> lock(reg.mu);
      bufs = reg.bufs;
    }
    for (auto& b : bufs) {
      std::lock_guard<std::mutex> lock(b->mu);
      if (b->spans.empty()) continue;
      all.insert(all.end(), std::make_move_iterator(b->spans.begin()),
                 std::make_move_iterator(b->spans.end()));
      b->spans.clear();
    }
  }
  for (size_t base = 0; base < all.size(); base += kMaxSpansPerPost) {
    size_t n = std::min(kMaxSpansPerPost, all.size() - base);
    std::vector<FinishedSpan> batch(all.begin() + static_cast<long>(base),
                                    all.begin() + static_cast<long>(base + n));
    if (s.transport == Transport::Grpc) {
      // size-bounded sub-batches: each Export request must fit the peer's
      // default HTTP/2 flow-control window (grpcx::kMaxRequestBytes)
      size_t start = 0, take = batch.size();
      while (start < batch.size()) {
        take = std::min(take, batch.size() - start);
        std::vector<FinishedSpan> sub(batch.begin() + static_cast<long>(start),
                                      batch.begin() + static_cast<long>(start + take));
        std::string msg = encode_spans_pb(s, sub);
        if (msg.size() > grpcx::kMaxRequestBytes && take > 1) {
          take /= 2;
          continue;
        }
        try {
          grpc_export(s, "/opentelemetry.proto.collector.trace.v1.TraceService/Export",
                      std::move(msg));
        } catch (const std::exception&) {
          logx::counter_add("monotonic_counter.otlp_export_failures", 1);
        }
        start += take;
      }
      continue;
    }
    if (s.transport == Transport::HttpProtobuf) {
      try {
        post_payload(s.endpoint + "/v1/traces", encode_spans_pb(s, batch),
                     "application/x-protobuf");
      } catch (const std::exception&) {
          logx::counter_add("monotonic_counter.otlp_export_failures", 1);
        }
      continue;
    }
    jsn::Value spans = jsn::Value::array();
    for (const auto& fs : batch) {
      jsn::Value sp = jsn::Value::object();
      sp["traceId"] = fs.trace_id;
      sp["spanId"] = fs.span_id;
      if (!fs.parent_id.empty()) sp["parentSpanId"] = fs.parent_id;
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
    } catch (const std::exception&) {
          logx::counter_add("monotonic_counter.otlp_export_failures", 1);
        }
  }

  // ---- metrics: the counter registry (monotonic counters + gauges) ----
  auto counters = logx::counters_snapshot();
  if (!counters.empty() && s.transport == Transport::Grpc) {
    try {
      grpc_export(s, "/opentelemetry.proto.collector.metrics.v1.MetricsService/Export",
                  encode_metrics_pb(s, counters, now_unix_ns()));
    } catch (const std::exception&) {
      logx::counter_add("monotonic_counter.otlp_export_failures", 1);
    }
  } else if (!counters.empty() && s.transport == Transport::HttpProtobuf) {
    try {
      post_payload(s.endpoint + "/v1/metrics",
                   encode_metrics_pb(s, counters, now_unix_ns()),
                   "application/x-protobuf");
    } catch (const std::exception&) {
      logx::counter_add("monotonic_counter.otlp_export_failures", 1);
    }
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
    } catch (const std::exception&) {
      logx::counter_add("monotonic_counter.otlp_export_failures", 1);
    }
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
  if (const char* iv = std::getenv("OTEL_METRIC_EXPORT_INTERVAL"); iv && *iv)
    s.interval_ms = std::atoi(iv);
  // Transport: explicit OTEL_EXPORTER_OTLP_PROTOCOL wins; otherwise a :4317
  // endpoint gets gRPC (the OTLP default port tonic/the reference target)
  // and anything else OTLP/HTTP+JSON.
  auto parsed = http::Url::parse(s.endpoint);
  if (parsed) {
    s.grpc_host = parsed->host;
    s.grpc_port = parsed->port;
  }
  const char* proto = std::getenv("OTEL_EXPORTER_OTLP_PROTOCOL");
  std::string p = proto ? proto : "";
  if (p == "grpc") s.transport = Transport::Grpc;
  else if (p == "http/protobuf") s.transport = Transport::HttpProtobuf;
  else if (p == "http/json") s.transport = Transport::HttpJson;
  else s.transport = (parsed && parsed->port == 4317) ? Transport::Grpc
                                                      : Transport::HttpJson;
  s.enabled.store(true);
  s.running.store(true);
  s.exporter = std::thread([&s] {
    std::unique_lock<std::mutex> lock(s.mu);
    while (!s.stop) {
      qx::cv_wait_for(s.cv, lock, std::chrono::milliseconds(s.interval_ms));
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

SpanContext current_context() {
  if (!t_ctx.stack.empty()) return {t_ctx.trace_id, t_ctx.stack.back()};
  if (!t_ctx.installed_parent.empty()) return {t_ctx.trace_id, t_ctx.installed_parent};
  return {};
}

SpanGuard::SpanGuard(const std::string& name) : name_(name), start_ns_(now_unix_ns()) {
  if (!enabled()) return;  // ids stay empty; destructor is a no-op
  span_id_ = rand_hex(8);
  if (!t_ctx.stack.empty()) {
    parent_id_ = t_ctx.stack.back();
    trace_id_ = t_ctx.trace_id;
  } else if (!t_ctx.installed_parent.empty()) {
    parent_id_ = t_ctx.installed_parent;
    trace_id_ = t_ctx.trace_id;
  } else {
    trace_id_ = rand_hex(16);  // root span: fresh trace
    t_ctx.trace_id = trace_id_;
  }
  t_ctx.stack.push_back(span_id_);
}

SpanGuard::~SpanGuard() {
  if (span_id_.empty()) return;  // was created before/without otlp::init
  if (!t_ctx.stack.empty() && t_ctx.stack.back() == span_id_) t_ctx.stack.pop_back();
  if (!enabled()) return;
  ThreadSpanBuf& buf = thread_buf();
  std::lock_guard<std::mutex> lock(buf.mu);
  if (buf.spans.size() >= kMaxPerThreadSpans) {
    // bounded buffer: a stalled exporter drops oldest spans instead of
    // growing the daemon heap without limit
    buf.spans.erase(buf.spans.begin(),
                    buf.spans.begin() + static_cast<long>(kMaxPerThreadSpans / 10));
    state().dropped_spans.fetch_add(kMaxPerThreadSpans / 10, std::memory_order_relaxed);
  }
  buf.spans.push_back({name_, trace_id_, span_id_, parent_id_, start_ns_, now_unix_ns()});
}

ContextGuard::ContextGuard(const SpanContext& parent) {
  if (parent.span_id.empty()) return;
  saved_trace_id_ = t_ctx.trace_id;
  saved_parent_ = t_ctx.installed_parent;
  t_ctx.trace_id = parent.trace_id;
  t_ctx.installed_parent = parent.span_id;
  installed_ = true;
}

ContextGuard::~ContextGuard() {
  if (!installed_) return;
  t_ctx.trace_id = saved_trace_id_;
  t_ctx.installed_parent = saved_parent_;
}

uint64_t delivered_batches() { return state().delivered.load(std::memory_order_relaxed); }

}  // namespace otlp
