// unit_main.cpp — native unit tests for the sanitizer CI tier.
//
// The Python suite covers behavior; this binary re-exercises the pure-logic
// core (JSON, query builder, scaling model, queue, CLI) so it can be built
// under -fsanitize=address,undefined / thread and run with zero Python in
// the loop (the Rust reference gets safety from the
// compiler + clippy; the C++ build gets ASan/TSan CI jobs as the
// equivalent).
#include <cassert>
#include <cstdio>
#include <thread>
#include <vector>

#include "../common/json.hpp"
#include "../common/queue.hpp"
#include "../common/threadpool.hpp"
#include "../common/strutil.hpp"
#include "../common/grpc_client.hpp"
#include "../exporter/sampler.hpp"
#include "../pruner/config.hpp"
#include "../pruner/promql.hpp"
#include "../pruner/resources.hpp"

using namespace pruner;

static int failures = 0;
#define CHECK(cond)                                                   \
  do {                                                                \
    if (!(cond)) {                                                    \
      std::fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond); \
      failures++;                                                     \
    }                                                                 \
  } while (0)

static void test_json() {
  jsn::Value v = jsn::parse(R"({"a": [1, 2.5, "x", null, true], "b": {"c": "dé"}})");
  CHECK(v.get("a").size() == 5);
  CHECK(v.at({"b", "c"}).as_string() == "d\xc3\xa9");
  CHECK(jsn::parse(v.dump()) == v);
  jsn::Value obj = jsn::parse(R"({"spec": {"replicas": 3, "keep": 1}})");
  obj.merge_patch(jsn::parse(R"({"spec": {"replicas": 0}})"));
  CHECK(obj.at({"spec", "replicas"}).as_int() == 0);
  CHECK(obj.at({"spec", "keep"}).as_int() == 1);
  obj.merge_patch(jsn::parse(R"({"spec": {"keep": null}})"));
  CHECK(!obj.get("spec").contains("keep"));
  bool threw = false;
  try {
    jsn::parse("{broken");
  } catch (const jsn::ParseError&) {
    threw = true;
  }
  CHECK(threw);
}

static void test_promql() {
  QueryArgs qa;
  qa.duration_min = 45;
  std::string q = build_idle_query(qa);
  CHECK(q.find("max_over_time(") != std::string::npos);
  CHECK(q.find("avg_over_time") == std::string::npos);
  CHECK(q.find("[45m]") != std::string::npos);
  CHECK(q.find("unless") == std::string::npos);
  qa.power_threshold_w = 150.0;
  qa.namespace_re = "ml-team";
  q = build_idle_query(qa);
  CHECK(q.find("unless on (exported_pod, exported_namespace)") != std::string::npos);
  CHECK(q.find(">= 150") != std::string::npos);
  CHECK(strutil::count_occurrences(q, "exported_namespace =~ \"ml-team\"") == 5);
}

static void test_resources() {
  CHECK(get_enabled_resources("drsin") == RK_ALL);
  CHECK(get_enabled_resources("xdqz") == RK_DEPLOYMENT);
  CHECK(get_enabled_resources("") == RK_NONE);
  ScaleKind a{Kind::Notebook, jsn::parse(R"({"metadata":{"name":"a","namespace":"ns","uid":"u1"}})")};
  ScaleKind b{Kind::Notebook, jsn::parse(R"({"metadata":{"name":"b","namespace":"ns","uid":"u1"}})")};
  CHECK(a == b);  // CRDs: uid equality
  ScaleKindSet set;
  set.insert(a);
  set.insert(b);
  CHECK(set.size() == 1);
  jsn::Value ev = generate_scale_event(a);
  CHECK(ev.get("reason").as_string() == "Pod ns::a was not using GPU");
  CHECK(strutil::starts_with(ev.at({"metadata", "name"}).as_string(), "gpuscaler-"));
  CHECK(object_path(Kind::InferenceService, "ns", "x") ==
        "/apis/serving.kserve.io/v1beta1/namespaces/ns/inferenceservices/x");
}

static void test_queue_mpmc() {
  // TSan target: hammer the bounded queue from several producers/consumers.
  qx::BoundedQueue<int> q(16);
  std::atomic<long> sum{0};
  std::vector<std::thread> threads;
  for (int p = 0; p < 4; p++)
    threads.emplace_back([&q, p] {
      for (int i = 0; i < 1000; i++) q.push(p * 1000 + i);
    });
  for (int c = 0; c < 4; c++)
    threads.emplace_back([&q, &sum] {
      while (auto v = q.pop()) sum.fetch_add(*v);
    });
  for (int p = 0; p < 4; p++) threads[static_cast<size_t>(p)].join();
  q.close();
  for (size_t c = 4; c < threads.size(); c++) threads[c].join();
  long expect = 0;
  for (int p = 0; p < 4; p++)
    for (int i = 0; i < 1000; i++) expect += p * 1000 + i;
  CHECK(sum.load() == expect);
}

static void test_threadpool() {
  // TSan target: bounded parallel_for correctness incl. caller participation
  std::atomic<long> sum{0};
  qx::ThreadPool pool(8);
  pool.parallel_for(1000, 16, [&](size_t i) { sum.fetch_add(static_cast<long>(i)); });
  CHECK(sum.load() == 999 * 1000 / 2);
  sum = 0;
  pool.parallel_for(5, 1, [&](size_t i) { sum.fetch_add(static_cast<long>(i) + 1); });
  CHECK(sum.load() == 15);
  pool.parallel_for(0, 8, [&](size_t) { CHECK(false); });
  // repeated use reuses the same workers
  for (int round = 0; round < 50; round++) {
    std::atomic<int> n{0};
    pool.parallel_for(64, 32, [&](size_t) { n.fetch_add(1); });
    CHECK(n.load() == 64);
  }
}

static void test_cli() {
  auto r = parse_cli({"--prometheus-url", "http://p", "-t", "15", "--run-mode",
                      "scale-down"});
  CHECK(!r.error);
  CHECK(r.config.duration_min == 15);
  CHECK(r.config.run_mode == RunMode::ScaleDown);
  CHECK(parse_cli({}).error.has_value());
  CHECK(parse_cli({"--prometheus-url", "http://p", "--bogus"}).error.has_value());
}

static void test_strutil() {
  double ts = 0;
  CHECK(strutil::parse_rfc3339("2026-01-02T03:04:05Z", &ts));
  CHECK(ts == 1767323045.0);  // date -u -d 2026-01-02T03:04:05Z +%s
  double ts2 = 0;
  CHECK(strutil::parse_rfc3339("2026-01-02T04:04:05+01:00", &ts2));
  CHECK(ts == ts2);  // same instant
  CHECK(!strutil::parse_rfc3339("not a date", &ts));
  CHECK(strutil::uuid4_simple().size() == 32);
  CHECK(strutil::uuid4_simple() != strutil::uuid4_simple());
}

static void test_activity_window() {
  using exporter::ActivityWindow;
  // time-weighted ratio over known segments, sliding window
  ActivityWindow w;
  for (int i = 0; i <= 10; i++) w.add(i, i % 2 == 0 ? 1.0 : 0.0, true);
  double known = 0;
  double r = w.ratio(10.0, 10.0, &known);
  CHECK(known > 9.99 && known < 10.01);
  CHECK(r > 0.3 && r < 0.7);
  // idempotent reads
  CHECK(w.ratio(10.0, 10.0) == w.ratio(10.0, 10.0));
  // unknown (failed-read) time contributes to neither side
  ActivityWindow u;
  u.add(0.0, 0.0, true);
  u.add(1.0, 1.0, true);
  for (int t = 2; t <= 8; t++) u.add(t, 0.0, false);
  r = u.ratio(8.0, 8.0, &known);
  CHECK(known > 0.99 && known < 1.01);
  CHECK(r > 0.99);
  // burst ages out of the window
  ActivityWindow b;
  b.add(0.0, 0.0, true);
  b.add(1.0, 1.0, true);
  for (int t = 2; t <= 40; t++) b.add(t, 0.0, true);
  CHECK(b.ratio(40.0, 5.0) == 0.0);
  // retention bounds memory
  ActivityWindow m;
  m.set_retention(10.0);
  for (int i = 0; i < 1000; i++) m.add(i, 0.5, true);
  CHECK(m.size() < 20);
  // concurrent readers while a poller appends (sampler mu_ serializes in
  // production; here the reads are on const snapshots via copies)
}

static void test_hpack_decoder() {
  // RFC 7541 Appendix C.4.1-C.4.3: three consecutive Huffman-coded header
  // blocks sharing one dynamic table — the RFC's own hex, decoded under the
  // sanitizer tiers (the bit-twiddling lives here, not in Python).
  auto unhex = [](const char* h) {
    std::string out;
    for (size_t i = 0; h[i] && h[i + 1]; i += 2) {
      auto nib = [](char c) -> int {
        return c <= '9' ? c - '0' : c - 'a' + 10;
      };
      out += static_cast<char>(nib(h[i]) * 16 + nib(h[i + 1]));
    }
    return out;
  };
  grpcx::HpackDecoder d;
  auto r1 = d.decode_block(unhex("828684418cf1e3c2e5f23a6ba0ab90f4ff"));
  CHECK(r1.size() == 4);
  CHECK(r1[0].first == ":method" && r1[0].second == "GET");
  CHECK(r1[3].first == ":authority" && r1[3].second == "www.example.com");
  auto r2 = d.decode_block(unhex("828684be5886a8eb10649cbf"));
  CHECK(r2.size() == 5);
  CHECK(r2[3].second == "www.example.com");  // dynamic-table hit
  CHECK(r2[4].first == "cache-control" && r2[4].second == "no-cache");
  auto r3 = d.decode_block(unhex("828785bf408825a849e95ba97d7f8925a849e95bb8e8b4bf"));
  CHECK(r3.size() == 5);
  CHECK(r3[4].first == "custom-key" && r3[4].second == "custom-value");

  // garbage in → GrpcError or clean result, never a crash/overread (the
  // sanitizer build is the point of this loop); fixed-seed LCG for repro
  uint64_t state = 0x9e3779b97f4a7c15ull;
  auto rnd = [&]() {
    state = state * 6364136223846793005ull + 1442695040888963407ull;
    return static_cast<uint8_t>(state >> 33);
  };
  for (int iter = 0; iter < 2000; iter++) {
    std::string blob;
    size_t n = rnd() % 64;
    for (size_t i = 0; i < n; i++) blob += static_cast<char>(rnd());
    grpcx::HpackDecoder fuzz;
    try {
      (void)fuzz.decode_block(blob);
    } catch (const grpcx::GrpcError&) {
      // expected for most random inputs
    }
  }
}

int main() {
  test_json();
  test_promql();
  test_resources();
  test_queue_mpmc();
  test_threadpool();
  test_cli();
  test_strutil();
  test_activity_window();
  test_hpack_decoder();
  if (failures == 0) std::puts("native unit tests: all passed");
  return failures == 0 ? 0 : 1;
}
