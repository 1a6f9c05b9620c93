// synthbench.hpp — native synthetic Prometheus + kube-apiserver backend for
// the benchmark harness.
//
// The pytest fixtures (gpu_pruner_amd/fixtures/) are independent Python
// implementations used for correctness; this C++ backend exists so
// `bench.py` measures the decision engine itself rather than a Python HTTP
// server's GIL. It builds the BASELINE.json synthetic cluster shape (N idle
// pods over mixed Deployment / StatefulSet+Notebook / InferenceService
// parents), serves both API surfaces the daemon needs, counts actuations,
// and can inject a per-request latency to emulate apiserver RTT.
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <string>

#include "../common/http_server.hpp"
#include "../common/json.hpp"

namespace pruner {

struct SynthOptions {
  int n_pods = 1000;
  int pods_per_parent = 2;
  int gpus_per_pod = 1;
  int n_namespaces = 4;
  std::string model_name = "AMD Instinct MI355X";
  int latency_us = 0;  // injected per-request service latency (apiserver RTT emulation)
};

class SyntheticBackend {
public:
  explicit SyntheticBackend(SynthOptions opts);
  ~SyntheticBackend();

  void start();
  void stop();

  std::string prom_url() const;
  std::string k8s_url() const;

  // The activity value carried by every series (0.0 = idle → candidates).
  void set_series_value(double v) { series_value_.store(v); }

  int64_t events_posted() const { return events_posted_.load(); }
  int64_t scale_patches() const { return scale_patches_.load(); }
  int64_t requests_served() const { return requests_.load(); }
  int64_t watch_streams() const { return watch_streams_.load(); }
  int expected_parents() const { return n_parents_; }

private:
  http::ServerResponse handle_prom(const http::ServerRequest& req);
  http::ServerResponse handle_k8s(const http::ServerRequest& req);
  void build_cluster();

  SynthOptions opts_;
  int n_parents_ = 0;
  struct StoredObject {
    jsn::Value obj;
    std::string cached_dump;  // invalidated on PATCH; GETs dominate 6:1
  };
  std::mutex mu_;
  // kind → ns → name → object
  std::map<std::string, std::map<std::string, std::map<std::string, StoredObject>>> objects_;
  std::string series_json_zero_;  // pre-rendered result vector (values patched in)
  // watch support: monotonic resourceVersion + event log (pre-serialized
  // lines), mirroring the Python fixture's semantics
  std::condition_variable event_cv_;
  uint64_t rv_ = 1;
  struct WatchEvent {
    uint64_t rv;
    std::string kind, ns, line;
  };
  std::deque<WatchEvent> watch_log_;
  std::atomic<bool> closing_{false};
  std::atomic<int64_t> watch_streams_{0};
  std::atomic<double> series_value_{0.0};
  std::atomic<int64_t> events_posted_{0};
  std::atomic<int64_t> scale_patches_{0};
  std::atomic<int64_t> requests_{0};
  std::unique_ptr<http::Server> prom_server_;
  std::unique_ptr<http::Server> k8s_server_;
};

}  // namespace pruner
