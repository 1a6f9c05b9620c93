#include "daemon.hpp"

#include <csignal>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <memory>
#include <mutex>
#include <sstream>
#include <thread>
#include <vector>

#include "../common/http_server.hpp"
#include "informer.hpp"
#include "leader.hpp"
#include "../common/log.hpp"
#include "../common/queue.hpp"
#include "engine.hpp"
#include "otlp.hpp"
#include "promql.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::daemon";

// Graceful shutdown: SIGTERM/SIGINT set the flag (only an atomic store —
// the async-signal-safe subset); the interval wait polls it in 100 ms
// slices so K8s pod termination interrupts a 180 s tick wait promptly,
// finishes in-flight scale actions, and exits 0.
std::atomic<bool> g_shutdown{false};

void on_shutdown_signal(int) { g_shutdown.store(true); }

// Sleep until `deadline` or shutdown, whichever first.
void interruptible_sleep_until(std::chrono::steady_clock::time_point deadline) {
  while (!g_shutdown.load(std::memory_order_relaxed)) {
    auto now = std::chrono::steady_clock::now();
    if (now >= deadline) return;
    auto slice = std::min<std::chrono::steady_clock::duration>(
        deadline - now, std::chrono::milliseconds(100));
    std::this_thread::sleep_for(slice);
  }
}

// Daemon self-metrics in Prometheus text format (--metrics-port): the six
// counters of SURVEY.md §5.5 plus a liveness endpoint.
std::string render_self_metrics() {
  std::ostringstream out;
  for (const auto& [name, value] : logx::counters_snapshot()) {
    bool monotonic = name.rfind("monotonic_counter.", 0) == 0;
    std::string short_name = "gpu_pruner_" + name.substr(name.find('.') + 1);
    if (monotonic) short_name += "_total";
    out << "# TYPE " << short_name << (monotonic ? " counter" : " gauge") << "\n";
    out << short_name << " " << value << "\n";
  }
  return out.str();
}

}  // namespace

int run_daemon(const Config& cfg) {
  const uint8_t enabled = get_enabled_resources(cfg.enabled_resources);
  {
    std::string names;
    for (Kind k : {Kind::Deployment, Kind::ReplicaSet, Kind::StatefulSet,
                   Kind::InferenceService, Kind::Notebook})
      if (enabled & kind_flag(k)) names += std::string(names.empty() ? "" : " | ") + kind_name(k);
    LOGI(TARGET, "Enabled resources: " + (names.empty() ? "(none)" : names));
  }

  // The query is built once at startup; per-tick work is the HTTP round-trips
  // (reference renders its template once at main.rs:280-282).
  const std::string query = build_idle_query(cfg.query_args());
  LOGI(TARGET, "Running w/ Query: " + query);

  qx::BoundedQueue<ScaleKind> queue(static_cast<size_t>(cfg.queue_capacity));
  std::atomic<int> exit_code{0};

  // Consumer pool draining the scale queue. The reference uses one serial
  // consumer (main.rs:332-367); each scale is 2 apiserver round-trips, so a
  // small pool keeps actuation off the critical path at 1000-pod scale. The
  // shared KubeClient's connection pool is reused across workers.
  std::shared_ptr<KubeClient> consumer_kube;
  std::mutex consumer_kube_mu;
  auto consume = [&] {
    while (auto sk = queue.pop()) {
      try {
        {
          std::lock_guard<std::mutex> lock(consumer_kube_mu);
          if (!consumer_kube)
            consumer_kube = std::make_shared<KubeClient>(KubeConfig::resolve());
        }
      } catch (const std::exception& e) {
        logx::counter_add("monotonic_counter.scale_failures", 1);
        LOGE(TARGET, std::string("Failed to build Kubernetes client: ") + e.what());
        continue;
      }
      // shared actuation path (enabled-mask check + scale + counters + log)
      scale_one(*consumer_kube, *sk, enabled);
    }
  };
  int n_consumers = std::max(1, std::min(cfg.max_concurrency, 8));
  std::vector<std::thread> consumers;
  consumers.reserve(static_cast<size_t>(n_consumers));
  for (int i = 0; i < n_consumers; i++) consumers.emplace_back(consume);

  g_shutdown.store(false);
  std::signal(SIGTERM, on_shutdown_signal);
  std::signal(SIGINT, on_shutdown_signal);

  // Optional self-metrics endpoint (not part of the reference's surface).
  std::unique_ptr<http::Server> metrics_server;
  if (cfg.metrics_port > 0) {
    metrics_server = std::make_unique<http::Server>(
        "0.0.0.0", static_cast<uint16_t>(cfg.metrics_port),
        [](const http::ServerRequest& req) {
          http::ServerResponse resp;
          if (req.path == "/metrics") {
            resp.body = render_self_metrics();
            resp.content_type = "text/plain; version=0.0.4; charset=utf-8";
          } else if (req.path == "/healthz") {
            resp.body = "ok\n";
          } else {
            resp.status = 404;
            resp.body = "not found\n";
          }
          return resp;
        });
    metrics_server->start();
    LOGI(TARGET, "Self-metrics on :" + std::to_string(metrics_server->port()));
  }

  // Optional Lease-based leader election (multi-replica deployments): only
  // the holder runs decision ticks; standbys keep renewing their candidacy
  // and take over when the lease expires or is released.
  std::unique_ptr<LeaderElector> elector;
  if (cfg.leader_elect && cfg.daemon_mode) {
    std::string identity;
    if (const char* pn = std::getenv("POD_NAME"); pn && *pn) identity = pn;
    if (identity.empty()) {
      char host[256] = {0};
      ::gethostname(host, sizeof host - 1);
      identity = std::string(host) + "-" + std::to_string(::getpid());
    }
    KubeConfig kc = KubeConfig::resolve();
    std::string lease_ns = kc.default_namespace;
    if (const char* pns = std::getenv("POD_NAMESPACE"); pns && *pns) lease_ns = pns;
    elector = std::make_unique<LeaderElector>(kc, lease_ns, "gpu-pruner", identity,
                                              cfg.leader_lease_duration_s,
                                              cfg.leader_renew_period_s);
    elector->start();
    LOGI(TARGET, "Leader election on (lease " + lease_ns + "/gpu-pruner, identity \"" +
                     identity + "\")");
  }

  int consecutive_failures = 0;
  bool was_leader = true;  // log standby transitions once
  auto next_tick = std::chrono::steady_clock::now();
  while (!g_shutdown.load()) {
    if (cfg.daemon_mode) {
      interruptible_sleep_until(next_tick);
      if (g_shutdown.load()) break;
      next_tick += std::chrono::seconds(cfg.check_interval_s);
    }
    if (elector && !elector->is_leader()) {
      if (was_leader) LOGI(TARGET, "Not the leader — standing by");
      was_leader = false;
      continue;
    }
    if (elector && !was_leader) {
      LOGI(TARGET, "Leadership acquired — resuming decision ticks");
      was_leader = true;
    }
    try {
      otlp::SpanGuard span("run_query_and_scale");
      // Clients are rebuilt every tick so rotated tokens are picked up
      // (reference main.rs:296,377-388).
      auto prom = build_prom_client(cfg);
      KubeClient kube(KubeConfig::resolve());
      QueryOutcome qr = run_query_and_scale(*prom, kube, query, cfg, &queue);
      consecutive_failures = 0;
      logx::counter_add("monotonic_counter.query_successes", 1);
      LOGI(TARGET, "Query succeeded");
      logx::gauge_set("counter.query_returned_candidates",
                      static_cast<int64_t>(qr.num_unique_pods));
      logx::gauge_set("counter.query_returned_shutdown_events",
                      static_cast<int64_t>(qr.shutdown_events));
      LOGI(TARGET, "Returned candidates: " + std::to_string(qr.num_unique_pods) +
                       ", shutdown events: " + std::to_string(qr.shutdown_events));
    } catch (const std::exception& e) {
      logx::counter_add("monotonic_counter.query_failures", 1);
      LOGE(TARGET, std::string("Failed to run query and scale down: ") + e.what());
      // Pre-increment comparison for reference parity (main.rs:310-321 reads
      // the counter before bumping it): with the default max of 5 the daemon
      // survives 6 consecutive failures and exits on the 7th.
      if (consecutive_failures++ > cfg.max_consecutive_failures) {
        LOGE(TARGET, "Too many consecutive failures, exiting");
        exit_code.store(1);
        break;
      }
    }
    if (!cfg.daemon_mode) break;
  }

  if (g_shutdown.load()) LOGI(TARGET, "Shutdown signal received, draining");
  queue.close();  // producer done: consumers drain and exit
  for (auto& c : consumers) c.join();
  if (metrics_server) metrics_server->stop();
  if (elector) elector->stop();  // releases the lease for instant failover
  InformerRegistry::global().stop_all();  // close watch streams (if any)
  return exit_code.load();
}

}  // namespace pruner
