#include "daemon.hpp"

#include <atomic>
#include <chrono>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "../common/log.hpp"
#include "../common/queue.hpp"
#include "engine.hpp"
#include "otlp.hpp"
#include "promql.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::daemon";
}

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
      if (!(enabled & kind_flag(sk->kind))) {
        LOGI(TARGET, "Skipping resource type " + sk->kind_str() + " because it is not enabled");
        continue;
      }
      try {
        {
          std::lock_guard<std::mutex> lock(consumer_kube_mu);
          if (!consumer_kube)
            consumer_kube = std::make_shared<KubeClient>(KubeConfig::resolve());
        }
        scale(*consumer_kube, *sk);
      } catch (const std::exception& e) {
        logx::counter_add("monotonic_counter.scale_failures", 1);
        LOGE(TARGET, std::string("Failed to scale resource! ") + e.what());
        continue;
      }
      logx::counter_add("monotonic_counter.scale_successes", 1);
      LOGI(TARGET, "Scaled Resource: [" + sk->kind_str() + "] - " +
                       sk->ns().value_or("default") + ":" + sk->name());
    }
  };
  int n_consumers = std::max(1, std::min(cfg.max_concurrency, 8));
  std::vector<std::thread> consumers;
  consumers.reserve(static_cast<size_t>(n_consumers));
  for (int i = 0; i < n_consumers; i++) consumers.emplace_back(consume);

  int consecutive_failures = 0;
  auto next_tick = std::chrono::steady_clock::now();
  while (true) {
    if (cfg.daemon_mode) {
      std::this_thread::sleep_until(next_tick);
      next_tick += std::chrono::seconds(cfg.check_interval_s);
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
      if (++consecutive_failures > cfg.max_consecutive_failures) {
        LOGE(TARGET, "Too many consecutive failures, exiting");
        exit_code.store(1);
        break;
      }
    }
    if (!cfg.daemon_mode) break;
  }

  queue.close();  // producer done: consumers drain and exit
  for (auto& c : consumers) c.join();
  return exit_code.load();
}

}  // namespace pruner
