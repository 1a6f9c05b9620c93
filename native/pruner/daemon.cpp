#include "daemon.hpp"

#include <atomic>
#include <chrono>
#include <thread>

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

  std::thread consumer([&] {
    // The consumer owns its own apiserver client (the reference builds a
    // second KubeClient in scale_down_task, main.rs:333).
    std::unique_ptr<KubeClient> kube;
    while (auto sk = queue.pop()) {
      if (!(enabled & kind_flag(sk->kind))) {
        LOGI(TARGET, "Skipping resource type " + sk->kind_str() + " because it is not enabled");
        continue;
      }
      try {
        if (!kube) kube = std::make_unique<KubeClient>(KubeConfig::resolve());
        scale(*kube, *sk);
      } catch (const std::exception& e) {
        logx::counter_add("monotonic_counter.scale_failures", 1);
        LOGE(TARGET, std::string("Failed to scale resource! ") + e.what());
        continue;
      }
      logx::counter_add("monotonic_counter.scale_successes", 1);
      LOGI(TARGET, "Scaled Resource: [" + sk->kind_str() + "] - " +
                       sk->ns().value_or("default") + ":" + sk->name());
    }
  });

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

  queue.close();  // producer done: consumer drains and exits
  consumer.join();
  return exit_code.load();
}

}  // namespace pruner
