#include "engine.hpp"

#include <atomic>
#include <chrono>
#include <mutex>
#include <set>
#include <thread>

#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../common/threadpool.hpp"
#include "otlp.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::engine";
}

std::optional<ScaleKind> find_root_object(KubeClient& kube, const jsn::Value& pod) {
  ObjectCache direct(kube, EvalStrategy::PerPodGet);
  return find_root_object(direct, pod);
}

std::optional<ScaleKind> find_root_object(ObjectCache& objs, const jsn::Value& pod) {
  otlp::SpanGuard span("find_root_object");
  const jsn::Value& meta = pod.get("metadata");
  std::string pod_name = meta.get("name").as_string();
  std::string ns = meta.get("namespace").as_string_or("");

  LOGI(TARGET, "Finding root object of \"" + pod_name + "\" for scale-down.");

  // KServe shortcut: the predictor pod carries the InferenceService name as a
  // label, skipping the two-hop owner walk entirely.
  const jsn::Value& labels = meta.get("labels");
  const jsn::Value& ks = labels.get("serving.kserve.io/inferenceservice");
  if (ks.is_string()) {
    auto is = objs.get_object(Kind::InferenceService, ns, ks.as_string());
    if (!is) {
      LOGW(TARGET, "KServe label points at missing InferenceService " + ks.as_string());
      return std::nullopt;
    }
    return ScaleKind{Kind::InferenceService, *is};
  }

  const jsn::Value& ors = meta.get("ownerReferences");
  if (ors.is_array()) {
    for (const auto& owner : ors.arr()) {
      std::string owner_kind = owner.get("kind").as_string();
      std::string owner_name = owner.get("name").as_string();
      if (owner_kind == "ReplicaSet") {
        LOGI(TARGET, "Found ReplicaSet!");
        // fetch errors fall through to the next owner reference (the
        // reference's `if let Ok(rs)` swallows them — lib.rs:464)
        std::optional<jsn::Value> rs;
        try {
          rs = objs.get_object(Kind::ReplicaSet, ns, owner_name);
        } catch (const std::exception& e) {
          LOGW(TARGET, "ReplicaSet " + owner_name + " fetch failed: " + e.what());
        }
        if (rs) {
          const jsn::Value& rs_ors = rs->at({"metadata", "ownerReferences"});
          if (rs_ors.is_array()) {
            for (const auto& rs_or : rs_ors.arr()) {
              if (rs_or.get("kind").as_string() == "Deployment") {
                LOGI(TARGET, "Found Deployment owning ReplicaSet!");
                auto dep = objs.get_object(Kind::Deployment, ns, rs_or.get("name").as_string());
                if (!dep) return std::nullopt;
                return ScaleKind{Kind::Deployment, *dep};
              }
            }
          }
          // ReplicaSet with no Deployment parent scales directly.
          return ScaleKind{Kind::ReplicaSet, *rs};
        }
      } else if (owner_kind == "StatefulSet") {
        LOGI(TARGET, "Found StatefulSet!");
        std::optional<jsn::Value> ss;
        try {
          ss = objs.get_object(Kind::StatefulSet, ns, owner_name);
        } catch (const std::exception& e) {
          LOGW(TARGET, "StatefulSet " + owner_name + " fetch failed: " + e.what());
        }
        if (ss) {
          const jsn::Value& ss_ors = ss->at({"metadata", "ownerReferences"});
          if (ss_ors.is_array()) {
            for (const auto& ss_or : ss_ors.arr()) {
              if (ss_or.get("kind").as_string() == "Notebook") {
                LOGI(TARGET, "Found Notebook owning StatefulSet!");
                auto nb = objs.get_object(Kind::Notebook, ns, ss_or.get("name").as_string());
                if (!nb) return std::nullopt;
                return ScaleKind{Kind::Notebook, *nb};
              }
            }
          }
          // StatefulSet with no Notebook parent scales directly.
          return ScaleKind{Kind::StatefulSet, *ss};
        }
      } else {
        LOGD(TARGET, "Ignoring unrecognized owner ref kind: " + owner_kind);
      }
    }
  }

  LOGW(TARGET, "no scalable root object found for pod \"" + pod_name + "\"");
  return std::nullopt;
}

void scale(KubeClient& kube, const ScaleKind& sk) {
  otlp::SpanGuard span("scale");
  auto ns = sk.ns();
  if (ns) {
    // Announce first; a failed Event post never blocks the scale itself
    // (reference lib.rs:339-349).
    jsn::Value event = generate_scale_event(sk);
    try {
      kube.create("/api/v1/namespaces/" + strutil::url_encode(*ns) + "/events", event);
      LOGD(TARGET, "Emitted scale event for " + sk.kind_str() + " " + sk.name());
    } catch (const std::exception& e) {
      LOGE(TARGET, std::string("Failed to push Event for scale down!: ") + e.what());
    }
  }

  std::string namespace_ = ns.value_or("default");
  switch (sk.kind) {
    case Kind::Deployment:
    case Kind::ReplicaSet:
    case Kind::StatefulSet: {
      // Built-in workloads: spec.replicas=0 through the /scale subresource.
      otlp::SpanGuard sub("scale_to_zero");
      jsn::Value patch = jsn::Value::object();
      patch["spec"]["replicas"] = 0;
      kube.patch_scale(sk.kind, namespace_, sk.name(), patch);
      break;
    }
    case Kind::Notebook: {
      // Kubeflow convention: the stop annotation, set to "now".
      otlp::SpanGuard sub("scale_notebook_to_zero");
      jsn::Value patch = jsn::Value::object();
      patch["metadata"]["annotations"]["kubeflow-resource-stopped"] = strutil::rfc3339_now();
      kube.merge_patch(object_path(Kind::Notebook, namespace_, sk.name()), patch);
      break;
    }
    case Kind::InferenceService: {
      // KServe scales the predictor down itself once minReplicas is 0 and
      // rescales on traffic; durable capacity needs a manual minReplicas
      // reset (same semantics as the reference, lib.rs:553-576).
      otlp::SpanGuard sub("scale_inference_service_to_zero");
      jsn::Value patch = jsn::Value::object();
      patch["spec"]["predictor"]["minReplicas"] = 0;
      kube.merge_patch(object_path(Kind::InferenceService, namespace_, sk.name()), patch);
      break;
    }
  }
}

bool scale_one(KubeClient& kube, const ScaleKind& sk, uint8_t enabled_mask) {
  if (!(enabled_mask & kind_flag(sk.kind))) {
    LOGI(TARGET, "Skipping resource type " + sk.kind_str() + " because it is not enabled");
    return false;
  }
  try {
    scale(kube, sk);
  } catch (const std::exception& e) {
    logx::counter_add("monotonic_counter.scale_failures", 1);
    LOGE(TARGET, std::string("Failed to scale resource! ") + e.what());
    return false;
  }
  logx::counter_add("monotonic_counter.scale_successes", 1);
  LOGI(TARGET, "Scaled Resource: [" + sk.kind_str() + "] - " +
                   sk.ns().value_or("default") + ":" + sk.name());
  return true;
}

size_t scale_all(KubeClient& kube, const std::vector<ScaleKind>& roots,
                 uint8_t enabled_mask, int concurrency) {
  std::atomic<size_t> scaled{0};
  // propagate the caller's span (run_query_and_scale) onto the pool threads
  // so scale → scale_to_zero trees nest under it
  otlp::SpanContext ctx = otlp::current_context();
  qx::ThreadPool::global().parallel_for(
      roots.size(), concurrency, [&](size_t i) {
        otlp::ContextGuard cg(ctx);
        if (scale_one(kube, roots[i], enabled_mask))
          scaled.fetch_add(1, std::memory_order_relaxed);
      });
  return scaled.load();
}

std::vector<ScaleKind> evaluate_candidates(KubeClient& kube, const jsn::Value& result_vector,
                                           const Config& cfg, QueryOutcome* outcome) {
  // Dedup series by (pod, namespace): multi-GPU pods emit one series per GPU
  // but the owner chain is resolved once per pod (reference main.rs:416-437).
  std::set<std::pair<std::string, std::string>> seen;
  std::vector<PodMetricData> unique_pods;
  size_t num_series = result_vector.is_array() ? result_vector.size() : 0;
  if (result_vector.is_array()) {
    for (const auto& series : result_vector.arr()) {
      try {
        PodMetricData pmd = parse_pod_metric(series);
        if (seen.emplace(pmd.name, pmd.ns).second) unique_pods.push_back(std::move(pmd));
      } catch (const PodConvertError& e) {
        LOGE(TARGET, std::string("Failed to unwrap pod fields: ") + e.what());
      }
    }
  }
  LOGI(TARGET, "Query returned " + std::to_string(num_series) + " series across " +
                   std::to_string(unique_pods.size()) + " unique pods");

  // LIST-vs-GET strategy: prefetch candidate namespaces' collections when
  // it pays (see objcache.hpp).
  ObjectCache objs(kube, cfg.eval_strategy);
  {
    std::map<std::string, int> ns_counts;
    for (const auto& pmd : unique_pods) ns_counts[pmd.ns]++;
    objs.prefetch(ns_counts, cfg.max_concurrency);
  }

  // A pod must predate the whole lookback window (+ grace) for the "no
  // activity over the window" signal to be trustworthy.
  const double lookback_s =
      static_cast<double>(cfg.duration_min) * 60.0 + static_cast<double>(cfg.grace_period_s);
  const double now_s =
      std::chrono::duration<double>(std::chrono::system_clock::now().time_since_epoch()).count();
  const double lookback_start = now_s - lookback_s;

  // Evaluate pods concurrently: each needs 1-3 apiserver round-trips (pod GET
  // + owner walk). Worker count is the --max-concurrency knob.
  std::vector<std::optional<ScaleKind>> results(unique_pods.size());
  otlp::SpanContext span_ctx = otlp::current_context();
  qx::ThreadPool::global().parallel_for(
      unique_pods.size(), cfg.max_concurrency, [&](size_t i) {
      otlp::ContextGuard cg(span_ctx);
      const PodMetricData& pmd = unique_pods[i];
      try {
        auto pod = objs.get_pod(pmd.ns, pmd.name);
        if (!pod) {
          LOGI(TARGET, "Skipping " + pmd.ns + ":" + pmd.name + ", pod no longer exists");
          return;
        }
        std::string phase = pod->at({"status", "phase"}).as_string_or("Unknown");
        if (phase == "Pending") {
          LOGI(TARGET, "Skipping pod " + pmd.ns + ":" + pmd.name + ", it's still pending");
          return;
        }
        const jsn::Value& created = pod->at({"metadata", "creationTimestamp"});
        if (!created.is_string()) {
          LOGW(TARGET,
               "Pod " + pmd.ns + ":" + pmd.name + " has no creation timestamp, skipping");
          return;
        }
        double created_s = 0;
        if (!strutil::parse_rfc3339(created.as_string(), &created_s)) {
          LOGW(TARGET, "Pod " + pmd.ns + ":" + pmd.name +
                           " has unparseable creation timestamp, skipping");
          return;
        }
        if (created_s >= lookback_start) return;  // too young for the window
        LOGI(TARGET, "Pod " + pmd.ns + ":" + pmd.name + " is idle and eligible for scaledown");
        results[i] = find_root_object(objs, *pod);
      } catch (const std::exception& e) {
        LOGE(TARGET,
             "Skipping " + pmd.ns + ":" + pmd.name + ", retrieval error: " + e.what());
      }
      });

  // Pods sharing a parent collapse to one scale action (uid-hash dedup,
  // reference main.rs:534).
  ScaleKindSet roots;
  for (auto& r : results)
    if (r) roots.insert(std::move(*r));

  if (outcome) {
    outcome->num_series = num_series;
    outcome->num_unique_pods = unique_pods.size();
    outcome->shutdown_events = roots.size();
  }
  return std::vector<ScaleKind>(std::make_move_iterator(roots.begin()),
                                std::make_move_iterator(roots.end()));
}

QueryOutcome run_query_and_scale(PromClient& prom, KubeClient& kube, const std::string& query,
                                 const Config& cfg, qx::BoundedQueue<ScaleKind>* tx) {
  jsn::Value result = prom.query_vector(query);
  QueryOutcome outcome;
  std::vector<ScaleKind> roots = evaluate_candidates(kube, result, cfg, &outcome);

  for (auto& obj : roots) {
    std::string label =
        "[" + obj.kind_str() + "] " + obj.ns().value_or("") + ":" + obj.name();
    if (cfg.run_mode == RunMode::DryRun) {
      LOGI(TARGET, "Dry-run: Would have sent " + label + " for scaledown");
      continue;
    }
    LOGI(TARGET, "Sending " + label + " for scaledown");
    if (tx && !tx->push(std::move(obj)))
      LOGE(TARGET, "Failed to send object for scaledown: queue closed");
  }
  return outcome;
}

}  // namespace pruner
