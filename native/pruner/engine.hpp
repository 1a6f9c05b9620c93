// engine.hpp — decision engine + scaling engine of the MI355X-native
// gpu-pruner.
//
// Covers the reference's layers L3-L5:
//   * series→pod dedup and eligibility filtering (missing / Pending /
//     no-creation-timestamp / younger-than-lookback pods skipped —
//     reference main.rs:411-532),
//   * the owner-reference walk to the scalable root (KServe label shortcut →
//     InferenceService; ReplicaSet → Deployment; StatefulSet → Notebook —
//     reference lib.rs:437-513),
//   * parent dedup via uid-hash set (reference main.rs:534),
//   * per-kind scale-to-zero with a K8s Event per action (reference
//     lib.rs:337-386,517-576).
//
// The hot loop runs on a configurable N-way worker pool with keep-alive
// apiserver connections (the reference hard-caps 10 in-flight —
// main.rs:530 — and opens per-request streams).
#pragma once

#include <functional>
#include <optional>
#include <string>
#include <vector>

#include "../common/queue.hpp"
#include "config.hpp"
#include "k8s.hpp"
#include "objcache.hpp"
#include "prom.hpp"
#include "resources.hpp"

namespace pruner {

// Walk the owner references of `pod` (a dynamic Pod object) to its scalable
// root. Returns nullopt (with a log) when nothing scalable is found —
// matching the reference's error path (lib.rs:509-512). The ObjectCache
// overload answers from prefetched collections when available.
std::optional<ScaleKind> find_root_object(ObjectCache& objs, const jsn::Value& pod);
std::optional<ScaleKind> find_root_object(KubeClient& kube, const jsn::Value& pod);

// Emit the scale Event (failure non-fatal) and apply the per-kind
// scale-to-zero patch. Throws KubeError on patch failure.
void scale(KubeClient& kube, const ScaleKind& sk);

// Actuate a batch of roots concurrently (each root = Event POST + patch, 2
// apiserver round-trips): skips kinds not in `enabled_mask`, counts
// successes, logs failures. The reference drains its scale channel with a
// single serial consumer (main.rs:332-367); with per-root RTTs this is the
// second fan-out that matters at 1000-pod scale.
// Single actuation step shared by the daemon's consumer pool and scale_all
// (one path for enabled-mask check + scale + counters).
// Returns true when the resource was actually scaled.
bool scale_one(KubeClient& kube, const ScaleKind& sk, uint8_t enabled_mask);

size_t scale_all(KubeClient& kube, const std::vector<ScaleKind>& roots,
                 uint8_t enabled_mask, int concurrency);

struct QueryOutcome {
  size_t num_series = 0;       // raw series returned by Prometheus
  size_t num_unique_pods = 0;  // after (pod, namespace) dedup
  size_t shutdown_events = 0;  // scalable roots after parent dedup
};

// One decision pass: run `query` against Prometheus, evaluate every unique
// pod concurrently, dedup shared parents, and either log (dry-run) or enqueue
// each root for the scale consumer. `tx` may be null in dry-run.
QueryOutcome run_query_and_scale(PromClient& prom, KubeClient& kube,
                                 const std::string& query, const Config& cfg,
                                 qx::BoundedQueue<ScaleKind>* tx);

// Pure decision core, exposed for tests and the benchmark: dedup + filter +
// owner-walk + parent-dedup over an already-parsed vector result.
std::vector<ScaleKind> evaluate_candidates(KubeClient& kube, const jsn::Value& result_vector,
                                           const Config& cfg, QueryOutcome* outcome);

}  // namespace pruner
