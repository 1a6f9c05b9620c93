// promql.hpp — idle-GPU PromQL query builder.
//
// The reference renders a compiled-in Jinja2 template once at startup
// (gpu-pruner/src/query.promql.j2, gpu-pruner/src/main.rs:280-282). Here the
// query is composed by a native builder with the same rendered contract,
// pinned by the 11 template tests ported from reference main.rs:572-740:
//
//  * `max_over_time` (never avg) of DCGM_FI_PROF_GR_ENGINE_ACTIVE (0-1) OR
//    DCGM_FI_DEV_GPU_UTIL / 100 over the `[<duration>m]` window,
//  * summed by (Hostname, container, pod, namespace, gpu, modelName) — with
//    the `exported_` prefix on pod/namespace/container unless honor_labels,
//  * node-type enrichment join against node_dmi_info with a bare fallback,
//  * `== 0` idle predicate,
//  * optional `unless` clause excluding pods whose peak power over the window
//    reached --power-threshold,
//  * optional namespace / modelName regex filters in every compute selector.
//
// The series themselves come from the first-party mi355-exporter
// (native/exporter/), which publishes exactly these DCGM-shaped names for
// gfx950 so the decision layers stay drop-in.
#pragma once

#include <optional>
#include <string>

namespace pruner {

struct QueryArgs {
  long duration_min = 30;
  std::optional<std::string> namespace_re;   // regex filter on namespace label
  std::optional<std::string> model_name_re;  // regex filter on modelName label
  std::optional<double> power_threshold_w;   // watts; enables the unless clause
  bool honor_labels = false;                 // native vs exported_* label names
};

std::string build_idle_query(const QueryArgs& args);

}  // namespace pruner
