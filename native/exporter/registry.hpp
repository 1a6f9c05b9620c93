// registry.hpp — Prometheus text exposition for the mi355-exporter.
//
// Series names and label shapes are deliberately DCGM-compatible
// (DCGM_FI_PROF_GR_ENGINE_ACTIVE / DCGM_FI_DEV_GPU_UTIL /
// DCGM_FI_DEV_POWER_USAGE, labels gpu / UUID / device / modelName / Hostname
// / pod / namespace / container) so the pruner's PromQL — and any dashboards
// written for the reference's dcgm-exporter metrics — are drop-in
// (SURVEY.md §2.4 "GPU-adjacent surface").
#pragma once

#include <map>
#include <string>
#include <vector>

#include "attrib.hpp"
#include "sampler.hpp"

namespace exporter {

struct RenderOptions {
  std::string hostname;
  // extra constant labels appended to every series (e.g. node_type)
  std::vector<std::pair<std::string, std::string>> const_labels;
};

// Render one scrape: all device samples (+ pod attributions keyed by device
// index) in Prometheus text exposition format 0.0.4.
std::string render_metrics(const std::vector<DeviceSample>& samples,
                           const std::map<uint32_t, PodAttribution>& attribs,
                           const RenderOptions& opts);

}  // namespace exporter
