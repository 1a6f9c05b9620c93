#include "registry.hpp"

#include <cstdio>

namespace exporter {

namespace {

std::string fmt(double v) {
  char buf[64];
  if (v == static_cast<long long>(v) && v < 1e15 && v > -1e15) {
    std::snprintf(buf, sizeof buf, "%lld", static_cast<long long>(v));
  } else {
    std::snprintf(buf, sizeof buf, "%.10g", v);
  }
  return buf;
}

std::string escape_label(const std::string& s) {
  std::string out;
  for (char c : s) {
    if (c == '\\' || c == '"') out += '\\';
    if (c == '\n') {
      out += "\\n";
      continue;
    }
    out += c;
  }
  return out;
}

}  // namespace

std::string render_metrics(const std::vector<DeviceSample>& samples,
                           const std::map<uint32_t, PodAttribution>& attribs,
                           const RenderOptions& opts) {
  struct Family {
    const char* name;
    const char* help;
    const char* type;
    double (*get)(const DeviceSample&);
  };
  // DCGM-shaped families (names/units per dcgm-exporter conventions the
  // pruner's PromQL consumes) + AMD-native extras.
  static const Family families[] = {
      {"DCGM_FI_PROF_GR_ENGINE_ACTIVE",
       "Ratio of time the graphics engine was active over the scrape window (0-1)",
       "gauge", [](const DeviceSample& d) { return d.gr_engine_active; }},
      {"DCGM_FI_DEV_GPU_UTIL", "GPU utilization (0-100)", "gauge",
       [](const DeviceSample& d) { return d.busy_percent; }},
      {"DCGM_FI_DEV_MEM_COPY_UTIL", "Memory controller utilization (0-100)", "gauge",
       [](const DeviceSample& d) { return d.mem_busy_percent; }},
      {"DCGM_FI_DEV_POWER_USAGE", "Board power draw (W)", "gauge",
       [](const DeviceSample& d) { return d.power_w; }},
      {"DCGM_FI_DEV_TOTAL_ENERGY_CONSUMPTION", "Total energy consumption (mJ)", "counter",
       [](const DeviceSample& d) { return d.energy_j * 1000.0; }},
      {"DCGM_FI_DEV_FB_USED", "Framebuffer used (MiB)", "gauge",
       [](const DeviceSample& d) { return d.vram_used_b / (1024.0 * 1024.0); }},
      {"DCGM_FI_DEV_FB_FREE", "Framebuffer free (MiB)", "gauge",
       [](const DeviceSample& d) { return (d.vram_total_b - d.vram_used_b) / (1024.0 * 1024.0); }},
      {"DCGM_FI_DEV_GPU_TEMP", "GPU edge temperature (C)", "gauge",
       [](const DeviceSample& d) { return d.temp_edge_c; }},
      {"DCGM_FI_DEV_SM_CLOCK", "Graphics clock (MHz)", "gauge",
       [](const DeviceSample& d) { return d.gfx_clock_mhz; }},
      // AMD-native extras: xGMI topology + accumulated link traffic
      {"mi355_xgmi_link_width", "xGMI link width (lanes)", "gauge",
       [](const DeviceSample& d) { return d.xgmi_link_width; }},
      {"mi355_xgmi_link_speed", "xGMI link speed", "gauge",
       [](const DeviceSample& d) { return d.xgmi_link_speed; }},
      {"mi355_xgmi_read_kb_total", "Accumulated xGMI reads across links (KiB)",
       "counter", [](const DeviceSample& d) { return d.xgmi_read_kb; }},
      {"mi355_xgmi_write_kb_total", "Accumulated xGMI writes across links (KiB)",
       "counter", [](const DeviceSample& d) { return d.xgmi_write_kb; }},
  };

  // Sampler read-health families (always emitted, healthy or not): alerting
  // surface for a device whose SMU/sysfs reads are failing.
  static const Family health_families[] = {
      {"mi355_sampler_healthy",
       "1 when device activity reads are succeeding; 0 after repeated failures "
       "(activity series are withheld while 0)",
       "gauge", [](const DeviceSample& d) { return d.healthy ? 1.0 : 0.0; }},
      {"mi355_sampler_last_good_read_age_seconds",
       "Seconds since the last successful activity read for this device",
       "gauge", [](const DeviceSample& d) { return d.staleness_s; }},
  };

  std::string out;
  out.reserve(4096);
  auto emit_family = [&](const Family& fam, bool healthy_only) {
    out += std::string("# HELP ") + fam.name + " " + fam.help + "\n";
    out += std::string("# TYPE ") + fam.name + " " + fam.type + "\n";
    for (const auto& d : samples) {
      // A device with a dead read path must not publish (stale) values:
      // a withheld series can never satisfy the culler's `== 0` idle
      // predicate, so broken telemetry fails safe (no cull).
      if (healthy_only && !d.healthy) continue;
      out += fam.name;
      out += "{gpu=\"" + std::to_string(d.index) + "\"";
      out += ",UUID=\"" + escape_label(d.unique_id) + "\"";
      out += ",device=\"renderD" + std::to_string(d.drm_render_minor) + "\"";
      out += ",modelName=\"" + escape_label(d.model_name) + "\"";
      out += ",Hostname=\"" + escape_label(opts.hostname) + "\"";
      if (auto it = attribs.find(d.index); it != attribs.end()) {
        out += ",pod=\"" + escape_label(it->second.pod) + "\"";
        out += ",namespace=\"" + escape_label(it->second.ns) + "\"";
        out += ",container=\"" + escape_label(it->second.container) + "\"";
      }
      for (const auto& [k, v] : opts.const_labels)
        out += "," + k + "=\"" + escape_label(v) + "\"";
      out += "} " + fmt(fam.get(d)) + "\n";
    }
  };
  for (const auto& fam : families) emit_family(fam, /*healthy_only=*/true);
  for (const auto& fam : health_families) emit_family(fam, /*healthy_only=*/false);
  return out;
}

}  // namespace exporter
