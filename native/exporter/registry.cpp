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

  std::string out;
  out.reserve(4096);
  for (const auto& fam : families) {
    out += std::string("# HELP ") + fam.name + " " + fam.help + "\n";
    out += std::string("# TYPE ") + fam.name + " " + fam.type + "\n";
    for (const auto& d : samples) {
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
  }
  return out;
}

}  // namespace exporter
