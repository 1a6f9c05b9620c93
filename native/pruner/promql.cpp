#include "promql.hpp"

#include <cstdio>

namespace pruner {

namespace {

// Escape a value for embedding inside a double-quoted PromQL string
// (regex matchers): backslashes and quotes. The reference interpolates the
// raw flag value into its template (query.promql.j2:12), so a quote in
// --namespace breaks its query; here it cannot.
std::string esc(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  for (char c : s) {
    if (c == '\\' || c == '"') out += '\\';
    out += c;
  }
  return out;
}

std::string fmt_num(double v) {
  char buf[64];
  // integral thresholds render without a decimal point ("150"), matching the
  // reference's minijinja float rendering asserted by its power tests.
  if (v == static_cast<long long>(v)) {
    std::snprintf(buf, sizeof buf, "%lld", static_cast<long long>(v));
  } else {
    std::snprintf(buf, sizeof buf, "%g", v);
  }
  return buf;
}

}  // namespace

std::string build_idle_query(const QueryArgs& a) {
  const std::string pl = a.honor_labels ? "pod" : "exported_pod";
  const std::string nl = a.honor_labels ? "namespace" : "exported_namespace";
  const std::string cl = a.honor_labels ? "container" : "exported_container";
  const std::string win = "[" + std::to_string(a.duration_min) + "m]";

  // Selector body shared by every compute metric: non-empty pod label plus
  // the optional namespace / model regex filters.
  auto selector = [&](bool with_model) {
    std::string s = "{\n      " + pl + " != \"\"";
    if (a.namespace_re) s += ", " + nl + " =~ \"" + esc(*a.namespace_re) + "\"";
    if (with_model && a.model_name_re)
      s += ", modelName =~ \"" + esc(*a.model_name_re) + "\"";
    s += "\n    }";
    return s;
  };

  // Peak activity per (node, container, pod, namespace, gpu, model) over the
  // window: the profiling-class activity ratio, with the 0-100 utilization
  // gauge normalized to 0-1 as a fallback for exporters that only publish it.
  const std::string group_by =
      "Hostname, " + cl + ", " + pl + ", " + nl + ", gpu, modelName";
  const std::string idle_gpus =
      "sum by (" + group_by + ") (\n"
      "    max_over_time(DCGM_FI_PROF_GR_ENGINE_ACTIVE" + selector(true) + win + ")\n"
      "    or\n"
      "    max_over_time(DCGM_FI_DEV_GPU_UTIL" + selector(true) + win + ") / 100\n"
      ")";

  // Enrich with the node hardware type from node_dmi_info (joined on
  // Hostname via label_replace), falling back to the bare series when the
  // node-exporter join has no match.
  std::string q =
      "(\n  " + idle_gpus + " * on (Hostname) group_left(node_type) (\n"
      "    label_replace(\n"
      "      label_replace(node_dmi_info,\n"
      "        \"Hostname\", \"$1\", \"instance\", \"(.+)\"\n"
      "      ),\n"
      "      \"node_type\", \"$1\", \"product_name\", \"(.+)\"\n"
      "    )\n"
      "  )\n"
      "  or on (" + group_by + ")\n  " + idle_gpus + "\n)\n== 0";

  if (a.power_threshold_w) {
    // Corroborating power signal: drop candidates whose peak board power over
    // the window reached the threshold even though compute activity was 0.
    q += "\nunless on (" + pl + ", " + nl + ")\n(\n"
         "  max_over_time(DCGM_FI_DEV_POWER_USAGE" + selector(false) + win + ")"
         " >= " + fmt_num(*a.power_threshold_w) + "\n)";
  }
  return q;
}

}  // namespace pruner
