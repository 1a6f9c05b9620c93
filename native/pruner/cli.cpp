#include <cstdlib>
#include <map>

#include "../common/strutil.hpp"
#include "config.hpp"

namespace pruner {

namespace {

const char* HELP = R"(gpu-pruner — MI355X-native idle-GPU culler

Prunes idle pods based on GPU utilization: queries Prometheus for a window of
per-pod GPU activity (published by mi355-exporter from gfx950 counters), walks
pod owner references to the scalable root (Deployment / ReplicaSet /
StatefulSet / Kubeflow Notebook / KServe InferenceService) and scales it to
zero, emitting a Kubernetes Event for each action.

USAGE: gpu-pruner [OPTIONS] --prometheus-url <PROMETHEUS_URL>

OPTIONS:
  -t, --duration <MIN>            minutes of no GPU activity required to prune [default: 30]
  -d, --daemon-mode               run indefinitely on --check-interval
  -e, --enabled-resources <STR>   letters d/r/s/i/n for Deployment, ReplicaSet,
                                  StatefulSet, InferenceService, Notebook [default: drsin]
  -c, --check-interval <SEC>      daemon-mode poll interval [default: 180]
  -n, --namespace <RE>            namespace regex filter (pushed into PromQL)
  -g, --grace-period <SEC>        metric-publication grace period [default: 300]
  -m, --model-name <RE>           GPU model regex filter, e.g. "AMD Instinct MI355X"
      --power-threshold <W>       exclude pods whose peak power over the window
                                  reached this many watts
      --honor-labels [BOOL]       scrape config uses honorLabels: true
                                  (native pod/namespace/container label names)
  -r, --run-mode <MODE>           scale-down | dry-run [default: dry-run]
      --prometheus-url <URL>      Prometheus base URL (required)
      --prometheus-token <TOK>    bearer token (default: $PROMETHEUS_TOKEN →
                                  service-account token → `oc whoami -t`)
      --prometheus-tls-mode <M>   verify | skip [default: verify]
      --prometheus-tls-cert <PEM> extra CA bundle for TLS verification
  -l, --log-format <FMT>          default | json | pretty [default: default]
      --max-concurrency <N>       in-flight pod evaluations [default: 32]
      --queue-capacity <N>        scale-event queue bound [default: 100]
      --max-failures <N>          abort after more than N consecutive query
                                  failures [default: 5]
      --metrics-port <PORT>       serve daemon self-metrics + /healthz on
                                  this port (0 = disabled) [default: 0]
      --eval-strategy <S>         candidate fetch strategy: watch (persistent
                                  informers, delta traffic), get (per-object
                                  GETs, reference-equivalent) | list
                                  (namespace LISTs) | auto [default: auto]
      --leader-elect              coordination.k8s.io Lease-based leader
                                  election: only the lease holder runs ticks
                                  (safe multi-replica deployments)
      --leader-elect-lease-duration <SEC>  [default: 15]
      --leader-elect-renew-period <SEC>    [default: 5]
  -h, --help                      print this help
)";

bool parse_bool(const std::string& v, bool* out) {
  std::string s = strutil::lower(v);
  if (s == "true" || s == "1" || s == "yes") { *out = true; return true; }
  if (s == "false" || s == "0" || s == "no") { *out = false; return true; }
  return false;
}

}  // namespace

std::string cli_help() { return HELP; }

CliResult parse_cli(const std::vector<std::string>& argv) {
  CliResult res;
  Config& c = res.config;

  // canonical long name for each alias
  static const std::map<std::string, std::string> alias = {
      {"-t", "duration"},          {"--duration", "duration"},
      {"-d", "daemon-mode"},       {"--daemon-mode", "daemon-mode"},
      {"-e", "enabled-resources"}, {"--enabled-resources", "enabled-resources"},
      {"-c", "check-interval"},    {"--check-interval", "check-interval"},
      {"-n", "namespace"},         {"--namespace", "namespace"},
      {"-g", "grace-period"},      {"--grace-period", "grace-period"},
      {"-m", "model-name"},        {"--model-name", "model-name"},
      {"--power-threshold", "power-threshold"},
      {"--honor-labels", "honor-labels"},
      {"-r", "run-mode"},          {"--run-mode", "run-mode"},
      {"--prometheus-url", "prometheus-url"},
      {"--prometheus-token", "prometheus-token"},
      {"--prometheus-tls-mode", "prometheus-tls-mode"},
      {"--prometheus-tls-cert", "prometheus-tls-cert"},
      {"-l", "log-format"},        {"--log-format", "log-format"},
      {"--max-concurrency", "max-concurrency"},
      {"--queue-capacity", "queue-capacity"},
      {"--max-failures", "max-failures"},
      {"--metrics-port", "metrics-port"},
      {"--eval-strategy", "eval-strategy"},
      {"--leader-elect", "leader-elect"},
      {"--leader-elect-lease-duration", "leader-elect-lease-duration"},
      {"--leader-elect-renew-period", "leader-elect-renew-period"},
      {"-h", "help"},              {"--help", "help"},
  };
  // flags that never take a value
  auto is_switch = [](const std::string& name) {
    return name == "daemon-mode" || name == "leader-elect" || name == "help";
  };
  // flags with an optional boolean value (clap bool with default_value)
  auto is_opt_bool = [](const std::string& name) { return name == "honor-labels"; };

  auto fail = [&](const std::string& msg) {
    res.error = msg;
    return res;
  };

  for (size_t i = 0; i < argv.size(); i++) {
    std::string tok = argv[i];
    std::string inline_val;
    bool has_inline = false;
    if (strutil::starts_with(tok, "--")) {
      size_t eq = tok.find('=');
      if (eq != std::string::npos) {
        inline_val = tok.substr(eq + 1);
        has_inline = true;
        tok = tok.substr(0, eq);
      }
    }
    auto it = alias.find(tok);
    if (it == alias.end()) return fail("unknown flag: " + tok);
    const std::string& name = it->second;

    auto take_value = [&](std::string* out) {
      if (has_inline) { *out = inline_val; return true; }
      if (i + 1 >= argv.size()) return false;
      *out = argv[++i];
      return true;
    };

    if (name == "help") { res.show_help = true; return res; }
    if (is_switch(name)) {
      if (name == "daemon-mode") c.daemon_mode = true;
      else if (name == "leader-elect") c.leader_elect = true;
      continue;
    }
    if (is_opt_bool(name)) {
      bool v = true;
      if (has_inline) {
        if (!parse_bool(inline_val, &v)) return fail("invalid bool for --" + name);
      } else if (i + 1 < argv.size()) {
        bool parsed;
        if (parse_bool(argv[i + 1], &parsed)) { v = parsed; i++; }
      }
      c.honor_labels = v;
      continue;
    }

    std::string val;
    if (!take_value(&val)) return fail("missing value for --" + name);

    try {
      if (name == "duration") c.duration_min = std::stol(val);
      else if (name == "enabled-resources") c.enabled_resources = val;
      else if (name == "leader-elect-lease-duration") c.leader_lease_duration_s = std::stoi(val);
      else if (name == "leader-elect-renew-period") c.leader_renew_period_s = std::stoi(val);
      else if (name == "check-interval") c.check_interval_s = std::stoul(val);
      else if (name == "namespace") c.namespace_ = val;
      else if (name == "grace-period") c.grace_period_s = std::stol(val);
      else if (name == "model-name") c.model_name = val;
      else if (name == "power-threshold") c.power_threshold = std::stod(val);
      else if (name == "run-mode") {
        std::string m = strutil::lower(val);
        if (m == "scale-down") c.run_mode = RunMode::ScaleDown;
        else if (m == "dry-run") c.run_mode = RunMode::DryRun;
        else return fail("invalid --run-mode (expected scale-down|dry-run): " + val);
      } else if (name == "prometheus-url") c.prometheus_url = val;
      else if (name == "prometheus-token") c.prometheus_token = val;
      else if (name == "prometheus-tls-mode") {
        std::string m = strutil::lower(val);
        if (m == "skip") c.prometheus_tls_mode = TlsModeOpt::Skip;
        else if (m == "verify") c.prometheus_tls_mode = TlsModeOpt::Verify;
        else return fail("invalid --prometheus-tls-mode (expected skip|verify): " + val);
      } else if (name == "prometheus-tls-cert") c.prometheus_tls_cert = val;
      else if (name == "log-format") {
        std::string m = strutil::lower(val);
        if (m == "json") c.log_format = LogFormatOpt::Json;
        else if (m == "pretty") c.log_format = LogFormatOpt::Pretty;
        else if (m == "default") c.log_format = LogFormatOpt::Default;
        else return fail("invalid --log-format (expected json|default|pretty): " + val);
      } else if (name == "max-concurrency") c.max_concurrency = std::stoi(val);
      else if (name == "queue-capacity") c.queue_capacity = std::stoi(val);
      else if (name == "max-failures") c.max_consecutive_failures = std::stoi(val);
      else if (name == "metrics-port") c.metrics_port = std::stoi(val);
      else if (name == "eval-strategy") {
        std::string m = strutil::lower(val);
        if (m == "get") c.eval_strategy = EvalStrategy::PerPodGet;
        else if (m == "list") c.eval_strategy = EvalStrategy::NamespaceList;
        else if (m == "auto") c.eval_strategy = EvalStrategy::Auto;
        else if (m == "watch") c.eval_strategy = EvalStrategy::Watch;
        else return fail("invalid --eval-strategy (expected get|list|auto|watch): " + val);
      }
    } catch (const std::exception&) {
      return fail("invalid value for --" + name + ": " + val);
    }
  }

  if (!res.show_help && c.prometheus_url.empty())
    return fail("--prometheus-url is required");
  if (c.max_concurrency < 1) c.max_concurrency = 1;
  if (c.queue_capacity < 1) c.queue_capacity = 1;
  return res;
}

}  // namespace pruner
