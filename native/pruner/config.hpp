// config.hpp — CLI / configuration surface of the MI355X-native gpu-pruner.
//
// Flag names, shorts, and defaults match the reference daemon's clap-derived
// CLI (reference gpu-pruner/src/main.rs:46-134) so deployments are drop-in,
// plus two knobs the reference hard-codes: --max-concurrency (the hot loop's
// in-flight pod evaluations; the reference pins 10 at main.rs:530) and
// --queue-capacity (the producer→consumer channel bound; 100 at main.rs:284).
#pragma once

#include <optional>
#include <string>
#include <vector>

#include "objcache.hpp"
#include "promql.hpp"

namespace pruner {

enum class RunMode { DryRun, ScaleDown };
enum class LogFormatOpt { Default, Json, Pretty };
enum class TlsModeOpt { Verify, Skip };

struct Config {
  long duration_min = 30;          // -t, --duration
  bool daemon_mode = false;        // -d, --daemon-mode
  std::string enabled_resources = "drsin";  // -e, --enabled-resources
  unsigned long check_interval_s = 180;     // -c, --check-interval
  std::optional<std::string> namespace_;    // -n, --namespace
  long grace_period_s = 300;       // -g, --grace-period
  std::optional<std::string> model_name;    // -m, --model-name
  std::optional<double> power_threshold;    // --power-threshold
  bool honor_labels = false;       // --honor-labels
  RunMode run_mode = RunMode::DryRun;       // -r, --run-mode
  std::string prometheus_url;      // --prometheus-url (required)
  std::optional<std::string> prometheus_token;     // --prometheus-token
  TlsModeOpt prometheus_tls_mode = TlsModeOpt::Verify;  // --prometheus-tls-mode
  std::optional<std::string> prometheus_tls_cert;  // --prometheus-tls-cert
  LogFormatOpt log_format = LogFormatOpt::Default; // -l, --log-format

  // MI355X-native additions (the reference hard-codes these):
  int max_concurrency = 32;        // --max-concurrency
  int queue_capacity = 100;        // --queue-capacity
  int max_consecutive_failures = 5;  // --max-failures (abort after more than N)
  int metrics_port = 0;            // --metrics-port (0 = disabled)
  // --eval-strategy: how candidate pods/owners are fetched —
  //   get  = per-object GETs (reference-equivalent, 1-3 RTTs per pod)
  //   list = namespace-collection LISTs (O(namespaces) RTTs per tick)
  //   auto = LIST namespaces with >= 10 candidates, GETs elsewhere
  EvalStrategy eval_strategy = EvalStrategy::Auto;
  // --leader-elect: coordination.k8s.io/v1 Lease-based leader election so
  // the daemon can run multi-replica without double-culling (the reference
  // is single-replica by convention). Only the lease holder runs ticks.
  bool leader_elect = false;
  int leader_lease_duration_s = 15;  // --leader-elect-lease-duration
  int leader_renew_period_s = 5;     // --leader-elect-renew-period

  QueryArgs query_args() const {
    QueryArgs qa;
    qa.duration_min = duration_min;
    qa.namespace_re = namespace_;
    qa.model_name_re = model_name;
    qa.power_threshold_w = power_threshold;
    qa.honor_labels = honor_labels;
    return qa;
  }
};

struct CliResult {
  Config config;
  bool show_help = false;
  std::optional<std::string> error;
};

CliResult parse_cli(const std::vector<std::string>& argv);
std::string cli_help();

}  // namespace pruner
