// prom.hpp — Prometheus HTTP API client + series parsing.
//
// C++ equivalent of the reference's prometheus-http-query + reqwest stack
// (SURVEY.md §2.1 "Query executor" / "Prom auth/TLS" / "Metric-series
// parser"; reference lib.rs:205-282, lib.rs:136-187, main.rs:377-410).
//
// Token resolution chain (reference lib.rs:205-230):
//   $PROMETHEUS_TOKEN → kube service-account token file → kube token env →
//   `oc whoami -t` subprocess.
#pragma once

#include <memory>
#include <optional>
#include <string>
#include <vector>

#include "../common/http.hpp"
#include "../common/json.hpp"
#include "config.hpp"

namespace pruner {

std::string get_prometheus_token();

// One series of an instant-vector result, parsed into the fields the decision
// engine needs. Labels honor the exported_* → native fallback.
struct PodMetricData {
  std::string name;       // exported_pod | pod
  std::string ns;         // exported_namespace | namespace
  std::string container;  // exported_container | container
  std::string node_type;  // default "unknown"
  std::string gpu_model;  // modelName (required)
  double value = 0.0;
};

class PodConvertError : public std::runtime_error {
public:
  explicit PodConvertError(const std::string& key)
      : std::runtime_error("the data for key `" + key + "` is not available") {}
};

// Parse one element of data.result (an instant-vector sample). Throws
// PodConvertError when a required label is missing.
PodMetricData parse_pod_metric(const jsn::Value& series);

class PromError : public std::runtime_error {
public:
  using std::runtime_error::runtime_error;
};

class PromClient {
public:
  // url: Prometheus base URL (http(s)://host:port[/prefix]).
  PromClient(const std::string& url, const std::string& token, TlsModeOpt tls_mode,
             const std::optional<std::string>& ca_file);

  // Run an instant query; returns the `data` object of the API response
  // ({"resultType": "vector", "result": [...]}) or throws PromError.
  jsn::Value query(const std::string& promql);

  // Convenience: instant query that must be a vector; returns data.result.
  jsn::Value query_vector(const std::string& promql);

private:
  std::unique_ptr<http::Client> http_;
  std::string prefix_;  // path prefix from the base URL, e.g. "" or "/prom"
};

// Build a client with the configured URL/TLS and resolved token
// (reference main.rs:377-388 — rebuilt every tick so tokens stay fresh).
std::unique_ptr<PromClient> build_prom_client(const Config& cfg);

}  // namespace pruner
