#include "prom.hpp"

#include <array>
#include <cstdio>
#include <fstream>

#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "k8s.hpp"

namespace pruner {

namespace {

constexpr const char* TARGET = "pruner::prom";

std::optional<std::string> read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return std::nullopt;
  std::string data((std::istreambuf_iterator<char>(f)), std::istreambuf_iterator<char>());
  return data;
}

}  // namespace

std::string get_prometheus_token() {
  if (const char* t = std::getenv("PROMETHEUS_TOKEN"); t && *t) {
    LOGD(TARGET, "Using token from PROMETHEUS_TOKEN");
    return t;
  }

  // Kubernetes-config-derived token (reference infers via kube::Config):
  // service-account token file, then explicit token env used by tests.
  std::string sa_dir = "/var/run/secrets/kubernetes.io/serviceaccount";
  if (const char* dir = std::getenv("GPU_PRUNER_SA_DIR"); dir && *dir) sa_dir = dir;
  if (const char* tf = std::getenv("GPU_PRUNER_K8S_TOKEN_FILE"); tf && *tf) {
    if (auto t = read_file(tf)) return strutil::trim(*t);
  }
  if (auto t = read_file(sa_dir + "/token")) {
    LOGI(TARGET, "Inferred Prometheus token from service-account token file");
    return strutil::trim(*t);
  }
  if (const char* t = std::getenv("GPU_PRUNER_K8S_TOKEN"); t && *t) {
    LOGI(TARGET, "Using Kubernetes token for Prometheus");
    return t;
  }
  // token from the inferred kube config (kubeconfig user token / token file
  // — reference lib.rs:210-222 reads the same via kube::Config::infer)
  try {
    KubeConfig kc = KubeConfig::resolve();
    if (kc.token_file) {
      if (auto t = read_file(*kc.token_file)) {
        LOGI(TARGET, "Inferred Prometheus token from kube config token file");
        return strutil::trim(*t);
      }
    }
    if (kc.token) {
      LOGI(TARGET, "Found K8s token");
      return *kc.token;
    }
  } catch (const std::exception&) {
    // no kube config at all — fall through to oc
  }

  // Last resort: the logged-in OpenShift user's token.
  LOGI(TARGET, "No token provided, trying `oc whoami -t` as last resort");
  std::string out;
  if (FILE* p = ::popen("oc whoami -t 2>/dev/null", "r")) {
    std::array<char, 256> buf{};
    size_t r;
    while ((r = std::fread(buf.data(), 1, buf.size(), p)) > 0) out.append(buf.data(), r);
    ::pclose(p);
  }
  return strutil::trim(out);
}

PodMetricData parse_pod_metric(const jsn::Value& series) {
  const jsn::Value& m = series.get("metric");
  auto pick = [&](const char* exported, const char* native) -> std::string {
    const jsn::Value& e = m.get(exported);
    if (e.is_string()) return e.as_string();
    const jsn::Value& n = m.get(native);
    if (n.is_string()) return n.as_string();
    throw PodConvertError(std::string(exported) + "/" + native);
  };

  PodMetricData pmd;
  pmd.name = pick("exported_pod", "pod");
  pmd.ns = pick("exported_namespace", "namespace");
  pmd.container = pick("exported_container", "container");
  pmd.node_type = m.get("node_type").as_string_or("unknown");
  const jsn::Value& model = m.get("modelName");
  if (!model.is_string()) throw PodConvertError("modelName");
  pmd.gpu_model = model.as_string();

  // instant-vector sample: "value": [<ts>, "<float>"]
  const jsn::Value& v = series.get("value");
  if (v.is_array() && v.size() == 2) {
    const jsn::Value& sample = v[1];
    pmd.value = sample.is_string() ? std::strtod(sample.as_string().c_str(), nullptr)
                                   : sample.as_double();
  }
  return pmd;
}

PromClient::PromClient(const std::string& url, const std::string& token, TlsModeOpt tls_mode,
                       const std::optional<std::string>& ca_file) {
  auto parsed = http::Url::parse(url);
  if (!parsed) throw PromError("invalid Prometheus URL: " + url);
  prefix_ = parsed->path == "/" ? "" : parsed->path;
  if (!prefix_.empty() && prefix_.back() == '/') prefix_.pop_back();

  http::ClientOptions opts;
  if (tls_mode == TlsModeOpt::Skip) {
    opts.tls = http::TlsVerify::Skip;
  } else if (ca_file) {
    opts.tls = http::TlsVerify::CustomCa;
    opts.ca_file = *ca_file;
  }
  http_ = std::make_unique<http::Client>(*parsed, opts);
  if (!token.empty()) http_->set_default_header("Authorization", "Bearer " + token);
}

jsn::Value PromClient::query(const std::string& promql) {
  // POST form-encoded — avoids URL-length limits on the composite idle query.
  http::Request r;
  r.method = "POST";
  r.path = prefix_ + "/api/v1/query";
  r.body = "query=" + strutil::url_encode(promql);
  r.headers.emplace_back("Content-Type", "application/x-www-form-urlencoded");
  http::Response resp = http_->request(r);
  if (resp.status < 200 || resp.status >= 300)
    throw PromError("Prometheus query failed: HTTP " + std::to_string(resp.status) + ": " +
                    resp.body.substr(0, 300));
  jsn::Value body = jsn::parse(resp.body);
  if (body.get("status").as_string() != "success")
    throw PromError("Prometheus query error: " + body.get("error").as_string_or(resp.body));
  return body.get("data");
}

jsn::Value PromClient::query_vector(const std::string& promql) {
  jsn::Value data = query(promql);
  if (data.get("resultType").as_string() != "vector")
    throw PromError("expected vector response from prometheus, got " +
                    data.get("resultType").as_string());
  return data.get("result");
}

std::unique_ptr<PromClient> build_prom_client(const Config& cfg) {
  std::string token =
      cfg.prometheus_token ? *cfg.prometheus_token : get_prometheus_token();
  return std::make_unique<PromClient>(cfg.prometheus_url, token, cfg.prometheus_tls_mode,
                                      cfg.prometheus_tls_cert);
}

}  // namespace pruner
