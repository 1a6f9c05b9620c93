#include "k8s.hpp"

#include <fstream>

#include "../common/log.hpp"
#include "../common/strutil.hpp"

namespace pruner {

namespace {

std::optional<std::string> read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return std::nullopt;
  std::string data((std::istreambuf_iterator<char>(f)), std::istreambuf_iterator<char>());
  return data;
}

const char* env(const char* name) {
  const char* v = std::getenv(name);
  return v && *v ? v : nullptr;
}

}  // namespace

KubeConfig KubeConfig::resolve() {
  KubeConfig cfg;

  if (const char* url = env("GPU_PRUNER_K8S_URL")) {
    cfg.url = url;
    if (const char* t = env("GPU_PRUNER_K8S_TOKEN")) cfg.token = t;
    if (const char* tf = env("GPU_PRUNER_K8S_TOKEN_FILE")) cfg.token_file = tf;
    if (const char* ca = env("GPU_PRUNER_K8S_CA")) cfg.ca_file = ca;
    if (const char* cc = env("GPU_PRUNER_K8S_CLIENT_CERT")) cfg.client_cert_file = cc;
    if (const char* ck = env("GPU_PRUNER_K8S_CLIENT_KEY")) cfg.client_key_file = ck;
    if (const char* skip = env("GPU_PRUNER_K8S_SKIP_TLS"))
      cfg.skip_tls = std::string(skip) != "0" && strutil::lower(skip) != "false";
    if (const char* ns = env("GPU_PRUNER_K8S_NAMESPACE")) cfg.default_namespace = ns;
    return cfg;
  }

  std::string sa_dir = "/var/run/secrets/kubernetes.io/serviceaccount";
  if (const char* dir = env("GPU_PRUNER_SA_DIR")) sa_dir = dir;

  const char* host = env("KUBERNETES_SERVICE_HOST");
  const char* port = env("KUBERNETES_SERVICE_PORT");
  if (host) {
    std::string h = host;
    if (h.find(':') != std::string::npos) h = "[" + h + "]";  // IPv6
    cfg.url = "https://" + h + ":" + (port ? port : "443");
    cfg.token_file = sa_dir + "/token";
    if (read_file(sa_dir + "/ca.crt")) cfg.ca_file = sa_dir + "/ca.crt";
    if (auto ns = read_file(sa_dir + "/namespace")) cfg.default_namespace = strutil::trim(*ns);
    return cfg;
  }

  throw std::runtime_error(
      "no Kubernetes config: set GPU_PRUNER_K8S_URL or run in-cluster "
      "(KUBERNETES_SERVICE_HOST)");
}

KubeClient::KubeClient(KubeConfig cfg) : cfg_(std::move(cfg)) {
  auto url = http::Url::parse(cfg_.url);
  if (!url) throw std::runtime_error("invalid Kubernetes API URL: " + cfg_.url);
  http::ClientOptions opts;
  if (cfg_.skip_tls) {
    opts.tls = http::TlsVerify::Skip;
  } else if (cfg_.ca_file) {
    opts.tls = http::TlsVerify::CustomCa;
    opts.ca_file = *cfg_.ca_file;
  }
  if (cfg_.client_cert_file && cfg_.client_key_file) {
    opts.client_cert_file = *cfg_.client_cert_file;
    opts.client_key_file = *cfg_.client_key_file;
  }
  http_ = std::make_unique<http::Client>(*url, opts);
}

std::string KubeClient::bearer() const {
  if (cfg_.token) return *cfg_.token;
  if (cfg_.token_file) {
    if (auto t = read_file(*cfg_.token_file)) return strutil::trim(*t);
  }
  return "";
}

http::Response KubeClient::authed(const http::Request& req) {
  http::Request r = req;
  std::string token = bearer();
  if (!token.empty()) r.headers.emplace_back("Authorization", "Bearer " + token);
  return http_->request(r);
}

std::optional<jsn::Value> KubeClient::get_opt(const std::string& path) {
  http::Request r;
  r.method = "GET";
  r.path = path;
  http::Response resp = authed(r);
  if (resp.status == 404) return std::nullopt;
  if (resp.status < 200 || resp.status >= 300)
    throw KubeError(resp.status, "GET " + path + " -> " + std::to_string(resp.status) + ": " +
                                     resp.body.substr(0, 300));
  return jsn::parse(resp.body);
}

jsn::Value KubeClient::get(const std::string& path) {
  auto v = get_opt(path);
  if (!v) throw KubeError(404, "GET " + path + " -> 404");
  return *v;
}

std::optional<jsn::Value> KubeClient::get_pod(const std::string& ns, const std::string& name) {
  return get_opt("/api/v1/namespaces/" + strutil::url_encode(ns) + "/pods/" +
                 strutil::url_encode(name));
}

std::optional<jsn::Value> KubeClient::get_object(Kind kind, const std::string& ns,
                                                 const std::string& name) {
  return get_opt(object_path(kind, ns, name));
}

jsn::Value KubeClient::merge_patch(const std::string& path, const jsn::Value& patch) {
  http::Request r;
  r.method = "PATCH";
  r.path = path;
  r.body = patch.dump();
  r.headers.emplace_back("Content-Type", "application/merge-patch+json");
  http::Response resp = authed(r);
  if (resp.status < 200 || resp.status >= 300)
    throw KubeError(resp.status, "PATCH " + path + " -> " + std::to_string(resp.status) + ": " +
                                     resp.body.substr(0, 300));
  return jsn::parse(resp.body);
}

jsn::Value KubeClient::patch_scale(Kind kind, const std::string& ns, const std::string& name,
                                   const jsn::Value& patch) {
  return merge_patch(object_path(kind, ns, name) + "/scale", patch);
}

jsn::Value KubeClient::create(const std::string& collection, const jsn::Value& obj) {
  http::Request r;
  r.method = "POST";
  r.path = collection;
  r.body = obj.dump();
  r.headers.emplace_back("Content-Type", "application/json");
  http::Response resp = authed(r);
  if (resp.status < 200 || resp.status >= 300)
    throw KubeError(resp.status, "POST " + collection + " -> " + std::to_string(resp.status) +
                                     ": " + resp.body.substr(0, 300));
  return jsn::parse(resp.body);
}

}  // namespace pruner
