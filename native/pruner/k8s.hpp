// k8s.hpp — Kubernetes API client for the MI355X-native gpu-pruner.
//
// From-scratch C++ equivalent of the reference's kube-rs client surface
// (SURVEY.md §2.1 "K8s client layer"; reference uses kube 3.x —
// Cargo.toml:23): in-cluster config resolution, typed GETs for the five
// scalable kinds + Pods, RFC 7386 merge-PATCH, the /scale subresource PATCH,
// and Event POST. CRDs are dynamic JSON (SURVEY.md §2.2 note).
//
// Config resolution order:
//   1. env override — GPU_PRUNER_K8S_URL (+ optional GPU_PRUNER_K8S_TOKEN /
//      GPU_PRUNER_K8S_TOKEN_FILE / GPU_PRUNER_K8S_CA / GPU_PRUNER_K8S_SKIP_TLS)
//      — used by tests and out-of-cluster runs;
//   2. in-cluster — KUBERNETES_SERVICE_HOST/PORT + the service-account dir
//      (token, ca.crt, namespace), overridable via GPU_PRUNER_SA_DIR.
#pragma once

#include <memory>
#include <optional>
#include <string>
#include <vector>

#include "../common/http.hpp"
#include "../common/json.hpp"
#include "resources.hpp"

namespace pruner {

// kubeconfig `user.exec` credential plugin (client.authentication.k8s.io):
// managed clusters (EKS aws-iam-authenticator, GKE gke-gcloud-auth-plugin,
// OpenShift oc) mint short-lived tokens through an external command whose
// stdout is an ExecCredential JSON. Tokens are cached until their
// expirationTimestamp.
struct ExecConfig {
  std::string command;
  std::vector<std::string> args;
  std::vector<std::pair<std::string, std::string>> env;
};

struct KubeConfig {
  std::string url;                 // https://host:port
  std::optional<std::string> token;
  std::optional<std::string> token_file;  // re-read per client build (token rotation)
  std::optional<std::string> ca_file;
  // mTLS client-certificate auth (kube client-cert users); both required
  std::optional<std::string> client_cert_file;
  std::optional<std::string> client_key_file;
  // Decoded kubeconfig base64 `-data` material, held IN MEMORY (never written
  // to /tmp: the daemon re-resolves config every tick and leaked key files
  // would accumulate unboundedly — ADVICE r1).
  std::optional<std::string> ca_data;
  std::optional<std::string> client_cert_data;
  std::optional<std::string> client_key_data;
  std::optional<ExecConfig> exec;  // credential plugin (run lazily, cached)
  bool skip_tls = false;
  std::string default_namespace = "default";

  // Resolve from the environment; throws std::runtime_error when neither an
  // override nor in-cluster config is present.
  static KubeConfig resolve();
};

// Drop cached exec-plugin tokens (tests exercise expiry/refresh).
void exec_cred_cache_clear_for_test();

class KubeError : public std::runtime_error {
public:
  KubeError(int http_status, const std::string& msg)
      : std::runtime_error(msg), status(http_status) {}
  int status;  // HTTP status; 0 for transport errors
};

class KubeClient {
public:
  explicit KubeClient(KubeConfig cfg);

  // GET that maps 404 → nullopt (the reference's Api::get_opt).
  std::optional<jsn::Value> get_opt(const std::string& path);
  // GET that throws KubeError on any non-2xx.
  jsn::Value get(const std::string& path);

  std::optional<jsn::Value> get_pod(const std::string& ns, const std::string& name);
  std::optional<jsn::Value> get_object(Kind kind, const std::string& ns,
                                       const std::string& name);

  // RFC 7386 merge patch on the object itself. The response body is not
  // parsed (every caller discards it; at 1000-pod scale the actuation path
  // issues ~1000 of these per tick and the echoed-object parse was pure
  // overhead).
  void merge_patch(const std::string& path, const jsn::Value& patch);
  // Merge patch on the /scale subresource (spec.replicas).
  void patch_scale(Kind kind, const std::string& ns, const std::string& name,
                   const jsn::Value& patch);

  void create(const std::string& collection_path, const jsn::Value& obj);

  // PUT replace with optimistic concurrency: sending metadata.resourceVersion
  // makes the apiserver reject stale writes with 409 Conflict (KubeError
  // status 409) — the primitive Lease-based leader election rests on.
  // Returns the stored object (carrying the new resourceVersion).
  jsn::Value replace(const std::string& path, const jsn::Value& obj);

  // Streaming GET (Kubernetes watch): returns once status+headers arrive.
  std::unique_ptr<http::BodyStream> open_stream(const std::string& path);

  const KubeConfig& config() const { return cfg_; }

private:
  std::string bearer() const;
  http::Response authed(const http::Request& req);

  KubeConfig cfg_;
  std::unique_ptr<http::Client> http_;
};

}  // namespace pruner
