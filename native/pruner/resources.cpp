#include "resources.hpp"

#include <cstdlib>
#include <functional>

#include "../common/strutil.hpp"
#include "otlp.hpp"

namespace pruner {

uint8_t get_enabled_resources(const std::string& s) {
  uint8_t out = RK_NONE;
  for (char c : s) {
    switch (c) {
      case 'd': out |= RK_DEPLOYMENT; break;
      case 'r': out |= RK_REPLICA_SET; break;
      case 's': out |= RK_STATEFUL_SET; break;
      case 'i': out |= RK_INFERENCE_SERVICE; break;
      case 'n': out |= RK_NOTEBOOK; break;
      default: break;  // unknown characters silently ignored
    }
  }
  return out;
}

const char* kind_name(Kind k) {
  switch (k) {
    case Kind::Deployment: return "Deployment";
    case Kind::ReplicaSet: return "ReplicaSet";
    case Kind::StatefulSet: return "StatefulSet";
    case Kind::InferenceService: return "InferenceService";
    case Kind::Notebook: return "Notebook";
  }
  return "?";
}

const char* kind_api_version(Kind k) {
  switch (k) {
    case Kind::Deployment:
    case Kind::ReplicaSet:
    case Kind::StatefulSet:
      return "apps/v1";
    // The reference reports the bare CRD *version* for the two CRDs
    // (lib.rs:308-316: "v1" for Notebook, "v1beta1" for InferenceService) and
    // its unit tests pin that; we match for drop-in Event compatibility.
    case Kind::Notebook: return "v1";
    case Kind::InferenceService: return "v1beta1";
  }
  return "?";
}

uint8_t kind_flag(Kind k) {
  switch (k) {
    case Kind::Deployment: return RK_DEPLOYMENT;
    case Kind::ReplicaSet: return RK_REPLICA_SET;
    case Kind::StatefulSet: return RK_STATEFUL_SET;
    case Kind::InferenceService: return RK_INFERENCE_SERVICE;
    case Kind::Notebook: return RK_NOTEBOOK;
  }
  return 0;
}

static std::optional<std::string> meta_str(const jsn::Value& obj, const char* field) {
  const jsn::Value& v = obj.at({"metadata"}).get(field);
  if (!v.is_string()) return std::nullopt;
  return v.as_string();
}

std::string ScaleKind::name() const { return meta_str(object, "name").value_or(""); }
std::optional<std::string> ScaleKind::ns() const { return meta_str(object, "namespace"); }
std::optional<std::string> ScaleKind::uid() const { return meta_str(object, "uid"); }
std::optional<std::string> ScaleKind::resource_version() const {
  return meta_str(object, "resourceVersion");
}

bool ScaleKind::operator==(const ScaleKind& o) const {
  if (kind != o.kind) return false;
  switch (kind) {
    // CRDs compare by uid only (reference lib.rs:50-52)…
    case Kind::InferenceService:
    case Kind::Notebook:
      return uid() == o.uid();
    // …built-in kinds by full object equality (lib.rs:47-49).
    default:
      return object == o.object;
  }
}

size_t ScaleKind::hash() const {
  size_t h = std::hash<int>{}(static_cast<int>(kind));
  auto u = uid();
  size_t hu = u ? std::hash<std::string>{}(*u) : 0x9e3779b97f4a7c15ull;
  return h ^ (hu + 0x9e3779b97f4a7c15ull + (h << 6) + (h >> 2));
}

jsn::Value generate_scale_event(const ScaleKind& sk) {
  otlp::SpanGuard span("generate_scale_event");
  std::string now = strutil::rfc3339_now();
  std::string now_micro = strutil::rfc3339_micro_now();

  const char* pod_name_env = std::getenv("POD_NAME");
  std::string reporting_instance = pod_name_env && *pod_name_env ? pod_name_env : "gpu_pruner";

  auto ns = sk.ns();

  jsn::Value ev = jsn::Value::object();
  ev["apiVersion"] = "v1";
  ev["kind"] = "Event";
  ev["metadata"] = jsn::Value::object();
  ev["metadata"]["name"] = "gpuscaler-" + strutil::uuid4_simple();
  if (ns) ev["metadata"]["namespace"] = *ns;
  ev["firstTimestamp"] = now;
  ev["lastTimestamp"] = now;
  ev["eventTime"] = now_micro;
  ev["action"] = "scale_down";
  ev["type"] = "Normal";
  ev["reason"] = "Pod " + ns.value_or("") + "::" + sk.name() + " was not using GPU";
  ev["reportingComponent"] = "gpu-pruner";
  ev["reportingInstance"] = reporting_instance;

  jsn::Value io = jsn::Value::object();
  io["apiVersion"] = sk.api_version();
  io["kind"] = sk.kind_str();
  io["name"] = sk.name();
  if (ns) io["namespace"] = *ns;
  if (auto rv = sk.resource_version()) io["resourceVersion"] = *rv;
  if (auto uid = sk.uid()) io["uid"] = *uid;
  ev["involvedObject"] = io;
  return ev;
}

std::string collection_path(Kind kind, const std::string& ns) {
  std::string e_ns = strutil::url_encode(ns);
  switch (kind) {
    case Kind::Deployment: return "/apis/apps/v1/namespaces/" + e_ns + "/deployments";
    case Kind::ReplicaSet: return "/apis/apps/v1/namespaces/" + e_ns + "/replicasets";
    case Kind::StatefulSet: return "/apis/apps/v1/namespaces/" + e_ns + "/statefulsets";
    case Kind::Notebook: return "/apis/kubeflow.org/v1/namespaces/" + e_ns + "/notebooks";
    case Kind::InferenceService:
      return "/apis/serving.kserve.io/v1beta1/namespaces/" + e_ns + "/inferenceservices";
  }
  return "";
}

std::string object_path(Kind kind, const std::string& ns, const std::string& name) {
  return collection_path(kind, ns) + "/" + strutil::url_encode(name);
}

}  // namespace pruner
