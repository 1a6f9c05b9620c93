// resources.hpp — scale-target model of the MI355X-native gpu-pruner.
//
// Re-implements the reference's scaling model (SURVEY.md §2.1 "Scale-target
// model" / "Resource-kind flags" / "Meta trait" / "Event generator";
// reference gpu-pruner/src/lib.rs:36-202,287-427) as a C++ tagged struct over
// dynamic JSON Kubernetes objects. The two CRDs (Kubeflow Notebook, KServe
// InferenceService) are deliberately NOT given generated typed bindings — the
// reference carries 31k generated lines of which only metadata and two patch
// paths are used.
#pragma once

#include <cstdint>
#include <optional>
#include <string>
#include <unordered_set>
#include <vector>

#include "../common/json.hpp"

namespace pruner {

// ---- resource-kind flags ----------------------------------------------------

enum ResourceKind : uint8_t {
  RK_NONE = 0,
  RK_DEPLOYMENT = 1u << 0,
  RK_REPLICA_SET = 1u << 1,
  RK_STATEFUL_SET = 1u << 2,
  RK_INFERENCE_SERVICE = 1u << 3,
  RK_NOTEBOOK = 1u << 4,
  RK_ALL = 0b11111,
};

// Parse the "drsin" flag string: d=Deployment r=ReplicaSet s=StatefulSet
// i=InferenceService n=Notebook; unknown characters are silently ignored
// (matches reference lib.rs:116-129).
uint8_t get_enabled_resources(const std::string& s);

// ---- scale-target variant ---------------------------------------------------

enum class Kind : uint8_t {
  Deployment,
  ReplicaSet,
  StatefulSet,
  InferenceService,
  Notebook,
};

const char* kind_name(Kind k);           // "Deployment" ...
const char* kind_api_version(Kind k);    // "apps/v1", "v1" (Notebook), "v1beta1" (IS)
uint8_t kind_flag(Kind k);               // ResourceKind bit for the variant

// A scalable root object: the kind tag plus the full (dynamic JSON) object as
// fetched from the apiserver. Equality follows the reference: built-in kinds
// compare by full object equality, CRDs by uid; hashing is kind + uid so a
// hash set dedups pods sharing a parent (reference lib.rs:45-82).
struct ScaleKind {
  Kind kind;
  jsn::Value object;

  std::string name() const;
  std::optional<std::string> ns() const;
  std::optional<std::string> uid() const;
  std::optional<std::string> resource_version() const;
  std::string api_version() const { return kind_api_version(kind); }
  std::string kind_str() const { return kind_name(kind); }

  bool operator==(const ScaleKind& o) const;
  size_t hash() const;
};

struct ScaleKindHash {
  size_t operator()(const ScaleKind& sk) const { return sk.hash(); }
};
using ScaleKindSet = std::unordered_set<ScaleKind, ScaleKindHash>;

// Build the K8s Event object announcing the scale-down — name
// "gpuscaler-<uuid4simple>", action "scale_down", type "Normal", reason
// "Pod <ns>::<name> was not using GPU", reporting_component "gpu-pruner",
// reporting_instance from $POD_NAME (reference lib.rs:388-427).
jsn::Value generate_scale_event(const ScaleKind& sk);

// REST path for the object collection of `kind` in `ns`
// (e.g. /apis/apps/v1/namespaces/ns/deployments).
std::string collection_path(Kind kind, const std::string& ns);
// REST path of a single named object.
std::string object_path(Kind kind, const std::string& ns, const std::string& name);

}  // namespace pruner
