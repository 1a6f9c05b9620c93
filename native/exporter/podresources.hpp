// podresources.hpp — kubelet PodResources API client (gRPC over unix socket).
//
// This is how dcgm-exporter learns which pod owns which GPU, and the
// "hard part" SURVEY.md §7 flags for the MI355X build: kubelet exposes
// `/var/lib/kubelet/pod-resources/kubelet.sock`, a gRPC service
// (v1.PodResourcesLister/List) reporting, per container, the device IDs the
// device plugin allocated. Unlike the KFD process registry (attrib.cpp),
// this covers *allocated-but-idle* GPUs — exactly the pods an idle culler
// must attribute even when no process has the device open.
//
// Implemented from scratch: a minimal HTTP/2-cleartext framing layer (unary
// call, static-table HPACK for the request, frame-level skip of response
// headers) plus a hand-rolled protobuf wire decoder for the response
// messages:
//
//   ListPodResourcesResponse { repeated PodResources pod_resources = 1; }
//   PodResources     { string name = 1; string namespace = 2;
//                      repeated ContainerResources containers = 3; }
//   ContainerResources { string name = 1; repeated ContainerDevices devices = 2; }
//   ContainerDevices { string resource_name = 1; repeated string device_ids = 2; }
//
// No grpc/protobuf library dependency.
#pragma once

#include <stdexcept>
#include <string>
#include <vector>

namespace exporter {

struct ContainerDevices {
  std::string resource_name;           // e.g. "amd.com/gpu"
  std::vector<std::string> device_ids;
};

struct PodResourcesEntry {
  std::string pod;
  std::string ns;
  std::string container;
  std::vector<ContainerDevices> devices;
};

class PodResourcesError : public std::runtime_error {
public:
  using std::runtime_error::runtime_error;
};

// One unary List call against the kubelet socket. Throws PodResourcesError
// on transport/protocol failure.
std::vector<PodResourcesEntry> list_pod_resources(const std::string& socket_path,
                                                  int timeout_ms = 5000);

// Exposed for unit tests: decode a serialized ListPodResourcesResponse.
std::vector<PodResourcesEntry> decode_list_response(const std::string& payload);

}  // namespace exporter
