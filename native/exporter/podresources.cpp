#include "podresources.hpp"

#include <poll.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstring>

#include "../common/grpc_client.hpp"
#include "../common/log.hpp"

namespace exporter {

namespace {

constexpr const char* TARGET = "exporter::podresources";

// ---------------- protobuf wire decoding ----------------

class PbReader {
public:
  PbReader(const char* p, size_t n) : p_(p), end_(p + n) {}
  bool done() const { return p_ >= end_; }

  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (p_ < end_) {
      uint8_t b = static_cast<uint8_t>(*p_++);
      v |= static_cast<uint64_t>(b & 0x7F) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift > 63) break;
    }
    throw PodResourcesError("malformed protobuf varint");
  }

  // returns field number; sets wire type
  uint32_t tag(uint32_t* wire) {
    uint64_t t = varint();
    *wire = static_cast<uint32_t>(t & 0x7);
    return static_cast<uint32_t>(t >> 3);
  }

  std::string bytes() {
    uint64_t len = varint();
    if (p_ + len > end_) throw PodResourcesError("malformed protobuf length");
    std::string out(p_, len);
    p_ += len;
    return out;
  }

  void skip(uint32_t wire) {
    switch (wire) {
      case 0: varint(); break;
      case 1: advance(8); break;
      case 2: bytes(); break;
      case 5: advance(4); break;
      default: throw PodResourcesError("unsupported protobuf wire type");
    }
  }

private:
  void advance(size_t n) {
    if (p_ + n > end_) throw PodResourcesError("malformed protobuf");
    p_ += n;
  }
  const char* p_;
  const char* end_;
};

ContainerDevices decode_devices(const std::string& buf) {
  ContainerDevices out;
  PbReader r(buf.data(), buf.size());
  while (!r.done()) {
    uint32_t wire;
    uint32_t field = r.tag(&wire);
    if (field == 1 && wire == 2) out.resource_name = r.bytes();
    else if (field == 2 && wire == 2) out.device_ids.push_back(r.bytes());
    else r.skip(wire);
  }
  return out;
}

void decode_container(const std::string& buf, const std::string& pod, const std::string& ns,
                      std::vector<PodResourcesEntry>* out) {
  PodResourcesEntry e;
  e.pod = pod;
  e.ns = ns;
  PbReader r(buf.data(), buf.size());
  while (!r.done()) {
    uint32_t wire;
    uint32_t field = r.tag(&wire);
    if (field == 1 && wire == 2) e.container = r.bytes();
    else if (field == 2 && wire == 2) e.devices.push_back(decode_devices(r.bytes()));
    else r.skip(wire);
  }
  out->push_back(std::move(e));
}

void decode_pod(const std::string& buf, std::vector<PodResourcesEntry>* out) {
  std::string name, ns;
  std::vector<std::string> containers;
  PbReader r(buf.data(), buf.size());
  while (!r.done()) {
    uint32_t wire;
    uint32_t field = r.tag(&wire);
    if (field == 1 && wire == 2) name = r.bytes();
    else if (field == 2 && wire == 2) ns = r.bytes();
    else if (field == 3 && wire == 2) containers.push_back(r.bytes());
    else r.skip(wire);
  }
  for (const auto& c : containers) decode_container(c, name, ns, out);
}

}  // namespace

std::vector<PodResourcesEntry> decode_list_response(const std::string& payload) {
  std::vector<PodResourcesEntry> out;
  PbReader r(payload.data(), payload.size());
  while (!r.done()) {
    uint32_t wire;
    uint32_t field = r.tag(&wire);
    if (field == 1 && wire == 2) decode_pod(r.bytes(), &out);
    else r.skip(wire);
  }
  return out;
}

std::vector<PodResourcesEntry> list_pod_resources(const std::string& socket_path,
                                                  int timeout_ms) {
  // Unary v1.PodResourcesLister/List over the kubelet's unix socket, via the
  // shared h2c gRPC client (native/common/grpc_client.cpp). The request is
  // the empty ListPodResourcesRequest.
  grpcx::Target t;
  t.unix_path = socket_path;
  std::string payload;
  try {
    payload = grpcx::unary_call(t, "/v1.PodResourcesLister/List", "", timeout_ms);
  } catch (const grpcx::GrpcError& e) {
    throw PodResourcesError(e.what());
  }
  std::vector<PodResourcesEntry> entries = decode_list_response(payload);
  LOGD(TARGET, "PodResources List: " + std::to_string(entries.size()) + " container entries");
  return entries;
}

}  // namespace exporter
