#include "podresources.hpp"

#include <poll.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstring>

#include "../common/log.hpp"

namespace exporter {

namespace {

constexpr const char* TARGET = "exporter::podresources";

// ---------------- raw socket with deadline ----------------

class Sock {
public:
  Sock(const std::string& path, int timeout_ms)
      : deadline_(std::chrono::steady_clock::now() + std::chrono::milliseconds(timeout_ms)) {
    fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd_ < 0) throw PodResourcesError("socket(AF_UNIX) failed");
    struct sockaddr_un addr {};
    addr.sun_family = AF_UNIX;
    if (path.size() >= sizeof(addr.sun_path)) {
      ::close(fd_);
      throw PodResourcesError("socket path too long: " + path);
    }
    std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd_, reinterpret_cast<struct sockaddr*>(&addr), sizeof addr) < 0) {
      ::close(fd_);
      throw PodResourcesError("connect to " + path + " failed: " + std::strerror(errno));
    }
  }
  ~Sock() {
    if (fd_ >= 0) ::close(fd_);
  }

  void write_all(const void* data, size_t n) {
    const char* p = static_cast<const char*>(data);
    size_t off = 0;
    while (off < n) {
      ssize_t w = ::send(fd_, p + off, n - off, MSG_NOSIGNAL);
      if (w < 0) {
        if (errno == EINTR) continue;
        if (errno == EAGAIN || errno == EWOULDBLOCK) {
          wait_io(false);
          continue;
        }
        throw PodResourcesError(std::string("write failed: ") + std::strerror(errno));
      }
      off += static_cast<size_t>(w);
    }
  }

  // Exactly n bytes or throw.
  void read_exact(void* buf, size_t n) {
    char* p = static_cast<char*>(buf);
    size_t off = 0;
    while (off < n) {
      ssize_t r = ::recv(fd_, p + off, n - off, MSG_DONTWAIT);
      if (r > 0) {
        off += static_cast<size_t>(r);
        continue;
      }
      if (r == 0) throw PodResourcesError("connection closed mid-frame");
      if (errno == EINTR) continue;
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        wait_io(true);
        continue;
      }
      throw PodResourcesError(std::string("read failed: ") + std::strerror(errno));
    }
  }

private:
  void wait_io(bool want_read) {
    auto now = std::chrono::steady_clock::now();
    if (now >= deadline_) throw PodResourcesError("PodResources call timed out");
    int ms = static_cast<int>(
        std::chrono::duration_cast<std::chrono::milliseconds>(deadline_ - now).count());
    struct pollfd pfd {fd_, static_cast<short>(want_read ? POLLIN : POLLOUT), 0};
    int rc = ::poll(&pfd, 1, std::max(ms, 1));
    if (rc == 0) throw PodResourcesError("PodResources call timed out");
    if (rc < 0 && errno != EINTR)
      throw PodResourcesError(std::string("poll failed: ") + std::strerror(errno));
  }

  int fd_ = -1;
  std::chrono::steady_clock::time_point deadline_;
};

// ---------------- HTTP/2 framing ----------------

enum FrameType : uint8_t {
  F_DATA = 0x0,
  F_HEADERS = 0x1,
  F_RST_STREAM = 0x3,
  F_SETTINGS = 0x4,
  F_PING = 0x6,
  F_GOAWAY = 0x7,
  F_WINDOW_UPDATE = 0x8,
  F_CONTINUATION = 0x9,
};

constexpr uint8_t FLAG_END_STREAM = 0x1;
constexpr uint8_t FLAG_ACK = 0x1;

void put_frame_header(std::string& out, size_t len, uint8_t type, uint8_t flags,
                      uint32_t stream) {
  out += static_cast<char>((len >> 16) & 0xFF);
  out += static_cast<char>((len >> 8) & 0xFF);
  out += static_cast<char>(len & 0xFF);
  out += static_cast<char>(type);
  out += static_cast<char>(flags);
  out += static_cast<char>((stream >> 24) & 0x7F);
  out += static_cast<char>((stream >> 16) & 0xFF);
  out += static_cast<char>((stream >> 8) & 0xFF);
  out += static_cast<char>(stream & 0xFF);
}

// HPACK emitters (request side only): static-table indexed fields and
// literal-without-indexing fields — no dynamic table, no Huffman.
void hpack_indexed(std::string& out, uint8_t index) {
  out += static_cast<char>(0x80 | index);
}
void hpack_str(std::string& out, const std::string& s) {
  // 7-bit length prefix, no Huffman; all our strings are < 127 bytes
  out += static_cast<char>(s.size() & 0x7F);
  out += s;
}
void hpack_literal_indexed_name(std::string& out, uint8_t name_index,
                                const std::string& value) {
  out += static_cast<char>(name_index & 0x0F);  // 0000xxxx: without indexing
  hpack_str(out, value);
}
void hpack_literal_new_name(std::string& out, const std::string& name,
                            const std::string& value) {
  out += static_cast<char>(0x00);
  hpack_str(out, name);
  hpack_str(out, value);
}

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
  Sock sock(socket_path, timeout_ms);

  // ---- connection preface + SETTINGS + generous connection window ----
  std::string out("PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n");
  // SETTINGS: INITIAL_WINDOW_SIZE (0x4) = 16 MiB so large responses flow
  // without per-stream WINDOW_UPDATE bookkeeping
  put_frame_header(out, 6, F_SETTINGS, 0, 0);
  out += static_cast<char>(0x00);
  out += static_cast<char>(0x04);
  uint32_t win = 1u << 24;
  out += static_cast<char>((win >> 24) & 0xFF);
  out += static_cast<char>((win >> 16) & 0xFF);
  out += static_cast<char>((win >> 8) & 0xFF);
  out += static_cast<char>(win & 0xFF);
  // connection-level WINDOW_UPDATE: +16 MiB
  put_frame_header(out, 4, F_WINDOW_UPDATE, 0, 0);
  out += static_cast<char>((win >> 24) & 0x7F);
  out += static_cast<char>((win >> 16) & 0xFF);
  out += static_cast<char>((win >> 8) & 0xFF);
  out += static_cast<char>(win & 0xFF);

  // ---- HEADERS (stream 1) ----
  std::string hdrs;
  hpack_indexed(hdrs, 3);                               // :method: POST
  hpack_indexed(hdrs, 6);                               // :scheme: http
  hpack_literal_indexed_name(hdrs, 4, "/v1.PodResourcesLister/List");  // :path
  hpack_literal_indexed_name(hdrs, 1, "localhost");     // :authority
  hpack_literal_new_name(hdrs, "content-type", "application/grpc");
  hpack_literal_new_name(hdrs, "te", "trailers");
  put_frame_header(out, hdrs.size(), F_HEADERS, 0x4 /*END_HEADERS*/, 1);
  out += hdrs;

  // ---- DATA: gRPC frame carrying the empty ListPodResourcesRequest ----
  const char grpc_empty[5] = {0, 0, 0, 0, 0};
  put_frame_header(out, 5, F_DATA, FLAG_END_STREAM, 1);
  out.append(grpc_empty, 5);

  sock.write_all(out.data(), out.size());

  // ---- read frames until END_STREAM on stream 1 ----
  std::string grpc_payload;
  bool stream_done = false;
  while (!stream_done) {
    uint8_t fh[9];
    sock.read_exact(fh, 9);
    size_t len = (static_cast<size_t>(fh[0]) << 16) | (static_cast<size_t>(fh[1]) << 8) | fh[2];
    uint8_t type = fh[3], flags = fh[4];
    uint32_t stream = (static_cast<uint32_t>(fh[5] & 0x7F) << 24) |
                      (static_cast<uint32_t>(fh[6]) << 16) |
                      (static_cast<uint32_t>(fh[7]) << 8) | fh[8];
    std::string payload(len, '\0');
    if (len) sock.read_exact(payload.data(), len);

    switch (type) {
      case F_SETTINGS:
        if (!(flags & FLAG_ACK)) {  // ack the server's settings
          std::string ack;
          put_frame_header(ack, 0, F_SETTINGS, FLAG_ACK, 0);
          sock.write_all(ack.data(), ack.size());
        }
        break;
      case F_PING:
        if (!(flags & FLAG_ACK)) {
          std::string pong;
          put_frame_header(pong, 8, F_PING, FLAG_ACK, 0);
          pong += payload;
          sock.write_all(pong.data(), pong.size());
        }
        break;
      case F_DATA:
        if (stream == 1) grpc_payload += payload;
        if (stream == 1 && (flags & FLAG_END_STREAM)) stream_done = true;
        break;
      case F_HEADERS:  // response headers / trailers — content not needed
        if (stream == 1 && (flags & FLAG_END_STREAM)) stream_done = true;
        break;
      case F_RST_STREAM:
        if (stream == 1) throw PodResourcesError("stream reset by kubelet");
        break;
      case F_GOAWAY:
        if (!stream_done && grpc_payload.empty())
          throw PodResourcesError("connection closed by kubelet (GOAWAY)");
        stream_done = true;
        break;
      default:
        break;  // WINDOW_UPDATE / CONTINUATION-free responses / unknown
    }
  }

  // ---- unwrap gRPC length-prefixed message(s) ----
  std::vector<PodResourcesEntry> entries;
  size_t pos = 0;
  while (pos + 5 <= grpc_payload.size()) {
    uint8_t compressed = static_cast<uint8_t>(grpc_payload[pos]);
    uint32_t mlen = (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 1])) << 24) |
                    (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 2])) << 16) |
                    (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 3])) << 8) |
                    static_cast<uint8_t>(grpc_payload[pos + 4]);
    if (compressed) throw PodResourcesError("compressed gRPC response unsupported");
    if (pos + 5 + mlen > grpc_payload.size())
      throw PodResourcesError("truncated gRPC message");
    auto part = decode_list_response(grpc_payload.substr(pos + 5, mlen));
    entries.insert(entries.end(), part.begin(), part.end());
    pos += 5 + mlen;
  }
  LOGD(TARGET, "PodResources List: " + std::to_string(entries.size()) + " container entries");
  return entries;
}

}  // namespace exporter
