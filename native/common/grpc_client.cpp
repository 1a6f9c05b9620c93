#include "grpc_client.hpp"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstring>

namespace grpcx {

namespace {

// ---------------- raw socket with deadline ----------------

class Sock {
public:
  Sock(const Target& t, int timeout_ms)
      : deadline_(std::chrono::steady_clock::now() + std::chrono::milliseconds(timeout_ms)) {
    if (!t.unix_path.empty()) {
      connect_unix(t.unix_path);
    } else {
      connect_tcp(t.host, t.port);
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
        throw GrpcError(std::string("write failed: ") + std::strerror(errno));
      }
      off += static_cast<size_t>(w);
    }
  }

  void read_exact(void* buf, size_t n) {
    char* p = static_cast<char*>(buf);
    size_t off = 0;
    while (off < n) {
      ssize_t r = ::recv(fd_, p + off, n - off, MSG_DONTWAIT);
      if (r > 0) {
        off += static_cast<size_t>(r);
        continue;
      }
      if (r == 0) throw GrpcError("connection closed mid-frame");
      if (errno == EINTR) continue;
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        wait_io(true);
        continue;
      }
      throw GrpcError(std::string("read failed: ") + std::strerror(errno));
    }
  }

private:
  void set_nonblock() { /* MSG_DONTWAIT covers reads; writes poll on EAGAIN */ }

  void connect_unix(const std::string& path) {
    fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd_ < 0) throw GrpcError("socket(AF_UNIX) failed");
    struct sockaddr_un addr {};
    addr.sun_family = AF_UNIX;
    if (path.size() >= sizeof(addr.sun_path)) {
      ::close(fd_);
      fd_ = -1;
      throw GrpcError("socket path too long: " + path);
    }
    std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd_, reinterpret_cast<struct sockaddr*>(&addr), sizeof addr) < 0) {
      int e = errno;
      ::close(fd_);
      fd_ = -1;
      throw GrpcError("connect to " + path + " failed: " + std::strerror(e));
    }
  }

  void connect_tcp(const std::string& host, uint16_t port) {
    struct addrinfo hints {};
    hints.ai_family = AF_UNSPEC;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    std::string port_s = std::to_string(port);
    int rc = ::getaddrinfo(host.c_str(), port_s.c_str(), &hints, &res);
    if (rc != 0) throw GrpcError("DNS resolution failed for " + host + ": " + gai_strerror(rc));
    std::string last_err = "no addresses";
    for (struct addrinfo* ai = res; ai; ai = ai->ai_next) {
      int fd = ::socket(ai->ai_family, ai->ai_socktype, ai->ai_protocol);
      if (fd < 0) continue;
      int one = 1;
      ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
      if (::connect(fd, ai->ai_addr, ai->ai_addrlen) == 0) {
        fd_ = fd;
        break;
      }
      last_err = std::strerror(errno);
      ::close(fd);
    }
    ::freeaddrinfo(res);
    if (fd_ < 0)
      throw GrpcError("connect to " + host + ":" + port_s + " failed: " + last_err);
  }

  void wait_io(bool want_read) {
    auto now = std::chrono::steady_clock::now();
    if (now >= deadline_) throw GrpcError("gRPC call timed out");
    int ms = static_cast<int>(
        std::chrono::duration_cast<std::chrono::milliseconds>(deadline_ - now).count());
    struct pollfd pfd {fd_, static_cast<short>(want_read ? POLLIN : POLLOUT), 0};
    int rc = ::poll(&pfd, 1, std::max(ms, 1));
    if (rc == 0) throw GrpcError("gRPC call timed out");
    if (rc < 0 && errno != EINTR)
      throw GrpcError(std::string("poll failed: ") + std::strerror(errno));
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
constexpr uint8_t FLAG_END_HEADERS = 0x4;
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
// literal-without-indexing fields — no dynamic table, no Huffman. String
// lengths use the full RFC 7541 §5.1 integer coding, so values of any
// length are legal (the round-1 client capped at 127 bytes).
void hpack_int(std::string& out, uint8_t first_byte_bits, uint8_t prefix_bits,
               size_t value) {
  const size_t max_prefix = (1u << prefix_bits) - 1;
  if (value < max_prefix) {
    out += static_cast<char>(first_byte_bits | value);
    return;
  }
  out += static_cast<char>(first_byte_bits | max_prefix);
  value -= max_prefix;
  while (value >= 128) {
    out += static_cast<char>((value & 0x7F) | 0x80);
    value >>= 7;
  }
  out += static_cast<char>(value);
}

void hpack_indexed(std::string& out, uint8_t index) {
  out += static_cast<char>(0x80 | index);
}

void hpack_str(std::string& out, const std::string& s) {
  hpack_int(out, 0x00, 7, s.size());  // H bit clear: raw, any length
  out += s;
}

void hpack_literal_indexed_name(std::string& out, uint8_t name_index,
                                const std::string& value) {
  hpack_int(out, 0x00, 4, name_index);  // 0000xxxx: literal w/o indexing
  hpack_str(out, value);
}

void hpack_literal_new_name(std::string& out, const std::string& name,
                            const std::string& value) {
  out += static_cast<char>(0x00);
  hpack_str(out, name);
  hpack_str(out, value);
}

}  // namespace

std::string unary_call(const Target& target, const std::string& method_path,
                       const std::string& request_msg, int timeout_ms) {
  if (request_msg.size() > kMaxRequestBytes)
    throw GrpcError("request of " + std::to_string(request_msg.size()) +
                    " bytes exceeds the flow-control-safe limit (" +
                    std::to_string(kMaxRequestBytes) + "); split into multiple calls");
  Sock sock(target, timeout_ms);

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
  hpack_indexed(hdrs, 3);  // :method: POST
  hpack_indexed(hdrs, 6);  // :scheme: http
  hpack_literal_indexed_name(hdrs, 4, method_path);     // :path
  hpack_literal_indexed_name(hdrs, 1, target.authority);  // :authority
  hpack_literal_new_name(hdrs, "content-type", "application/grpc");
  hpack_literal_new_name(hdrs, "te", "trailers");
  put_frame_header(out, hdrs.size(), F_HEADERS, FLAG_END_HEADERS, 1);
  out += hdrs;

  // ---- DATA: gRPC length-prefixed message (uncompressed) ----
  std::string grpc_frame;
  grpc_frame += static_cast<char>(0);  // no compression
  uint32_t mlen = static_cast<uint32_t>(request_msg.size());
  grpc_frame += static_cast<char>((mlen >> 24) & 0xFF);
  grpc_frame += static_cast<char>((mlen >> 16) & 0xFF);
  grpc_frame += static_cast<char>((mlen >> 8) & 0xFF);
  grpc_frame += static_cast<char>(mlen & 0xFF);
  grpc_frame += request_msg;
  // split into DATA frames if beyond the peer's default 16 KiB max frame size
  constexpr size_t kMaxFrame = 16384;
  for (size_t off = 0; off < grpc_frame.size() || off == 0; off += kMaxFrame) {
    size_t n = std::min(kMaxFrame, grpc_frame.size() - off);
    bool last = off + n >= grpc_frame.size();
    put_frame_header(out, n, F_DATA, last ? FLAG_END_STREAM : 0, 1);
    out.append(grpc_frame, off, n);
    if (last) break;
  }

  sock.write_all(out.data(), out.size());

  // ---- read frames until END_STREAM on stream 1 ----
  // CONTINUATION handling: a HEADERS frame may carry END_STREAM but not
  // END_HEADERS; the stream then ends only after the final CONTINUATION.
  std::string grpc_payload;
  bool stream_done = false;
  bool headers_pending_end_stream = false;
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
        if (!(flags & FLAG_ACK)) {
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
      case F_HEADERS:  // response headers / trailers — content not parsed
        if (stream == 1 && (flags & FLAG_END_STREAM)) {
          if (flags & FLAG_END_HEADERS) stream_done = true;
          else headers_pending_end_stream = true;
        }
        break;
      case F_CONTINUATION:
        if (stream == 1 && headers_pending_end_stream && (flags & FLAG_END_HEADERS))
          stream_done = true;
        break;
      case F_RST_STREAM:
        if (stream == 1) throw GrpcError("stream reset by server");
        break;
      case F_GOAWAY:
        if (!stream_done && grpc_payload.empty())
          throw GrpcError("connection closed by server (GOAWAY)");
        stream_done = true;
        break;
      default:
        break;  // WINDOW_UPDATE / unknown
    }
  }

  // ---- unwrap gRPC length-prefixed message(s) ----
  std::string msg;
  size_t pos = 0;
  while (pos + 5 <= grpc_payload.size()) {
    uint8_t compressed = static_cast<uint8_t>(grpc_payload[pos]);
    uint32_t rlen = (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 1])) << 24) |
                    (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 2])) << 16) |
                    (static_cast<uint32_t>(static_cast<uint8_t>(grpc_payload[pos + 3])) << 8) |
                    static_cast<uint8_t>(grpc_payload[pos + 4]);
    if (compressed) throw GrpcError("compressed gRPC response unsupported");
    if (pos + 5 + rlen > grpc_payload.size()) throw GrpcError("truncated gRPC message");
    msg.append(grpc_payload, pos + 5, rlen);
    pos += 5 + rlen;
  }
  return msg;
}

}  // namespace grpcx
