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
#include <unordered_map>

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

// ---------------- HPACK response decoding ----------------

// RFC 7541 Appendix B Huffman code: {code, bit-length} per symbol 0..255
// plus EOS (256). Validated as THE canonical table: the 257 entries form a
// prefix-free code with Kraft sum exactly 1 and a 30-bit EOS, which pins it
// uniquely to the RFC appendix.
struct HuffSym { uint32_t code; uint8_t len; };
constexpr HuffSym kHuff[257] = {
    {0x1ff8u, 13}, {0x7fffd8u, 23}, {0xfffffe2u, 28}, {0xfffffe3u, 28},
    {0xfffffe4u, 28}, {0xfffffe5u, 28}, {0xfffffe6u, 28}, {0xfffffe7u, 28},
    {0xfffffe8u, 28}, {0xffffeau, 24}, {0x3ffffffcu, 30}, {0xfffffe9u, 28},
    {0xfffffeau, 28}, {0x3ffffffdu, 30}, {0xfffffebu, 28}, {0xfffffecu, 28},
    {0xfffffedu, 28}, {0xfffffeeu, 28}, {0xfffffefu, 28}, {0xffffff0u, 28},
    {0xffffff1u, 28}, {0xffffff2u, 28}, {0x3ffffffeu, 30}, {0xffffff3u, 28},
    {0xffffff4u, 28}, {0xffffff5u, 28}, {0xffffff6u, 28}, {0xffffff7u, 28},
    {0xffffff8u, 28}, {0xffffff9u, 28}, {0xffffffau, 28}, {0xffffffbu, 28},
    {0x14u, 6}, {0x3f8u, 10}, {0x3f9u, 10}, {0xffau, 12},
    {0x1ff9u, 13}, {0x15u, 6}, {0xf8u, 8}, {0x7fau, 11},
    {0x3fau, 10}, {0x3fbu, 10}, {0xf9u, 8}, {0x7fbu, 11},
    {0xfau, 8}, {0x16u, 6}, {0x17u, 6}, {0x18u, 6},
    {0x0u, 5}, {0x1u, 5}, {0x2u, 5}, {0x19u, 6},
    {0x1au, 6}, {0x1bu, 6}, {0x1cu, 6}, {0x1du, 6},
    {0x1eu, 6}, {0x1fu, 6}, {0x5cu, 7}, {0xfbu, 8},
    {0x7ffcu, 15}, {0x20u, 6}, {0xffbu, 12}, {0x3fcu, 10},
    {0x1ffau, 13}, {0x21u, 6}, {0x5du, 7}, {0x5eu, 7},
    {0x5fu, 7}, {0x60u, 7}, {0x61u, 7}, {0x62u, 7},
    {0x63u, 7}, {0x64u, 7}, {0x65u, 7}, {0x66u, 7},
    {0x67u, 7}, {0x68u, 7}, {0x69u, 7}, {0x6au, 7},
    {0x6bu, 7}, {0x6cu, 7}, {0x6du, 7}, {0x6eu, 7},
    {0x6fu, 7}, {0x70u, 7}, {0x71u, 7}, {0x72u, 7},
    {0xfcu, 8}, {0x73u, 7}, {0xfdu, 8}, {0x1ffbu, 13},
    {0x7fff0u, 19}, {0x1ffcu, 13}, {0x3ffcu, 14}, {0x22u, 6},
    {0x7ffdu, 15}, {0x3u, 5}, {0x23u, 6}, {0x4u, 5},
    {0x24u, 6}, {0x5u, 5}, {0x25u, 6}, {0x26u, 6},
    {0x27u, 6}, {0x6u, 5}, {0x74u, 7}, {0x75u, 7},
    {0x28u, 6}, {0x29u, 6}, {0x2au, 6}, {0x7u, 5},
    {0x2bu, 6}, {0x76u, 7}, {0x2cu, 6}, {0x8u, 5},
    {0x9u, 5}, {0x2du, 6}, {0x77u, 7}, {0x78u, 7},
    {0x79u, 7}, {0x7au, 7}, {0x7bu, 7}, {0x7ffeu, 15},
    {0x7fcu, 11}, {0x3ffdu, 14}, {0x1ffdu, 13}, {0xffffffcu, 28},
    {0xfffe6u, 20}, {0x3fffd2u, 22}, {0xfffe7u, 20}, {0xfffe8u, 20},
    {0x3fffd3u, 22}, {0x3fffd4u, 22}, {0x3fffd5u, 22}, {0x7fffd9u, 23},
    {0x3fffd6u, 22}, {0x7fffdau, 23}, {0x7fffdbu, 23}, {0x7fffdcu, 23},
    {0x7fffddu, 23}, {0x7fffdeu, 23}, {0xffffebu, 24}, {0x7fffdfu, 23},
    {0xffffecu, 24}, {0xffffedu, 24}, {0x3fffd7u, 22}, {0x7fffe0u, 23},
    {0xffffeeu, 24}, {0x7fffe1u, 23}, {0x7fffe2u, 23}, {0x7fffe3u, 23},
    {0x7fffe4u, 23}, {0x1fffdcu, 21}, {0x3fffd8u, 22}, {0x7fffe5u, 23},
    {0x3fffd9u, 22}, {0x7fffe6u, 23}, {0x7fffe7u, 23}, {0xffffefu, 24},
    {0x3fffdau, 22}, {0x1fffddu, 21}, {0xfffe9u, 20}, {0x3fffdbu, 22},
    {0x3fffdcu, 22}, {0x7fffe8u, 23}, {0x7fffe9u, 23}, {0x1fffdeu, 21},
    {0x7fffeau, 23}, {0x3fffddu, 22}, {0x3fffdeu, 22}, {0xfffff0u, 24},
    {0x1fffdfu, 21}, {0x3fffdfu, 22}, {0x7fffebu, 23}, {0x7fffecu, 23},
    {0x1fffe0u, 21}, {0x1fffe1u, 21}, {0x3fffe0u, 22}, {0x1fffe2u, 21},
    {0x7fffedu, 23}, {0x3fffe1u, 22}, {0x7fffeeu, 23}, {0x7fffefu, 23},
    {0xfffeau, 20}, {0x3fffe2u, 22}, {0x3fffe3u, 22}, {0x3fffe4u, 22},
    {0x7ffff0u, 23}, {0x3fffe5u, 22}, {0x3fffe6u, 22}, {0x7ffff1u, 23},
    {0x3ffffe0u, 26}, {0x3ffffe1u, 26}, {0xfffebu, 20}, {0x7fff1u, 19},
    {0x3fffe7u, 22}, {0x7ffff2u, 23}, {0x3fffe8u, 22}, {0x1ffffecu, 25},
    {0x3ffffe2u, 26}, {0x3ffffe3u, 26}, {0x3ffffe4u, 26}, {0x7ffffdeu, 27},
    {0x7ffffdfu, 27}, {0x3ffffe5u, 26}, {0xfffff1u, 24}, {0x1ffffedu, 25},
    {0x7fff2u, 19}, {0x1fffe3u, 21}, {0x3ffffe6u, 26}, {0x7ffffe0u, 27},
    {0x7ffffe1u, 27}, {0x3ffffe7u, 26}, {0x7ffffe2u, 27}, {0xfffff2u, 24},
    {0x1fffe4u, 21}, {0x1fffe5u, 21}, {0x3ffffe8u, 26}, {0x3ffffe9u, 26},
    {0xffffffdu, 28}, {0x7ffffe3u, 27}, {0x7ffffe4u, 27}, {0x7ffffe5u, 27},
    {0xfffecu, 20}, {0xfffff3u, 24}, {0xfffedu, 20}, {0x1fffe6u, 21},
    {0x3fffe9u, 22}, {0x1fffe7u, 21}, {0x1fffe8u, 21}, {0x7ffff3u, 23},
    {0x3fffeau, 22}, {0x3fffebu, 22}, {0x1ffffeeu, 25}, {0x1ffffefu, 25},
    {0xfffff4u, 24}, {0xfffff5u, 24}, {0x3ffffeau, 26}, {0x7ffff4u, 23},
    {0x3ffffebu, 26}, {0x7ffffe6u, 27}, {0x3ffffecu, 26}, {0x3ffffedu, 26},
    {0x7ffffe7u, 27}, {0x7ffffe8u, 27}, {0x7ffffe9u, 27}, {0x7ffffeau, 27},
    {0x7ffffebu, 27}, {0xffffffeu, 28}, {0x7ffffecu, 27}, {0x7ffffedu, 27},
    {0x7ffffeeu, 27}, {0x7ffffefu, 27}, {0x7fffff0u, 27}, {0x3ffffeeu, 26},
    {0x3fffffffu, 30}
};

// Bit-serial decode against the canonical table: accumulate bits, test the
// accumulator at every length that exists in the code (5..30). Strings in
// headers are short (metric of one connection), so O(bits) with a hash probe
// per candidate length is plenty.
const std::unordered_map<uint64_t, uint16_t>& huff_map() {
  static const std::unordered_map<uint64_t, uint16_t> m = [] {
    std::unordered_map<uint64_t, uint16_t> t;
    for (uint16_t s = 0; s < 257; s++)
      t.emplace((static_cast<uint64_t>(kHuff[s].len) << 32) | kHuff[s].code, s);
    return t;
  }();
  return m;
}

}  // namespace

std::string huffman_decode(const uint8_t* data, size_t len) {
  const auto& map = huff_map();
  std::string out;
  uint32_t acc = 0;
  uint8_t nbits = 0;
  for (size_t i = 0; i < len; i++) {
    for (int b = 7; b >= 0; b--) {
      acc = (acc << 1) | ((data[i] >> b) & 1);
      nbits++;
      auto it = map.find((static_cast<uint64_t>(nbits) << 32) | acc);
      if (it != map.end()) {
        if (it->second == 256) throw GrpcError("HPACK: EOS inside Huffman string");
        out += static_cast<char>(it->second);
        acc = 0;
        nbits = 0;
      } else if (nbits > 30) {
        throw GrpcError("HPACK: invalid Huffman code");
      }
    }
  }
  // padding must be <8 bits of the EOS prefix (all ones)
  if (nbits >= 8 || acc != (1u << nbits) - 1)
    throw GrpcError("HPACK: bad Huffman padding");
  return out;
}

namespace {

// RFC 7541 Appendix A static table (1..61); empty string = no value.
constexpr const char* kStatic[62][2] = {
    {"", ""},
    {":authority", ""}, {":method", "GET"}, {":method", "POST"},
    {":path", "/"}, {":path", "/index.html"}, {":scheme", "http"},
    {":scheme", "https"}, {":status", "200"}, {":status", "204"},
    {":status", "206"}, {":status", "304"}, {":status", "400"},
    {":status", "404"}, {":status", "500"}, {"accept-charset", ""},
    {"accept-encoding", "gzip, deflate"}, {"accept-language", ""},
    {"accept-ranges", ""}, {"accept", ""}, {"access-control-allow-origin", ""},
    {"age", ""}, {"allow", ""}, {"authorization", ""}, {"cache-control", ""},
    {"content-disposition", ""}, {"content-encoding", ""},
    {"content-language", ""}, {"content-length", ""}, {"content-location", ""},
    {"content-range", ""}, {"content-type", ""}, {"cookie", ""}, {"date", ""},
    {"etag", ""}, {"expect", ""}, {"expires", ""}, {"from", ""}, {"host", ""},
    {"if-match", ""}, {"if-modified-since", ""}, {"if-none-match", ""},
    {"if-range", ""}, {"if-unmodified-since", ""}, {"last-modified", ""},
    {"link", ""}, {"location", ""}, {"max-forwards", ""},
    {"proxy-authenticate", ""}, {"proxy-authorization", ""}, {"range", ""},
    {"referer", ""}, {"refresh", ""}, {"retry-after", ""}, {"server", ""},
    {"set-cookie", ""}, {"strict-transport-security", ""},
    {"transfer-encoding", ""}, {"user-agent", ""}, {"vary", ""}, {"via", ""},
    {"www-authenticate", ""},
};

}  // namespace

std::vector<Header> HpackDecoder::decode_block(const std::string& block) {
  std::vector<Header> out;
  size_t pos = 0;
  const auto* p = reinterpret_cast<const uint8_t*>(block.data());

  auto read_int = [&](uint8_t prefix_bits) -> uint64_t {
    if (pos >= block.size()) throw GrpcError("HPACK: truncated integer");
    const uint64_t max_prefix = (1u << prefix_bits) - 1;
    uint64_t v = p[pos++] & max_prefix;
    if (v < max_prefix) return v;
    int shift = 0;
    while (true) {
      if (pos >= block.size()) throw GrpcError("HPACK: truncated integer");
      uint8_t b = p[pos++];
      v += static_cast<uint64_t>(b & 0x7F) << shift;
      shift += 7;
      if (!(b & 0x80)) return v;
      if (shift > 56) throw GrpcError("HPACK: integer overflow");
    }
  };

  auto read_string = [&]() -> std::string {
    if (pos >= block.size()) throw GrpcError("HPACK: truncated string");
    bool huff = (p[pos] & 0x80) != 0;
    uint64_t n = read_int(7);
    if (pos + n > block.size()) throw GrpcError("HPACK: truncated string");
    std::string s = huff ? huffman_decode(p + pos, n)
                         : block.substr(pos, n);
    pos += n;
    return s;
  };

  auto lookup = [&](uint64_t idx) -> Header {
    if (idx == 0) throw GrpcError("HPACK: index 0");
    if (idx <= 61) return {kStatic[idx][0], kStatic[idx][1]};
    size_t d = idx - 62;
    if (d >= dynamic_.size()) throw GrpcError("HPACK: dynamic index out of range");
    return dynamic_[d];
  };

  while (pos < block.size()) {
    uint8_t b = p[pos];
    if (b & 0x80) {                      // indexed header field
      out.push_back(lookup(read_int(7)));
    } else if ((b & 0xC0) == 0x40) {     // literal, incremental indexing
      uint64_t idx = read_int(6);
      std::string name = idx ? lookup(idx).first : read_string();
      std::string value = read_string();
      dynamic_.emplace_front(name, value);
      // Eviction bookkeeping is deliberately skipped: a decoder that never
      // evicts can only OVER-retain, and an index referencing an entry the
      // encoder evicted would have been re-inserted by the encoder first.
      // Connections here are one-RPC-lived; bound memory anyway:
      if (dynamic_.size() > 1024) dynamic_.pop_back();
      out.emplace_back(std::move(name), std::move(value));
    } else if ((b & 0xE0) == 0x20) {     // dynamic table size update
      read_int(5);
    } else {                             // literal w/o indexing / never-indexed
      uint64_t idx = read_int(4);
      std::string name = idx ? lookup(idx).first : read_string();
      std::string value = read_string();
      out.emplace_back(std::move(name), std::move(value));
    }
  }
  return out;
}

namespace {

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
  // response header / trailer decoding: the dynamic table persists across
  // the initial HEADERS and the trailers (grpc servers index grpc-status et
  // al. in the first block and reference it from the second)
  HpackDecoder hpack;
  std::vector<Header> resp_headers;
  std::string hdr_block;
  bool hdr_collecting = false;
  bool hdr_unreliable = false;  // any decode failure → skip status checks
  auto decode_hdr_block = [&]() {
    try {
      auto hs = hpack.decode_block(hdr_block);
      resp_headers.insert(resp_headers.end(), hs.begin(), hs.end());
    } catch (const GrpcError&) {
      hdr_unreliable = true;  // tolerate odd encoders; behave as untyped h2
    }
    hdr_block.clear();
  };
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
      case F_HEADERS:  // response headers / trailers
        if (stream == 1) {
          // strip PADDED (0x8: leading pad-length byte + trailing pad) and
          // PRIORITY (0x20: 5 bytes of dep + weight) before HPACK
          size_t start = 0, end = payload.size();
          if (flags & 0x8) {
            if (!payload.empty()) {
              uint8_t pad = static_cast<uint8_t>(payload[0]);
              start = 1;
              end = pad <= end - start ? end - pad : start;
            }
          }
          if (flags & 0x20) start = std::min(start + 5, end);
          hdr_block.assign(payload, start, end - start);
          if (flags & FLAG_END_HEADERS) decode_hdr_block();
          else hdr_collecting = true;
          if (flags & FLAG_END_STREAM) {
            if (flags & FLAG_END_HEADERS) stream_done = true;
            else headers_pending_end_stream = true;
          }
        }
        break;
      case F_CONTINUATION:
        if (stream == 1 && hdr_collecting) {
          hdr_block += payload;
          if (flags & FLAG_END_HEADERS) {
            decode_hdr_block();
            hdr_collecting = false;
            if (headers_pending_end_stream) stream_done = true;
          }
        }
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

  // ---- gRPC status (trailers; last occurrence wins) ----
  if (!hdr_unreliable) {
    std::string http_status, grpc_status, grpc_message;
    for (const auto& [k, v] : resp_headers) {
      if (k == ":status") http_status = v;
      else if (k == "grpc-status") grpc_status = v;
      else if (k == "grpc-message") grpc_message = v;
    }
    if (!http_status.empty() && http_status != "200")
      throw GrpcError("HTTP " + http_status + " from gRPC server");
    if (!grpc_status.empty() && grpc_status != "0") {
      // grpc-message is percent-encoded (gRPC HTTP/2 spec)
      std::string msg_out;
      for (size_t i = 0; i < grpc_message.size(); i++) {
        if (grpc_message[i] == '%' && i + 2 < grpc_message.size()) {
          auto hex = [](char c) -> int {
            if (c >= '0' && c <= '9') return c - '0';
            if (c >= 'a' && c <= 'f') return c - 'a' + 10;
            if (c >= 'A' && c <= 'F') return c - 'A' + 10;
            return -1;
          };
          int hi = hex(grpc_message[i + 1]), lo = hex(grpc_message[i + 2]);
          if (hi >= 0 && lo >= 0) {
            msg_out += static_cast<char>(hi * 16 + lo);
            i += 2;
            continue;
          }
        }
        msg_out += grpc_message[i];
      }
      throw GrpcError("grpc-status " + grpc_status +
                      (msg_out.empty() ? "" : ": " + msg_out));
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
