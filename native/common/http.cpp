#include "http.hpp"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/err.h>
#include <openssl/pem.h>
#include <openssl/ssl.h>
#include <openssl/x509.h>
#include <poll.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <atomic>
#include <cstring>

#include "strutil.hpp"

namespace http {

// ----------------------------- Url ------------------------------------------

std::optional<Url> Url::parse(const std::string& s) {
  Url u;
  size_t sep = s.find("://");
  if (sep == std::string::npos) return std::nullopt;
  u.scheme = strutil::lower(s.substr(0, sep));
  std::string rest = s.substr(sep + 3);
  if (u.scheme == "unix") {
    // unix:///run/foo.sock — everything is the socket path; requests carry their own path.
    u.host = rest.empty() ? "" : "/" + rest;
    // tolerate unix://%2Frun%2Ffoo.sock style too (not needed internally)
    if (!rest.empty() && rest[0] == '/') u.host = rest;
    u.path = "/";
    return u;
  }
  if (u.scheme != "http" && u.scheme != "https") return std::nullopt;
  size_t slash = rest.find('/');
  std::string hostport = slash == std::string::npos ? rest : rest.substr(0, slash);
  u.path = slash == std::string::npos ? "/" : rest.substr(slash);
  if (!hostport.empty() && hostport[0] == '[') {  // [v6]:port
    size_t close = hostport.find(']');
    if (close == std::string::npos) return std::nullopt;
    u.host = hostport.substr(1, close - 1);
    if (close + 1 < hostport.size() && hostport[close + 1] == ':')
      u.port = static_cast<uint16_t>(std::stoi(hostport.substr(close + 2)));
  } else {
    size_t colon = hostport.rfind(':');
    if (colon == std::string::npos) {
      u.host = hostport;
    } else {
      u.host = hostport.substr(0, colon);
      try {
        u.port = static_cast<uint16_t>(std::stoi(hostport.substr(colon + 1)));
      } catch (...) {
        return std::nullopt;
      }
    }
  }
  if (u.port == 0) u.port = u.scheme == "https" ? 443 : 80;
  if (u.host.empty()) return std::nullopt;
  return u;
}

std::string Url::origin() const {
  if (scheme == "unix") return "unix://" + host;
  return scheme + "://" + host + ":" + std::to_string(port);
}

// --------------------------- Connection --------------------------------------

namespace {

struct SslInit {
  SslInit() {
    SSL_library_init();
    SSL_load_error_strings();
  }
};

void ensure_ssl_init() { static SslInit init; }

std::string ssl_err_string() {
  unsigned long e = ERR_get_error();
  char buf[256];
  ERR_error_string_n(e, buf, sizeof buf);
  return buf;
}

// Add every certificate in an in-memory PEM bundle to the ctx's trust store.
bool add_ca_pem(SSL_CTX* ctx, const std::string& pem) {
  BIO* bio = BIO_new_mem_buf(pem.data(), static_cast<int>(pem.size()));
  if (!bio) return false;
  X509_STORE* store = SSL_CTX_get_cert_store(ctx);
  int added = 0;
  while (X509* cert = PEM_read_bio_X509(bio, nullptr, nullptr, nullptr)) {
    if (X509_STORE_add_cert(store, cert) == 1) added++;
    X509_free(cert);
  }
  ERR_clear_error();  // trailing-garbage error from the final read attempt
  BIO_free(bio);
  return added > 0;
}

// Load an in-memory client certificate chain + private key (kubeconfig
// `-data` material — kept off the filesystem by design).
bool use_client_pem(SSL_CTX* ctx, const std::string& cert_pem, const std::string& key_pem) {
  BIO* cbio = BIO_new_mem_buf(cert_pem.data(), static_cast<int>(cert_pem.size()));
  if (!cbio) return false;
  X509* leaf = PEM_read_bio_X509(cbio, nullptr, nullptr, nullptr);
  bool ok = leaf && SSL_CTX_use_certificate(ctx, leaf) == 1;
  if (leaf) X509_free(leaf);
  while (ok) {  // remaining certs in the bundle form the chain
    X509* extra = PEM_read_bio_X509(cbio, nullptr, nullptr, nullptr);
    if (!extra) break;
    if (SSL_CTX_add_extra_chain_cert(ctx, extra) != 1) {  // ctx owns on success
      X509_free(extra);
      ok = false;
    }
  }
  ERR_clear_error();
  BIO_free(cbio);
  if (!ok) return false;
  BIO* kbio = BIO_new_mem_buf(key_pem.data(), static_cast<int>(key_pem.size()));
  if (!kbio) return false;
  EVP_PKEY* key = PEM_read_bio_PrivateKey(kbio, nullptr, nullptr, nullptr);
  ok = key && SSL_CTX_use_PrivateKey(ctx, key) == 1;
  if (key) EVP_PKEY_free(key);
  BIO_free(kbio);
  return ok && SSL_CTX_check_private_key(ctx) == 1;
}

}  // namespace

class Connection {
public:
  Connection(const Url& u, const ClientOptions& opts, SSL_CTX* ctx) : opts_(opts) {
    if (u.scheme == "unix") {
      connect_unix(u.host);
    } else {
      connect_tcp(u.host, u.port);
      if (u.scheme == "https") start_tls(ctx, u.host);
    }
  }

  ~Connection() {
    if (ssl_) {
      SSL_shutdown(ssl_);
      SSL_free(ssl_);
    }
    if (fd_ >= 0) ::close(fd_);
  }

  bool healthy() const { return fd_ >= 0 && !broken_; }
  void mark_broken() { broken_ = true; }

  void write_all(const char* data, size_t n) {
    size_t off = 0;
    while (off < n) {
      ssize_t w;
      if (ssl_) {
        w = SSL_write(ssl_, data + off, static_cast<int>(n - off));
        if (w <= 0) {
          int err = SSL_get_error(ssl_, static_cast<int>(w));
          if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) {
            wait_io(err == SSL_ERROR_WANT_READ);
            continue;
          }
          broken_ = true;
          throw Error("TLS write failed: " + ssl_err_string());
        }
      } else {
        w = ::send(fd_, data + off, n - off, MSG_NOSIGNAL);
        if (w < 0) {
          if (errno == EINTR) continue;
          if (errno == EAGAIN || errno == EWOULDBLOCK) {
            wait_io(false);
            continue;
          }
          broken_ = true;
          throw Error(std::string("write failed: ") + std::strerror(errno));
        }
      }
      off += static_cast<size_t>(w);
    }
  }

  // Returns bytes read (>0), 0 on orderly EOF. Throws on error/timeout.
  size_t read_some(char* buf, size_t cap) {
    while (true) {
      ssize_t r;
      if (ssl_) {
        r = SSL_read(ssl_, buf, static_cast<int>(cap));
        if (r <= 0) {
          int err = SSL_get_error(ssl_, static_cast<int>(r));
          if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) {
            wait_io(err == SSL_ERROR_WANT_READ);
            continue;
          }
          if (err == SSL_ERROR_ZERO_RETURN) return 0;
          broken_ = true;
          if (err == SSL_ERROR_SYSCALL && r == 0) return 0;  // unclean EOF
          throw Error("TLS read failed: " + ssl_err_string());
        }
        return static_cast<size_t>(r);
      }
      r = ::recv(fd_, buf, cap, 0);
      if (r < 0) {
        if (errno == EINTR) continue;
        if (errno == EAGAIN || errno == EWOULDBLOCK) {
          wait_io(true);
          continue;
        }
        broken_ = true;
        throw Error(std::string("read failed: ") + std::strerror(errno));
      }
      return static_cast<size_t>(r);
    }
  }

private:
  void set_nonblock(int fd) { fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK); }

  void wait_io(bool want_read) {
    struct pollfd pfd {};
    pfd.fd = fd_;
    pfd.events = want_read ? POLLIN : POLLOUT;
    int rc = ::poll(&pfd, 1, opts_.io_timeout_ms);
    if (rc == 0) {
      broken_ = true;
      throw Error("I/O timeout");
    }
    if (rc < 0 && errno != EINTR) {
      broken_ = true;
      throw Error(std::string("poll failed: ") + std::strerror(errno));
    }
  }

  void connect_tcp(const std::string& host, uint16_t port) {
    struct addrinfo hints {};
    hints.ai_family = AF_UNSPEC;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    std::string port_s = std::to_string(port);
    int rc = ::getaddrinfo(host.c_str(), port_s.c_str(), &hints, &res);
    if (rc != 0) throw Error("DNS resolution failed for " + host + ": " + gai_strerror(rc));
    std::string last_err = "no addresses";
    for (struct addrinfo* ai = res; ai; ai = ai->ai_next) {
      int fd = ::socket(ai->ai_family, ai->ai_socktype, ai->ai_protocol);
      if (fd < 0) continue;
      set_nonblock(fd);
      int one = 1;
      ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
      rc = ::connect(fd, ai->ai_addr, ai->ai_addrlen);
      if (rc < 0 && errno == EINPROGRESS) {
        struct pollfd pfd {};
        pfd.fd = fd;
        pfd.events = POLLOUT;
        rc = ::poll(&pfd, 1, opts_.connect_timeout_ms);
        if (rc > 0) {
          int soerr = 0;
          socklen_t len = sizeof soerr;
          ::getsockopt(fd, SOL_SOCKET, SO_ERROR, &soerr, &len);
          if (soerr == 0) {
            fd_ = fd;
            break;
          }
          last_err = std::strerror(soerr);
        } else {
          last_err = rc == 0 ? "connect timeout" : std::strerror(errno);
        }
      } else if (rc == 0) {
        fd_ = fd;
        break;
      } else {
        last_err = std::strerror(errno);
      }
      ::close(fd);
    }
    ::freeaddrinfo(res);
    if (fd_ < 0) throw Error("connect to " + host + ":" + port_s + " failed: " + last_err);
  }

  void connect_unix(const std::string& path) {
    int fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) throw Error(std::string("socket(AF_UNIX): ") + std::strerror(errno));
    struct sockaddr_un addr {};
    addr.sun_family = AF_UNIX;
    if (path.size() >= sizeof(addr.sun_path)) {
      ::close(fd);
      throw Error("unix socket path too long: " + path);
    }
    std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd, reinterpret_cast<struct sockaddr*>(&addr), sizeof addr) < 0) {
      ::close(fd);
      throw Error("connect to unix socket " + path + " failed: " + std::strerror(errno));
    }
    set_nonblock(fd);
    fd_ = fd;
  }

  static bool is_ip_literal(const std::string& host) {
    unsigned char buf[sizeof(struct in6_addr)];
    return inet_pton(AF_INET, host.c_str(), buf) == 1 ||
           inet_pton(AF_INET6, host.c_str(), buf) == 1;
  }

  void start_tls(SSL_CTX* ctx, const std::string& host) {
    ensure_ssl_init();
    ssl_ = SSL_new(ctx);
    if (!ssl_) throw Error("SSL_new failed");
    SSL_set_fd(ssl_, fd_);
    // Peer identity check (when verify enabled on the ctx). An IP-literal host
    // (the in-cluster apiserver is https://$KUBERNETES_SERVICE_HOST — a
    // ClusterIP) must be matched against IP SANs, not DNS names, and RFC 6066
    // forbids SNI for IP literals.
    X509_VERIFY_PARAM* param = SSL_get0_param(ssl_);
    if (is_ip_literal(host)) {
      X509_VERIFY_PARAM_set1_ip_asc(param, host.c_str());
    } else {
      SSL_set_tlsext_host_name(ssl_, host.c_str());
      X509_VERIFY_PARAM_set1_host(param, host.c_str(), 0);
    }
    while (true) {
      int rc = SSL_connect(ssl_);
      if (rc == 1) break;
      int err = SSL_get_error(ssl_, rc);
      if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) {
        wait_io(err == SSL_ERROR_WANT_READ);
        continue;
      }
      long vr = SSL_get_verify_result(ssl_);
      std::string why = ssl_err_string();
      if (vr != X509_V_OK)
        why += std::string(" (verify: ") + X509_verify_cert_error_string(vr) + ")";
      throw Error("TLS handshake with " + host + " failed: " + why);
    }
  }

public:
  // Abort a blocked read from another thread (watch-stream shutdown).
  void shutdown_socket() {
    broken_ = true;
    if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
  }

  ClientOptions opts_;
  int fd_ = -1;
  SSL* ssl_ = nullptr;
  // written by shutdown_socket() from other threads (watch-stream abort)
  std::atomic<bool> broken_{false};
};

namespace {

// Serialize a request against `base` with `default_headers` (shared by the
// buffered and streaming paths).
std::string serialize_request(
    const Url& base, const std::vector<std::pair<std::string, std::string>>& default_headers,
    const Request& req) {
  std::string host_hdr =
      base.scheme == "unix" ? "localhost" : base.host + ":" + std::to_string(base.port);
  std::string out;
  out.reserve(512 + req.body.size());
  out += req.method + " " + (req.path.empty() ? "/" : req.path) + " HTTP/1.1\r\n";
  out += "Host: " + host_hdr + "\r\n";
  bool have_accept = false, have_ct = false;
  auto append_hdr = [&](const std::string& k, const std::string& v) {
    std::string lk = strutil::lower(k);
    if (lk == "accept") have_accept = true;
    if (lk == "content-type") have_ct = true;
    out += k + ": " + v + "\r\n";
  };
  for (const auto& [k, v] : default_headers) {
    bool overridden = false;
    for (const auto& [rk, rv] : req.headers)
      if (strutil::lower(rk) == strutil::lower(k)) overridden = true;
    if (!overridden) append_hdr(k, v);
  }
  for (const auto& [k, v] : req.headers) append_hdr(k, v);
  if (!have_accept) out += "Accept: application/json\r\n";
  if (!req.body.empty() && !have_ct) out += "Content-Type: application/json\r\n";
  if (!req.body.empty() || req.method == "POST" || req.method == "PUT" || req.method == "PATCH")
    out += "Content-Length: " + std::to_string(req.body.size()) + "\r\n";
  out += "Connection: keep-alive\r\n\r\n";
  out += req.body;
  return out;
}

// Read status line + headers (skipping interim 1xx). Leaves any body bytes
// already received in *leftover.
void read_response_head(Connection& c, Response* resp, std::string* leftover) {
  std::string buf;
  std::string rest;
  while (true) {
    char tmp[8192];
    while (buf.find("\r\n\r\n") == std::string::npos) {
      size_t r = c.read_some(tmp, sizeof tmp);
      if (r == 0) {
        c.mark_broken();
        throw Error("connection closed before response headers");
      }
      buf.append(tmp, r);
      if (buf.size() > (1u << 20)) {
        c.mark_broken();
        throw Error("response headers too large");
      }
    }
    *resp = Response{};
    size_t hdr_end = buf.find("\r\n\r\n");
    std::string head = buf.substr(0, hdr_end);
    rest = buf.substr(hdr_end + 4);
    auto lines = strutil::split(head, '\n');
    if (lines.empty()) throw Error("malformed response");
    {
      std::string status_line = strutil::trim(lines[0]);
      size_t sp1 = status_line.find(' ');
      if (sp1 == std::string::npos) throw Error("malformed status line: " + status_line);
      resp->status = std::atoi(status_line.c_str() + sp1 + 1);
    }
    for (size_t i = 1; i < lines.size(); i++) {
      std::string line = strutil::trim(lines[i]);
      size_t colon = line.find(':');
      if (colon == std::string::npos) continue;
      resp->headers[strutil::lower(line.substr(0, colon))] =
          strutil::trim(line.substr(colon + 1));
    }
    if (resp->status >= 100 && resp->status < 200) {
      buf = rest;
      continue;
    }
    break;
  }
  *leftover = std::move(rest);
}

}  // namespace

// ------------------------------ Client ---------------------------------------

Client::Client(Url base, ClientOptions opts) : base_(std::move(base)), opts_(std::move(opts)) {
  if (base_.scheme == "https") {
    ensure_ssl_init();
    SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
    if (!ctx) throw Error("SSL_CTX_new failed");
    SSL_CTX_set_min_proto_version(ctx, TLS1_2_VERSION);
    switch (opts_.tls) {
      case TlsVerify::Skip:
        SSL_CTX_set_verify(ctx, SSL_VERIFY_NONE, nullptr);
        break;
      case TlsVerify::Verify:
        SSL_CTX_set_verify(ctx, SSL_VERIFY_PEER, nullptr);
        SSL_CTX_set_default_verify_paths(ctx);
        break;
      case TlsVerify::CustomCa:
        SSL_CTX_set_verify(ctx, SSL_VERIFY_PEER, nullptr);
        SSL_CTX_set_default_verify_paths(ctx);
        if (!opts_.ca_file.empty() &&
            SSL_CTX_load_verify_locations(ctx, opts_.ca_file.c_str(), nullptr) != 1) {
          SSL_CTX_free(ctx);
          throw Error("failed to load CA bundle " + opts_.ca_file + ": " + ssl_err_string());
        }
        if (!opts_.ca_pem.empty() && !add_ca_pem(ctx, opts_.ca_pem)) {
          SSL_CTX_free(ctx);
          throw Error("failed to load in-memory CA bundle: " + ssl_err_string());
        }
        break;
    }
    // mTLS client authentication (kube client-certificate auth)
    if (!opts_.client_cert_file.empty()) {
      if (SSL_CTX_use_certificate_chain_file(ctx, opts_.client_cert_file.c_str()) != 1 ||
          SSL_CTX_use_PrivateKey_file(ctx, opts_.client_key_file.c_str(),
                                      SSL_FILETYPE_PEM) != 1 ||
          SSL_CTX_check_private_key(ctx) != 1) {
        SSL_CTX_free(ctx);
        throw Error("failed to load client certificate/key (" + opts_.client_cert_file +
                    ", " + opts_.client_key_file + "): " + ssl_err_string());
      }
    } else if (!opts_.client_cert_pem.empty()) {
      if (!use_client_pem(ctx, opts_.client_cert_pem, opts_.client_key_pem)) {
        SSL_CTX_free(ctx);
        throw Error("failed to load in-memory client certificate/key: " + ssl_err_string());
      }
    }
    ssl_ctx_ = ctx;
  }
}

Client::~Client() {
  {
    std::lock_guard<std::mutex> lock(pool_mu_);
    pool_.clear();  // destroy connections before the shared SSL_CTX
  }
  if (ssl_ctx_) SSL_CTX_free(static_cast<SSL_CTX*>(ssl_ctx_));
}

void Client::set_default_header(const std::string& k, const std::string& v) {
  for (auto& [hk, hv] : default_headers_) {
    if (strutil::lower(hk) == strutil::lower(k)) {
      hv = v;
      return;
    }
  }
  default_headers_.emplace_back(k, v);
}

std::unique_ptr<Connection> Client::checkout() {
  {
    std::lock_guard<std::mutex> lock(pool_mu_);
    while (!pool_.empty()) {
      auto c = std::move(pool_.back());
      pool_.pop_back();
      if (c->healthy()) return c;
    }
  }
  return std::make_unique<Connection>(base_, opts_, static_cast<SSL_CTX*>(ssl_ctx_));
}

void Client::checkin(std::unique_ptr<Connection> c) {
  if (!c || !c->healthy()) return;
  std::lock_guard<std::mutex> lock(pool_mu_);
  if (pool_.size() < static_cast<size_t>(opts_.max_pool_per_origin)) pool_.push_back(std::move(c));
}

Response Client::request(const Request& req) {
  // Retry once on a stale pooled connection (peer closed keep-alive).
  for (int attempt = 0;; attempt++) {
    auto conn = checkout();
    try {
      Response r = do_request_on(*conn, req);
      checkin(std::move(conn));
      return r;
    } catch (const Error&) {
      if (attempt >= 1) throw;
      // retry with a fresh connection
    }
  }
}

Response Client::do_request_on(Connection& c, const Request& req) {
  std::string out = serialize_request(base_, default_headers_, req);
  c.write_all(out.data(), out.size());

  Response resp;
  std::string rest;
  read_response_head(c, &resp, &rest);

  bool keep_alive = true;
  {
    auto it = resp.headers.find("connection");
    if (it != resp.headers.end() && strutil::lower(it->second) == "close") keep_alive = false;
  }

  auto read_n = [&](size_t need) {
    char tmp[16384];
    while (rest.size() < need) {
      size_t r = c.read_some(tmp, sizeof tmp);
      if (r == 0) {
        c.mark_broken();
        throw Error("connection closed mid-body");
      }
      rest.append(tmp, r);
    }
  };

  auto it_te = resp.headers.find("transfer-encoding");
  if (it_te != resp.headers.end() &&
      strutil::lower(it_te->second).find("chunked") != std::string::npos) {
    // chunked decoding
    std::string body;
    size_t pos = 0;
    while (true) {
      // find CRLF after chunk size
      size_t crlf;
      while ((crlf = rest.find("\r\n", pos)) == std::string::npos) {
        char tmp[16384];
        size_t r = c.read_some(tmp, sizeof tmp);
        if (r == 0) {
          c.mark_broken();
          throw Error("connection closed mid-chunk-size");
        }
        rest.append(tmp, r);
      }
      size_t chunk_len = std::strtoul(rest.substr(pos, crlf - pos).c_str(), nullptr, 16);
      size_t data_start = crlf + 2;
      read_n(data_start + chunk_len + 2);
      if (chunk_len == 0) break;
      body.append(rest, data_start, chunk_len);
      pos = data_start + chunk_len + 2;  // skip trailing CRLF
    }
    resp.body = std::move(body);
  } else if (resp.headers.count("content-length")) {
    size_t len = std::strtoul(resp.headers["content-length"].c_str(), nullptr, 10);
    read_n(len);
    resp.body = rest.substr(0, len);
  } else if (req.method == "HEAD" || resp.status == 204 || resp.status == 304) {
    resp.body.clear();
  } else {
    // read until close
    char tmp[16384];
    while (true) {
      size_t r;
      try {
        r = c.read_some(tmp, sizeof tmp);
      } catch (const Error&) {
        break;
      }
      if (r == 0) break;
      rest.append(tmp, r);
    }
    resp.body = rest;
    keep_alive = false;
  }

  if (!keep_alive) c.mark_broken();
  return resp;
}

Response Client::get(const std::string& path,
                     const std::vector<std::pair<std::string, std::string>>& headers) {
  Request r;
  r.method = "GET";
  r.path = path;
  r.headers = headers;
  return request(r);
}

Response Client::post(const std::string& path, const std::string& body,
                      const std::string& content_type,
                      const std::vector<std::pair<std::string, std::string>>& headers) {
  Request r;
  r.method = "POST";
  r.path = path;
  r.body = body;
  r.headers = headers;
  r.headers.emplace_back("Content-Type", content_type);
  return request(r);
}

Response Client::patch(const std::string& path, const std::string& body,
                       const std::string& content_type,
                       const std::vector<std::pair<std::string, std::string>>& headers) {
  Request r;
  r.method = "PATCH";
  r.path = path;
  r.body = body;
  r.headers = headers;
  r.headers.emplace_back("Content-Type", content_type);
  return request(r);
}

// ---------------------------- BodyStream -------------------------------------

BodyStream::BodyStream(std::unique_ptr<Connection> conn, int status,
                       std::map<std::string, std::string> headers, std::string initial,
                       bool chunked)
    : conn_(std::move(conn)), status_(status), headers_(std::move(headers)),
      raw_(std::move(initial)), chunked_(chunked) {}

BodyStream::~BodyStream() = default;

void BodyStream::shutdown() {
  if (conn_) conn_->shutdown_socket();
}

bool BodyStream::fill() {
  // move decoded bytes from raw_ (and the socket) into buf_
  while (true) {
    if (!chunked_) {
      if (!raw_.empty()) {
        buf_ += raw_;
        raw_.clear();
        return true;
      }
    } else {
      // decode whatever complete chunk data we have
      bool progressed = false;
      while (true) {
        if (chunk_remaining_ > 0) {
          size_t take = std::min(chunk_remaining_, raw_.size());
          if (take == 0) break;
          buf_.append(raw_, 0, take);
          raw_.erase(0, take);
          chunk_remaining_ -= take;
          progressed = true;
          if (chunk_remaining_ == 0) {
            // consume the trailing CRLF when it arrives
            if (raw_.size() >= 2) raw_.erase(0, 2);
            else chunk_remaining_ = 0;  // CRLF split across reads: handled below
          }
          continue;
        }
        // skip a stray CRLF left from a chunk boundary
        while (!raw_.empty() && (raw_[0] == '\r' || raw_[0] == '\n')) raw_.erase(0, 1);
        size_t crlf = raw_.find("\r\n");
        if (crlf == std::string::npos) break;  // need more bytes for the size line
        size_t len = std::strtoul(raw_.substr(0, crlf).c_str(), nullptr, 16);
        raw_.erase(0, crlf + 2);
        if (len == 0) {  // final chunk
          eof_ = true;
          return progressed;
        }
        chunk_remaining_ = len;
      }
      if (progressed) return true;
    }
    if (eof_) return false;
    char tmp[16384];
    size_t r;
    try {
      r = conn_->read_some(tmp, sizeof tmp);
    } catch (const Error&) {
      eof_ = true;
      throw;
    }
    if (r == 0) {
      eof_ = true;
      return !raw_.empty() || !buf_.empty();
    }
    raw_.append(tmp, r);
  }
}

bool BodyStream::read_line(std::string* line) {
  while (true) {
    size_t nl = buf_.find('\n');
    if (nl != std::string::npos) {
      *line = buf_.substr(0, nl);
      if (!line->empty() && line->back() == '\r') line->pop_back();
      buf_.erase(0, nl + 1);
      return true;
    }
    if (eof_) {
      if (buf_.empty()) return false;
      *line = std::move(buf_);  // unterminated trailing line
      buf_.clear();
      return true;
    }
    if (!fill() && eof_ && buf_.empty()) return false;
  }
}

std::unique_ptr<BodyStream> Client::open_stream(const Request& req) {
  auto conn = std::make_unique<Connection>(base_, opts_, static_cast<SSL_CTX*>(ssl_ctx_));
  std::string out = serialize_request(base_, default_headers_, req);
  conn->write_all(out.data(), out.size());
  Response head;
  std::string leftover;
  read_response_head(*conn, &head, &leftover);
  bool chunked = false;
  if (auto it = head.headers.find("transfer-encoding"); it != head.headers.end())
    chunked = strutil::lower(it->second).find("chunked") != std::string::npos;
  return std::make_unique<BodyStream>(std::move(conn), head.status,
                                      std::move(head.headers), std::move(leftover),
                                      chunked);
}

Response fetch(const std::string& url, const Request& req, const ClientOptions& opts) {
  auto parsed = Url::parse(url);
  if (!parsed) throw Error("invalid URL: " + url);
  Client client(*parsed, opts);
  Request r = req;
  if (r.path == "/" && parsed->path != "/") r.path = parsed->path;
  return client.request(r);
}

}  // namespace http
