// http.hpp — minimal HTTP/1.1 client with TLS, used for both the Prometheus
// and kube-apiserver data paths of the MI355X-native gpu-pruner.
//
// The reference daemon reaches Prometheus through reqwest and the apiserver
// through kube-rs (SURVEY.md §2.1 "Query executor" / "K8s client"); this is
// the from-scratch C++ equivalent: blocking sockets + OpenSSL, a per-endpoint
// keep-alive connection pool (the decision engine's N-way concurrent pod
// evaluation checks connections in and out), chunked transfer decoding, and
// unix-domain-socket support (kubelet PodResources / test fixtures).
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <vector>

namespace http {

enum class TlsVerify {
  Skip,     // accept any certificate (--prometheus-tls-mode=skip)
  Verify,   // system roots (default)
  CustomCa  // verify against a provided PEM bundle
};

struct Url {
  std::string scheme;  // http | https | unix
  std::string host;    // or unix socket path when scheme == "unix"
  uint16_t port = 0;
  std::string path;  // begins with '/' (may include query)

  // Parses "http://host[:port][/path]", "https://...", "unix:///run/x.sock".
  static std::optional<Url> parse(const std::string& s);
  std::string origin() const;  // scheme://host:port — connection pool key
};

struct Request {
  std::string method = "GET";
  std::string path = "/";
  std::vector<std::pair<std::string, std::string>> headers;
  std::string body;
};

struct Response {
  int status = 0;
  std::map<std::string, std::string> headers;  // lower-cased keys
  std::string body;
};

struct ClientOptions {
  TlsVerify tls = TlsVerify::Verify;
  std::string ca_file;          // PEM bundle for TlsVerify::CustomCa
  std::string client_cert_file; // mTLS client certificate (PEM), optional
  std::string client_key_file;  // mTLS client private key (PEM)
  // In-memory PEM alternatives (kubeconfig base64 `-data` material): loaded
  // via BIOs so key material never touches the filesystem.
  std::string ca_pem;           // additional trusted roots for CustomCa
  std::string client_cert_pem;  // mTLS client certificate chain
  std::string client_key_pem;   // mTLS client private key
  int connect_timeout_ms = 5000;
  int io_timeout_ms = 30000;    // per-request read/write deadline
  int max_pool_per_origin = 256;
};

class Connection;  // opaque: one TCP/TLS (or unix) stream

// Incrementally-read response body (Kubernetes watch streams: chunked JSON
// events that arrive over minutes). Owns its connection; the connection is
// NOT returned to the pool (watch streams are not reusable).
class BodyStream {
public:
  BodyStream(std::unique_ptr<Connection> conn, int status,
             std::map<std::string, std::string> headers, std::string initial,
             bool chunked);
  ~BodyStream();

  int status() const { return status_; }
  const std::map<std::string, std::string>& headers() const { return headers_; }

  // Next newline-terminated line of the decoded body (without the '\n').
  // Returns false on orderly end of stream. Throws Error on transport
  // errors/timeouts (io_timeout_ms of the owning client's options).
  bool read_line(std::string* line);

  // Abort from another thread: shuts the socket down so a blocked
  // read_line returns/throws promptly.
  void shutdown();

private:
  bool fill();  // read more decoded bytes into buf_; false on EOF

  std::unique_ptr<Connection> conn_;
  int status_;
  std::map<std::string, std::string> headers_;
  std::string raw_;      // undecoded (possibly chunked) bytes
  std::string buf_;      // decoded body bytes not yet consumed
  bool chunked_;
  bool eof_ = false;
  size_t chunk_remaining_ = 0;  // bytes left in the current chunk's data
};

// Thread-safe HTTP client for one origin (scheme+host+port). Connections are
// pooled and reused across requests; a request that finds the pooled
// connection stale (server closed keep-alive) is retried once on a fresh one.
class Client {
public:
  Client(Url base, ClientOptions opts);
  ~Client();

  Client(const Client&) = delete;
  Client& operator=(const Client&) = delete;

  // `path` overrides base.path; headers are appended to defaults.
  Response request(const Request& req);

  // Open a streaming request (Kubernetes watch): returns after status +
  // headers arrive; body bytes are pulled incrementally via BodyStream.
  std::unique_ptr<BodyStream> open_stream(const Request& req);

  // convenience
  Response get(const std::string& path,
               const std::vector<std::pair<std::string, std::string>>& headers = {});
  Response post(const std::string& path, const std::string& body,
                const std::string& content_type,
                const std::vector<std::pair<std::string, std::string>>& headers = {});
  Response patch(const std::string& path, const std::string& body,
                 const std::string& content_type,
                 const std::vector<std::pair<std::string, std::string>>& headers = {});

  void set_default_header(const std::string& k, const std::string& v);
  const Url& base() const { return base_; }

private:
  std::unique_ptr<Connection> checkout();
  void checkin(std::unique_ptr<Connection> c);
  Response do_request_on(Connection& c, const Request& req);

  Url base_;
  ClientOptions opts_;
  std::vector<std::pair<std::string, std::string>> default_headers_;
  std::mutex pool_mu_;
  std::vector<std::unique_ptr<Connection>> pool_;
  void* ssl_ctx_ = nullptr;  // SSL_CTX*, shared by all connections of this client
};

// One-shot helper for tests/tools.
Response fetch(const std::string& url, const Request& req = {},
               const ClientOptions& opts = {});

class Error : public std::runtime_error {
public:
  using std::runtime_error::runtime_error;
};

}  // namespace http
