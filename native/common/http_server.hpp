// http_server.hpp — minimal threaded HTTP/1.1 server.
//
// Serves the mi355-exporter's /metrics endpoint (Prometheus text exposition)
// and the pruner's debug endpoints. Plain TCP only — TLS termination is the
// scrape infrastructure's job, as with the reference's dcgm-exporter setup.
#pragma once

#include <atomic>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

namespace http {

struct ConnRegistry;

struct ServerRequest {
  std::string method;
  std::string path;    // path only, query split off
  std::string query;   // raw query string (no '?')
  std::map<std::string, std::string> headers;  // lower-cased keys
  std::string body;
};

struct ServerResponse {
  int status = 200;
  std::string content_type = "text/plain; charset=utf-8";
  std::string body;
  // Streaming mode (Kubernetes watch): when set, `body` is ignored and the
  // response is sent with chunked transfer encoding — the function is
  // called repeatedly from the connection thread; each non-empty *chunk is
  // written as one chunk; returning false ends the stream (after writing
  // any final chunk). The connection closes after a streamed response.
  // Implementations must bound their internal waits so server stop() can
  // drain (they are invoked on detached connection threads).
  std::function<bool(std::string* chunk)> streamer;
};

class Server {
public:
  using Handler = std::function<ServerResponse(const ServerRequest&)>;

  // bind_addr e.g. "0.0.0.0" or "127.0.0.1"; port 0 picks an ephemeral port.
  Server(const std::string& bind_addr, uint16_t port, Handler handler);
  ~Server();

  void start();  // spawns the accept loop; throws on bind failure
  void stop();
  uint16_t port() const { return port_; }

private:
  void accept_loop();
  void handle_conn(int fd);

  std::string bind_addr_;
  uint16_t port_;
  Handler handler_;
  int listen_fd_ = -1;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::shared_ptr<ConnRegistry> registry_;
};

}  // namespace http
