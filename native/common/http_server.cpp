#include "http_server.hpp"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <condition_variable>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <set>
#include <stdexcept>

#include "strutil.hpp"
#include "tsan_compat.hpp"

namespace http {

// Connection bookkeeping shared with detached handler threads: stop() must be
// able to shut down every live keep-alive socket without joining threads that
// are blocked in recv().
struct ConnRegistry {
  std::mutex mu;
  std::condition_variable cv;
  std::set<int> fds;
  int active = 0;

  void add(int fd) {
    std::lock_guard<std::mutex> lock(mu);
    fds.insert(fd);
    active++;
  }
  void remove(int fd) {
    std::lock_guard<std::mutex> lock(mu);
    fds.erase(fd);
    active--;
    cv.notify_all();
  }
  void shutdown_all() {
    std::lock_guard<std::mutex> lock(mu);
    for (int fd : fds) ::shutdown(fd, SHUT_RDWR);
  }
  void wait_drained(int timeout_ms) {
    std::unique_lock<std::mutex> lock(mu);
    qx::cv_wait_for(cv, lock, std::chrono::milliseconds(timeout_ms), [&] { return active == 0; });
  }
};

Server::Server(const std::string& bind_addr, uint16_t port, Handler handler)
    : bind_addr_(bind_addr), port_(port), handler_(std::move(handler)) {
  registry_ = std::make_shared<ConnRegistry>();
}

Server::~Server() { stop(); }

void Server::start() {
  listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
  if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
  int one = 1;
  ::setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  struct sockaddr_in addr {};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(port_);
  if (::inet_pton(AF_INET, bind_addr_.c_str(), &addr.sin_addr) != 1)
    throw std::runtime_error("invalid bind address: " + bind_addr_);
  if (::bind(listen_fd_, reinterpret_cast<struct sockaddr*>(&addr), sizeof addr) < 0) {
    ::close(listen_fd_);
    listen_fd_ = -1;
    throw std::runtime_error("bind to " + bind_addr_ + ":" + std::to_string(port_) +
                             " failed: " + std::strerror(errno));
  }
  if (port_ == 0) {
    socklen_t len = sizeof addr;
    ::getsockname(listen_fd_, reinterpret_cast<struct sockaddr*>(&addr), &len);
    port_ = ntohs(addr.sin_port);
  }
  if (::listen(listen_fd_, 256) < 0) throw std::runtime_error("listen() failed");
  running_.store(true);
  accept_thread_ = std::thread([this] { accept_loop(); });
}

void Server::stop() {
  if (!running_.exchange(false)) return;
  ::shutdown(listen_fd_, SHUT_RDWR);
  ::close(listen_fd_);
  if (accept_thread_.joinable()) accept_thread_.join();
  registry_->shutdown_all();
  registry_->wait_drained(5000);
}

void Server::accept_loop() {
  while (running_.load()) {
    int fd = ::accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) {
      if (!running_.load()) break;
      continue;
    }
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    registry_->add(fd);
    // Detached per-connection handler; stop() tears sockets down via the
    // registry instead of joining (a join would block on live keep-alives).
    auto reg = registry_;
    std::thread([this, fd, reg] {
      handle_conn(fd);
      reg->remove(fd);
    }).detach();
  }
}

void Server::handle_conn(int fd) {
  struct timeval tv {30, 0};
  ::setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
  ::setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);

  std::string buf;
  char tmp[16384];
  while (running_.load()) {
    // ---- read request head ----
    size_t hdr_end;
    while ((hdr_end = buf.find("\r\n\r\n")) == std::string::npos) {
      ssize_t r = ::recv(fd, tmp, sizeof tmp, 0);
      if (r <= 0) {
        ::close(fd);
        return;
      }
      buf.append(tmp, static_cast<size_t>(r));
      if (buf.size() > (1u << 20)) {
        ::close(fd);
        return;
      }
    }
    ServerRequest req;
    {
      std::string head = buf.substr(0, hdr_end);
      auto lines = strutil::split(head, '\n');
      auto parts = strutil::split(strutil::trim(lines[0]), ' ');
      if (parts.size() < 2) {
        ::close(fd);
        return;
      }
      req.method = parts[0];
      std::string target = parts[1];
      size_t q = target.find('?');
      req.path = q == std::string::npos ? target : target.substr(0, q);
      req.query = q == std::string::npos ? "" : target.substr(q + 1);
      for (size_t i = 1; i < lines.size(); i++) {
        std::string line = strutil::trim(lines[i]);
        size_t colon = line.find(':');
        if (colon != std::string::npos)
          req.headers[strutil::lower(line.substr(0, colon))] =
              strutil::trim(line.substr(colon + 1));
      }
    }
    size_t content_len = 0;
    if (auto it = req.headers.find("content-length"); it != req.headers.end())
      content_len = std::strtoul(it->second.c_str(), nullptr, 10);
    size_t body_start = hdr_end + 4;
    while (buf.size() < body_start + content_len) {
      ssize_t r = ::recv(fd, tmp, sizeof tmp, 0);
      if (r <= 0) {
        ::close(fd);
        return;
      }
      buf.append(tmp, static_cast<size_t>(r));
    }
    req.body = buf.substr(body_start, content_len);
    buf.erase(0, body_start + content_len);

    ServerResponse resp;
    try {
      resp = handler_(req);
    } catch (const std::exception& e) {
      resp.status = 500;
      resp.body = std::string("internal error: ") + e.what();
    }

    const char* reason = resp.status == 200   ? "OK"
                         : resp.status == 201 ? "Created"
                         : resp.status == 404 ? "Not Found"
                         : resp.status == 400 ? "Bad Request"
                                              : "Status";
    auto send_all = [&](const std::string& data) {
      size_t off = 0;
      while (off < data.size()) {
        ssize_t w = ::send(fd, data.data() + off, data.size() - off, MSG_NOSIGNAL);
        if (w <= 0) return false;
        off += static_cast<size_t>(w);
      }
      return true;
    };

    if (resp.streamer) {
      // chunked streaming response (watch): one chunk per streamer call
      std::string head = "HTTP/1.1 " + std::to_string(resp.status) + " " + reason + "\r\n";
      head += "Content-Type: " + resp.content_type + "\r\n";
      head += "Transfer-Encoding: chunked\r\n";
      head += "Connection: close\r\n\r\n";
      if (!send_all(head)) {
        ::close(fd);
        return;
      }
      bool more = true;
      while (more && running_.load()) {
        std::string chunk;
        more = resp.streamer(&chunk);
        if (!chunk.empty()) {
          char sz[24];  // 16 hex digits (64-bit size_t) + CRLF + NUL
          std::snprintf(sz, sizeof sz, "%zx\r\n", chunk.size());
          if (!send_all(std::string(sz) + chunk + "\r\n")) {
            ::close(fd);
            return;
          }
        }
      }
      send_all("0\r\n\r\n");
      ::close(fd);
      return;
    }

    std::string out = "HTTP/1.1 " + std::to_string(resp.status) + " " + reason + "\r\n";
    out += "Content-Type: " + resp.content_type + "\r\n";
    out += "Content-Length: " + std::to_string(resp.body.size()) + "\r\n";
    out += "Connection: keep-alive\r\n\r\n";
    out += resp.body;
    if (!send_all(out)) {
      ::close(fd);
      return;
    }
  }
  ::close(fd);
}

}  // namespace http
