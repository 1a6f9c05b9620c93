#include "informer.hpp"

#include <chrono>

#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../common/tsan_compat.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::informer";
// Server-side watch timeout: the apiserver closes the stream gracefully
// after this long; must stay below the HTTP client's io_timeout (30 s) so
// an idle watch ends in a clean EOF + reopen, not a broken connection.
constexpr int kWatchTimeoutS = 25;
}  // namespace

Informer::Informer(KubeConfig cfg, std::string collection_path, std::string kind_key)
    : cfg_(std::move(cfg)), path_(std::move(collection_path)), kind_key_(std::move(kind_key)) {}

Informer::~Informer() { stop(); }

void Informer::start() {
  std::lock_guard<std::mutex> lock(mu_);
  if (thread_.joinable()) return;
  stop_ = false;
  thread_ = std::thread([this] { run(); });
}

void Informer::stop() {
  {
    std::lock_guard<std::mutex> lock(mu_);
    if (!thread_.joinable()) return;
    stop_ = true;
  }
  cv_.notify_all();
  {
    // abort a blocked watch read so the thread exits promptly
    std::lock_guard<std::mutex> lock(stream_mu_);
    if (active_stream_) active_stream_->shutdown();
  }
  thread_.join();
}

bool Informer::wait_synced(int timeout_ms) {
  std::unique_lock<std::mutex> lock(mu_);
  qx::cv_wait_for(cv_, lock, std::chrono::milliseconds(timeout_ms),
                  [this] { return synced_ || stop_.load(); });
  return synced_;
}

bool Informer::synced() const {
  std::lock_guard<std::mutex> lock(mu_);
  return synced_;
}

std::optional<jsn::Value> Informer::get(const std::string& name) const {
  std::lock_guard<std::mutex> lock(mu_);
  auto it = store_.find(name);
  if (it == store_.end()) return std::nullopt;
  return it->second;
}

std::map<std::string, jsn::Value> Informer::store_snapshot() const {
  std::lock_guard<std::mutex> lock(mu_);
  return store_;
}

size_t Informer::size() const {
  std::lock_guard<std::mutex> lock(mu_);
  return store_.size();
}

bool Informer::do_list(KubeClient& kube) {
  auto list = kube.get_opt(path_);
  if (!list) {
    // collection/API group absent (e.g. CRD not installed): an EMPTY synced
    // store — every lookup resolves to "gone", same as a 404 on GET
    std::lock_guard<std::mutex> lock(mu_);
    store_.clear();
    resource_version_.clear();
    synced_ = true;
    cv_.notify_all();
    return false;  // nothing to watch
  }
  std::map<std::string, jsn::Value> fresh;
  const jsn::Value& items = list->get("items");
  if (items.is_array())
    for (const auto& obj : items.arr()) {
      std::string name = obj.at({"metadata", "name"}).as_string();
      if (!name.empty()) fresh.emplace(std::move(name), obj);
    }
  std::string rv = list->at({"metadata", "resourceVersion"}).as_string_or("");
  {
    std::lock_guard<std::mutex> lock(mu_);
    store_ = std::move(fresh);
    resource_version_ = rv;
    synced_ = true;
    lists_issued_++;
  }
  cv_.notify_all();
  LOGD(TARGET, "LIST " + path_ + ": " + std::to_string(size()) + " objects @rv=" + rv);
  return !rv.empty();
}

void Informer::watch_once(KubeClient& kube) {
  std::string rv;
  {
    std::lock_guard<std::mutex> lock(mu_);
    rv = resource_version_;
  }
  std::string sep = path_.find('?') == std::string::npos ? "?" : "&";
  std::string watch_path = path_ + sep + "watch=true&resourceVersion=" + rv +
                           "&allowWatchBookmarks=true&timeoutSeconds=" +
                           std::to_string(kWatchTimeoutS);
  auto stream = kube.open_stream(watch_path);
  if (stream->status() == 410) {  // resourceVersion too old → relist
    std::lock_guard<std::mutex> lock(mu_);
    synced_ = false;
    return;
  }
  if (stream->status() < 200 || stream->status() >= 300)
    throw KubeError(stream->status(),
                    "watch " + watch_path + " -> " + std::to_string(stream->status()));
  {
    std::lock_guard<std::mutex> lock(stream_mu_);
    active_stream_ = stream.get();
  }
  std::string line;
  try {
    while (!stop_ && stream->read_line(&line)) {
      if (strutil::trim(line).empty()) continue;
      jsn::Value ev = jsn::parse(line);
      std::string type = ev.get("type").as_string();
      const jsn::Value& obj = ev.get("object");
      std::string name = obj.at({"metadata", "name"}).as_string();
      std::string obj_rv = obj.at({"metadata", "resourceVersion"}).as_string_or("");
      std::lock_guard<std::mutex> lock(mu_);
      events_seen_++;
      if (type == "ADDED" || type == "MODIFIED") {
        if (!name.empty()) store_[name] = obj;
      } else if (type == "DELETED") {
        store_.erase(name);
      } else if (type == "ERROR") {
        // 410 Gone delivered in-band: invalidate and relist
        synced_ = false;
        break;
      }
      // BOOKMARK (and every event) advances the resume point
      if (!obj_rv.empty()) resource_version_ = obj_rv;
    }
  } catch (...) {
    std::lock_guard<std::mutex> lock(stream_mu_);
    active_stream_ = nullptr;
    throw;
  }
  std::lock_guard<std::mutex> lock(stream_mu_);
  active_stream_ = nullptr;
}

void Informer::run() {
  int backoff_ms = 200;
  while (true) {
    {
      std::lock_guard<std::mutex> lock(mu_);
      if (stop_) return;
    }
    try {
      KubeClient kube(cfg_);
      bool watchable = true;
      {
        std::lock_guard<std::mutex> lock(mu_);
        watchable = synced_ && !resource_version_.empty();
      }
      if (!watchable) watchable = do_list(kube);
      while (watchable) {
        {
          std::lock_guard<std::mutex> lock(mu_);
          if (stop_ || !synced_) break;
        }
        watch_once(kube);  // returns on server timeout (EOF) or invalidation
      }
      if (!watchable) {
        // nothing to watch (collection absent): poll the LIST occasionally
        std::unique_lock<std::mutex> lock(mu_);
        qx::cv_wait_for(cv_, lock, std::chrono::seconds(5), [this] { return stop_.load(); });
        if (stop_) return;
        synced_ = false;  // retry the LIST (the CRD may have appeared)
      }
      backoff_ms = 200;
    } catch (const std::exception& e) {
      {
        std::lock_guard<std::mutex> lock(mu_);
        if (stop_) return;
        synced_ = false;  // force a fresh LIST on reconnect
      }
      LOGW(TARGET, "watch loop for " + path_ + " failed (" + e.what() + "), retrying in " +
                       std::to_string(backoff_ms) + " ms");
      std::unique_lock<std::mutex> lock(mu_);
      qx::cv_wait_for(cv_, lock, std::chrono::milliseconds(backoff_ms), [this] { return stop_.load(); });
      if (stop_) return;
      backoff_ms = std::min(backoff_ms * 2, 10000);
    }
  }
}

// ---------------------------- registry ---------------------------------------

InformerRegistry& InformerRegistry::global() {
  static InformerRegistry r;
  return r;
}

Informer& InformerRegistry::get_or_create(const KubeConfig& cfg, const std::string& ns,
                                          const std::string& kind_key,
                                          const std::string& path) {
  (void)ns;
  std::lock_guard<std::mutex> lock(mu_);
  auto key = std::make_pair(cfg.url, path);
  auto it = informers_.find(key);
  if (it == informers_.end()) {
    auto inf = std::make_unique<Informer>(cfg, path, kind_key);
    inf->start();
    it = informers_.emplace(key, std::move(inf)).first;
  }
  return *it->second;
}

void InformerRegistry::stop_all() {
  std::lock_guard<std::mutex> lock(mu_);
  for (auto& [_, inf] : informers_) inf->stop();
  informers_.clear();
}

}  // namespace pruner
