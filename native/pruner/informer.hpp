// informer.hpp — watch-based incremental object cache (daemon mode).
//
// The LIST strategy (objcache.hpp) re-fetches whole collections every tick;
// at 1000 pods and a 180 s check interval that is ~6 full-collection LISTs
// per namespace per tick, forever. An informer does what kube controllers
// do instead: one initial LIST capturing the collection resourceVersion,
// then a long-lived `?watch=true&resourceVersion=RV` stream whose
// ADDED/MODIFIED/DELETED/BOOKMARK events keep an in-memory store current —
// steady-state per-tick apiserver traffic becomes O(changes), not
// O(objects).
//
// Lifecycle: the registry (and its watch threads) persists across daemon
// ticks; `ObjectCache` with EvalStrategy::Watch reads through it. A watch
// stream that closes (server timeout, network) is reopened from the last
// seen resourceVersion; a 410 Gone (resourceVersion too old) triggers a
// fresh LIST. Decision semantics match LIST: an object absent from the
// synced store is exactly as gone as a 404.
#pragma once

#include <atomic>
#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <thread>

#include "k8s.hpp"
#include "resources.hpp"

namespace pruner {

class Informer {
public:
  // `collection_path` e.g. "/api/v1/namespaces/ml/pods". The informer owns
  // a dedicated KubeClient (the watch connection is held open).
  Informer(KubeConfig cfg, std::string collection_path, std::string kind_key);
  ~Informer();

  void start();
  void stop();

  // Block until the initial LIST completed (or timeout). Returns synced().
  bool wait_synced(int timeout_ms);
  bool synced() const;

  // nullopt when the object is not in the store (== gone, when synced).
  std::optional<jsn::Value> get(const std::string& name) const;
  // Consistent copy of the whole store (values are COW-shared — ~one
  // refcount bump per object). The per-tick snapshot every decision pass
  // reads from: lookups inside a tick must all see ONE resourceVersion of
  // each object or reference-parity struct-equality dedup double-counts.
  std::map<std::string, jsn::Value> store_snapshot() const;
  size_t size() const;
  uint64_t lists_issued() const { return lists_issued_; }
  uint64_t events_seen() const { return events_seen_; }

private:
  void run();
  bool do_list(KubeClient& kube);
  void watch_once(KubeClient& kube);

  KubeConfig cfg_;
  std::string path_;
  std::string kind_key_;

  mutable std::mutex mu_;
  std::condition_variable cv_;
  std::map<std::string, jsn::Value> store_;
  std::string resource_version_;
  bool synced_ = false;
  std::atomic<bool> stop_{false};
  uint64_t lists_issued_ = 0;
  uint64_t events_seen_ = 0;

  std::thread thread_;
  std::mutex stream_mu_;
  http::BodyStream* active_stream_ = nullptr;  // for shutdown()
};

// Process-wide informer registry keyed by (namespace, kind). Informers are
// created lazily on first use and persist until process exit (or reset).
class InformerRegistry {
public:
  static InformerRegistry& global();

  // Returns a started informer for the collection (creating it if needed).
  Informer& get_or_create(const KubeConfig& cfg, const std::string& ns,
                          const std::string& kind_key, const std::string& path);

  void stop_all();  // also used by tests to reset state

private:
  std::mutex mu_;
  // key: apiserver URL + collection path (different clusters/test fixtures
  // must never share an informer)
  std::map<std::pair<std::string, std::string>, std::unique_ptr<Informer>> informers_;
};

}  // namespace pruner
