#include "objcache.hpp"

#include <atomic>
#include <mutex>

#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../common/threadpool.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::objcache";

struct ListJob {
  std::string ns;
  std::string kind_key;  // "Pod" or kind_name(kind)
  std::string path;      // collection path
};
}  // namespace

void ObjectCache::prefetch(const std::map<std::string, int>& ns_counts, int concurrency) {
  if (strategy_ == EvalStrategy::PerPodGet) return;

  if (strategy_ == EvalStrategy::Watch) {
    // Persistent informers: created on the first tick that sees a namespace
    // (each LISTs once in its own thread), free from the second tick on.
    auto& reg = InformerRegistry::global();
    for (const auto& [ns, count] : ns_counts) {
      (void)count;
      informers_[{ns, "Pod"}] = &reg.get_or_create(
          kube_.config(), ns, "Pod",
          "/api/v1/namespaces/" + strutil::url_encode(ns) + "/pods");
      for (Kind k : {Kind::ReplicaSet, Kind::Deployment, Kind::StatefulSet, Kind::Notebook,
                     Kind::InferenceService})
        informers_[{ns, kind_name(k)}] =
            &reg.get_or_create(kube_.config(), ns, kind_name(k), collection_path(k, ns));
    }
    // wait for initial syncs (in parallel across informer threads), then
    // SNAPSHOT each synced store into the per-tick cache: every lookup in
    // this tick sees one consistent resourceVersion of each object (the
    // informer keeps moving underneath — our own previous patches stream
    // back as MODIFIED events — and reference-parity struct-equality dedup
    // must not see two versions of one parent). A not-yet-synced informer
    // stays uncached → GET fallback this tick.
    for (auto& [key, inf] : informers_) {
      if (!inf->wait_synced(10000)) {
        LOGW(TARGET, "informer for " + key.first + "/" + key.second +
                         " not synced; falling back to GETs this tick");
        continue;
      }
      cache_[key.first].by_kind[key.second] = inf->store_snapshot();
    }
    return;
  }

  std::vector<std::string> namespaces;
  for (const auto& [ns, count] : ns_counts) {
    if (strategy_ == EvalStrategy::NamespaceList || count >= auto_threshold_)
      namespaces.push_back(ns);
  }
  if (namespaces.empty()) return;

  std::vector<ListJob> jobs;
  for (const auto& ns : namespaces) {
    jobs.push_back({ns, "Pod", "/api/v1/namespaces/" + strutil::url_encode(ns) + "/pods"});
    for (Kind k : {Kind::ReplicaSet, Kind::Deployment, Kind::StatefulSet, Kind::Notebook,
                   Kind::InferenceService})
      jobs.push_back({ns, kind_name(k), collection_path(k, ns)});
  }

  std::mutex mu;
  qx::ThreadPool::global().parallel_for(
      jobs.size(), std::max(concurrency, 1), [&](size_t i) {
        const ListJob& job = jobs[i];
        try {
          auto list = kube_.get_opt(job.path);
          if (!list) return;  // CRD/API group absent → GET fallback for this kind
          const jsn::Value& items = list->get("items");
          if (!items.is_array()) return;
          std::map<std::string, jsn::Value> by_name;
          for (const auto& obj : items.arr()) {
            std::string name = obj.at({"metadata", "name"}).as_string();
            if (!name.empty()) by_name.emplace(std::move(name), obj);
          }
          std::lock_guard<std::mutex> lock(mu);
          cache_[job.ns].by_kind[job.kind_key] = std::move(by_name);
          lists_issued_++;
        } catch (const std::exception& e) {
          // leave the kind uncached → lookups fall through to GETs
          LOGW(TARGET, "LIST " + job.path + " failed (" + e.what() +
                           "), falling back to GETs");
        }
      });
  LOGD(TARGET, "prefetched " + std::to_string(lists_issued_) + " collections across " +
                   std::to_string(namespaces.size()) + " namespaces");
}

std::optional<jsn::Value> ObjectCache::get_pod(const std::string& ns,
                                               const std::string& name) {
  if (auto nit = cache_.find(ns); nit != cache_.end()) {
    if (auto kit = nit->second.by_kind.find("Pod"); kit != nit->second.by_kind.end()) {
      auto oit = kit->second.find(name);
      if (oit == kit->second.end()) return std::nullopt;  // fresh LIST: pod gone
      return oit->second;
    }
  }
  return kube_.get_pod(ns, name);
}

std::optional<jsn::Value> ObjectCache::get_object(Kind kind, const std::string& ns,
                                                  const std::string& name) {
  if (auto nit = cache_.find(ns); nit != cache_.end()) {
    auto kit = nit->second.by_kind.find(kind_name(kind));
    if (kit != nit->second.by_kind.end()) {
      auto oit = kit->second.find(name);
      if (oit == kit->second.end()) return std::nullopt;
      return oit->second;
    }
  }
  return kube_.get_object(kind, ns, name);
}

}  // namespace pruner
