#include "synthbench.hpp"

#include <time.h>

#include <algorithm>
#include <chrono>
#include <cstdio>

#include "../common/strutil.hpp"
#include "../common/tsan_compat.hpp"
#include "resources.hpp"

namespace pruner {

namespace {

std::string uid_for(const std::string& kind, const std::string& ns, const std::string& name) {
  return "uid-" + kind + "-" + ns + "-" + name;
}

jsn::Value meta(const std::string& name, const std::string& ns, const std::string& kind,
                double age_s = 7200.0) {
  jsn::Value m = jsn::Value::object();
  m["name"] = name;
  m["namespace"] = ns;
  m["uid"] = uid_for(kind, ns, name);
  m["resourceVersion"] = "1";
  // creationTimestamp well before any lookback window
  time_t t = ::time(nullptr) - static_cast<time_t>(age_s);
  struct tm tm {};
  gmtime_r(&t, &tm);
  char buf[40];
  std::snprintf(buf, sizeof buf, "%04d-%02d-%02dT%02d:%02d:%02dZ", tm.tm_year + 1900,
                tm.tm_mon + 1, tm.tm_mday, tm.tm_hour, tm.tm_min, tm.tm_sec);
  m["creationTimestamp"] = buf;
  return m;
}

}  // namespace

SyntheticBackend::SyntheticBackend(SynthOptions opts) : opts_(std::move(opts)) {
  build_cluster();
}

SyntheticBackend::~SyntheticBackend() { stop(); }

void SyntheticBackend::build_cluster() {
  // Mirrors gpu_pruner_amd/fixtures/synth.py: parents rotate
  // Deployment → Notebook-owned StatefulSet → InferenceService.
  n_parents_ = (opts_.n_pods + opts_.pods_per_parent - 1) / opts_.pods_per_parent;
  struct Parent {
    std::string owner_kind;  // pod's ownerReference kind ("" = KServe label)
    std::string owner_name;
    std::string ns;
  };
  std::vector<Parent> parents;
  for (int p = 0; p < n_parents_; p++) {
    std::string ns = "ml-team-" + std::to_string(p % opts_.n_namespaces);
    int flavor = p % 3;
    if (flavor == 0) {
      std::string dep = "dep-" + std::to_string(p);
      std::string rs = dep + "-rs";
      jsn::Value d = jsn::Value::object();
      d["apiVersion"] = "apps/v1";
      d["kind"] = "Deployment";
      d["metadata"] = meta(dep, ns, "Deployment");
      d["spec"]["replicas"] = 1;
      objects_["Deployment"][ns][dep] = {d, d.dump()};
      jsn::Value r = jsn::Value::object();
      r["apiVersion"] = "apps/v1";
      r["kind"] = "ReplicaSet";
      r["metadata"] = meta(rs, ns, "ReplicaSet");
      jsn::Value owner = jsn::Value::object();
      owner["apiVersion"] = "apps/v1";
      owner["kind"] = "Deployment";
      owner["name"] = dep;
      owner["uid"] = uid_for("Deployment", ns, dep);
      r["metadata"]["ownerReferences"] = jsn::Value(jsn::Array{owner});
      r["spec"]["replicas"] = 1;
      objects_["ReplicaSet"][ns][rs] = {r, r.dump()};
      parents.push_back({"ReplicaSet", rs, ns});
    } else if (flavor == 1) {
      std::string nb = "nb-" + std::to_string(p);
      std::string ss = nb + "-ss";
      jsn::Value n = jsn::Value::object();
      n["apiVersion"] = "kubeflow.org/v1";
      n["kind"] = "Notebook";
      n["metadata"] = meta(nb, ns, "Notebook");
      n["spec"]["template"] = nullptr;
      objects_["Notebook"][ns][nb] = {n, n.dump()};
      jsn::Value s = jsn::Value::object();
      s["apiVersion"] = "apps/v1";
      s["kind"] = "StatefulSet";
      s["metadata"] = meta(ss, ns, "StatefulSet");
      jsn::Value owner = jsn::Value::object();
      owner["apiVersion"] = "kubeflow.org/v1";
      owner["kind"] = "Notebook";
      owner["name"] = nb;
      owner["uid"] = uid_for("Notebook", ns, nb);
      s["metadata"]["ownerReferences"] = jsn::Value(jsn::Array{owner});
      s["spec"]["replicas"] = 1;
      objects_["StatefulSet"][ns][ss] = {s, s.dump()};
      parents.push_back({"StatefulSet", ss, ns});
    } else {
      std::string isvc = "isvc-" + std::to_string(p);
      jsn::Value v = jsn::Value::object();
      v["apiVersion"] = "serving.kserve.io/v1beta1";
      v["kind"] = "InferenceService";
      v["metadata"] = meta(isvc, ns, "InferenceService");
      v["spec"]["predictor"]["minReplicas"] = 1;
      objects_["InferenceService"][ns][isvc] = {v, v.dump()};
      parents.push_back({"", isvc, ns});
    }
  }

  // Pods + the pre-rendered Prometheus result vector (value patched at
  // serve time from series_value_).
  std::string series;
  series.reserve(static_cast<size_t>(opts_.n_pods) * opts_.gpus_per_pod * 256);
  series += "[";
  bool first = true;
  for (int i = 0; i < opts_.n_pods; i++) {
    const Parent& par = parents[static_cast<size_t>(i) / opts_.pods_per_parent];
    std::string pod = "pod-" + std::to_string(i);
    jsn::Value p = jsn::Value::object();
    p["apiVersion"] = "v1";
    p["kind"] = "Pod";
    p["metadata"] = meta(pod, par.ns, "Pod");
    if (par.owner_kind.empty()) {
      p["metadata"]["labels"]["serving.kserve.io/inferenceservice"] = par.owner_name;
    } else {
      jsn::Value owner = jsn::Value::object();
      owner["apiVersion"] = "apps/v1";
      owner["kind"] = par.owner_kind;
      owner["name"] = par.owner_name;
      owner["uid"] = uid_for(par.owner_kind, par.ns, par.owner_name);
      p["metadata"]["ownerReferences"] = jsn::Value(jsn::Array{owner});
    }
    p["status"]["phase"] = "Running";
    objects_["Pod"][par.ns][pod] = {p, p.dump()};

    for (int g = 0; g < opts_.gpus_per_pod; g++) {
      if (!first) series += ",";
      first = false;
      series += "{\"metric\":{\"Hostname\":\"mi355-node-0\",\"exported_pod\":\"" + pod +
                "\",\"exported_namespace\":\"" + par.ns +
                "\",\"exported_container\":\"main\",\"gpu\":\"" + std::to_string(g) +
                "\",\"modelName\":\"" + opts_.model_name +
                "\",\"node_type\":\"amd-mi355x\"},\"value\":[1700000000,\"%V%\"]}";
    }
  }
  series += "]";
  series_json_zero_ = std::move(series);
}

void SyntheticBackend::start() {
  prom_server_ = std::make_unique<http::Server>(
      "127.0.0.1", 0, [this](const http::ServerRequest& r) { return handle_prom(r); });
  k8s_server_ = std::make_unique<http::Server>(
      "127.0.0.1", 0, [this](const http::ServerRequest& r) { return handle_k8s(r); });
  prom_server_->start();
  k8s_server_->start();
}

void SyntheticBackend::stop() {
  closing_.store(true);
  {
    std::lock_guard<std::mutex> lock(mu_);  // pair with streamer waits
  }
  event_cv_.notify_all();
  if (prom_server_) prom_server_->stop();
  if (k8s_server_) k8s_server_->stop();
}

std::string SyntheticBackend::prom_url() const {
  return "http://127.0.0.1:" + std::to_string(prom_server_->port());
}

std::string SyntheticBackend::k8s_url() const {
  return "http://127.0.0.1:" + std::to_string(k8s_server_->port());
}

http::ServerResponse SyntheticBackend::handle_prom(const http::ServerRequest& req) {
  requests_.fetch_add(1, std::memory_order_relaxed);
  http::ServerResponse resp;
  if (req.path.find("/api/v1/query") == std::string::npos) {
    resp.status = 404;
    resp.body = "{}";
    return resp;
  }
  // Emulate the idle query's `== 0` predicate: Prometheus only returns
  // series whose peak activity over the window is zero, so a non-zero
  // activity value (busy GPU) yields an empty vector — no candidates.
  double v = series_value_.load();
  if (v != 0.0) {
    resp.content_type = "application/json";
    resp.body = "{\"status\":\"success\",\"data\":{\"resultType\":\"vector\",\"result\":[]}}";
    return resp;
  }
  char val[32];
  std::snprintf(val, sizeof val, "%g", v);
  std::string result = series_json_zero_;
  // patch the placeholder value into every series
  std::string out;
  out.reserve(result.size());
  size_t pos = 0;
  while (true) {
    size_t ph = result.find("%V%", pos);
    if (ph == std::string::npos) {
      out.append(result, pos, std::string::npos);
      break;
    }
    out.append(result, pos, ph - pos);
    out += val;
    pos = ph + 3;
  }
  resp.content_type = "application/json";
  resp.body = "{\"status\":\"success\",\"data\":{\"resultType\":\"vector\",\"result\":" + out +
              "}}";
  return resp;
}

http::ServerResponse SyntheticBackend::handle_k8s(const http::ServerRequest& req) {
  requests_.fetch_add(1, std::memory_order_relaxed);
  if (opts_.latency_us > 0) {
    struct timespec ts {0, opts_.latency_us * 1000L};
    nanosleep(&ts, nullptr);
  }
  http::ServerResponse resp;
  resp.content_type = "application/json";

  // parse /api(s)/<group...>/namespaces/<ns>/<plural>/<name>[/scale]
  auto parts = strutil::split(req.path, '/');
  // e.g. ["", "apis", "apps", "v1", "namespaces", ns, plural, name, ("scale")]
  std::string ns, plural, name;
  bool is_scale = false;
  for (size_t i = 0; i + 1 < parts.size(); i++) {
    if (parts[i] == "namespaces") {
      ns = parts[i + 1];
      if (i + 2 < parts.size()) plural = parts[i + 2];
      if (i + 3 < parts.size()) name = parts[i + 3];
      if (i + 4 < parts.size() && parts[i + 4] == "scale") is_scale = true;
      break;
    }
  }
  static const std::map<std::string, std::string> plural_kind = {
      {"pods", "Pod"},           {"deployments", "Deployment"},
      {"replicasets", "ReplicaSet"}, {"statefulsets", "StatefulSet"},
      {"notebooks", "Notebook"}, {"inferenceservices", "InferenceService"},
      {"events", "Event"}};
  auto pk = plural_kind.find(plural);
  if (pk == plural_kind.end()) {
    resp.status = 404;
    resp.body = "{\"kind\":\"Status\",\"code\":404}";
    return resp;
  }
  const std::string& kind = pk->second;

  if (req.method == "POST" && kind == "Event") {
    events_posted_.fetch_add(1, std::memory_order_relaxed);
    resp.status = 201;
    resp.body = req.body;
    return resp;
  }

  // Kubernetes watch: chunked stream of events after ?resourceVersion,
  // served by the generic streamer path (informer / --eval-strategy watch).
  if (req.method == "GET" && name.empty() &&
      req.query.find("watch=true") != std::string::npos) {
    uint64_t since = 0;
    if (size_t p = req.query.find("resourceVersion="); p != std::string::npos)
      since = std::strtoull(req.query.c_str() + p + 16, nullptr, 10);
    double timeout_s = 30.0;
    if (size_t p = req.query.find("timeoutSeconds="); p != std::string::npos)
      timeout_s = std::strtod(req.query.c_str() + p + 15, nullptr);
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::duration_cast<std::chrono::steady_clock::duration>(
                        std::chrono::duration<double>(timeout_s));
    watch_streams_.fetch_add(1, std::memory_order_relaxed);
    bool bookmark_sent = false;
    resp.streamer = [this, kind, ns, since, deadline,
                     bookmark_sent](std::string* chunk) mutable {
      std::unique_lock<std::mutex> lock(mu_);
      while (true) {
        // log rvs are strictly increasing: binary-search the resume point
        auto it = std::lower_bound(watch_log_.begin(), watch_log_.end(), since + 1,
                                   [](const WatchEvent& e, uint64_t rv) { return e.rv < rv; });
        for (; it != watch_log_.end(); ++it) {
          if (it->kind == kind && it->ns == ns) {
            since = it->rv;
            *chunk = it->line + "\n";
            return true;
          }
          since = it->rv;
        }
        if (bookmark_sent) return false;
        if (closing_.load() || std::chrono::steady_clock::now() >= deadline) {
          *chunk = "{\"type\":\"BOOKMARK\",\"object\":{\"kind\":\"" + kind +
                   "\",\"metadata\":{\"resourceVersion\":\"" + std::to_string(rv_) +
                   "\"}}}\n";
          bookmark_sent = true;
          return true;  // deliver the bookmark; next call ends the stream
        }
        qx::cv_wait_for(event_cv_, lock, std::chrono::milliseconds(25));
      }
    };
    return resp;
  }

  // parse any PATCH body BEFORE taking the global lock: at 1000-pod scale
  // the actuation phase sends ~500 patches and the fixture must not
  // serialize the engine on its own mutex more than necessary
  jsn::Value patch;
  if (req.method == "PATCH") patch = jsn::parse(req.body);

  std::lock_guard<std::mutex> lock(mu_);
  auto kit = objects_.find(kind);
  // namespaced collection LIST (no object name)
  if (req.method == "GET" && name.empty()) {
    std::string body = "{\"kind\":\"" + kind + "List\",\"metadata\":{\"resourceVersion\":\"" +
                       std::to_string(rv_) + "\"},\"items\":[";
    bool first = true;
    if (kit != objects_.end()) {
      auto nit = kit->second.find(ns);
      if (nit != kit->second.end()) {
        for (auto& [oname, stored] : nit->second) {
          if (!first) body += ",";
          first = false;
          if (stored.cached_dump.empty()) stored.cached_dump = stored.obj.dump();
          body += stored.cached_dump;
        }
      }
    }
    body += "]}";
    resp.body = std::move(body);
    return resp;
  }
  StoredObject* obj = nullptr;
  if (kit != objects_.end()) {
    auto nit = kit->second.find(ns);
    if (nit != kit->second.end()) {
      auto oit = nit->second.find(name);
      if (oit != nit->second.end()) obj = &oit->second;
    }
  }
  if (!obj) {
    resp.status = 404;
    resp.body = "{\"kind\":\"Status\",\"code\":404}";
    return resp;
  }
  if (req.method == "GET") {
    if (obj->cached_dump.empty()) obj->cached_dump = obj->obj.dump();
    resp.body = obj->cached_dump;
    return resp;
  }
  if (req.method == "PATCH") {
    auto record_modified = [&] {
      rv_++;
      obj->obj["metadata"]["resourceVersion"] = std::to_string(rv_);
      obj->cached_dump.clear();
      obj->cached_dump = obj->obj.dump();
      watch_log_.push_back(
          {rv_, kind, ns, "{\"type\":\"MODIFIED\",\"object\":" + obj->cached_dump + "}"});
      while (watch_log_.size() > 20000) watch_log_.pop_front();
      // NO per-event notify: with ~500 patches per bench tick a notify_all
      // here wakes every watch streamer 500x24 times per tick (measured
      // scheduler churn under the CPU quota). Streamers poll at 25 ms —
      // bounded delivery latency, which the static bench cluster and the
      // informer semantics tolerate.
    };
    if (is_scale) {
      scale_patches_.fetch_add(1, std::memory_order_relaxed);
      obj->obj["spec"]["replicas"] = patch.at({"spec", "replicas"});
      record_modified();
      resp.body = "{\"kind\":\"Scale\",\"spec\":" + patch.get("spec").dump() + "}";
      return resp;
    }
    scale_patches_.fetch_add(1, std::memory_order_relaxed);
    obj->obj.merge_patch(patch);
    record_modified();
    resp.body = obj->obj.dump();
    return resp;
  }
  resp.status = 405;
  resp.body = "{}";
  return resp;
}

}  // namespace pruner
