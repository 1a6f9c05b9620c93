#include "leader.hpp"

#include <chrono>
#include <ctime>

#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../common/tsan_compat.hpp"

namespace pruner {

namespace {
constexpr const char* TARGET = "pruner::leader";

double now_unix() {
  return std::chrono::duration<double>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}
}  // namespace

LeaderElector::LeaderElector(KubeConfig cfg, std::string ns, std::string lease_name,
                             std::string identity, int lease_duration_s,
                             int renew_period_s)
    : cfg_(std::move(cfg)), ns_(std::move(ns)), name_(std::move(lease_name)),
      identity_(std::move(identity)), lease_duration_s_(lease_duration_s),
      renew_period_s_(renew_period_s) {}

LeaderElector::~LeaderElector() { stop(); }

std::string LeaderElector::lease_path() const {
  return "/apis/coordination.k8s.io/v1/namespaces/" + strutil::url_encode(ns_) +
         "/leases/" + strutil::url_encode(name_);
}

jsn::Value LeaderElector::lease_body(const std::string& holder, const std::string& rv,
                                     const std::string& acquire_time,
                                     int64_t transitions) const {
  jsn::Value l = jsn::Value::object();
  l["apiVersion"] = "coordination.k8s.io/v1";
  l["kind"] = "Lease";
  l["metadata"]["name"] = name_;
  l["metadata"]["namespace"] = ns_;
  if (!rv.empty()) l["metadata"]["resourceVersion"] = rv;
  l["spec"]["holderIdentity"] = holder;
  l["spec"]["leaseDurationSeconds"] = lease_duration_s_;
  l["spec"]["acquireTime"] = acquire_time;
  l["spec"]["renewTime"] = strutil::rfc3339_micro_now();
  l["spec"]["leaseTransitions"] = transitions;
  return l;
}

bool LeaderElector::try_acquire_or_renew() {
  try {
    KubeClient kube(cfg_);
    auto lease = kube.get_opt(lease_path());
    if (!lease) {
      // no lease yet: race to create it (409 AlreadyExists = lost)
      try {
        kube.create("/apis/coordination.k8s.io/v1/namespaces/" +
                        strutil::url_encode(ns_) + "/leases",
                    lease_body(identity_, "", strutil::rfc3339_micro_now(), 0));
        leading_.store(true);
        LOGI(TARGET, "Acquired leadership (created lease " + ns_ + "/" + name_ + ")");
        return true;
      } catch (const KubeError& e) {
        if (e.status == 409) {
          leading_.store(false);
          return false;  // another replica created it first
        }
        throw;
      }
    }

    std::string holder = lease->at({"spec", "holderIdentity"}).as_string_or("");
    std::string rv = lease->at({"metadata", "resourceVersion"}).as_string_or("");
    std::string acquire_time =
        lease->at({"spec", "acquireTime"}).as_string_or(strutil::rfc3339_micro_now());
    int64_t transitions = lease->at({"spec", "leaseTransitions"}).as_int(0);
    double renew_s = 0.0;
    std::string renew = lease->at({"spec", "renewTime"}).as_string_or("");
    bool have_renew = strutil::parse_rfc3339(renew, &renew_s);
    int64_t duration =
        lease->at({"spec", "leaseDurationSeconds"}).as_int(lease_duration_s_);
    bool expired = !have_renew || now_unix() > renew_s + static_cast<double>(duration);

    if (holder == identity_) {
      // renew our own lease (rv-fenced: someone force-updating or deleting
      // the Lease between our GET and this PUT means our claim is stale —
      // treat it as lost and re-contest next tick, don't keep acting)
      try {
        kube.replace(lease_path(), lease_body(identity_, rv, acquire_time, transitions));
      } catch (const KubeError& e) {
        if (e.status == 409 || e.status == 404) {
          if (leading_.exchange(false))
            LOGW(TARGET, "Lost leadership of " + ns_ + "/" + name_ + " (lease " +
                             (e.status == 409 ? "conflict" : "deleted") +
                             " during renew)");
          return false;
        }
        throw;
      }
      if (!leading_.exchange(true))
        LOGI(TARGET, "Re-acquired leadership of " + ns_ + "/" + name_);
      return true;
    }
    if (!holder.empty() && !expired) {
      if (leading_.exchange(false))
        LOGW(TARGET, "Lost leadership to \"" + holder + "\"");
      return false;  // someone else holds a live lease
    }
    // vacant or expired: take over, fenced by resourceVersion (409 = lost race)
    try {
      kube.replace(lease_path(), lease_body(identity_, rv,
                                            strutil::rfc3339_micro_now(),
                                            transitions + 1));
      leading_.store(true);
      LOGI(TARGET, "Acquired leadership of " + ns_ + "/" + name_ +
                       (holder.empty() ? " (released lease)"
                                       : " (expired holder \"" + holder + "\")"));
      return true;
    } catch (const KubeError& e) {
      if (e.status == 409) {
        leading_.store(false);
        return false;
      }
      throw;
    }
  } catch (const std::exception& e) {
    // apiserver unreachable: FAIL SAFE — do not keep acting on a lease we
    // can no longer renew (another replica may take over meanwhile)
    if (leading_.exchange(false))
      LOGW(TARGET, std::string("Dropping leadership (lease unreachable: ") + e.what() +
                       ")");
    return false;
  }
}

void LeaderElector::start() {
  if (thread_.joinable()) return;
  stop_.store(false);
  try_acquire_or_renew();  // synchronous first attempt: fast startup verdict
  thread_ = std::thread([this] { run(); });
}

void LeaderElector::run() {
  while (!stop_.load()) {
    {
      std::unique_lock<std::mutex> lock(mu_);
      qx::cv_wait_for(cv_, lock, std::chrono::seconds(renew_period_s_),
                      [this] { return stop_.load(); });
    }
    if (stop_.load()) return;
    try_acquire_or_renew();
  }
}

void LeaderElector::stop() {
  if (!thread_.joinable()) {
    stop_.store(true);
    return;
  }
  stop_.store(true);
  cv_.notify_all();
  thread_.join();
  if (leading_.exchange(false)) {
    // release: blank the holder so a standby can take over immediately
    try {
      KubeClient kube(cfg_);
      auto lease = kube.get_opt(lease_path());
      if (lease && lease->at({"spec", "holderIdentity"}).as_string_or("") == identity_) {
        std::string rv = lease->at({"metadata", "resourceVersion"}).as_string_or("");
        int64_t transitions = lease->at({"spec", "leaseTransitions"}).as_int(0);
        kube.replace(lease_path(),
                     lease_body("", rv, strutil::rfc3339_micro_now(), transitions));
        LOGI(TARGET, "Released lease " + ns_ + "/" + name_);
      }
    } catch (const std::exception& e) {
      LOGW(TARGET, std::string("Failed to release lease on shutdown: ") + e.what());
    }
  }
}

}  // namespace pruner
