// leader.hpp — Kubernetes Lease-based leader election (--leader-elect).
//
// The reference runs a single replica and relies on `restartPolicy: Always`
// (hack/deployment.yaml:38); running two replicas would double-cull and
// double-post Events. This optional elector closes that gap the standard
// Kubernetes way: a coordination.k8s.io/v1 Lease named `gpu-pruner` in the
// daemon's namespace, acquired/renewed with resourceVersion-preconditioned
// PUTs (409 Conflict = lost the race). Only the holder runs decision ticks;
// non-holders keep re-checking and take over once
// renewTime + leaseDurationSeconds has passed. On clean shutdown the holder
// releases the lease so failover is immediate.
//
// Semantics mirror client-go's leaderelection/resourcelock (spec fields
// holderIdentity / leaseDurationSeconds / acquireTime / renewTime /
// leaseTransitions), interoperable with kubectl `describe lease`.
#pragma once

#include <atomic>
#include <condition_variable>
#include <mutex>
#include <string>
#include <thread>

#include "k8s.hpp"

namespace pruner {

class LeaderElector {
public:
  // identity: this replica's holderIdentity (POD_NAME, or hostname-pid).
  LeaderElector(KubeConfig cfg, std::string ns, std::string lease_name,
                std::string identity, int lease_duration_s = 15,
                int renew_period_s = 5);
  ~LeaderElector();

  void start();
  void stop();  // releases the lease when currently leading (best effort)

  bool is_leader() const { return leading_.load(std::memory_order_relaxed); }
  const std::string& identity() const { return identity_; }

  // One acquire-or-renew attempt (also used by the background thread;
  // public for tests). Returns the new leadership state.
  bool try_acquire_or_renew();

private:
  void run();
  std::string lease_path() const;
  jsn::Value lease_body(const std::string& holder, const std::string& rv,
                        const std::string& acquire_time, int64_t transitions) const;

  KubeConfig cfg_;
  std::string ns_;
  std::string name_;
  std::string identity_;
  int lease_duration_s_;
  int renew_period_s_;

  std::atomic<bool> leading_{false};
  std::atomic<bool> stop_{false};
  std::mutex mu_;
  std::condition_variable cv_;
  std::thread thread_;
};

}  // namespace pruner
