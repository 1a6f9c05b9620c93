// threadpool.hpp — persistent worker pool with bounded parallel_for.
//
// The decision engine fans out twice per tick (pod evaluation, root
// actuation). Spawning N std::threads per phase per tick costs ~1-2 ms at
// N=32-128 and inverts the concurrency curve at low apiserver RTT
// (profiles/raw/sweep2.jsonl: conc 10 beat conc 128 at 0 latency). This pool
// spawns its workers once; parallel_for bounds *effective* concurrency per
// call while sharing the same threads.
#pragma once

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace qx {

class ThreadPool {
public:
  // Global pool sized for the biggest useful fan-out; lazily started.
  static ThreadPool& global() {
    static ThreadPool pool(default_size());
    return pool;
  }

  explicit ThreadPool(size_t n_threads) {
    workers_.reserve(n_threads);
    for (size_t i = 0; i < n_threads; i++)
      workers_.emplace_back([this] { worker_loop(); });
  }

  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> lock(mu_);
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& w : workers_) w.join();
  }

  size_t size() const { return workers_.size(); }

  // Run body(i) for i in [0, n) with at most max_par concurrent executions.
  // The calling thread participates, so this works even with a size-0 pool
  // and never deadlocks on pool exhaustion (each submitted driver drains the
  // shared index until empty).
  void parallel_for(size_t n, int max_par, const std::function<void(size_t)>& body) {
    if (n == 0) return;
    size_t par = std::min<size_t>(std::max(max_par, 1), n);
    if (par <= 1) {
      for (size_t i = 0; i < n; i++) body(i);
      return;
    }
    auto ctx = std::make_shared<ForCtx>();
    ctx->n = n;
    ctx->body = &body;
    size_t helpers = std::min(par - 1, workers_.size());
    ctx->active.store(static_cast<int>(helpers) + 1);
    {
      std::lock_guard<std::mutex> lock(mu_);
      for (size_t h = 0; h < helpers; h++) tasks_.push_back([ctx] { drive(*ctx); });
    }
    cv_.notify_all();
    drive(*ctx);  // caller drains too
    std::unique_lock<std::mutex> lock(ctx->done_mu);
    ctx->done_cv.wait(lock, [&] { return ctx->active.load() == 0; });
  }

private:
  struct ForCtx {
    size_t n = 0;
    const std::function<void(size_t)>* body = nullptr;
    std::atomic<size_t> next{0};
    std::atomic<int> active{0};
    std::mutex done_mu;
    std::condition_variable done_cv;
  };

  static void drive(ForCtx& ctx) {
    while (true) {
      size_t i = ctx.next.fetch_add(1, std::memory_order_relaxed);
      if (i >= ctx.n) break;
      (*ctx.body)(i);
    }
    if (ctx.active.fetch_sub(1) == 1) {
      std::lock_guard<std::mutex> lock(ctx.done_mu);
      ctx.done_cv.notify_all();
    }
  }

  // Effective CPU allowance: the cgroup quota when set (containers often
  // cap far below the node's core count — a 256-core box with a 16-CPU
  // quota CFS-throttles oversized thread herds into synchronized ~80 ms
  // stalls), else hardware_concurrency.
  static size_t allowed_cpus() {
    auto read_quota = [](const char* path, const char* fmt) -> double {
      if (FILE* f = std::fopen(path, "r")) {
        long long quota = -1, period = -1;
        char buf[64] = {0};
        if (std::fgets(buf, sizeof buf, f)) {
          if (buf[0] == 'm') {  // "max <period>" = no quota
            std::fclose(f);
            return 0;
          }
          std::sscanf(buf, fmt, &quota, &period);
        }
        std::fclose(f);
        if (quota > 0 && period > 0) return static_cast<double>(quota) / period;
      }
      return 0;
    };
    double cpus = read_quota("/sys/fs/cgroup/cpu.max", "%lld %lld");  // v2
    if (cpus <= 0) {  // v1 pair
      long long q = -1, p = -1;
      if (FILE* f = std::fopen("/sys/fs/cgroup/cpu/cpu.cfs_quota_us", "r")) {
        if (std::fscanf(f, "%lld", &q) != 1) q = -1;
        std::fclose(f);
      }
      if (FILE* f = std::fopen("/sys/fs/cgroup/cpu/cpu.cfs_period_us", "r")) {
        if (std::fscanf(f, "%lld", &p) != 1) p = -1;
        std::fclose(f);
      }
      if (q > 0 && p > 0) cpus = static_cast<double>(q) / p;
    }
    if (cpus <= 0) cpus = static_cast<double>(std::thread::hardware_concurrency());
    return cpus < 1 ? 1 : static_cast<size_t>(cpus);
  }

  static size_t default_size() {
    // The fan-out is I/O-bound (apiserver round-trips), so worker count can
    // exceed the CPU allowance — blocked threads cost no quota — but herds
    // far beyond it burn quota on wakeups and trip CFS throttling. 4× the
    // allowance balances both; override with GPU_PRUNER_POOL_SIZE.
    if (const char* env = std::getenv("GPU_PRUNER_POOL_SIZE"); env && *env) {
      long v = std::strtol(env, nullptr, 10);
      if (v > 0 && v <= 4096) return static_cast<size_t>(v);
    }
    size_t cpus = allowed_cpus();
    size_t size = cpus * 4;
    if (size < 32) size = 32;
    if (size > 256) size = 256;
    return size;
  }

  void worker_loop() {
    while (true) {
      std::function<void()> task;
      {
        std::unique_lock<std::mutex> lock(mu_);
        cv_.wait(lock, [&] { return stop_ || !tasks_.empty(); });
        if (stop_ && tasks_.empty()) return;
        task = std::move(tasks_.front());
        tasks_.pop_front();
      }
      task();
    }
  }

  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::function<void()>> tasks_;
  bool stop_ = false;
  std::vector<std::thread> workers_;
};

}  // namespace qx
