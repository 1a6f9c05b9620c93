// sampler.hpp — first-party ROCm/gfx950 GPU activity sampler.
//
// This is the MI355X-native replacement for the NVIDIA dcgm-exporter the
// reference depends on: it enumerates the node's GPUs
// through rocm_smi_lib and samples, per device:
//   * instantaneous busy percent (GRBM-busy-derived, 0-100),
//   * a true *windowed* graphics-activity ratio in [0,1] — time-weighted
//     between polls, and cross-checked against the firmware's accumulated
//     gfx_activity_acc counter from the gpu_metrics table when available —
//     this is the DCGM_FI_PROF_GR_ENGINE_ACTIVE analog, and it must read
//     exactly 0.0 on a truly idle device or the pruner's `== 0` PromQL
//     predicate silently never fires (SURVEY.md §7 "Counter semantics
//     parity"),
//   * socket power (W), VRAM used/total, memory-controller activity,
//     edge temperature, gfx clock,
//   * per-device read health: consecutive SMU/sysfs read failures and the
//     age of the last successful read. A device whose reads fail must NEVER
//     look idle — the exporter drops its activity series instead (so the
//     `== 0` predicate cannot fire from a dead read path) and publishes
//     mi355_sampler_healthy / last-good-read-age series for alerting.
//
// The sampler owns a background polling thread (default 1 s cadence); the
// /metrics handler renders the latest snapshot. Scrapes are IDEMPOTENT:
// the windowed ratio is computed over a fixed sliding window (default 30 s)
// from a retained poll history, so any number of concurrent scrapers (HA
// Prometheus pairs, debug curls) observe the same value and none can
// truncate the window another scraper sees.
#pragma once

#include <cstdint>
#include <deque>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace exporter {

struct DeviceSample {
  uint32_t index = 0;
  std::string model_name;       // e.g. "AMD Instinct MI355X"
  std::string unique_id;        // hex unique id (or serial)
  std::string pci_bdf;          // 0000:0a:00.0
  uint32_t drm_render_minor = 0;  // /dev/dri/renderD<minor>
  uint64_t kfd_gpu_id = 0;        // KFD topology gpu_id (pod attribution key)

  double busy_percent = 0.0;       // 0-100, instantaneous
  double gr_engine_active = 0.0;   // 0-1, ratio over the sliding window
  double mem_busy_percent = 0.0;   // 0-100 memory-controller activity
  double power_w = 0.0;
  double vram_used_b = 0.0;
  double vram_total_b = 0.0;
  double temp_edge_c = 0.0;
  double gfx_clock_mhz = 0.0;
  double energy_j = 0.0;           // accumulated energy (J) when supported
  bool metrics_table_ok = false;   // gpu_metrics table was readable
  uint64_t gfx_activity_acc = 0;   // raw accumulated activity counter
  uint64_t firmware_timestamp = 0; // ns on gfx950 (header says 10 ns)

  // Read health: a failing SMU must not masquerade as idle.
  bool read_ok = true;             // this poll's activity read succeeded
  bool healthy = true;             // < failure threshold of consecutive bad polls
  double staleness_s = 0.0;        // age of the last successful activity read

  // xGMI topology/traffic (topology awareness only — this
  // daemon moves no tensors): per-device link width/speed and the firmware's
  // accumulated per-link transfer counters, summed (KiB units per rocm_smi).
  double xgmi_link_width = 0.0;
  double xgmi_link_speed = 0.0;
  double xgmi_read_kb = 0.0;
  double xgmi_write_kb = 0.0;
};

class SamplerError : public std::runtime_error {
public:
  using std::runtime_error::runtime_error;
};

// ActivityWindow — pure sliding-window integrator (no I/O; unit-tested on
// CPU via _gpumon.ActivityWindow). Each poll appends a segment
// [prev_t, t) carrying the activity ratio observed over that span, or an
// "unknown" segment when the read failed. ratio(now, window) integrates the
// KNOWN segments inside [now-window, now]; unknown time contributes to
// neither numerator nor denominator, so a failing device degrades to "no
// data" (and the health flag drops) instead of decaying toward a false 0.
struct ActivityWindow {
  struct Segment {
    double t0 = 0.0, t1 = 0.0;
    double ratio = 0.0;
    bool known = false;
  };

  // Observation at time t (monotonic seconds): activity ratio since the
  // previous observation, or unknown if the read failed.
  void add(double t, double ratio, bool known) {
    if (has_last_ && t > last_t_) segments_.push_back({last_t_, t, ratio, known});
    last_t_ = t;
    has_last_ = true;
    // retain enough history for the longest supported window (+slack)
    while (!segments_.empty() && segments_.front().t1 < t - retention_s_)
      segments_.pop_front();
  }

  // Time-weighted mean over known segments intersecting [now-window_s, now].
  // Returns 0.0 with *known_s == 0 when no known time is in the window.
  double ratio(double now, double window_s, double* known_s = nullptr) const {
    double lo = now - window_s;
    double busy = 0.0, known = 0.0;
    for (const auto& s : segments_) {
      if (s.t1 <= lo || !s.known) continue;
      double a = s.t0 < lo ? lo : s.t0;
      double b = s.t1 > now ? now : s.t1;
      if (b <= a) continue;
      busy += s.ratio * (b - a);
      known += b - a;
    }
    if (known_s) *known_s = known;
    return known > 0.0 ? busy / known : 0.0;
  }

  void set_retention(double s) { retention_s_ = s; }
  size_t size() const { return segments_.size(); }

private:
  std::deque<Segment> segments_;
  double last_t_ = 0.0;
  bool has_last_ = false;
  double retention_s_ = 120.0;
};

class Sampler {
public:
  // poll_interval_ms: cadence of the background poll used to integrate the
  // windowed activity ratio. idle_epsilon: windowed ratios below this are
  // reported as exactly 0.0 — idle MI355X silicon emits sporadic
  // firmware/driver housekeeping blips of ~0.02-0.03% activity
  // (profiles/raw/winsem_debug.log) which would otherwise poison the
  // culler's `== 0` predicate over long windows with false negatives (the
  // same trap NVIDIA's DCGM PROF metrics have; divergence documented in
  // PARITY.md — set 0 for strict drop-in behavior). window_s: length of the
  // sliding activity window every scrape reads (scrape-idempotent; 30 s
  // matches a typical Prometheus scrape interval).
  explicit Sampler(int poll_interval_ms = 1000, double idle_epsilon = 0.005,
                   double window_s = 30.0);
  ~Sampler();

  // Initialize rocm_smi and enumerate devices. Throws SamplerError when the
  // ROCm stack / amdgpu driver is unavailable (no silent fallback: on a GPU
  // host a failure here must be loud).
  void init();
  void start();  // launch the polling thread
  void stop();

  size_t device_count() const { return static_cast<size_t>(n_devices_); }

  // Read-only, idempotent snapshot of the latest samples; the windowed
  // activity ratio is evaluated over the trailing window_s at call time.
  std::vector<DeviceSample> snapshot();

  // Poll once synchronously (also used by the background thread).
  void poll_once();

  // Consecutive failed activity reads before a device is marked unhealthy.
  static constexpr int kUnhealthyAfter = 3;

private:
  struct DevState {
    DeviceSample last;
    ActivityWindow win;
    double win_prev_t = 0.0;      // previous poll time (segment start)
    uint64_t prev_acc = 0;        // gfx_activity_acc at previous poll
    uint64_t prev_fw_ts = 0;      // firmware_timestamp at previous poll
    bool have_prev_acc = false;
    int consecutive_failures = 0;
    double last_good_monotonic = 0.0;
  };

  void poll_device(uint32_t i);

  int poll_interval_ms_;
  double idle_epsilon_;
  double window_s_;
  uint32_t n_devices_ = 0;
  bool initialized_ = false;
  // poll_mu_ serializes pollers; mu_ guards only the stored state so
  // snapshot() never waits behind a slow firmware (SMU) read — concurrent
  // rsmi gpu_metrics reads from several processes can stall ~80 ms
  // (observed under the 4-rank bench) and must not block consumers.
  std::mutex poll_mu_;
  mutable std::mutex mu_;
  std::vector<DevState> devices_;
  std::thread poller_;
  bool running_ = false;
};

}  // namespace exporter
