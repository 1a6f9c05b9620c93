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
//     edge temperature, gfx clock.
//
// The sampler owns a background polling thread (default 1 s cadence); the
// /metrics handler renders the latest snapshot.
#pragma once

#include <cstdint>
#include <mutex>
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
  double gr_engine_active = 0.0;   // 0-1, windowed ratio since previous scrape
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

class Sampler {
public:
  // poll_interval_ms: cadence of the background poll used to integrate the
  // windowed activity ratio. idle_epsilon: windowed ratios below this are
  // reported as exactly 0.0 — idle MI355X silicon emits sporadic
  // firmware/driver housekeeping blips of ~0.02-0.03% activity
  // (profiles/raw/winsem_debug.log) which would otherwise poison the
  // culler's `== 0` predicate over long windows with false negatives (the
  // same trap NVIDIA's DCGM PROF metrics have). 0.5% is far above the
  // noise floor and far below any real workload.
  explicit Sampler(int poll_interval_ms = 1000, double idle_epsilon = 0.005);
  ~Sampler();

  // Initialize rocm_smi and enumerate devices. Throws SamplerError when the
  // ROCm stack / amdgpu driver is unavailable (no silent fallback: on a GPU
  // host a failure here must be loud).
  void init();
  void start();  // launch the polling thread
  void stop();

  size_t device_count() const { return static_cast<size_t>(n_devices_); }

  // Snapshot of the latest samples; `reset_window` folds the activity
  // integrator (scrape semantics: each scrape reads the ratio since the
  // previous scrape).
  std::vector<DeviceSample> snapshot(bool reset_window = false);

  // Poll once synchronously (also used by the background thread).
  void poll_once();

private:
  struct DevState {
    DeviceSample last;
    // window integrator: busy-seconds and wall-seconds since last scrape
    double busy_seconds = 0.0;
    double wall_seconds = 0.0;
    uint64_t prev_acc = 0;        // gfx_activity_acc at window start
    uint64_t prev_fw_ts = 0;      // firmware_timestamp at window start
    bool have_prev_acc = false;
    double prev_poll_monotonic = 0.0;
  };

  void poll_device(uint32_t i);

  int poll_interval_ms_;
  double idle_epsilon_;
  uint32_t n_devices_ = 0;
  bool initialized_ = false;
  // poll_mu_ serializes pollers; mu_ guards only the stored state so
  // snapshot() never waits behind a slow firmware (SMU) read — concurrent
  // rsmi gpu_metrics reads from several processes can stall ~80 ms
  // (observed under the 4-rank bench) and must not block consumers.
  std::mutex poll_mu_;
  std::mutex mu_;
  std::vector<DevState> devices_;
  std::thread poller_;
  bool running_ = false;
};

}  // namespace exporter
