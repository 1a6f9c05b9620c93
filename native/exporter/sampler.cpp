#include "sampler.hpp"

#include <rocm_smi/rocm_smi.h>

#include <chrono>
#include <cstdio>
#include <cstring>
#include <fstream>

#include "../common/log.hpp"

namespace exporter {

namespace {

constexpr const char* TARGET = "exporter::sampler";

double monotonic_s() {
  return std::chrono::duration<double>(std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

std::string rsmi_err(rsmi_status_t st) {
  const char* msg = nullptr;
  rsmi_status_string(st, &msg);
  return msg ? msg : ("rsmi error " + std::to_string(st));
}

// KFD topology: map a PCI BDF id to the KFD gpu_id used under
// /sys/class/kfd/kfd/proc/<pid>/ for pod attribution.
uint64_t kfd_gpu_id_for_bdf(uint64_t bdfid) {
  const char* root = std::getenv("GPU_EXPORTER_SYSFS_ROOT");
  std::string base = std::string(root && *root ? root : "") + "/sys/class/kfd/kfd/topology/nodes";
  for (int n = 0; n < 64; n++) {
    std::string dir = base + "/" + std::to_string(n);
    std::ifstream props(dir + "/properties");
    if (!props) break;
    uint64_t domain = ~0ull, location_id = ~0ull;
    std::string key;
    uint64_t value;
    while (props >> key >> value) {
      if (key == "domain") domain = value;
      else if (key == "location_id") location_id = value;
    }
    // rsmi bdfid: ((domain & 0xffffffff) << 32) | (bus << 8) | (dev << 3) | func
    uint64_t node_bdf = ((domain & 0xffffffff) << 32) | (location_id & 0xffffffff);
    if (node_bdf == bdfid) {
      std::ifstream gid(dir + "/gpu_id");
      uint64_t id = 0;
      gid >> id;
      return id;
    }
  }
  return 0;
}

}  // namespace

Sampler::Sampler(int poll_interval_ms, double idle_epsilon, double window_s)
    : poll_interval_ms_(poll_interval_ms), idle_epsilon_(idle_epsilon), window_s_(window_s) {}

Sampler::~Sampler() {
  stop();
  if (initialized_) rsmi_shut_down();
}

void Sampler::init() {
  rsmi_status_t st = rsmi_init(0);
  if (st != RSMI_STATUS_SUCCESS)
    throw SamplerError("rsmi_init failed (" + rsmi_err(st) +
                       ") — is the amdgpu driver loaded?");
  initialized_ = true;
  st = rsmi_num_monitor_devices(&n_devices_);
  if (st != RSMI_STATUS_SUCCESS)
    throw SamplerError("rsmi_num_monitor_devices failed: " + rsmi_err(st));
  if (n_devices_ == 0) throw SamplerError("no AMD GPUs enumerated by rocm_smi");

  devices_.resize(n_devices_);
  for (uint32_t i = 0; i < n_devices_; i++) {
    devices_[i].win.set_retention(window_s_ * 2.0 + 10.0);
    DeviceSample& d = devices_[i].last;
    d.index = i;
    char name[256] = {0};
    if (rsmi_dev_name_get(i, name, sizeof name) == RSMI_STATUS_SUCCESS) d.model_name = name;
    if (d.model_name.empty() || d.model_name.rfind("0x", 0) == 0) {
      // some firmware returns a hex id; fall back to the brand string
      char brand[256] = {0};
      if (rsmi_dev_brand_get(i, brand, sizeof brand) == RSMI_STATUS_SUCCESS && brand[0])
        d.model_name = brand;
    }
    uint64_t uid = 0;
    if (rsmi_dev_unique_id_get(i, &uid) == RSMI_STATUS_SUCCESS) {
      char buf[32];
      std::snprintf(buf, sizeof buf, "%016lx", static_cast<unsigned long>(uid));
      d.unique_id = buf;
    }
    uint64_t bdfid = 0;
    if (rsmi_dev_pci_id_get(i, &bdfid) == RSMI_STATUS_SUCCESS) {
      char buf[32];
      std::snprintf(buf, sizeof buf, "%04lx:%02lx:%02lx.%lx",
                    static_cast<unsigned long>((bdfid >> 32) & 0xffffffff),
                    static_cast<unsigned long>((bdfid >> 8) & 0xff),
                    static_cast<unsigned long>((bdfid >> 3) & 0x1f),
                    static_cast<unsigned long>(bdfid & 0x7));
      d.pci_bdf = buf;
      d.kfd_gpu_id = kfd_gpu_id_for_bdf(bdfid);
    }
    uint32_t minor = 0;
    if (rsmi_dev_drm_render_minor_get(i, &minor) == RSMI_STATUS_SUCCESS)
      d.drm_render_minor = minor;
    uint64_t total = 0;
    if (rsmi_dev_memory_total_get(i, RSMI_MEM_TYPE_VRAM, &total) == RSMI_STATUS_SUCCESS)
      d.vram_total_b = static_cast<double>(total);
  }
  LOGI(TARGET, "Enumerated " + std::to_string(n_devices_) + " AMD GPU(s); model=\"" +
                   devices_[0].last.model_name + "\"");
  poll_once();  // establish baselines so the first scrape has data
}

void Sampler::start() {
  if (running_) return;
  running_ = true;
  poller_ = std::thread([this] {
    while (running_) {
      std::this_thread::sleep_for(std::chrono::milliseconds(poll_interval_ms_));
      if (!running_) break;
      try {
        poll_once();
      } catch (const std::exception& e) {
        LOGE(TARGET, std::string("poll failed: ") + e.what());
      }
    }
  });
}

void Sampler::stop() {
  if (!running_) return;
  running_ = false;
  if (poller_.joinable()) poller_.join();
}

void Sampler::poll_once() {
  std::lock_guard<std::mutex> poll_lock(poll_mu_);
  for (uint32_t i = 0; i < n_devices_; i++) poll_device(i);
}

void Sampler::poll_device(uint32_t i) {
  // rsmi/SMU reads happen on a local copy, without holding the state lock
  DeviceSample d;
  {
    std::lock_guard<std::mutex> lock(mu_);
    d = devices_[i].last;
  }
  d.metrics_table_ok = false;
  double now = monotonic_s();

  uint32_t busy = 0;
  bool busy_read_ok = rsmi_dev_busy_percent_get(i, &busy) == RSMI_STATUS_SUCCESS;
  if (busy_read_ok) d.busy_percent = static_cast<double>(busy);

  uint32_t mem_busy = 0;
  if (rsmi_dev_memory_busy_percent_get(i, &mem_busy) == RSMI_STATUS_SUCCESS)
    d.mem_busy_percent = static_cast<double>(mem_busy);

  uint64_t power = 0;
  RSMI_POWER_TYPE ptype = RSMI_INVALID_POWER;
  if (rsmi_dev_power_get(i, &power, &ptype) == RSMI_STATUS_SUCCESS)
    d.power_w = static_cast<double>(power) / 1e6;  // µW → W

  uint64_t used = 0;
  if (rsmi_dev_memory_usage_get(i, RSMI_MEM_TYPE_VRAM, &used) == RSMI_STATUS_SUCCESS)
    d.vram_used_b = static_cast<double>(used);

  int64_t temp = 0;
  if (rsmi_dev_temp_metric_get(i, RSMI_TEMP_TYPE_EDGE, RSMI_TEMP_CURRENT, &temp) ==
      RSMI_STATUS_SUCCESS)
    d.temp_edge_c = static_cast<double>(temp) / 1000.0;

  // gpu_metrics table: the firmware's own accumulated activity counter.
  rsmi_gpu_metrics_t gm;
  std::memset(&gm, 0, sizeof gm);
  if (rsmi_dev_gpu_metrics_info_get(i, &gm) == RSMI_STATUS_SUCCESS) {
    d.metrics_table_ok = true;
    d.gfx_activity_acc = gm.gfx_activity_acc;
    d.firmware_timestamp = gm.firmware_timestamp;
    d.gfx_clock_mhz = gm.current_gfxclk;
    if (gm.current_socket_power) d.power_w = gm.current_socket_power;
    // energy_accumulator counts 15.259 µJ per unit
    d.energy_j = static_cast<double>(gm.energy_accumulator) * 15.259e-6;
    d.xgmi_link_width = gm.xgmi_link_width;
    d.xgmi_link_speed = gm.xgmi_link_speed;
    double rd = 0, wr = 0;
    for (int l = 0; l < RSMI_MAX_NUM_XGMI_LINKS; l++) {
      rd += static_cast<double>(gm.xgmi_read_data_acc[l]);
      wr += static_cast<double>(gm.xgmi_write_data_acc[l]);
    }
    d.xgmi_read_kb = rd;
    d.xgmi_write_kb = wr;
    // prefer the firmware activity percentage when the busy-percent sysfs
    // read is unsupported
    if (d.busy_percent == 0.0 && gm.average_gfx_activity > 0 &&
        gm.average_gfx_activity <= 100)
      d.busy_percent = gm.average_gfx_activity;
  }

  // An activity observation exists when either the busy-percent sysfs read
  // or the firmware gpu_metrics table succeeded. When BOTH fail the span
  // since the previous poll is recorded as UNKNOWN — it never contributes
  // idle time — and the device's health counter advances (a
  // dying SMU read path must not make a busy GPU look permanently idle).
  bool activity_ok = busy_read_ok || d.metrics_table_ok;
  d.read_ok = activity_ok;

  // ---- append the poll segment to the sliding window (brief state lock) ----
  std::lock_guard<std::mutex> lock(mu_);
  DevState& st = devices_[i];
  {
    double ratio = d.busy_percent / 100.0;
    // When the firmware accumulator advanced, derive the exact ratio over
    // the poll interval from Δacc/Δt. Units calibrated on MI355X silicon
    // (profiles/raw/calibration.log): firmware_timestamp ticks in ns on
    // gfx950 (the rocm_smi header documents 10 ns — we auto-detect by
    // comparing against the host monotonic delta), and gfx_activity_acc
    // accumulates 100,000 counts per second at 100% busy.
    double dt = now - st.win_prev_t;
    if (d.metrics_table_ok && st.have_prev_acc && st.win_prev_t > 0.0 && dt > 0 &&
        d.firmware_timestamp > st.prev_fw_ts) {
      double fw_dt = static_cast<double>(d.firmware_timestamp - st.prev_fw_ts);
      double fw_dt_ns = fw_dt * 1e-9, fw_dt_10ns = fw_dt * 1e-8;
      double fw_dt_s =
          std::abs(fw_dt_ns - dt) <= std::abs(fw_dt_10ns - dt) ? fw_dt_ns : fw_dt_10ns;
      constexpr double kAccFullRate = 100000.0;  // counts/s at 100% (measured)
      if (fw_dt_s > 0) {
        double acc_ratio = static_cast<double>(d.gfx_activity_acc - st.prev_acc) /
                           (fw_dt_s * kAccFullRate);
        if (acc_ratio >= 0.0 && acc_ratio <= 1.5) ratio = std::min(acc_ratio, 1.0);
      }
    }
    st.win_prev_t = now;
    st.win.add(now, ratio, activity_ok);
  }
  st.prev_acc = d.gfx_activity_acc;
  st.prev_fw_ts = d.firmware_timestamp;
  st.have_prev_acc = d.metrics_table_ok;

  if (activity_ok) {
    st.consecutive_failures = 0;
    st.last_good_monotonic = now;
  } else {
    st.consecutive_failures++;
    if (st.consecutive_failures == kUnhealthyAfter)
      LOGE(TARGET, "device " + std::to_string(i) + ": " +
                       std::to_string(st.consecutive_failures) +
                       " consecutive failed activity reads — marking unhealthy "
                       "(activity series will be withheld)");
  }
  st.last = d;
}

std::vector<DeviceSample> Sampler::snapshot() {
  std::lock_guard<std::mutex> lock(mu_);
  double now = monotonic_s();
  std::vector<DeviceSample> out;
  out.reserve(devices_.size());
  for (auto& st : devices_) {
    DeviceSample d = st.last;
    double known_s = 0.0;
    double window_ratio = std::min(st.win.ratio(now, window_s_, &known_s), 1.0);
    // Floor sub-noise ratios to an exact 0 (see ctor comment): the idle
    // predicate is `== 0`, and housekeeping blips are not workload.
    d.gr_engine_active = window_ratio < idle_epsilon_ ? 0.0 : window_ratio;
    d.healthy = st.consecutive_failures < kUnhealthyAfter;
    d.staleness_s = st.last_good_monotonic > 0.0 ? now - st.last_good_monotonic : 0.0;
    out.push_back(std::move(d));
  }
  return out;
}

}  // namespace exporter
