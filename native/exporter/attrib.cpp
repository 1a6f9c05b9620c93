#include "attrib.hpp"

#include <dirent.h>

#include <chrono>
#include <cstring>
#include <fstream>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "../pruner/k8s.hpp"
#include "podresources.hpp"
#include "sampler.hpp"

#include <sys/stat.h>

namespace exporter {

namespace {

constexpr const char* TARGET = "exporter::attrib";

std::string root(const char* env_name) {
  const char* v = std::getenv(env_name);
  return v && *v ? v : "";
}

double now_s() {
  return std::chrono::duration<double>(std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// "a1b2c3d4_e5f6..." (systemd escapes '-' as '_') → canonical dashed uuid
std::string canonical_uid(std::string uid) {
  for (auto& c : uid)
    if (c == '_') c = '-';
  return strutil::lower(uid);
}

bool is_hex_uuid(const std::string& s) {
  size_t hex = 0;
  for (char c : s) {
    if (std::isxdigit(static_cast<unsigned char>(c))) hex++;
    else if (c != '-') return false;
  }
  return hex == 32;
}

}  // namespace

std::optional<std::string> pod_uid_from_cgroup(const std::string& cgroup_text) {
  // Look for "pod<uid>" path components in any hierarchy line. Handles:
  //   v1:  .../kubepods/besteffort/pod8f7e…-…/…
  //   v2:  .../kubepods.slice/kubepods-burstable.slice/
  //          kubepods-burstable-pod8f7e…_….slice/cri-….scope
  size_t pos = 0;
  while ((pos = cgroup_text.find("pod", pos)) != std::string::npos) {
    size_t start = pos + 3;
    size_t end = start;
    while (end < cgroup_text.size() &&
           (std::isxdigit(static_cast<unsigned char>(cgroup_text[end])) ||
            cgroup_text[end] == '-' || cgroup_text[end] == '_'))
      end++;
    std::string cand = cgroup_text.substr(start, end - start);
    // strip trailing separators picked up greedily
    while (!cand.empty() && (cand.back() == '-' || cand.back() == '_')) cand.pop_back();
    std::string canon = canonical_uid(cand);
    if (is_hex_uuid(canon)) return canon;
    pos = end;
  }
  return std::nullopt;
}

std::map<uint64_t, std::vector<int>> kfd_gpu_pids() {
  std::map<uint64_t, std::vector<int>> out;
  std::string base = root("GPU_EXPORTER_SYSFS_ROOT") + "/sys/class/kfd/kfd/proc";
  DIR* d = ::opendir(base.c_str());
  if (!d) return out;
  while (struct dirent* e = ::readdir(d)) {
    if (e->d_name[0] == '.') continue;
    char* endp = nullptr;
    long pid = std::strtol(e->d_name, &endp, 10);
    if (!endp || *endp != '\0' || pid <= 0) continue;
    std::string pid_dir = base + "/" + e->d_name;
    DIR* pd = ::opendir(pid_dir.c_str());
    if (!pd) continue;
    while (struct dirent* pe = ::readdir(pd)) {
      // per-GPU usage files: vram_<gpu_id>, sdma_<gpu_id>
      if (std::strncmp(pe->d_name, "vram_", 5) == 0) {
        uint64_t gpu_id = std::strtoull(pe->d_name + 5, nullptr, 10);
        if (gpu_id) out[gpu_id].push_back(static_cast<int>(pid));
      }
    }
    ::closedir(pd);
  }
  ::closedir(d);
  return out;
}

Attributor::Attributor() {
  const char* map_file = std::getenv("GPU_EXPORTER_POD_MAP_FILE");
  if (map_file && *map_file) {
    std::ifstream f(map_file);
    if (f) {
      std::string data((std::istreambuf_iterator<char>(f)), std::istreambuf_iterator<char>());
      try {
        jsn::Value v = jsn::parse(data);
        for (const auto& [uid, m] : v.obj()) {
          static_map_[canonical_uid(uid)] = PodAttribution{
              m.get("pod").as_string(), m.get("namespace").as_string(),
              m.get("container").as_string_or("")};
        }
        LOGI(TARGET, "Loaded " + std::to_string(static_map_.size()) +
                         " static pod attributions from " + map_file);
      } catch (const std::exception& e) {
        LOGE(TARGET, std::string("Failed to parse pod map file: ") + e.what());
      }
    }
  }
}

void Attributor::maybe_refresh_apiserver_cache() {
  if (!checked_k8s_) {
    checked_k8s_ = true;
    have_k8s_ = std::getenv("GPU_PRUNER_K8S_URL") || std::getenv("KUBERNETES_SERVICE_HOST");
  }
  if (!have_k8s_) return;
  double t = now_s();
  if (t - last_refresh_s < refresh_s) return;
  last_refresh_s = t;
  try {
    pruner::KubeClient kube(pruner::KubeConfig::resolve());
    // Pods on this node; NODE_NAME is pushed down in the DaemonSet manifest.
    std::string path = "/api/v1/pods";
    if (const char* node = std::getenv("NODE_NAME"); node && *node)
      path += "?fieldSelector=" + strutil::url_encode(std::string("spec.nodeName=") + node);
    jsn::Value list = kube.get(path);
    std::map<std::string, PodAttribution> fresh;
    if (list.get("items").is_array()) {
      for (const auto& pod : list.get("items").arr()) {
        const jsn::Value& meta = pod.get("metadata");
        std::string uid = meta.get("uid").as_string();
        if (uid.empty()) continue;
        std::string container;
        const jsn::Value& containers = pod.at({"spec", "containers"});
        if (containers.is_array() && containers.size() > 0)
          container = containers[0].get("name").as_string();
        fresh[canonical_uid(uid)] = PodAttribution{
            meta.get("name").as_string(), meta.get("namespace").as_string(), container};
      }
    }
    cluster_map_ = std::move(fresh);
    LOGD(TARGET, "Refreshed apiserver pod cache: " + std::to_string(cluster_map_.size()));
  } catch (const std::exception& e) {
    LOGW(TARGET, std::string("apiserver pod cache refresh failed: ") + e.what());
  }
}

std::optional<PodAttribution> Attributor::lookup_uid(const std::string& uid) {
  std::string canon = canonical_uid(uid);
  if (auto it = static_map_.find(canon); it != static_map_.end()) return it->second;
  maybe_refresh_apiserver_cache();
  if (auto it = cluster_map_.find(canon); it != cluster_map_.end()) return it->second;
  return std::nullopt;
}

std::map<uint32_t, PodAttribution> Attributor::resolve(
    const std::vector<std::pair<uint32_t, uint64_t>>& index_to_kfd_id) {
  std::map<uint32_t, PodAttribution> out;
  auto gpu_pids = kfd_gpu_pids();
  std::string proc_root = root("GPU_EXPORTER_PROCFS_ROOT") + "/proc";
  for (const auto& [index, kfd_id] : index_to_kfd_id) {
    auto it = gpu_pids.find(kfd_id);
    if (it == gpu_pids.end()) continue;
    for (int pid : it->second) {
      std::ifstream cg(proc_root + "/" + std::to_string(pid) + "/cgroup");
      if (!cg) continue;
      std::string text((std::istreambuf_iterator<char>(cg)),
                       std::istreambuf_iterator<char>());
      auto uid = pod_uid_from_cgroup(text);
      if (!uid) continue;
      if (auto attr = lookup_uid(*uid)) {
        out[index] = *attr;
        break;  // first attributed pid wins for this GPU
      }
    }
  }
  return out;
}

}  // namespace exporter

// ---- PodResources-based allocation attribution -----------------------------
// (kept at the end of the file: depends on sampler.hpp's DeviceSample)


namespace exporter {

namespace {

// AMD device-plugin device IDs vary by plugin version/config; match a
// reported id against every stable identity the sampler knows for a device.
// Known forms (ROCm/k8s-device-plugin keys its devices by the PCI address
// from /sys/module/amdgpu/drivers/pci:amdgpu; other stacks report KFD
// gpu_id, the 64-bit unique id, DRM node names, or a bare index):
//   "0000:23:00.0" / "0000:23:00"  — PCI BDF, with/without function
//   "23:00.0"                      — BDF without the PCI domain
//   "56525"                        — KFD topology gpu_id
//   "32da0b77724e0fe2" / "0x32da…" — unique id (zero-padding optional)
//   "card2" / "renderD152" / "/dev/dri/renderD152"
//   "0"                            — device index
bool device_id_matches(const std::string& raw_id, const DeviceSample& d) {
  std::string id = strutil::lower(strutil::trim(raw_id));
  if (id.empty()) return false;
  if (id == std::to_string(d.kfd_gpu_id)) return true;
  std::string uid = strutil::lower(d.unique_id);
  if (!uid.empty()) {
    if (id == uid || id == "0x" + uid) return true;
    // unpadded hex: compare with leading zeros stripped from both sides
    auto strip = [](std::string s) {
      if (s.rfind("0x", 0) == 0) s.erase(0, 2);
      size_t nz = s.find_first_not_of('0');
      return nz == std::string::npos ? std::string("0") : s.substr(nz);
    };
    if (strip(id) == strip(uid) && id.find_first_not_of("0123456789abcdefx") ==
                                       std::string::npos)
      return true;
  }
  if (id == "renderd" + std::to_string(d.drm_render_minor)) return true;
  if (id == "/dev/dri/renderd" + std::to_string(d.drm_render_minor)) return true;
  // card index convention: render minor 128+N ↔ cardN
  if (d.drm_render_minor >= 128 &&
      id == "card" + std::to_string(d.drm_render_minor - 128))
    return true;
  std::string bdf = strutil::lower(d.pci_bdf);
  if (!bdf.empty()) {
    if (id == bdf) return true;
    // without the function suffix ("0000:23:00")
    if (size_t dot = bdf.rfind('.'); dot != std::string::npos && id == bdf.substr(0, dot))
      return true;
    // without the PCI domain ("23:00.0")
    if (size_t colon = bdf.find(':'); colon != std::string::npos &&
                                      id == bdf.substr(colon + 1))
      return true;
  }
  if (id == std::to_string(d.index)) return true;
  return false;
}

bool is_gpu_resource(const std::string& resource_name) {
  return resource_name.find("gpu") != std::string::npos ||
         resource_name.find("amd.com") != std::string::npos;
}

}  // namespace

std::map<uint32_t, PodAttribution> Attributor::resolve_full(
    const std::vector<DeviceSample>& devices) {
  std::map<uint32_t, PodAttribution> out;

  std::string sock = "/var/lib/kubelet/pod-resources/kubelet.sock";
  if (const char* env = std::getenv("GPU_EXPORTER_PODRESOURCES_SOCKET"); env && *env)
    sock = env;
  struct stat st {};
  if (::stat(sock.c_str(), &st) == 0) {
    try {
      auto entries = list_pod_resources(sock);
      for (const auto& e : entries) {
        for (const auto& cd : e.devices) {
          if (!is_gpu_resource(cd.resource_name)) continue;
          for (const auto& id : cd.device_ids) {
            for (const auto& d : devices) {
              if (out.count(d.index) == 0 && device_id_matches(id, d))
                out[d.index] = PodAttribution{e.pod, e.ns, e.container};
            }
          }
        }
      }
    } catch (const std::exception& ex) {
      LOGW("exporter::attrib",
           std::string("PodResources attribution failed, falling back to KFD: ") +
               ex.what());
    }
  }

  // KFD usage-based fallback for devices PodResources did not cover.
  std::vector<std::pair<uint32_t, uint64_t>> unmatched;
  for (const auto& d : devices)
    if (out.count(d.index) == 0) unmatched.emplace_back(d.index, d.kfd_gpu_id);
  if (!unmatched.empty()) {
    auto kfd = resolve(unmatched);
    for (auto& [idx, attr] : kfd) out[idx] = std::move(attr);
  }
  return out;
}

}  // namespace exporter
