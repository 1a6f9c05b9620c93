// attrib.hpp — GPU → pod attribution for the mi355-exporter.
//
// The reference leans on dcgm-exporter's kubelet PodResources integration for
// its pod/namespace/container labels (SURVEY.md §7 "Hard parts"). This
// first-party implementation walks the amdgpu KFD process registry instead:
//
//   /sys/class/kfd/kfd/proc/<pid>/vram_<gpu_id>   — which pid uses which GPU
//   /proc/<pid>/cgroup                            — kubepods[-...]-pod<uid>
//                                                   → the owning pod's UID
//   pod UID → (pod, namespace, container)         — via the apiserver's pod
//     list for this node (cached), or a static JSON map file
//     ($GPU_EXPORTER_POD_MAP_FILE) for tests / non-K8s hosts.
//
// Both /sys and /proc roots are overridable ($GPU_EXPORTER_SYSFS_ROOT /
// $GPU_EXPORTER_PROCFS_ROOT) so the whole chain is unit-testable without a
// GPU or a kubelet.
#pragma once

#include <cstdint>
#include <map>
#include <optional>
#include <string>
#include <vector>

namespace exporter {

struct PodAttribution {
  std::string pod;
  std::string ns;
  std::string container;
};

// pid → pod UID from its cgroup file; handles cgroup v1 and v2 kubepods
// layouts (…/kubepods/burstable/pod<uid>/…, kubepods-besteffort-pod<uid>.slice).
// Returned UID uses the canonical dashed form.
std::optional<std::string> pod_uid_from_cgroup(const std::string& cgroup_text);

// Scan the KFD process registry: kfd_gpu_id → pids with that GPU open.
std::map<uint64_t, std::vector<int>> kfd_gpu_pids();

struct DeviceSample;  // sampler.hpp

class Attributor {
public:
  Attributor();

  // Resolve attributions for the sampled devices (keyed by device index).
  // kfd ids come from the sampler's topology mapping (usage-based KFD path).
  std::map<uint32_t, PodAttribution> resolve(
      const std::vector<std::pair<uint32_t, uint64_t>>& index_to_kfd_id);

  // Full chain: kubelet PodResources allocations first (covers
  // allocated-but-idle GPUs — the culler's target case), KFD process
  // registry as fallback for unmatched devices. The PodResources socket is
  // $GPU_EXPORTER_PODRESOURCES_SOCKET (default
  // /var/lib/kubelet/pod-resources/kubelet.sock); absent socket → fallback.
  std::map<uint32_t, PodAttribution> resolve_full(
      const std::vector<DeviceSample>& devices);

  // pod UID → attribution; consults the static map file first, then the
  // apiserver cache (refreshing at most every `refresh_s`).
  std::optional<PodAttribution> lookup_uid(const std::string& uid);

private:
  void maybe_refresh_apiserver_cache();

  std::map<std::string, PodAttribution> static_map_;   // from map file
  std::map<std::string, PodAttribution> cluster_map_;  // from apiserver
  double last_refresh_s = 0.0;
  int refresh_s = 30;
  bool have_k8s_ = false;
  bool checked_k8s_ = false;
};

}  // namespace exporter
