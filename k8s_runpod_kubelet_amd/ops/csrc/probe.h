#pragma once

#include <algorithm>
#include <cstdint>
#include <string>
#include <vector>

namespace amdvk {

struct XgmiLink {
  int peer_kfd_node = -1;
  int peer_gpu_index = -1;
  uint32_t weight = 0;
  uint64_t min_bandwidth_mbs = 0;
  uint64_t max_bandwidth_mbs = 0;
};

struct GpuInfo {
  int index = -1;          // dense GPU index (HIP device order = KFD order)
  int kfd_node = -1;       // KFD topology node id
  int render_minor = -1;   // /dev/dri/renderD<minor>
  std::string gpu_id;      // KFD gpu_id
  std::string unique_id;   // amdgpu unique_id (stable across reboots)
  uint64_t gfx_target_version = 0;  // e.g. 90500 for gfx950
  uint32_t device_id = 0;
  uint32_t location_id = 0;
  uint64_t cu_count = 0;
  uint64_t max_engine_clk_mhz = 0;
  uint64_t vram_total_bytes = 0;
  uint64_t vram_used_bytes = 0;
  int busy_percent = -1;
  int64_t temperature_mc = -1;  // millidegrees C
  uint64_t ras_uncorrectable = 0;
  bool healthy = true;
  std::vector<XgmiLink> xgmi_links;
};

struct GpuDynamic {
  uint64_t vram_used_bytes = 0;
  uint64_t vram_total_bytes = 0;
  int busy_percent = -1;
  int64_t temperature_mc = -1;
};

// Enumerate GPUs from KFD topology + amdgpu DRM sysfs under sysfs_root.
std::vector<GpuInfo> EnumerateGpus(const std::string& sysfs_root);

// Cheap per-tick refresh of one GPU's live counters.
GpuDynamic ReadGpuDynamic(const std::string& sysfs_root, int render_minor);

}  // namespace amdvk
