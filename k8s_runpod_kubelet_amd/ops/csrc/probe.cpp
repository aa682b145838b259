// GPU inventory probe — MI355X-native replacement for the reference's cloud
// GPU catalog (reference pkg/virtual_kubelet/runpod_client.go:429-520
// GetGPUTypes, which queries RunPod's GraphQL API). Here the "catalog" is the
// local node: KFD topology (/sys/class/kfd/kfd/topology/nodes/*) for GPU
// identity, gfx target, render-node minor and the xGMI io_link matrix, plus
// amdgpu DRM sysfs (/sys/class/drm/renderD*/device/) for live VRAM
// used/total, busy percent and temperature. Reading sysfs directly is the
// fast path: one inventory refresh is a few dozen small file reads (~100 us)
// versus ~100 ms for shelling out to amd-smi/rocm-smi — this feeds the
// status-reconcile loop, so it must be cheap enough to call per tick.
//
// The sysfs root is a parameter so hermetic tests can point it at a fixture
// tree (no GPU required).

#include "probe.h"

#include <dirent.h>
#include <fcntl.h>
#include <unistd.h>

#include <cerrno>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <map>
#include <sstream>

namespace amdvk {

namespace {

bool ReadFileString(const std::string& path, std::string* out) {
  std::ifstream f(path);
  if (!f.good()) return false;
  std::ostringstream ss;
  ss << f.rdbuf();
  *out = ss.str();
  while (!out->empty() && (out->back() == '\n' || out->back() == ' ')) out->pop_back();
  return true;
}

bool ReadFileU64(const std::string& path, uint64_t* out) {
  std::string s;
  if (!ReadFileString(path, &s)) return false;
  errno = 0;
  char* end = nullptr;
  uint64_t v = strtoull(s.c_str(), &end, 10);
  if (errno != 0 || end == s.c_str()) return false;
  *out = v;
  return true;
}

// KFD "properties" files are lines of "key value".
std::map<std::string, uint64_t> ReadProperties(const std::string& path) {
  std::map<std::string, uint64_t> props;
  std::ifstream f(path);
  std::string key;
  uint64_t value;
  while (f >> key >> value) props[key] = value;
  return props;
}

std::vector<int> ListNumericDir(const std::string& path) {
  std::vector<int> out;
  DIR* d = opendir(path.c_str());
  if (!d) return out;
  while (dirent* e = readdir(d)) {
    if (e->d_name[0] == '.') continue;
    char* end = nullptr;
    long v = strtol(e->d_name, &end, 10);
    if (end && *end == '\0') out.push_back(static_cast<int>(v));
  }
  closedir(d);
  std::sort(out.begin(), out.end());
  return out;
}

// First hwmon subdirectory under <dev>/hwmon, or "".
std::string FindHwmon(const std::string& device_dir) {
  std::string hw = device_dir + "/hwmon";
  DIR* d = opendir(hw.c_str());
  if (!d) return "";
  std::string found;
  while (dirent* e = readdir(d)) {
    if (strncmp(e->d_name, "hwmon", 5) == 0) {
      found = hw + "/" + e->d_name;
      break;
    }
  }
  closedir(d);
  return found;
}

constexpr uint32_t kIoLinkTypeXgmi = 11;  // KFD_IOLINK_TYPE_XGMI
constexpr uint32_t kHeapTypeFbPublic = 1;
constexpr uint32_t kHeapTypeFbPrivate = 2;

}  // namespace

std::vector<GpuInfo> EnumerateGpus(const std::string& sysfs_root) {
  std::vector<GpuInfo> gpus;
  const std::string topo = sysfs_root + "/class/kfd/kfd/topology/nodes";
  std::vector<int> nodes = ListNumericDir(topo);

  // KFD node id -> gpu_id (CPU nodes have simd_count == 0 and are skipped,
  // but they still occupy node ids, so io_link node_to must be mapped).
  std::map<int, int> kfd_to_index;

  for (int node : nodes) {
    const std::string ndir = topo + "/" + std::to_string(node);
    auto props = ReadProperties(ndir + "/properties");
    if (props["simd_count"] == 0) continue;  // CPU/APU memory node

    GpuInfo g;
    g.kfd_node = node;
    g.index = static_cast<int>(gpus.size());
    g.gfx_target_version = props["gfx_target_version"];
    g.render_minor = static_cast<int>(props["drm_render_minor"]);
    g.device_id = static_cast<uint32_t>(props["device_id"]);
    g.location_id = static_cast<uint32_t>(props["location_id"]);
    g.max_engine_clk_mhz = props["max_engine_clk_fcompute"];
    g.cu_count = props["simd_count"] / (props["simd_per_cu"] ? props["simd_per_cu"] : 4);

    std::string uid;
    if (ReadFileString(ndir + "/gpu_id", &uid)) g.gpu_id = uid;

    // VRAM total from KFD mem banks (heap type FB public/private).
    for (int bank : ListNumericDir(ndir + "/mem_banks")) {
      auto bp = ReadProperties(ndir + "/mem_banks/" + std::to_string(bank) + "/properties");
      uint32_t heap = static_cast<uint32_t>(bp["heap_type"]);
      if (heap == kHeapTypeFbPublic || heap == kHeapTypeFbPrivate)
        g.vram_total_bytes += bp["size_in_bytes"];
    }

    // xGMI links to peer KFD nodes.
    for (int link : ListNumericDir(ndir + "/io_links")) {
      auto lp = ReadProperties(ndir + "/io_links/" + std::to_string(link) + "/properties");
      if (static_cast<uint32_t>(lp["type"]) != kIoLinkTypeXgmi) continue;
      XgmiLink xl;
      xl.peer_kfd_node = static_cast<int>(lp["node_to"]);
      xl.weight = static_cast<uint32_t>(lp["weight"]);
      xl.min_bandwidth_mbs = lp["min_bandwidth"];
      xl.max_bandwidth_mbs = lp["max_bandwidth"];
      g.xgmi_links.push_back(xl);
    }

    kfd_to_index[node] = g.index;
    gpus.push_back(std::move(g));
  }

  // Live counters from amdgpu DRM sysfs, addressed by render minor.
  for (auto& g : gpus) {
    if (g.render_minor < 0) continue;
    const std::string dev =
        sysfs_root + "/class/drm/renderD" + std::to_string(g.render_minor) + "/device";
    uint64_t v = 0;
    if (ReadFileU64(dev + "/mem_info_vram_total", &v) && v > 0) g.vram_total_bytes = v;
    if (ReadFileU64(dev + "/mem_info_vram_used", &v)) g.vram_used_bytes = v;
    if (ReadFileU64(dev + "/gpu_busy_percent", &v)) g.busy_percent = static_cast<int>(v);
    std::string uid;
    if (ReadFileString(dev + "/unique_id", &uid) && !uid.empty()) g.unique_id = uid;

    const std::string hwmon = FindHwmon(dev);
    if (!hwmon.empty() && ReadFileU64(hwmon + "/temp1_input", &v))
      g.temperature_mc = static_cast<int64_t>(v);

    // RAS health: a readable ras/ dir with nonzero uncorrectable counts marks
    // the GPU unhealthy (reference analogue: the global runpodAvailable flag,
    // kubelet.go:320-331 — here health is per-GPU).
    uint64_t ue = 0;
    std::string ras;
    if (ReadFileString(dev + "/ras/umc_err_count", &ras)) {
      // format: "ue: N\nce: M"
      const char* p = strstr(ras.c_str(), "ue:");
      if (p) ue = strtoull(p + 3, nullptr, 10);
    }
    g.ras_uncorrectable = ue;
    g.healthy = (ue == 0);
  }

  // Remap xGMI peer ids from KFD node numbering to GPU index numbering.
  for (auto& g : gpus) {
    for (auto& l : g.xgmi_links) {
      auto it = kfd_to_index.find(l.peer_kfd_node);
      l.peer_gpu_index = (it == kfd_to_index.end()) ? -1 : it->second;
    }
  }
  return gpus;
}

GpuDynamic ReadGpuDynamic(const std::string& sysfs_root, int render_minor) {
  GpuDynamic d;
  const std::string dev =
      sysfs_root + "/class/drm/renderD" + std::to_string(render_minor) + "/device";
  uint64_t v = 0;
  if (ReadFileU64(dev + "/mem_info_vram_used", &v)) d.vram_used_bytes = v;
  if (ReadFileU64(dev + "/mem_info_vram_total", &v)) d.vram_total_bytes = v;
  if (ReadFileU64(dev + "/gpu_busy_percent", &v)) d.busy_percent = static_cast<int>(v);
  const std::string hwmon = FindHwmon(dev);
  if (!hwmon.empty() && ReadFileU64(hwmon + "/temp1_input", &v))
    d.temperature_mc = static_cast<int64_t>(v);
  return d;
}

}  // namespace amdvk
