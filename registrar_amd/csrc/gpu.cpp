// gpu.cpp — KFD topology discovery + health presets (see gpu.hpp).
#include "gpu.hpp"

#include <dirent.h>
#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <map>
#include <sstream>

namespace registrar {
namespace gpu {

namespace {

// KFD properties files are "key value" lines (one per property).
std::map<std::string, uint64_t> read_properties(const std::string& path) {
  std::map<std::string, uint64_t> props;
  std::ifstream f(path);
  std::string key;
  uint64_t value;
  while (f >> key >> value) props[key] = value;
  return props;
}

std::string read_line(const std::string& path) {
  std::ifstream f(path);
  std::string line;
  std::getline(f, line);
  return line;
}

bool file_exists(const std::string& path) {
  struct stat st;
  return stat(path.c_str(), &st) == 0;
}

}  // namespace

std::vector<GpuTopoEntry> discover_gpus(const std::string& root) {
  std::vector<GpuTopoEntry> gpus;
  DIR* d = opendir(root.c_str());
  if (!d) return gpus;
  std::vector<int> node_ids;
  while (struct dirent* e = readdir(d)) {
    if (e->d_name[0] == '.') continue;
    char* end = nullptr;
    long id = strtol(e->d_name, &end, 10);
    if (end && *end == '\0') node_ids.push_back(static_cast<int>(id));
  }
  closedir(d);
  std::sort(node_ids.begin(), node_ids.end());

  for (int id : node_ids) {
    std::string base = root + "/" + std::to_string(id);
    auto props = read_properties(base + "/properties");
    auto it = props.find("simd_count");
    if (it == props.end() || it->second == 0) continue;  // CPU node
    GpuTopoEntry g;
    g.kfd_node = id;
    g.device_index = static_cast<int>(gpus.size());
    auto hive = props.find("hive_id");
    if (hive != props.end()) g.hive_id = hive->second;
    auto loc = props.find("location_id");
    if (loc != props.end()) g.location_id = static_cast<uint32_t>(loc->second);
    auto uid = props.find("unique_id");
    if (uid != props.end() && uid->second != 0) {
      char buf[32];
      snprintf(buf, sizeof(buf), "GPU-%016llx", static_cast<unsigned long long>(uid->second));
      g.uuid = buf;
    }
    g.name = read_line(base + "/name");
    gpus.push_back(g);
  }

  // xGMI-local rank: position within the hive, KFD enumeration order (matches
  // HIP device order on single-node boxes).
  std::map<uint64_t, int> hive_counters;
  for (auto& g : gpus) {
    if (g.hive_id != 0) {
      g.xgmi_rank = hive_counters[g.hive_id]++;
    } else {
      g.xgmi_rank = 0;  // standalone GPU: rank 0 of a hive of one
    }
  }
  return gpus;
}

int gpu_count() { return static_cast<int>(discover_gpus().size()); }

int xgmi_local_rank(int device_index) {
  auto gpus = discover_gpus();
  if (device_index < 0 || device_index >= static_cast<int>(gpus.size())) return -1;
  return gpus[static_cast<size_t>(device_index)].xgmi_rank;
}

bool gpu_alive(int device_index) {
  auto gpus = discover_gpus();
  return device_index >= 0 && device_index < static_cast<int>(gpus.size());
}

std::string gpu_health_command(int device_index) {
  // Prefer the SMI tools: they exercise the full driver/ioctl path, which is
  // what "this GPU can still serve" actually means. Fall back to a KFD sysfs
  // presence probe on minimal images.
  std::string idx = std::to_string(device_index);
  if (file_exists("/opt/rocm/bin/rocm-smi"))
    return "/opt/rocm/bin/rocm-smi -d " + idx + " --showid >/dev/null";
  if (file_exists("/usr/bin/rocm-smi")) return "/usr/bin/rocm-smi -d " + idx + " --showid >/dev/null";
  if (file_exists("/opt/rocm/bin/amd-smi")) return "/opt/rocm/bin/amd-smi list -g " + idx + " >/dev/null";
  // sysfs fallback: the KFD GPU node for this device exists
  return "test -n \"$(grep -l '^simd_count [1-9]' /sys/class/kfd/kfd/topology/nodes/*/properties 2>/dev/null | "
         "sed -n " +
         std::to_string(device_index + 1) + "p)\"";
}

}  // namespace gpu
}  // namespace registrar
