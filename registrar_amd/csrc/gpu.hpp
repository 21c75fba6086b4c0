// gpu.hpp — MI355X host-side integration: GPU discovery, xGMI topology rank,
// and GPU-liveness health command presets.
//
// This is the MI355X-native content of the build (BASELINE.json north star;
// SURVEY.md §2.3): the registrar itself is a host-side control-plane daemon —
// no kernels, no collectives — but (a) its health check gates registration on
// GPU liveness via rocm-smi/amdsmi, and (b) the host-record payload
// advertises the GPU's xGMI-local rank so consumers can prefer xGMI-local
// peers (xGMI is 7 point-to-point links per GPU; the rank identifies the
// GPU's position inside its xGMI hive).
//
// Discovery reads the ROCm KFD topology directly from sysfs
// (/sys/class/kfd/kfd/topology/nodes/*/properties): GPU nodes are those with
// simd_count > 0; hive_id groups the xGMI hive; the xGMI-local rank is the
// GPU's index within its hive in KFD enumeration order (which matches HIP
// device order). sysfs parsing avoids spawning a subprocess on the hot path;
// rocm-smi/amd-smi are used only inside the (cold) health-check command.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace registrar {
namespace gpu {

struct GpuTopoEntry {
  int kfd_node = -1;       // KFD topology node index
  int device_index = -1;   // HIP-ordered GPU index (0..N-1)
  uint64_t hive_id = 0;    // xGMI hive id (0 = not in a hive)
  int xgmi_rank = -1;      // rank within the hive (0 if not in one)
  uint32_t location_id = 0;
  std::string name;        // marketing/gfx name when available
  std::string uuid;
};

// All GPUs visible through KFD, in device-index order. Empty on a GPU-less
// host. `root` overrides the sysfs base for tests.
std::vector<GpuTopoEntry> discover_gpus(const std::string& root = "/sys/class/kfd/kfd/topology/nodes");

// Number of GPUs (0 on CPU-only hosts).
int gpu_count();

// xGMI-local rank of `device_index`, or -1 if no such GPU.
int xgmi_local_rank(int device_index);

// Shell command for the health-check gate on GPU `device_index`:
// prefers `rocm-smi` / `amd-smi` when installed (they exercise the full
// driver path), falling back to a KFD sysfs presence test. Intended as the
// healthCheck.command for per-GPU registrar processes (BASELINE config 3).
std::string gpu_health_command(int device_index);

// Fast in-process liveness check: KFD node for the device exists and exposes
// a nonzero simd_count.
bool gpu_alive(int device_index);

}  // namespace gpu
}  // namespace registrar
