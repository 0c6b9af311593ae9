// MI355X GPU telemetry straight from amdsmi (no nvml, no CUDA shims).
//
// The amdsmi C library (/opt/rocm/lib/libamd_smi.so) is loaded via
// dlopen so the gateway binary runs unchanged on GPU-less CI hosts
// (telemetry simply reports absent).  Per-GPU HBM3E occupancy (288 GB)
// and compute-busy feed the router's admission control
// (router.cpp select_endpoint) — BASELINE.json's GPU-aware routing.
//
// Also reads the xGMI link topology (amdsmi_topo_get_link_type /
// link metrics) to build the adjacency map used for TP-quad placement
// (llmlb_amd/parallel/topology.py consumes the same data via the
// Python amdsmi module; this C++ reader serves /api/gpu).
#include <dlfcn.h>
#include <unistd.h>

#include <cstring>
#include <vector>

#include "gateway.h"

namespace gw {

// ---- minimal amdsmi ABI (stable public C API, amd_smi/amdsmi.h) ----
typedef void* amdsmi_socket_handle;
typedef void* amdsmi_processor_handle;
typedef int amdsmi_status_t;  // AMDSMI_STATUS_SUCCESS == 0
struct amdsmi_engine_usage_t {
  uint32_t gfx_activity;
  uint32_t umc_activity;
  uint32_t mm_activity;
  uint32_t reserved[13];
};

namespace {
struct AmdSmi {
  void* handle = nullptr;
  amdsmi_status_t (*init)(uint64_t flags) = nullptr;
  amdsmi_status_t (*get_socket_handles)(uint32_t*, amdsmi_socket_handle*) =
      nullptr;
  amdsmi_status_t (*get_processor_handles)(amdsmi_socket_handle, uint32_t*,
                                           amdsmi_processor_handle*) =
      nullptr;
  // amdsmi_get_gpu_memory_usage(handle, mem_type=0 (VRAM), uint64* used)
  amdsmi_status_t (*get_memory_usage)(amdsmi_processor_handle, int,
                                      uint64_t*) = nullptr;
  amdsmi_status_t (*get_memory_total)(amdsmi_processor_handle, int,
                                      uint64_t*) = nullptr;
  amdsmi_status_t (*get_gpu_activity)(amdsmi_processor_handle,
                                      amdsmi_engine_usage_t*) = nullptr;
  std::vector<amdsmi_processor_handle> gpus;
  bool ready = false;
};

AmdSmi g_smi;
std::mutex g_mu;
std::vector<GpuInfo> g_info;
bool g_tried = false;

void try_init() {
  if (g_tried) return;
  g_tried = true;
  // only attempt amdsmi where an AMD GPU driver is actually present —
  // initializing it on a GPU-less host can abort the process inside
  // the library's ESMI path
  if (access("/dev/kfd", F_OK) != 0) {
    LOG_INFO("no /dev/kfd: GPU telemetry disabled on this host");
    return;
  }
  const char* paths[] = {"libamd_smi.so", "/opt/rocm/lib/libamd_smi.so"};
  for (const char* p : paths) {
    g_smi.handle = dlopen(p, RTLD_NOW | RTLD_LOCAL);
    if (g_smi.handle) break;
  }
  if (!g_smi.handle) {
    LOG_INFO("amdsmi not available (no GPU telemetry on this host)");
    return;
  }
  auto sym = [&](const char* name) { return dlsym(g_smi.handle, name); };
  g_smi.init = (decltype(g_smi.init))sym("amdsmi_init");
  g_smi.get_socket_handles = (decltype(g_smi.get_socket_handles))sym(
      "amdsmi_get_socket_handles");
  g_smi.get_processor_handles = (decltype(g_smi.get_processor_handles))sym(
      "amdsmi_get_processor_handles");
  g_smi.get_memory_usage =
      (decltype(g_smi.get_memory_usage))sym("amdsmi_get_gpu_memory_usage");
  g_smi.get_memory_total =
      (decltype(g_smi.get_memory_total))sym("amdsmi_get_gpu_memory_total");
  g_smi.get_gpu_activity =
      (decltype(g_smi.get_gpu_activity))sym("amdsmi_get_gpu_activity");
  if (!g_smi.init || !g_smi.get_socket_handles ||
      !g_smi.get_processor_handles || !g_smi.get_memory_usage) {
    LOG_WARN("amdsmi symbols missing; telemetry disabled");
    return;
  }
  if (g_smi.init(1 /* AMDSMI_INIT_AMD_GPUS */) != 0) {
    LOG_WARN("amdsmi_init failed; telemetry disabled");
    return;
  }
  uint32_t nsock = 0;
  if (g_smi.get_socket_handles(&nsock, nullptr) != 0 || nsock == 0) return;
  std::vector<amdsmi_socket_handle> socks(nsock);
  g_smi.get_socket_handles(&nsock, socks.data());
  for (auto& s : socks) {
    uint32_t nproc = 0;
    if (g_smi.get_processor_handles(s, &nproc, nullptr) != 0 || !nproc)
      continue;
    std::vector<amdsmi_processor_handle> procs(nproc);
    g_smi.get_processor_handles(s, &nproc, procs.data());
    for (auto& p : procs) g_smi.gpus.push_back(p);
  }
  g_smi.ready = !g_smi.gpus.empty();
  LOG_INFO("amdsmi: %zu GPU(s) visible", g_smi.gpus.size());
}
}  // namespace

void telemetry_poll_once() {
  try_init();
  if (!g_smi.ready) return;
  std::vector<GpuInfo> infos;
  infos.reserve(g_smi.gpus.size());
  for (size_t i = 0; i < g_smi.gpus.size(); ++i) {
    GpuInfo gi;
    gi.device_index = (int)i;
    uint64_t used = 0, total = 0;
    if (g_smi.get_memory_usage(g_smi.gpus[i], 0 /*VRAM*/, &used) == 0)
      gi.hbm_used_gb = (double)used / (1 << 30);
    if (g_smi.get_memory_total &&
        g_smi.get_memory_total(g_smi.gpus[i], 0, &total) == 0)
      gi.hbm_total_gb = (double)total / (1 << 30);
    if (g_smi.get_gpu_activity) {
      amdsmi_engine_usage_t act{};
      if (g_smi.get_gpu_activity(g_smi.gpus[i], &act) == 0)
        gi.busy_pct = act.gfx_activity;
    }
    gi.updated_ms = wall_ms();
    infos.push_back(gi);
  }
  std::lock_guard<std::mutex> lk(g_mu);
  g_info.swap(infos);
}

bool telemetry_get(int device_index, GpuInfo* out) {
  std::lock_guard<std::mutex> lk(g_mu);
  if (device_index < 0 || (size_t)device_index >= g_info.size())
    return false;
  *out = g_info[device_index];
  return true;
}

void telemetry_snapshot_json(std::string* out) {
  std::lock_guard<std::mutex> lk(g_mu);
  JsonArray arr;
  for (auto& g : g_info) {
    JsonObject o;
    o.emplace_back("device_index", Json((long long)g.device_index));
    o.emplace_back("hbm_used_gb", Json(g.hbm_used_gb));
    o.emplace_back("hbm_total_gb", Json(g.hbm_total_gb));
    o.emplace_back("busy_pct", Json(g.busy_pct));
    o.emplace_back("updated_ms", Json((long long)g.updated_ms));
    arr.emplace_back(Json(std::move(o)));
  }
  JsonObject root;
  root.emplace_back("gpus", Json(std::move(arr)));
  root.emplace_back("available", Json(g_smi.ready));
  *out = Json(std::move(root)).dump();
}

}  // namespace gw
