// _amdsmi: MI355X device enumeration through libamd_smi (pybind11).
//
// The reference enumerates GPUs with NVML cgo (ref: pkg/operator/base.go:19-75,
// returning UUID/index/memory). This binding is the MI355X-native equivalent
// and returns everything the agent's placement + materialization layers need:
// HIP enumeration index, UUID, VRAM bytes, DRM render/card minors, CU count,
// NUMA node, compute-partition mode (SPX/DPX/QPX/CPX) and the xGMI peer table
// (peers = 1-hop XGMI links) used for topology-aware preferred allocation.
//
// Deliberately NO fallback path: if libamd_smi or the GPUs are absent, calls
// raise RuntimeError.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <amd_smi/amdsmi.h>

#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

void check(amdsmi_status_t st, const char* what) {
  if (st != AMDSMI_STATUS_SUCCESS) {
    const char* s = nullptr;
    amdsmi_status_code_to_string(st, &s);
    throw std::runtime_error(std::string("amdsmi: ") + what + " failed: " +
                             (s ? s : std::to_string(static_cast<int>(st))));
  }
}

struct SmiSession {
  SmiSession() { check(amdsmi_init(AMDSMI_INIT_AMD_GPUS), "amdsmi_init"); }
  ~SmiSession() { amdsmi_shut_down(); }
  SmiSession(const SmiSession&) = delete;
  SmiSession& operator=(const SmiSession&) = delete;
};

std::vector<amdsmi_processor_handle> gpu_handles() {
  uint32_t socket_count = 0;
  check(amdsmi_get_socket_handles(&socket_count, nullptr), "get_socket_handles(count)");
  std::vector<amdsmi_socket_handle> sockets(socket_count);
  check(amdsmi_get_socket_handles(&socket_count, sockets.data()), "get_socket_handles");

  std::vector<amdsmi_processor_handle> gpus;
  for (auto sock : sockets) {
    uint32_t n = 0;
    check(amdsmi_get_processor_handles(sock, &n, nullptr), "get_processor_handles(count)");
    std::vector<amdsmi_processor_handle> procs(n);
    check(amdsmi_get_processor_handles(sock, &n, procs.data()), "get_processor_handles");
    for (auto p : procs) {
      processor_type_t t{};
      check(amdsmi_get_processor_type(p, &t), "get_processor_type");
      if (t == AMDSMI_PROCESSOR_TYPE_AMD_GPU) gpus.push_back(p);
    }
  }
  return gpus;
}

int xcds_for_partition(const std::string& partition) {
  // gfx950 SPX exposes all 8 XCDs as one device; partition modes split them.
  if (partition == "DPX") return 4;
  if (partition == "QPX") return 2;
  if (partition == "CPX") return 1;
  return 8;  // SPX / unknown
}

py::list enumerate_gpus() {
  SmiSession session;
  auto gpus = gpu_handles();

  // Collect per-handle info first (order of handles is amdsmi's, we re-sort
  // by HIP id so agent indexes match what workloads see from HIP).
  struct Info {
    uint32_t hip_id = 0, drm_render = 0, drm_card = 0, numa = 0, cus = 0;
    uint64_t vram_bytes = 0;
    std::string uuid, partition;
    size_t pos;  // position in `gpus` for the topology pass
  };
  std::vector<Info> infos;
  infos.reserve(gpus.size());

  for (size_t i = 0; i < gpus.size(); ++i) {
    auto h = gpus[i];
    Info inf;
    inf.pos = i;

    amdsmi_enumeration_info_t en{};
    check(amdsmi_get_gpu_enumeration_info(h, &en), "get_gpu_enumeration_info");
    inf.hip_id = en.hip_id;
    inf.drm_render = en.drm_render;
    inf.drm_card = en.drm_card;

    unsigned int uuid_len = AMDSMI_GPU_UUID_SIZE;
    char uuid_buf[AMDSMI_GPU_UUID_SIZE] = {0};
    check(amdsmi_get_gpu_device_uuid(h, &uuid_len, uuid_buf), "get_gpu_device_uuid");
    inf.uuid = uuid_buf;

    uint64_t total = 0;
    check(amdsmi_get_gpu_memory_total(h, AMDSMI_MEM_TYPE_VRAM, &total), "get_gpu_memory_total");
    inf.vram_bytes = total;

    amdsmi_asic_info_t asic{};
    if (amdsmi_get_gpu_asic_info(h, &asic) == AMDSMI_STATUS_SUCCESS &&
        asic.num_of_compute_units != 0xFFFFFFFFu) {
      inf.cus = asic.num_of_compute_units;
    }

    uint32_t numa = 0;
    if (amdsmi_topo_get_numa_node_number(h, &numa) == AMDSMI_STATUS_SUCCESS) inf.numa = numa;

    char part[32] = {0};
    if (amdsmi_get_gpu_compute_partition(h, part, sizeof(part)) == AMDSMI_STATUS_SUCCESS &&
        part[0] != '\0') {
      inf.partition = part;
    } else {
      inf.partition = "SPX";
    }
    infos.push_back(std::move(inf));
  }

  // xGMI peer table: 1-hop XGMI links between GPU pairs.
  std::vector<std::vector<uint32_t>> peers(infos.size());
  for (size_t a = 0; a < infos.size(); ++a) {
    for (size_t b = 0; b < infos.size(); ++b) {
      if (a == b) continue;
      uint64_t hops = 0;
      amdsmi_link_type_t type{};
      if (amdsmi_topo_get_link_type(gpus[infos[a].pos], gpus[infos[b].pos], &hops, &type) ==
              AMDSMI_STATUS_SUCCESS &&
          type == AMDSMI_LINK_TYPE_XGMI && hops <= 1) {
        peers[a].push_back(infos[b].hip_id);
      }
    }
  }

  py::list out;
  for (size_t i = 0; i < infos.size(); ++i) {
    const auto& inf = infos[i];
    py::dict d;
    d["index"] = inf.hip_id;
    d["uuid"] = inf.uuid;
    d["memory_bytes"] = inf.vram_bytes;
    d["drm_render"] = inf.drm_render;
    d["drm_card"] = inf.drm_card;
    d["cu_count"] = inf.cus ? inf.cus : 256;
    d["xcd_count"] = xcds_for_partition(inf.partition);
    d["numa_node"] = inf.numa;
    d["xgmi_peers"] = peers[i];
    d["compute_partition"] = inf.partition;
    out.append(d);
  }
  return out;
}

// ---- occupancy / utilization (per-pod occupancy reporting) -----------------

amdsmi_processor_handle handle_for_hip_index(uint32_t index,
                                             std::vector<amdsmi_processor_handle>& gpus) {
  for (auto h : gpus) {
    amdsmi_enumeration_info_t en{};
    if (amdsmi_get_gpu_enumeration_info(h, &en) == AMDSMI_STATUS_SUCCESS &&
        en.hip_id == index)
      return h;
  }
  throw std::runtime_error("no GPU with HIP index " + std::to_string(index));
}

py::dict gpu_utilization(uint32_t index) {
  SmiSession session;
  auto gpus = gpu_handles();
  auto h = handle_for_hip_index(index, gpus);
  py::dict d;
  amdsmi_engine_usage_t usage{};
  if (amdsmi_get_gpu_activity(h, &usage) == AMDSMI_STATUS_SUCCESS) {
    d["gfx_busy_percent"] = usage.gfx_activity;
    d["mem_busy_percent"] = usage.umc_activity;
  }
  amdsmi_vram_usage_t vram{};
  if (amdsmi_get_gpu_vram_usage(h, &vram) == AMDSMI_STATUS_SUCCESS) {
    d["vram_used_mb"] = vram.vram_used;
    d["vram_total_mb"] = vram.vram_total;
  }
  return d;
}

py::list gpu_processes(uint32_t index) {
  SmiSession session;
  auto gpus = gpu_handles();
  auto h = handle_for_hip_index(index, gpus);
  uint32_t n = 0;
  amdsmi_status_t st = amdsmi_get_gpu_process_list(h, &n, nullptr);
  py::list out;
  if (st != AMDSMI_STATUS_SUCCESS || n == 0) return out;
  std::vector<amdsmi_proc_info_t> procs(n);
  st = amdsmi_get_gpu_process_list(h, &n, procs.data());
  if (st != AMDSMI_STATUS_SUCCESS) return out;
  for (uint32_t i = 0; i < n; ++i) {
    py::dict p;
    p["pid"] = static_cast<uint64_t>(procs[i].pid);
    p["name"] = std::string(procs[i].name);
    p["vram_bytes"] = procs[i].memory_usage.vram_mem;
    p["gfx_busy_ns"] = procs[i].engine_usage.gfx;
    p["cu_occupancy"] = procs[i].cu_occupancy;  // CUs this process occupies NOW
    p["evicted_ms"] = procs[i].evicted_time;
    out.append(p);
  }
  return out;
}

// ---- compute partition control (SPX/DPX/QPX/CPX) ---------------------------

std::string get_compute_partition(uint32_t index) {
  SmiSession session;
  auto gpus = gpu_handles();
  auto h = handle_for_hip_index(index, gpus);
  char buf[32] = {0};
  check(amdsmi_get_gpu_compute_partition(h, buf, sizeof(buf)), "get_compute_partition");
  return buf;
}

void set_compute_partition(uint32_t index, const std::string& mode) {
  amdsmi_compute_partition_type_t t;
  if (mode == "SPX") t = AMDSMI_COMPUTE_PARTITION_SPX;
  else if (mode == "DPX") t = AMDSMI_COMPUTE_PARTITION_DPX;
  else if (mode == "QPX") t = AMDSMI_COMPUTE_PARTITION_QPX;
  else if (mode == "CPX") t = AMDSMI_COMPUTE_PARTITION_CPX;
  else throw std::runtime_error("unknown partition mode " + mode +
                                " (SPX|DPX|QPX|CPX)");
  SmiSession session;
  auto gpus = gpu_handles();
  auto h = handle_for_hip_index(index, gpus);
  check(amdsmi_set_gpu_compute_partition(h, t), "set_compute_partition");
}

}  // namespace

PYBIND11_MODULE(_amdsmi, m) {
  m.doc() = "MI355X enumeration + occupancy via libamd_smi";
  m.def("get_compute_partition", &get_compute_partition,
        "Current compute-partition mode of one GPU (by HIP index)");
  m.def("set_compute_partition", &set_compute_partition,
        "Set SPX/DPX/QPX/CPX on one GPU (device must be idle; re-enumeration "
        "follows via the agent's periodic refresh)");
  m.def("enumerate_gpus", &enumerate_gpus,
        "Enumerate AMD GPUs; returns list of dicts sorted by HIP id");
  m.def("gpu_utilization", &gpu_utilization,
        "Engine busy % + VRAM usage for one GPU (by HIP index)");
  m.def("gpu_processes", &gpu_processes,
        "Per-process VRAM / CU-occupancy list for one GPU (by HIP index)");
}
