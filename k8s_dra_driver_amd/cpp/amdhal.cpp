// _amdhal — native binding to AMD SMI for MI355X device management.
//
// The NVML-replacement native boundary (reference
// binds NVML via cgo at cmd/nvidia-dra-plugin/nvlib.go:59-63). Links
// libamd_smi directly: every node this driver runs on has the ROCm stack,
// and a missing library must fail loudly at import, never fall back
// silently to fake hardware.
//
// Exposed surface (consumed by k8s_dra_driver_amd.hal.amdsmi):
//   init() / shutdown()
//   enumerate()                -> list[dict] per physical GPU
//   set_compute_partition(i, "SPX|DPX|TPX|QPX|CPX")
//   set_memory_partition(i, "NPS1|NPS2|NPS4|NPS8")
//
// Build: hipcc/amdclang++ -shared -fPIC $(python -m pybind11 --includes)
//        amdhal.cpp -lamd_smi -o _amdhal.so   (driven by setup.py)

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <amd_smi/amdsmi.h>

#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

std::mutex g_mutex;
bool g_initialized = false;
std::vector<amdsmi_processor_handle> g_processors;

[[noreturn]] void throw_status(const char* what, amdsmi_status_t st) {
  const char* msg = nullptr;
  amdsmi_status_code_to_string(st, &msg);
  throw std::runtime_error(std::string(what) + ": " +
                           (msg ? msg : ("status " + std::to_string(st))));
}

void check(const char* what, amdsmi_status_t st) {
  if (st != AMDSMI_STATUS_SUCCESS) throw_status(what, st);
}

void ensure_init() {
  if (!g_initialized) throw std::runtime_error("_amdhal: init() not called");
}

void hal_init() {
  std::lock_guard<std::mutex> lock(g_mutex);
  if (g_initialized) return;
  check("amdsmi_init", amdsmi_init(AMDSMI_INIT_AMD_GPUS));
  // Enumerate sockets -> processors once; refreshed only via re-init
  // (partition changes alter the processor list; callers re-init after).
  uint32_t socket_count = 0;
  check("amdsmi_get_socket_handles(count)",
        amdsmi_get_socket_handles(&socket_count, nullptr));
  std::vector<amdsmi_socket_handle> sockets(socket_count);
  check("amdsmi_get_socket_handles",
        amdsmi_get_socket_handles(&socket_count, sockets.data()));
  g_processors.clear();
  for (auto sock : sockets) {
    uint32_t n = 0;
    if (amdsmi_get_processor_handles(sock, &n, nullptr) !=
        AMDSMI_STATUS_SUCCESS)
      continue;
    std::vector<amdsmi_processor_handle> procs(n);
    if (amdsmi_get_processor_handles(sock, &n, procs.data()) !=
        AMDSMI_STATUS_SUCCESS)
      continue;
    for (auto p : procs) g_processors.push_back(p);
  }
  g_initialized = true;
}

void hal_shutdown() {
  std::lock_guard<std::mutex> lock(g_mutex);
  if (!g_initialized) return;
  g_processors.clear();
  amdsmi_shut_down();
  g_initialized = false;
}

void hal_reinit() {
  hal_shutdown();
  hal_init();
}

std::string bdf_to_string(const amdsmi_bdf_t& bdf) {
  char buf[32];
  std::snprintf(buf, sizeof(buf), "%04lx:%02x:%02x.%x",
                static_cast<unsigned long>(bdf.domain_number),
                bdf.bus_number, bdf.device_number, bdf.function_number);
  return std::string(buf);
}

py::dict describe_processor(amdsmi_processor_handle h, size_t index) {
  py::dict out;
  out["index"] = index;

  unsigned int uuid_len = AMDSMI_GPU_UUID_SIZE;
  char uuid[AMDSMI_GPU_UUID_SIZE] = {0};
  if (amdsmi_get_gpu_device_uuid(h, &uuid_len, uuid) == AMDSMI_STATUS_SUCCESS)
    out["uuid"] = std::string(uuid);

  amdsmi_asic_info_t asic{};
  if (amdsmi_get_gpu_asic_info(h, &asic) == AMDSMI_STATUS_SUCCESS) {
    out["market_name"] = std::string(asic.market_name);
    out["vendor_id"] = asic.vendor_id;
    out["device_id"] = asic.device_id;
    out["rev_id"] = asic.rev_id;
    out["asic_serial"] = std::string(asic.asic_serial);
    out["oam_id"] = asic.oam_id;
    out["num_compute_units"] = asic.num_of_compute_units;
    out["target_graphics_version"] = asic.target_graphics_version;
  }

  amdsmi_kfd_info_t kfd{};
  if (amdsmi_get_gpu_kfd_info(h, &kfd) == AMDSMI_STATUS_SUCCESS) {
    out["kfd_id"] = kfd.kfd_id;
    out["kfd_node_id"] = kfd.node_id;
    out["current_partition_id"] = kfd.current_partition_id;
  }

  amdsmi_vram_info_t vram{};
  if (amdsmi_get_gpu_vram_info(h, &vram) == AMDSMI_STATUS_SUCCESS) {
    out["vram_size_mb"] = vram.vram_size;
    out["vram_type"] = static_cast<int>(vram.vram_type);
    out["vram_bit_width"] = vram.vram_bit_width;
    out["vram_max_bandwidth_gbs"] = vram.vram_max_bandwidth;
    out["vram_vendor"] = std::string(vram.vram_vendor);
  }

  amdsmi_driver_info_t drv{};
  if (amdsmi_get_gpu_driver_info(h, &drv) == AMDSMI_STATUS_SUCCESS) {
    out["driver_version"] = std::string(drv.driver_version);
    out["driver_name"] = std::string(drv.driver_name);
  }

  amdsmi_board_info_t board{};
  if (amdsmi_get_gpu_board_info(h, &board) == AMDSMI_STATUS_SUCCESS)
    out["product_name"] = std::string(board.product_name);

  amdsmi_bdf_t bdf{};
  if (amdsmi_get_gpu_device_bdf(h, &bdf) == AMDSMI_STATUS_SUCCESS)
    out["bdf"] = bdf_to_string(bdf);

  char part[16] = {0};
  if (amdsmi_get_gpu_compute_partition(h, part, sizeof(part)) ==
      AMDSMI_STATUS_SUCCESS)
    out["compute_partition"] = std::string(part);

  char mem_part[16] = {0};
  if (amdsmi_get_gpu_memory_partition(h, mem_part, sizeof(mem_part)) ==
      AMDSMI_STATUS_SUCCESS)
    out["memory_partition"] = std::string(mem_part);

  amdsmi_memory_partition_config_t mpc{};
  if (amdsmi_get_gpu_memory_partition_config(h, &mpc) ==
      AMDSMI_STATUS_SUCCESS) {
    py::list caps;
    if (mpc.partition_caps.nps_flags.nps1_cap) caps.append("NPS1");
    if (mpc.partition_caps.nps_flags.nps2_cap) caps.append("NPS2");
    if (mpc.partition_caps.nps_flags.nps4_cap) caps.append("NPS4");
    if (mpc.partition_caps.nps_flags.nps8_cap) caps.append("NPS8");
    out["nps_caps"] = caps;
  }

  amdsmi_xgmi_info_t xgmi{};
  if (amdsmi_get_xgmi_info(h, &xgmi) == AMDSMI_STATUS_SUCCESS) {
    out["xgmi_hive_id"] = xgmi.xgmi_hive_id;
    out["xgmi_node_id"] = xgmi.xgmi_node_id;
    out["xgmi_lanes"] = static_cast<int>(xgmi.xgmi_lanes);
  }

  amdsmi_link_metrics_t links{};
  if (amdsmi_get_link_metrics(h, &links) == AMDSMI_STATUS_SUCCESS) {
    py::list lst;
    for (uint32_t i = 0; i < links.num_links &&
                         i < AMDSMI_MAX_NUM_XGMI_PHYSICAL_LINK;
         ++i) {
      const auto& l = links.links[i];
      py::dict ld;
      ld["bdf"] = bdf_to_string(l.bdf);
      ld["bit_rate_gbs"] = l.bit_rate;
      ld["max_bandwidth_gbs"] = l.max_bandwidth;
      ld["link_type"] =
          l.link_type == AMDSMI_LINK_TYPE_XGMI
              ? "XGMI"
              : (l.link_type == AMDSMI_LINK_TYPE_PCIE ? "PCIE" : "OTHER");
      lst.append(ld);
    }
    out["links"] = lst;
  }

  return out;
}

py::list hal_enumerate() {
  std::lock_guard<std::mutex> lock(g_mutex);
  ensure_init();
  py::list out;
  for (size_t i = 0; i < g_processors.size(); ++i)
    out.append(describe_processor(g_processors[i], i));
  return out;
}

amdsmi_compute_partition_type_t compute_mode_from_string(const std::string& s) {
  if (s == "SPX") return AMDSMI_COMPUTE_PARTITION_SPX;
  if (s == "DPX") return AMDSMI_COMPUTE_PARTITION_DPX;
  if (s == "TPX") return AMDSMI_COMPUTE_PARTITION_TPX;
  if (s == "QPX") return AMDSMI_COMPUTE_PARTITION_QPX;
  if (s == "CPX") return AMDSMI_COMPUTE_PARTITION_CPX;
  throw std::invalid_argument("unknown compute partition mode: " + s);
}

amdsmi_memory_partition_type_t memory_mode_from_string(const std::string& s) {
  if (s == "NPS1") return AMDSMI_MEMORY_PARTITION_NPS1;
  if (s == "NPS2") return AMDSMI_MEMORY_PARTITION_NPS2;
  if (s == "NPS4") return AMDSMI_MEMORY_PARTITION_NPS4;
  if (s == "NPS8") return AMDSMI_MEMORY_PARTITION_NPS8;
  throw std::invalid_argument("unknown memory partition (NPS) mode: " + s);
}

amdsmi_processor_handle processor_at(size_t index) {
  ensure_init();
  if (index >= g_processors.size())
    throw std::out_of_range("gpu index " + std::to_string(index) +
                            " out of range (have " +
                            std::to_string(g_processors.size()) + ")");
  return g_processors[index];
}

void hal_set_compute_partition(size_t index, const std::string& mode) {
  std::lock_guard<std::mutex> lock(g_mutex);
  check("amdsmi_set_gpu_compute_partition",
        amdsmi_set_gpu_compute_partition(processor_at(index),
                                         compute_mode_from_string(mode)));
}

void hal_set_memory_partition(size_t index, const std::string& mode) {
  std::lock_guard<std::mutex> lock(g_mutex);
  // NOTE: may take seconds (driver re-init semantics on some stacks);
  // callers hold the per-GPU lock.
  check("amdsmi_set_gpu_memory_partition",
        amdsmi_set_gpu_memory_partition(processor_at(index),
                                        memory_mode_from_string(mode)));
}

py::dict hal_lib_version() {
  py::dict out;
  amdsmi_version_t v{};
  if (amdsmi_get_lib_version(&v) == AMDSMI_STATUS_SUCCESS) {
    out["major"] = v.major;
    out["minor"] = v.minor;
    out["release"] = v.release;
  }
  return out;
}

}  // namespace

// PYBIND11_MODULE_NAME lets the ASan CI flavor build this same source as
// `_amdhal_asan` (module init name must match the .so filename).
#ifndef PYBIND11_MODULE_NAME
#define PYBIND11_MODULE_NAME _amdhal
#endif

PYBIND11_MODULE(PYBIND11_MODULE_NAME, m) {
  m.doc() = "AMD SMI native binding for the MI355X DRA driver";
  m.def("init", &hal_init, "Initialize AMD SMI (idempotent)");
  m.def("shutdown", &hal_shutdown, "Shut down AMD SMI (idempotent)");
  m.def("reinit", &hal_reinit,
        "Re-initialize (refreshes the processor list after repartition)");
  m.def("enumerate", &hal_enumerate,
        "Describe every GPU processor visible to AMD SMI");
  m.def("set_compute_partition", &hal_set_compute_partition,
        py::arg("index"), py::arg("mode"));
  m.def("set_memory_partition", &hal_set_memory_partition,
        py::arg("index"), py::arg("mode"));
  m.def("lib_version", &hal_lib_version);
}
