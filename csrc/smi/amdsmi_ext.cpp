// gpud_amd native SMI binding: pybind11 over ROCm's libamd_smi.
//
// This is the AMD-first replacement for the reference's go-nvml cgo boundary
// (reference: pkg/nvidia/nvml/instance.go, lib/lib.go — all GPU telemetry
// flows through one native binding). Design points:
//
//  * one shared amdsmi session for the whole daemon (init once, enumerate
//    once; reference keeps one nvml.Instance for all components);
//  * the per-GPU poll hot path is ONE call — metrics_snapshot() — which
//    gathers temperature/power/clocks/activity/VRAM/ECC/throttle/xGMI in
//    C++ with the GIL released, so a full 8-GPU telemetry sweep costs eight
//    Python→C++ transitions, not ~100;
//  * individual getters are also exposed for targeted checks and tests.
//
// Built in-tree as gpud_amd/smi/_amdsmi.so (see csrc/smi/setup.py).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <amd_smi/amdsmi.h>

#include <chrono>
#include <cstdlib>
#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

std::string status_str(amdsmi_status_t st) {
  const char* s = nullptr;
  if (amdsmi_status_code_to_string(st, &s) == AMDSMI_STATUS_SUCCESS && s) {
    return std::string(s);
  }
  return "status=" + std::to_string(static_cast<int>(st));
}

struct SmiError : std::runtime_error {
  amdsmi_status_t status;
  SmiError(const std::string& what, amdsmi_status_t st)
      : std::runtime_error(what + ": " + status_str(st)), status(st) {}
};

void check(amdsmi_status_t st, const char* what) {
  if (st != AMDSMI_STATUS_SUCCESS) throw SmiError(what, st);
}

// ---------------------------------------------------------------------------
// session: init + device enumeration (cached handles)
// ---------------------------------------------------------------------------

std::mutex g_mu;
// Serializes amdsmi entry from concurrent Python threads: the library's
// thread-safety is undocumented, and the daemon's component tickers can
// overlap distinct calls. Held inside the GIL-released scope so waiting
// never blocks the interpreter; per-cycle cost is nil (one caller
// dominates each poll cycle via the shared snapshot).
std::mutex g_call_mu;

// GPUD_AMDSMI_NO_CALL_MUTEX=1 disables the serialization ABOVE the library
// — an instrumentation knob for the ASan/TSan race hunt (docs/ROADMAP.md
// "Observed-but-unresolved" heap corruption): running the concurrency
// stress with the mutex off under ASan distinguishes a race in THIS
// binding / libamd_smi from corruption elsewhere. Never set in production.
bool call_mutex_disabled() {
  static const bool disabled = [] {
    const char* v = std::getenv("GPUD_AMDSMI_NO_CALL_MUTEX");
    return v != nullptr && v[0] == '1';
  }();
  return disabled;
}

// lock_guard that honors the kill switch
struct CallLock {
  std::unique_lock<std::mutex> lk;
  explicit CallLock(std::mutex& m) {
    if (!call_mutex_disabled()) lk = std::unique_lock<std::mutex>(m);
  }
};

bool g_initialized = false;
std::vector<amdsmi_processor_handle> g_handles;

void enumerate_locked() {
  g_handles.clear();
  uint32_t socket_count = 0;
  check(amdsmi_get_socket_handles(&socket_count, nullptr),
        "amdsmi_get_socket_handles(count)");
  std::vector<amdsmi_socket_handle> sockets(socket_count);
  check(amdsmi_get_socket_handles(&socket_count, sockets.data()),
        "amdsmi_get_socket_handles");
  for (auto& sock : sockets) {
    uint32_t dev_count = 0;
    auto st = amdsmi_get_processor_handles(sock, &dev_count, nullptr);
    if (st != AMDSMI_STATUS_SUCCESS) continue;
    std::vector<amdsmi_processor_handle> procs(dev_count);
    st = amdsmi_get_processor_handles(sock, &dev_count, procs.data());
    if (st != AMDSMI_STATUS_SUCCESS) continue;
    for (auto& p : procs) {
      processor_type_t ptype;
      if (amdsmi_get_processor_type(p, &ptype) != AMDSMI_STATUS_SUCCESS)
        continue;
      if (ptype == AMDSMI_PROCESSOR_TYPE_AMD_GPU) g_handles.push_back(p);
    }
  }
}

void smi_init() {
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_initialized) return;
  check(amdsmi_init(AMDSMI_INIT_AMD_GPUS), "amdsmi_init");
  g_initialized = true;
  enumerate_locked();
}

void smi_shutdown() {
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  std::lock_guard<std::mutex> lk(g_mu);
  if (!g_initialized) return;
  amdsmi_shut_down();
  g_initialized = false;
  g_handles.clear();
}

int device_count() {
  std::lock_guard<std::mutex> lk(g_mu);
  return static_cast<int>(g_handles.size());
}

amdsmi_processor_handle handle_at(int index) {
  std::lock_guard<std::mutex> lk(g_mu);
  if (!g_initialized) throw std::runtime_error("amdsmi not initialized");
  if (index < 0 || static_cast<size_t>(index) >= g_handles.size())
    throw std::out_of_range("gpu index out of range");
  return g_handles[static_cast<size_t>(index)];
}

// ---------------------------------------------------------------------------
// identity / static info
// ---------------------------------------------------------------------------

std::string device_uuid(int index) {
  auto h = handle_at(index);
  unsigned int len = AMDSMI_MAX_STRING_LENGTH;
  char buf[AMDSMI_MAX_STRING_LENGTH] = {0};
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  check(amdsmi_get_gpu_device_uuid(h, &len, buf), "amdsmi_get_gpu_device_uuid");
  return std::string(buf);
}

std::string device_bdf(int index) {
  auto h = handle_at(index);
  uint64_t bdfid = 0;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_bdf_id(h, &bdfid), "amdsmi_get_gpu_bdf_id");
  }
  // bdfid packs: [63:32] domain, [15:8] bus, [7:3] device, [2:0] function
  char buf[32];
  std::snprintf(buf, sizeof(buf), "%04x:%02x:%02x.%x",
                static_cast<unsigned>((bdfid >> 32) & 0xffffffff),
                static_cast<unsigned>((bdfid >> 8) & 0xff),
                static_cast<unsigned>((bdfid >> 3) & 0x1f),
                static_cast<unsigned>(bdfid & 0x7));
  return std::string(buf);
}

py::dict asic_info(int index) {
  auto h = handle_at(index);
  amdsmi_asic_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_asic_info(h, &info), "amdsmi_get_gpu_asic_info");
  }
  py::dict d;
  d["market_name"] = std::string(info.market_name);
  d["vendor_id"] = info.vendor_id;
  d["device_id"] = info.device_id;
  d["rev_id"] = info.rev_id;
  d["asic_serial"] = std::string(info.asic_serial);
  d["oam_id"] = info.oam_id;
  d["num_compute_units"] = info.num_of_compute_units;
  d["target_graphics_version"] = info.target_graphics_version;
  return d;
}

py::dict board_info(int index) {
  auto h = handle_at(index);
  amdsmi_board_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_board_info(h, &info), "amdsmi_get_gpu_board_info");
  }
  py::dict d;
  d["model_number"] = std::string(info.model_number);
  d["product_serial"] = std::string(info.product_serial);
  d["fru_id"] = std::string(info.fru_id);
  d["product_name"] = std::string(info.product_name);
  d["manufacturer_name"] = std::string(info.manufacturer_name);
  return d;
}

py::dict driver_info(int index) {
  auto h = handle_at(index);
  amdsmi_driver_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_driver_info(h, &info), "amdsmi_get_gpu_driver_info");
  }
  py::dict d;
  d["driver_version"] = std::string(info.driver_version);
  d["driver_date"] = std::string(info.driver_date);
  d["driver_name"] = std::string(info.driver_name);
  return d;
}

py::dict vbios_info(int index) {
  auto h = handle_at(index);
  amdsmi_vbios_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_vbios_info(h, &info), "amdsmi_get_gpu_vbios_info");
  }
  py::dict d;
  d["name"] = std::string(info.name);
  d["version"] = std::string(info.version);
  d["part_number"] = std::string(info.part_number);
  d["build_date"] = std::string(info.build_date);
  return d;
}

py::dict vram_info(int index) {
  auto h = handle_at(index);
  amdsmi_vram_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_vram_info(h, &info), "amdsmi_get_gpu_vram_info");
  }
  py::dict d;
  d["vram_type"] = static_cast<int>(info.vram_type);
  d["vram_vendor"] = std::string(info.vram_vendor);
  d["vram_size_bytes"] = info.vram_size;
  d["vram_bit_width"] = info.vram_bit_width;
  d["vram_max_bandwidth"] = info.vram_max_bandwidth;
  return d;
}

// ---------------------------------------------------------------------------
// dynamic telemetry — individual getters
// ---------------------------------------------------------------------------

int64_t temp_metric(int index, int sensor_type, int metric) {
  auto h = handle_at(index);
  int64_t v = 0;
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  check(amdsmi_get_temp_metric(
            h, static_cast<amdsmi_temperature_type_t>(sensor_type),
            static_cast<amdsmi_temperature_metric_t>(metric), &v),
        "amdsmi_get_temp_metric");
  return v;  // Celsius
}

py::dict power_info(int index) {
  auto h = handle_at(index);
  amdsmi_power_info_t info;
  std::memset(&info, 0, sizeof(info));
  amdsmi_power_cap_info_t cap;
  std::memset(&cap, 0, sizeof(cap));
  amdsmi_status_t st_cap;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_power_info(h, &info), "amdsmi_get_power_info");
    st_cap = amdsmi_get_power_cap_info(h, 0, &cap);
  }
  py::dict d;
  d["socket_power_w"] = info.socket_power;
  d["current_socket_power_w"] = info.current_socket_power;
  d["average_socket_power_w"] = info.average_socket_power;
  d["gfx_voltage_mv"] = info.gfx_voltage;
  d["power_limit_w"] = info.power_limit;
  if (st_cap == AMDSMI_STATUS_SUCCESS) {
    d["power_cap_uw"] = cap.power_cap;
    d["default_power_cap_uw"] = cap.default_power_cap;
    d["min_power_cap_uw"] = cap.min_power_cap;
    d["max_power_cap_uw"] = cap.max_power_cap;
  }
  return d;
}

py::dict clock_info(int index, int clk_type) {
  auto h = handle_at(index);
  amdsmi_clk_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_clock_info(h, static_cast<amdsmi_clk_type_t>(clk_type),
                                &info),
          "amdsmi_get_clock_info");
  }
  py::dict d;
  d["clk_mhz"] = info.clk;
  d["min_clk_mhz"] = info.min_clk;
  d["max_clk_mhz"] = info.max_clk;
  d["clk_locked"] = static_cast<int>(info.clk_locked);
  d["clk_deep_sleep"] = static_cast<int>(info.clk_deep_sleep);
  return d;
}

py::dict activity(int index) {
  auto h = handle_at(index);
  amdsmi_engine_usage_t u;
  std::memset(&u, 0, sizeof(u));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_activity(h, &u), "amdsmi_get_gpu_activity");
  }
  py::dict d;
  d["gfx_activity_pct"] = u.gfx_activity;
  d["umc_activity_pct"] = u.umc_activity;
  d["mm_activity_pct"] = u.mm_activity;
  return d;
}

py::dict vram_usage(int index) {
  auto h = handle_at(index);
  amdsmi_vram_usage_t u;
  std::memset(&u, 0, sizeof(u));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_vram_usage(h, &u), "amdsmi_get_gpu_vram_usage");
  }
  py::dict d;
  d["vram_total_mb"] = u.vram_total;
  d["vram_used_mb"] = u.vram_used;
  return d;
}

py::dict ecc_count_total(int index) {
  auto h = handle_at(index);
  amdsmi_error_count_t ec;
  std::memset(&ec, 0, sizeof(ec));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_total_ecc_count(h, &ec),
          "amdsmi_get_gpu_total_ecc_count");
  }
  py::dict d;
  d["correctable"] = ec.correctable_count;
  d["uncorrectable"] = ec.uncorrectable_count;
  d["deferred"] = ec.deferred_count;
  return d;
}

py::dict ecc_count_block(int index, uint64_t block) {
  auto h = handle_at(index);
  amdsmi_error_count_t ec;
  std::memset(&ec, 0, sizeof(ec));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_ecc_count(h, static_cast<amdsmi_gpu_block_t>(block),
                                   &ec),
          "amdsmi_get_gpu_ecc_count");
  }
  py::dict d;
  d["correctable"] = ec.correctable_count;
  d["uncorrectable"] = ec.uncorrectable_count;
  d["deferred"] = ec.deferred_count;
  return d;
}

py::dict bad_page_info(int index) {
  auto h = handle_at(index);
  uint32_t num = 0;
  amdsmi_status_t st;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    st = amdsmi_get_gpu_bad_page_info(h, &num, nullptr);
  }
  py::dict d;
  if (st != AMDSMI_STATUS_SUCCESS) throw SmiError("amdsmi_get_gpu_bad_page_info", st);
  std::vector<amdsmi_retired_page_record_t> recs(num);
  if (num > 0) {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_bad_page_info(h, &num, recs.data()),
          "amdsmi_get_gpu_bad_page_info(records)");
  }
  uint32_t reserved = 0, pending = 0, unreservable = 0;
  for (uint32_t i = 0; i < num; ++i) {
    switch (recs[i].status) {
      case AMDSMI_MEM_PAGE_STATUS_RESERVED: reserved++; break;
      case AMDSMI_MEM_PAGE_STATUS_PENDING: pending++; break;
      case AMDSMI_MEM_PAGE_STATUS_UNRESERVABLE: unreservable++; break;
    }
  }
  uint32_t threshold = 0;
  amdsmi_status_t st_thr;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    st_thr = amdsmi_get_gpu_bad_page_threshold(h, &threshold);
  }
  d["total"] = num;
  d["reserved"] = reserved;
  d["pending"] = pending;
  d["unreservable"] = unreservable;
  if (st_thr == AMDSMI_STATUS_SUCCESS) d["threshold"] = threshold;
  return d;
}

py::list process_list(int index) {
  auto h = handle_at(index);
  uint32_t n = 0;
  amdsmi_status_t st;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    st = amdsmi_get_gpu_process_list(h, &n, nullptr);
  }
  py::list out;
  if (st != AMDSMI_STATUS_SUCCESS && st != AMDSMI_STATUS_OUT_OF_RESOURCES)
    throw SmiError("amdsmi_get_gpu_process_list(count)", st);
  if (n == 0) return out;
  std::vector<amdsmi_proc_info_t> procs(n);
  std::memset(procs.data(), 0, sizeof(amdsmi_proc_info_t) * n);
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_process_list(h, &n, procs.data()),
          "amdsmi_get_gpu_process_list");
  }
  for (uint32_t i = 0; i < n; ++i) {
    py::dict p;
    p["name"] = std::string(procs[i].name);
    p["pid"] = procs[i].pid;
    p["mem_bytes"] = procs[i].mem;
    p["vram_mem_bytes"] = procs[i].memory_usage.vram_mem;
    p["gtt_mem_bytes"] = procs[i].memory_usage.gtt_mem;
    p["gfx_usage"] = procs[i].engine_usage.gfx;
    p["cu_occupancy"] = procs[i].cu_occupancy;
    out.append(p);
  }
  return out;
}

py::dict violation_status(int index) {
  auto h = handle_at(index);
  amdsmi_violation_status_t v;
  std::memset(&v, 0, sizeof(v));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_violation_status(h, &v), "amdsmi_get_violation_status");
  }
  py::dict d;
  d["acc_counter"] = v.acc_counter;
  d["acc_prochot_thrm"] = v.acc_prochot_thrm;
  d["acc_ppt_pwr"] = v.acc_ppt_pwr;
  d["acc_socket_thrm"] = v.acc_socket_thrm;
  d["acc_vr_thrm"] = v.acc_vr_thrm;
  d["acc_hbm_thrm"] = v.acc_hbm_thrm;
  d["per_prochot_thrm"] = v.per_prochot_thrm;
  d["per_ppt_pwr"] = v.per_ppt_pwr;
  d["per_socket_thrm"] = v.per_socket_thrm;
  d["per_vr_thrm"] = v.per_vr_thrm;
  d["per_hbm_thrm"] = v.per_hbm_thrm;
  d["active_prochot_thrm"] = static_cast<int>(v.active_prochot_thrm);
  d["active_ppt_pwr"] = static_cast<int>(v.active_ppt_pwr);
  d["active_socket_thrm"] = static_cast<int>(v.active_socket_thrm);
  d["active_vr_thrm"] = static_cast<int>(v.active_vr_thrm);
  d["active_hbm_thrm"] = static_cast<int>(v.active_hbm_thrm);
  return d;
}

py::dict xgmi_link_status(int index) {
  auto h = handle_at(index);
  amdsmi_xgmi_link_status_t s;
  std::memset(&s, 0, sizeof(s));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_gpu_xgmi_link_status(h, &s),
          "amdsmi_get_gpu_xgmi_link_status");
  }
  py::dict d;
  d["total_links"] = s.total_links;
  py::list states;
  for (uint32_t i = 0; i < s.total_links && i < AMDSMI_MAX_NUM_XGMI_LINKS; ++i)
    states.append(static_cast<int>(s.status[i]));  // 0 down, 1 up, 2 disabled
  d["states"] = states;
  return d;
}

int xgmi_error_status(int index) {
  auto h = handle_at(index);
  amdsmi_xgmi_status_t st;
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  check(amdsmi_gpu_xgmi_error_status(h, &st), "amdsmi_gpu_xgmi_error_status");
  return static_cast<int>(st);
}

py::dict xgmi_info(int index) {
  auto h = handle_at(index);
  amdsmi_xgmi_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_xgmi_info(h, &info), "amdsmi_get_xgmi_info");
  }
  py::dict d;
  d["xgmi_lanes"] = static_cast<int>(info.xgmi_lanes);
  d["xgmi_hive_id"] = info.xgmi_hive_id;
  d["xgmi_node_id"] = info.xgmi_node_id;
  d["index"] = info.index;
  return d;
}

py::dict link_metrics(int index) {
  auto h = handle_at(index);
  amdsmi_link_metrics_t lm;
  std::memset(&lm, 0, sizeof(lm));
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_link_metrics(h, &lm), "amdsmi_get_link_metrics");
  }
  py::dict d;
  d["num_links"] = lm.num_links;
  py::list links;
  for (uint32_t i = 0; i < lm.num_links && i < AMDSMI_MAX_NUM_XGMI_PHYSICAL_LINK;
       ++i) {
    py::dict l;
    l["bit_rate"] = lm.links[i].bit_rate;
    l["max_bandwidth"] = lm.links[i].max_bandwidth;
    l["link_type"] = static_cast<int>(lm.links[i].link_type);
    l["read_kb"] = lm.links[i].read;
    l["write_kb"] = lm.links[i].write;
    char bdf[32];
    std::snprintf(bdf, sizeof(bdf), "%04x:%02x:%02x.%x",
                  static_cast<unsigned>(lm.links[i].bdf.bdf.domain_number),
                  static_cast<unsigned>(lm.links[i].bdf.bdf.bus_number),
                  static_cast<unsigned>(lm.links[i].bdf.bdf.device_number),
                  static_cast<unsigned>(lm.links[i].bdf.bdf.function_number));
    l["bdf"] = std::string(bdf);
    links.append(l);
  }
  d["links"] = links;
  return d;
}

py::dict energy_count(int index) {
  auto h = handle_at(index);
  uint64_t acc = 0, ts = 0;
  float res = 0.f;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    check(amdsmi_get_energy_count(h, &acc, &res, &ts),
          "amdsmi_get_energy_count");
  }
  py::dict d;
  d["energy_accumulator"] = acc;
  d["counter_resolution_uj"] = res;
  d["timestamp"] = ts;
  return d;
}

// PCIe link health: width/speed plus the replay / L0-recovery / NAK
// counters — the PCIe analog of the reference's NVLink replay/recovery/CRC
// error counters (components/accelerator/nvidia/nvlink/nvlink.go:86-93).
py::dict pcie_info(int index) {
  auto h = handle_at(index);
  amdsmi_pcie_info_t info;
  std::memset(&info, 0, sizeof(info));
  {
    py::gil_scoped_release nogil;
    CallLock call_lk(g_call_mu);
    check(amdsmi_get_pcie_info(h, &info), "amdsmi_get_pcie_info");
  }
  py::dict d;
  d["max_width"] = info.pcie_static.max_pcie_width;
  d["max_speed_gts"] = info.pcie_static.max_pcie_speed;
  d["interface_version"] = info.pcie_static.pcie_interface_version;
  d["width"] = info.pcie_metric.pcie_width;
  d["speed_mts"] = info.pcie_metric.pcie_speed;
  d["bandwidth_mbps"] = info.pcie_metric.pcie_bandwidth;
  d["replay_count"] = info.pcie_metric.pcie_replay_count;
  d["l0_to_recovery_count"] = info.pcie_metric.pcie_l0_to_recovery_count;
  d["replay_rollover_count"] = info.pcie_metric.pcie_replay_roll_over_count;
  d["nak_sent_count"] = info.pcie_metric.pcie_nak_sent_count;
  d["nak_received_count"] = info.pcie_metric.pcie_nak_received_count;
  return d;
}

// ---------------------------------------------------------------------------
// partitioning + CPER RAS records
// ---------------------------------------------------------------------------

// Compute/memory partition mode (SPX/DPX/.../NPS1...) plus the accelerator
// partition profile. Read-only surface for the partition component — gpud is
// a monitor, never a partition setter. Reference analog: none (NVML MIG is
// not monitored by gpud); this is MI355X-specific coverage (SURVEY.md
// ROADMAP: partition-aware enumeration).
py::dict partition_info(int index) {
  auto h = handle_at(index);
  py::dict d;
  char buf[64] = {0};
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    if (amdsmi_get_gpu_compute_partition(h, buf, sizeof(buf)) ==
        AMDSMI_STATUS_SUCCESS) {
      // re-acquire handled after block
    } else {
      buf[0] = '\0';
    }
  }
  if (buf[0]) d["compute_partition"] = std::string(buf);
  char mbuf[64] = {0};
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    if (amdsmi_get_gpu_memory_partition(h, mbuf, sizeof(mbuf)) !=
        AMDSMI_STATUS_SUCCESS)
      mbuf[0] = '\0';
  }
  if (mbuf[0]) d["memory_partition"] = std::string(mbuf);
  amdsmi_accelerator_partition_profile_t prof;
  std::memset(&prof, 0, sizeof(prof));
  uint32_t part_ids[AMDSMI_MAX_ACCELERATOR_PARTITIONS] = {0};
  amdsmi_status_t prc;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    prc = amdsmi_get_gpu_accelerator_partition_profile(h, &prof, part_ids);
  }
  if (prc == AMDSMI_STATUS_SUCCESS) {
    static const char* kTypes[] = {"INVALID", "SPX", "DPX", "TPX", "QPX",
                                   "CPX"};
    int t = static_cast<int>(prof.profile_type);
    d["accelerator_profile_type"] =
        (t >= 0 && t <= 5) ? std::string(kTypes[t]) : std::to_string(t);
    d["num_partitions"] = prof.num_partitions;
    d["partition_id"] = part_ids[0];
  }
  return d;
}

// CPER (Common Platform Error Record) entries cached by the amdgpu driver —
// structured RAS with severity + notify-type GUID, richer than dmesg text.
// Cursor-based drain: the caller passes the cursor from the previous call
// (0 at daemon start) and we loop while the library reports MORE_DATA.
// Returns {"entries": [...], "cursor": next_cursor}.
py::dict cper_entries(int index, uint32_t severity_mask, uint64_t cursor,
                      int max_rounds) {
  auto h = handle_at(index);
  py::list entries;
  std::vector<char> data(1 << 20);  // 1 MiB CPER payload buffer per round
  std::vector<amdsmi_cper_hdr_t*> hdrs(256);
  bool supported = true;
  for (int round = 0; round < max_rounds; ++round) {
    uint64_t buf_size = data.size();
    uint64_t entry_count = hdrs.size();
    amdsmi_status_t rc;
    {
      py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
      rc = amdsmi_get_gpu_cper_entries(h, severity_mask, data.data(),
                                       &buf_size, hdrs.data(), &entry_count,
                                       &cursor);
    }
    if (rc != AMDSMI_STATUS_SUCCESS && rc != AMDSMI_STATUS_MORE_DATA) {
      if (round == 0) supported = false;
      break;
    }
    for (uint64_t i = 0; i < entry_count; ++i) {
      const amdsmi_cper_hdr_t* hd = hdrs[i];
      if (hd == nullptr) continue;
      py::dict e;
      e["severity"] = static_cast<int>(hd->error_severity);
      static const char* kSev[] = {"non_fatal_uncorrected", "fatal",
                                   "non_fatal_corrected"};
      int sv = static_cast<int>(hd->error_severity);
      e["severity_name"] =
          (sv >= 0 && sv <= 2) ? std::string(kSev[sv]) : std::to_string(sv);
      e["section_count"] = hd->sec_cnt;
      e["record_length"] = hd->record_length;
      e["record_id"] = std::string(hd->record_id, strnlen(hd->record_id, 8));
      if (hd->cper_valid_bits.valid_bits.timestamp) {
        char ts[40];
        const amdsmi_cper_timestamp_t& t = hd->timestamp;
        std::snprintf(ts, sizeof(ts), "%02u%02u-%02u-%02uT%02u:%02u:%02u",
                      t.century, t.year, t.month, t.day, t.hours, t.minutes,
                      t.seconds);
        e["timestamp"] = std::string(ts);
      }
      // notify-type GUID: first 8 bytes little-endian match the
      // amdsmi_cper_notify_type_t discriminants
      uint64_t ng = 0;
      std::memcpy(&ng, hd->notify_type.b, 8);
      const char* nt = nullptr;
      switch (ng) {
        case 0x450eBDD72DCE8BB1ull: nt = "CMC"; break;
        case 0x4a55D8434E292F96ull: nt = "CPE"; break;
        case 0x4cc5919CE8F56FFEull: nt = "MCE"; break;
        case 0x4dfc1A16CF93C01Full: nt = "PCIE"; break;
        case 0x454a9308CC5263E8ull: nt = "INIT"; break;
        case 0x42c9B7E65BAD89FFull: nt = "NMI"; break;
        case 0x409aAB403D61A466ull: nt = "BOOT"; break;
        case 0x4c27C6B3667DD791ull: nt = "DMAR"; break;
        case 0x11E4BBE89A78788Aull: nt = "SEA"; break;
        case 0x4E87B0AE5C284C81ull: nt = "SEI"; break;
        case 0x4214520409A9D5ACull: nt = "PEI"; break;
        case 0x49A341DF69293BC9ull: nt = "CXL"; break;
        default: break;
      }
      if (nt != nullptr) e["notify_type"] = std::string(nt);
      // Section descriptors (UEFI spec appendix N: 72-byte descriptors
      // following the 128-byte record header) — decoded for FRU-level
      // attribution: section type, section severity, FRU id/text when the
      // descriptor's validation bits say they are present.
      {
        const char* rec = reinterpret_cast<const char*>(hd);
        const char* buf_end = data.data() + data.size();
        const uint32_t rec_len = hd->record_length;
        const size_t hdr_sz = sizeof(amdsmi_cper_hdr_t);
        constexpr size_t kDescSz = 72;
        py::list sections;
        for (uint16_t sidx = 0; sidx < hd->sec_cnt && sidx < 16; ++sidx) {
          const char* desc = rec + hdr_sz + (size_t)sidx * kDescSz;
          if (desc + kDescSz > buf_end) break;
          if (rec_len && desc + kDescSz > rec + rec_len) break;
          py::dict s;
          uint32_t soff = 0, slen = 0, ssev = 0;
          std::memcpy(&soff, desc + 0, 4);
          std::memcpy(&slen, desc + 4, 4);
          std::memcpy(&ssev, desc + 48, 4);
          uint8_t valid = *(const uint8_t*)(desc + 10);
          s["offset"] = soff;
          s["length"] = slen;
          s["severity"] = ssev;
          // section type GUID (mixed-endian text form)
          const uint8_t* g = (const uint8_t*)(desc + 16);
          char gs[40];
          std::snprintf(gs, sizeof(gs),
                        "%02x%02x%02x%02x-%02x%02x-%02x%02x-%02x%02x-"
                        "%02x%02x%02x%02x%02x%02x",
                        g[3], g[2], g[1], g[0], g[5], g[4], g[7], g[6],
                        g[8], g[9], g[10], g[11], g[12], g[13], g[14],
                        g[15]);
          std::string guid(gs);
          s["type_guid"] = guid;
          // well-known UEFI CPER section types
          const char* tname = nullptr;
          if (guid == "a5bc1114-6f64-4ede-b863-3e83ed7c83b1")
            tname = "memory";
          else if (guid == "d995e954-bbc1-430f-ad91-b44dcb3c6f35")
            tname = "pcie";
          else if (guid == "9876ccad-47b4-4bdb-b65e-16f193c4f3db")
            tname = "processor_generic";
          else if (guid == "81212a96-09ed-4996-9471-8d729c8e69ed")
            tname = "firmware";
          else if (guid == "5b51fef7-c79d-4434-8f1b-aa62de3e2c64")
            tname = "dmar_generic";
          if (tname != nullptr) s["type_name"] = std::string(tname);
          if (valid & 0x1) {  // FRU id GUID valid
            const uint8_t* f = (const uint8_t*)(desc + 32);
            char fs[40];
            std::snprintf(fs, sizeof(fs),
                          "%02x%02x%02x%02x-%02x%02x-%02x%02x-%02x%02x-"
                          "%02x%02x%02x%02x%02x%02x",
                          f[3], f[2], f[1], f[0], f[5], f[4], f[7], f[6],
                          f[8], f[9], f[10], f[11], f[12], f[13], f[14],
                          f[15]);
            s["fru_id"] = std::string(fs);
          }
          if (valid & 0x2) {  // FRU text valid (20-byte ASCII)
            s["fru_text"] =
                std::string(desc + 52, strnlen(desc + 52, 20));
          }
          sections.append(s);
        }
        if (py::len(sections) > 0) e["sections"] = sections;
      }
      entries.append(e);
    }
    if (rc != AMDSMI_STATUS_MORE_DATA) break;
  }
  py::dict out;
  out["supported"] = supported;
  out["entries"] = entries;
  out["cursor"] = cursor;
  return out;
}

// ---------------------------------------------------------------------------
// the poll hot path: everything a telemetry sweep needs, in one native call
// ---------------------------------------------------------------------------

struct Snapshot {
  // filled flags let Python distinguish "not supported" from zero
  bool ok_temp = false, ok_power = false, ok_clock = false,
       ok_activity = false, ok_vram = false, ok_ecc = false,
       ok_throttle = false, ok_xgmi = false, ok_metrics = false;
  int64_t temp_edge = 0, temp_hotspot = 0, temp_vram = 0;
  int64_t temp_edge_limit = 0, temp_hotspot_limit = 0, temp_vram_limit = 0;
  int64_t temp_hotspot_shutdown = 0;
  uint32_t power_w = 0, avg_power_w = 0, power_limit_w = 0;
  uint64_t power_cap_uw = 0;
  uint16_t gfx_mhz = 0, mem_mhz = 0, gfx_max_mhz = 0, mem_max_mhz = 0;
  uint16_t gfx_activity = 0, umc_activity = 0, mm_activity = 0;
  amdsmi_vram_usage_t vram;
  amdsmi_error_count_t ecc;
  uint32_t xgmi_total_links = 0;
  int xgmi_states[AMDSMI_MAX_NUM_XGMI_LINKS] = {0};
  int xgmi_err = -1;
  uint32_t throttle_status = 0;
  uint64_t indep_throttle_status = 0;
  bool ok_bad_pages = false;
  uint32_t bp_total = 0, bp_reserved = 0, bp_pending = 0, bp_unreservable = 0;
  uint32_t bp_threshold = 0;
  bool ok_bp_threshold = false;
  // throttle residency accumulators from gpu_metrics (violation analog)
  uint64_t thr_acc_counter = 0, thr_prochot = 0, thr_ppt = 0, thr_socket = 0,
           thr_vr = 0, thr_hbm = 0;
  // per-XCC (per-XCD) instantaneous gfx busy from xcp_stats — the CDNA
  // per-engine-cluster utilization breakdown (8 XCDs on MI355X); a sick
  // XCD shows as an outlier against its siblings
  uint32_t xcc_busy[AMDSMI_MAX_NUM_XCP * AMDSMI_MAX_NUM_XCC] = {0};
  uint32_t n_xcc = 0;
};

// Static per-device values (temperature limits, power caps, max clocks, the
// bad-page threshold) never change at runtime — they are fetched once per
// handle and cached, keeping the per-cycle snapshot down to the dynamic
// ioctls only (~1 ms/GPU instead of ~3 ms; amdsmi_get_violation_status at
// ~100 ms stays off this path entirely, see NOTE below).
struct StaticInfo {
  bool filled = false;
  int64_t temp_edge_limit = 0, temp_hotspot_limit = 0, temp_vram_limit = 0;
  int64_t temp_hotspot_shutdown = 0;
  uint32_t power_limit_w = 0;
  uint64_t power_cap_uw = 0;
  uint16_t gfx_max_mhz = 0, mem_max_mhz = 0;
  uint32_t bp_threshold = 0;
  bool has_bp_threshold = false;
};

std::mutex g_static_mu;
std::map<amdsmi_processor_handle, StaticInfo> g_static;

const StaticInfo& static_info_for(amdsmi_processor_handle h) {
  {
    std::lock_guard<std::mutex> lk(g_static_mu);
    auto it = g_static.find(h);
    if (it != g_static.end() && it->second.filled) return it->second;
  }
  StaticInfo si;
  amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_EDGE, AMDSMI_TEMP_CRITICAL,
                         &si.temp_edge_limit);
  amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_HOTSPOT,
                         AMDSMI_TEMP_CRITICAL, &si.temp_hotspot_limit);
  amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_VRAM, AMDSMI_TEMP_CRITICAL,
                         &si.temp_vram_limit);
  amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_HOTSPOT,
                         AMDSMI_TEMP_SHUTDOWN, &si.temp_hotspot_shutdown);
  amdsmi_power_cap_info_t cap;
  std::memset(&cap, 0, sizeof(cap));
  if (amdsmi_get_power_cap_info(h, 0, &cap) == AMDSMI_STATUS_SUCCESS)
    si.power_cap_uw = cap.power_cap;
  amdsmi_power_info_t pw;
  std::memset(&pw, 0, sizeof(pw));
  if (amdsmi_get_power_info(h, &pw) == AMDSMI_STATUS_SUCCESS) {
    // observed on ROCm 7.2/MI355X: power_limit reported in MICROwatts
    // (1400000000 for a 1400 W board) — normalize to watts
    si.power_limit_w = pw.power_limit > 100000
                           ? (uint32_t)(pw.power_limit / 1000000)
                           : pw.power_limit;
  }
  if (si.power_limit_w == 0 && si.power_cap_uw > 0)
    si.power_limit_w = (uint32_t)(si.power_cap_uw / 1000000);
  amdsmi_clk_info_t ci;
  std::memset(&ci, 0, sizeof(ci));
  if (amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_GFX, &ci) ==
      AMDSMI_STATUS_SUCCESS)
    si.gfx_max_mhz = (uint16_t)ci.max_clk;
  std::memset(&ci, 0, sizeof(ci));
  if (amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_MEM, &ci) ==
      AMDSMI_STATUS_SUCCESS)
    si.mem_max_mhz = (uint16_t)ci.max_clk;
  si.has_bp_threshold = amdsmi_get_gpu_bad_page_threshold(
                            h, &si.bp_threshold) == AMDSMI_STATUS_SUCCESS;
  si.filled = true;
  std::lock_guard<std::mutex> lk(g_static_mu);
  auto& slot = g_static[h];
  slot = si;
  return slot;
}

inline bool gm_valid16(uint16_t v) { return v != 0 && v != 0xFFFF; }

void take_snapshot(amdsmi_processor_handle h, Snapshot& s) {
  const StaticInfo& si = static_info_for(h);
  s.temp_edge_limit = si.temp_edge_limit;
  s.temp_hotspot_limit = si.temp_hotspot_limit;
  s.temp_vram_limit = si.temp_vram_limit;
  s.temp_hotspot_shutdown = si.temp_hotspot_shutdown;
  s.power_limit_w = si.power_limit_w;
  s.power_cap_uw = si.power_cap_uw;
  s.gfx_max_mhz = si.gfx_max_mhz;
  s.mem_max_mhz = si.mem_max_mhz;

  // ONE gpu_metrics ioctl covers temps, activity, power, current clocks,
  // throttle residencies and xGMI link state
  amdsmi_gpu_metrics_t gm;
  std::memset(&gm, 0, sizeof(gm));
  if (amdsmi_get_gpu_metrics_info(h, &gm) == AMDSMI_STATUS_SUCCESS) {
    s.ok_metrics = true;
    s.throttle_status = gm.throttle_status;
    s.indep_throttle_status = gm.indep_throttle_status;
    if (gm_valid16(gm.temperature_hotspot) || gm_valid16(gm.temperature_edge) ||
        gm_valid16(gm.temperature_mem)) {
      s.ok_temp = true;
      s.temp_edge = gm_valid16(gm.temperature_edge) ? gm.temperature_edge
                                                    : gm.temperature_hotspot;
      s.temp_hotspot = gm.temperature_hotspot;
      s.temp_vram = gm.temperature_mem;
    }
    s.ok_activity = true;
    s.gfx_activity = gm.average_gfx_activity;
    s.umc_activity = gm.average_umc_activity;
    s.mm_activity = gm.average_mm_activity;
    // per-XCC busy: xcp_stats gfx_busy_inst, UINT32_MAX = N/A sentinel
    for (uint32_t p = 0; p < AMDSMI_MAX_NUM_XCP; ++p) {
      for (uint32_t x = 0; x < AMDSMI_MAX_NUM_XCC; ++x) {
        uint32_t v = gm.xcp_stats[p].gfx_busy_inst[x];
        if (v == UINT32_MAX) continue;
        if (s.n_xcc < AMDSMI_MAX_NUM_XCP * AMDSMI_MAX_NUM_XCC)
          s.xcc_busy[s.n_xcc++] = v;
      }
    }
    s.ok_power = true;
    s.power_w = gm_valid16(gm.current_socket_power)
                    ? gm.current_socket_power
                    : gm.average_socket_power;
    // 0xFFFF = not-supported sentinel in the metrics table
    s.avg_power_w = gm_valid16(gm.average_socket_power)
                        ? gm.average_socket_power
                        : s.power_w;
    s.ok_clock = true;
    s.gfx_mhz = gm_valid16(gm.current_gfxclk) ? gm.current_gfxclk
                                              : gm.average_gfxclk_frequency;
    s.mem_mhz = gm_valid16(gm.current_uclk) ? gm.current_uclk
                                            : gm.average_uclk_frequency;
    s.ok_throttle = true;
    s.thr_acc_counter = gm.accumulation_counter;
    s.thr_prochot = gm.prochot_residency_acc;
    s.thr_ppt = gm.ppt_residency_acc;
    s.thr_socket = gm.socket_thm_residency_acc;
    s.thr_vr = gm.vr_thm_residency_acc;
    s.thr_hbm = gm.hbm_thm_residency_acc;
    // xGMI link state straight from the metrics table (up/down per link)
    uint32_t nlinks = 0;
    for (uint32_t i = 0; i < AMDSMI_MAX_NUM_XGMI_LINKS; ++i) {
      uint16_t st = gm.xgmi_link_status[i];
      if (st == 0xFFFF) break;
      s.xgmi_states[nlinks++] = (int)st;
    }
    if (nlinks > 0) {
      s.ok_xgmi = true;
      s.xgmi_total_links = nlinks;
    }
  }
  // fallbacks for boards whose metrics table lacks a field
  if (!s.ok_temp) {
    int64_t t = 0;
    if (amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_HOTSPOT,
                               AMDSMI_TEMP_CURRENT, &t) ==
        AMDSMI_STATUS_SUCCESS) {
      s.ok_temp = true;
      s.temp_hotspot = t;
      s.temp_edge = t;
      amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_VRAM,
                             AMDSMI_TEMP_CURRENT, &s.temp_vram);
    }
  }
  if (!s.ok_xgmi) {
    amdsmi_xgmi_link_status_t xs;
    std::memset(&xs, 0, sizeof(xs));
    if (amdsmi_get_gpu_xgmi_link_status(h, &xs) == AMDSMI_STATUS_SUCCESS) {
      s.ok_xgmi = true;
      s.xgmi_total_links = xs.total_links;
      for (uint32_t i = 0;
           i < xs.total_links && i < AMDSMI_MAX_NUM_XGMI_LINKS; ++i)
        s.xgmi_states[i] = (int)xs.status[i];
    }
  }
  amdsmi_xgmi_status_t xe;
  if (amdsmi_gpu_xgmi_error_status(h, &xe) == AMDSMI_STATUS_SUCCESS)
    s.xgmi_err = static_cast<int>(xe);
  std::memset(&s.vram, 0, sizeof(s.vram));
  s.ok_vram = amdsmi_get_gpu_vram_usage(h, &s.vram) == AMDSMI_STATUS_SUCCESS;
  std::memset(&s.ecc, 0, sizeof(s.ecc));
  s.ok_ecc =
      amdsmi_get_gpu_total_ecc_count(h, &s.ecc) == AMDSMI_STATUS_SUCCESS;
  // NOTE: amdsmi_get_violation_status is NOT called here — it blocks ~100 ms
  // per GPU (double-samples internally to compute per_* rates). The same
  // throttle residency accumulators come from gpu_metrics above; the
  // throttle component derives activity from deltas across its own polls.
  // The explicit violation_status() getter remains for manual diags.

  // bad pages (retired HBM rows — the remapped-rows analog)
  uint32_t bp_num = 0;
  if (amdsmi_get_gpu_bad_page_info(h, &bp_num, nullptr) ==
      AMDSMI_STATUS_SUCCESS) {
    s.ok_bad_pages = true;
    s.bp_total = bp_num;
    if (bp_num > 0 && bp_num <= 65536) {
      std::vector<amdsmi_retired_page_record_t> recs(bp_num);
      if (amdsmi_get_gpu_bad_page_info(h, &bp_num, recs.data()) ==
          AMDSMI_STATUS_SUCCESS) {
        for (uint32_t i = 0; i < bp_num; ++i) {
          switch (recs[i].status) {
            case AMDSMI_MEM_PAGE_STATUS_RESERVED: s.bp_reserved++; break;
            case AMDSMI_MEM_PAGE_STATUS_PENDING: s.bp_pending++; break;
            case AMDSMI_MEM_PAGE_STATUS_UNRESERVABLE: s.bp_unreservable++; break;
          }
        }
      }
    }
    s.ok_bp_threshold = si.has_bp_threshold;
    s.bp_threshold = si.bp_threshold;
  }
}

py::dict snapshot_to_dict(const Snapshot& s) {
  py::dict d;
  if (s.ok_temp) {
    py::dict t;
    t["edge_c"] = s.temp_edge;
    t["hotspot_c"] = s.temp_hotspot;
    t["vram_c"] = s.temp_vram;
    t["edge_limit_c"] = s.temp_edge_limit;
    t["hotspot_limit_c"] = s.temp_hotspot_limit;
    t["vram_limit_c"] = s.temp_vram_limit;
    t["hotspot_shutdown_c"] = s.temp_hotspot_shutdown;
    d["temperature"] = t;
  }
  if (s.ok_power) {
    py::dict p;
    p["socket_power_w"] = s.power_w;
    p["current_socket_power_w"] = s.power_w;
    p["average_socket_power_w"] = s.avg_power_w;
    p["power_limit_w"] = s.power_limit_w;
    if (s.power_cap_uw) p["power_cap_uw"] = s.power_cap_uw;
    d["power"] = p;
  }
  if (s.ok_clock) {
    py::dict c;
    c["gfx_mhz"] = s.gfx_mhz;
    c["gfx_max_mhz"] = s.gfx_max_mhz;
    c["mem_mhz"] = s.mem_mhz;
    c["mem_max_mhz"] = s.mem_max_mhz;
    d["clock"] = c;
  }
  if (s.ok_activity) {
    py::dict a;
    a["gfx_activity_pct"] = s.gfx_activity;
    a["umc_activity_pct"] = s.umc_activity;
    a["mm_activity_pct"] = s.mm_activity;
    if (s.n_xcc > 0) {
      py::list xb;
      for (uint32_t i = 0; i < s.n_xcc; ++i) xb.append(s.xcc_busy[i]);
      a["xcc_busy_pct"] = xb;
    }
    d["activity"] = a;
  }
  if (s.ok_vram) {
    py::dict v;
    v["vram_total_mb"] = s.vram.vram_total;
    v["vram_used_mb"] = s.vram.vram_used;
    d["vram"] = v;
  }
  if (s.ok_ecc) {
    py::dict e;
    e["correctable"] = s.ecc.correctable_count;
    e["uncorrectable"] = s.ecc.uncorrectable_count;
    e["deferred"] = s.ecc.deferred_count;
    d["ecc"] = e;
  }
  if (s.ok_throttle) {
    // residency accumulators from gpu_metrics; the throttle component
    // derives activity/rates from deltas between its own polls
    py::dict v;
    v["acc_counter"] = s.thr_acc_counter;
    v["acc_prochot_thrm"] = s.thr_prochot;
    v["acc_ppt_pwr"] = s.thr_ppt;
    v["acc_socket_thrm"] = s.thr_socket;
    v["acc_vr_thrm"] = s.thr_vr;
    v["acc_hbm_thrm"] = s.thr_hbm;
    d["violation"] = v;
  }
  if (s.ok_xgmi) {
    py::dict x;
    x["total_links"] = s.xgmi_total_links;
    py::list states;
    for (uint32_t i = 0; i < s.xgmi_total_links; ++i)
      states.append(s.xgmi_states[i]);
    x["states"] = states;
    d["xgmi_link_status"] = x;
  }
  if (s.xgmi_err >= 0) d["xgmi_error_status"] = s.xgmi_err;
  if (s.ok_bad_pages) {
    py::dict b;
    b["total"] = s.bp_total;
    b["reserved"] = s.bp_reserved;
    b["pending"] = s.bp_pending;
    b["unreservable"] = s.bp_unreservable;
    if (s.ok_bp_threshold) b["threshold"] = s.bp_threshold;
    d["bad_pages"] = b;
  }
  if (s.ok_metrics) {
    py::dict m;
    m["throttle_status"] = s.throttle_status;
    m["indep_throttle_status"] = s.indep_throttle_status;
    m["current_gfxclk_mhz"] = s.gfx_mhz;
    m["current_uclk_mhz"] = s.mem_mhz;
    m["average_socket_power_w"] = s.avg_power_w;
    d["gpu_metrics"] = m;
  }
  return d;
}


bool power_management_enabled(int index) {
  auto h = handle_at(index);
  bool enabled = false;
  py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
  check(amdsmi_is_gpu_power_management_enabled(h, &enabled),
        "amdsmi_is_gpu_power_management_enabled");
  return enabled;
}

py::dict metrics_snapshot(int index) {
  auto h = handle_at(index);
  Snapshot s;
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    take_snapshot(h, s);
  }
  return snapshot_to_dict(s);
}

// per-call wall-time breakdown of one snapshot — the poll-latency
// instrument used to pick which SMI calls stay on the fast path
py::dict snapshot_timings(int index) {
  auto h = handle_at(index);
  py::dict out;
  auto time_call = [&](const char* name, auto&& fn) {
    auto t0 = std::chrono::steady_clock::now();
    fn();
    auto us = std::chrono::duration_cast<std::chrono::microseconds>(
                  std::chrono::steady_clock::now() - t0)
                  .count();
    out[name] = (double)us;
  };
  {
    int64_t tv;
    time_call("temp_edge", [&] {
      amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_EDGE,
                             AMDSMI_TEMP_CURRENT, &tv);
    });
    time_call("temp_hotspot", [&] {
      amdsmi_get_temp_metric(h, AMDSMI_TEMPERATURE_TYPE_HOTSPOT,
                             AMDSMI_TEMP_CURRENT, &tv);
    });
    amdsmi_power_info_t pw;
    time_call("power_info", [&] { amdsmi_get_power_info(h, &pw); });
    amdsmi_power_cap_info_t pc;
    time_call("power_cap", [&] { amdsmi_get_power_cap_info(h, 0, &pc); });
    amdsmi_clk_info_t ci;
    time_call("clock_gfx", [&] {
      amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_GFX, &ci);
    });
    time_call("clock_mem", [&] {
      amdsmi_get_clock_info(h, AMDSMI_CLK_TYPE_MEM, &ci);
    });
    amdsmi_engine_usage_t eu;
    time_call("activity", [&] { amdsmi_get_gpu_activity(h, &eu); });
    amdsmi_vram_usage_t vu;
    time_call("vram_usage", [&] { amdsmi_get_gpu_vram_usage(h, &vu); });
    amdsmi_error_count_t ec;
    time_call("ecc_total", [&] { amdsmi_get_gpu_total_ecc_count(h, &ec); });
    amdsmi_violation_status_t vs;
    time_call("violation", [&] { amdsmi_get_violation_status(h, &vs); });
    amdsmi_xgmi_link_status_t xs;
    time_call("xgmi_link_status",
              [&] { amdsmi_get_gpu_xgmi_link_status(h, &xs); });
    amdsmi_xgmi_status_t xe;
    time_call("xgmi_error", [&] { amdsmi_gpu_xgmi_error_status(h, &xe); });
    uint32_t bp = 0;
    time_call("bad_pages_count",
              [&] { amdsmi_get_gpu_bad_page_info(h, &bp, nullptr); });
    uint32_t thr = 0;
    time_call("bad_page_threshold",
              [&] { amdsmi_get_gpu_bad_page_threshold(h, &thr); });
    amdsmi_gpu_metrics_t gm;
    time_call("gpu_metrics", [&] { amdsmi_get_gpu_metrics_info(h, &gm); });
  }
  return out;
}

py::list metrics_snapshot_all() {
  std::vector<amdsmi_processor_handle> handles;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    if (!g_initialized) throw std::runtime_error("amdsmi not initialized");
    handles = g_handles;
  }
  std::vector<Snapshot> snaps(handles.size());
  {
    py::gil_scoped_release nogil;
  CallLock call_lk(g_call_mu);
    if (handles.size() <= 1) {
      for (size_t i = 0; i < handles.size(); ++i)
        take_snapshot(handles[i], snaps[i]);
    } else {
      // one thread per GPU: amdsmi getters are thread-safe reads, and the
      // per-device ioctls dominate, so an 8-GPU sweep costs ~one GPU's
      // latency instead of 8x (keeps the single-daemon poll cycle flat
      // to 8 GPUs — SURVEY.md §7 hard parts)
      std::vector<std::thread> ts;
      ts.reserve(handles.size());
      for (size_t i = 0; i < handles.size(); ++i) {
        ts.emplace_back([&, i] { take_snapshot(handles[i], snaps[i]); });
      }
      for (auto& t : ts) t.join();
    }
  }
  py::list out;
  for (auto& s : snaps) out.append(snapshot_to_dict(s));
  return out;
}

}  // namespace

PYBIND11_MODULE(_amdsmi, m) {
  m.doc() = "gpud_amd native binding of ROCm libamd_smi (MI355X telemetry)";
  static py::exception<SmiError> exc(m, "SmiError");
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) std::rethrow_exception(p);
    } catch (const SmiError& e) {
      py::set_error(exc, e.what());
    }
  });
  m.def("init", &smi_init, "Initialize amdsmi and enumerate GPUs");
  m.def("shutdown", &smi_shutdown);
  m.def("device_count", &device_count);
  m.def("device_uuid", &device_uuid, py::arg("index"));
  m.def("device_bdf", &device_bdf, py::arg("index"));
  m.def("asic_info", &asic_info, py::arg("index"));
  m.def("board_info", &board_info, py::arg("index"));
  m.def("driver_info", &driver_info, py::arg("index"));
  m.def("vbios_info", &vbios_info, py::arg("index"));
  m.def("vram_info", &vram_info, py::arg("index"));
  m.def("temp_metric", &temp_metric, py::arg("index"), py::arg("sensor_type"),
        py::arg("metric"));
  m.def("power_info", &power_info, py::arg("index"));
  m.def("clock_info", &clock_info, py::arg("index"), py::arg("clk_type"));
  m.def("activity", &activity, py::arg("index"));
  m.def("vram_usage", &vram_usage, py::arg("index"));
  m.def("ecc_count_total", &ecc_count_total, py::arg("index"));
  m.def("ecc_count_block", &ecc_count_block, py::arg("index"), py::arg("block"));
  m.def("bad_page_info", &bad_page_info, py::arg("index"));
  m.def("process_list", &process_list, py::arg("index"));
  m.def("violation_status", &violation_status, py::arg("index"));
  m.def("power_management_enabled", &power_management_enabled,
        py::arg("index"));
  m.def("xgmi_link_status", &xgmi_link_status, py::arg("index"));
  m.def("xgmi_error_status", &xgmi_error_status, py::arg("index"));
  m.def("xgmi_info", &xgmi_info, py::arg("index"));
  m.def("link_metrics", &link_metrics, py::arg("index"));
  m.def("energy_count", &energy_count, py::arg("index"));
  m.def("pcie_info", &pcie_info, py::arg("index"),
        "PCIe link width/speed + replay/recovery/NAK counters");
  m.def("partition_info", &partition_info, py::arg("index"),
        "Compute/memory partition mode and accelerator partition profile");
  m.def("cper_entries", &cper_entries, py::arg("index"),
        py::arg("severity_mask") = 0xffffffffu, py::arg("cursor") = 0,
        py::arg("max_rounds") = 8,
        "Drain CPER RAS records cached by the driver (cursor-based)");
  m.def("metrics_snapshot", &metrics_snapshot, py::arg("index"),
        "Full telemetry snapshot for one GPU in a single native call");
  m.def("snapshot_timings", &snapshot_timings, py::arg("index"),
        "Per-SMI-call microsecond timing of one snapshot (poll tuning)");
  m.def("metrics_snapshot_all", &metrics_snapshot_all,
        "Telemetry snapshots for every GPU, GIL released for the whole sweep");

  // enum constants used from Python
  m.attr("TEMP_EDGE") = static_cast<int>(AMDSMI_TEMPERATURE_TYPE_EDGE);
  m.attr("TEMP_HOTSPOT") = static_cast<int>(AMDSMI_TEMPERATURE_TYPE_HOTSPOT);
  m.attr("TEMP_VRAM") = static_cast<int>(AMDSMI_TEMPERATURE_TYPE_VRAM);
  m.attr("TEMP_CURRENT") = static_cast<int>(AMDSMI_TEMP_CURRENT);
  m.attr("TEMP_CRITICAL") = static_cast<int>(AMDSMI_TEMP_CRITICAL);
  m.attr("TEMP_SHUTDOWN") = static_cast<int>(AMDSMI_TEMP_SHUTDOWN);
  m.attr("CLK_GFX") = static_cast<int>(AMDSMI_CLK_TYPE_GFX);
  m.attr("CLK_MEM") = static_cast<int>(AMDSMI_CLK_TYPE_MEM);
  m.attr("GPU_BLOCK_UMC") = static_cast<uint64_t>(AMDSMI_GPU_BLOCK_UMC);
  m.attr("GPU_BLOCK_GFX") = static_cast<uint64_t>(AMDSMI_GPU_BLOCK_GFX);
  m.attr("GPU_BLOCK_SDMA") = static_cast<uint64_t>(AMDSMI_GPU_BLOCK_SDMA);
  m.attr("GPU_BLOCK_MMHUB") = static_cast<uint64_t>(AMDSMI_GPU_BLOCK_MMHUB);
  m.attr("GPU_BLOCK_XGMI_WAFL") =
      static_cast<uint64_t>(AMDSMI_GPU_BLOCK_XGMI_WAFL);
  m.attr("XGMI_LINK_DOWN") = 0;
  m.attr("XGMI_LINK_UP") = 1;
  m.attr("XGMI_LINK_DISABLE") = 2;
  m.attr("XGMI_STATUS_NO_ERRORS") = 0;
  m.attr("XGMI_STATUS_ERROR") = 1;
  m.attr("XGMI_STATUS_MULTIPLE_ERRORS") = 2;
}
