// _amdsmi — lazy dlopen binding over libamd_smi.so.
//
// The AMD-native equivalent of the reference's NVML cgo binding
// (reference: vendor/.../nvml/nvml_dl.go:29-36 dlopens
// libnvidia-ml.so.1 at runtime with unresolved symbols allowed, so the
// daemon runs on nodes without the driver stack).  Same contract here:
// the extension always imports; dlopen/dlsym happen on first use and
// failures surface as Python RuntimeError / available() == false.
//
// Exposes exactly the surface the plugin needs (mirroring how the
// reference uses only Init/Shutdown/GetDeviceCount/NewDeviceLite and
// the XID event set): device enumeration with BDF +
// UUID, xGMI hive info, ECC totals, and GPU event notifications
// (reset/thermal/vmfault — the XID-critical analogue).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <dlfcn.h>

#include <cstdint>
#include <cstdio>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include <amd_smi/amdsmi.h>

namespace py = pybind11;

namespace {

void *g_lib = nullptr;
std::mutex g_mutex;
std::vector<amdsmi_processor_handle> g_processors;
std::map<amdsmi_processor_handle, uint32_t> g_handle_to_index;
bool g_inited = false;

void *try_dlopen() {
  if (g_lib) return g_lib;
  static const char *names[] = {
      "libamd_smi.so",
      "libamd_smi.so.26",
      "/opt/rocm/lib/libamd_smi.so",
      nullptr,
  };
  for (const char **n = names; *n; ++n) {
    g_lib = dlopen(*n, RTLD_LAZY | RTLD_GLOBAL);
    if (g_lib) break;
  }
  return g_lib;
}

template <typename Fn>
Fn sym(const char *name) {
  if (!try_dlopen())
    throw std::runtime_error(std::string("libamd_smi.so not loadable: ") +
                             (dlerror() ?: "unknown error"));
  void *p = dlsym(g_lib, name);
  if (!p)
    throw std::runtime_error(std::string("symbol not found: ") + name);
  return reinterpret_cast<Fn>(p);
}

std::string status_str(amdsmi_status_t st) {
  using Fn = amdsmi_status_t (*)(amdsmi_status_t, const char **);
  try {
    const char *s = nullptr;
    if (sym<Fn>("amdsmi_status_code_to_string")(st, &s) ==
            AMDSMI_STATUS_SUCCESS && s)
      return s;
  } catch (const std::runtime_error &) {
  }
  char buf[32];
  snprintf(buf, sizeof buf, "status %d", static_cast<int>(st));
  return buf;
}

void check(amdsmi_status_t st, const char *what) {
  if (st != AMDSMI_STATUS_SUCCESS)
    throw std::runtime_error(std::string(what) + ": " + status_str(st));
}

std::string bdf_str(const amdsmi_bdf_t &bdf) {
  char buf[32];
  snprintf(buf, sizeof buf, "%04lx:%02lx:%02lx.%lx",
           static_cast<unsigned long>(bdf.domain_number),
           static_cast<unsigned long>(bdf.bus_number),
           static_cast<unsigned long>(bdf.device_number),
           static_cast<unsigned long>(bdf.function_number));
  return buf;
}

bool available() {
  std::lock_guard<std::mutex> lk(g_mutex);
  return try_dlopen() != nullptr;
}

void smi_shutdown_locked() {
  if (!g_inited) return;
  using Fn = amdsmi_status_t (*)(void);
  sym<Fn>("amdsmi_shut_down")();
  g_inited = false;
  g_processors.clear();
  g_handle_to_index.clear();
}

void smi_init() {
  std::lock_guard<std::mutex> lk(g_mutex);
  if (g_inited) return;
  using InitFn = amdsmi_status_t (*)(uint64_t);
  check(sym<InitFn>("amdsmi_init")(AMDSMI_INIT_AMD_GPUS), "amdsmi_init");
  g_inited = true;

  using SockFn = amdsmi_status_t (*)(uint32_t *, amdsmi_socket_handle *);
  using ProcFn = amdsmi_status_t (*)(amdsmi_socket_handle, uint32_t *,
                                     amdsmi_processor_handle *);
  auto get_sockets = sym<SockFn>("amdsmi_get_socket_handles");
  auto get_procs = sym<ProcFn>("amdsmi_get_processor_handles");

  uint32_t nsock = 0;
  // A failure mid-enumeration must not leave a half-initialized
  // library behind — undo the init before rethrowing.
  try {
    check(get_sockets(&nsock, nullptr), "amdsmi_get_socket_handles");
  } catch (...) {
    smi_shutdown_locked();
    throw;
  }
  std::vector<amdsmi_socket_handle> sockets(nsock);
  try {
    check(get_sockets(&nsock, sockets.data()),
          "amdsmi_get_socket_handles");
  } catch (...) {
    smi_shutdown_locked();
    throw;
  }

  g_processors.clear();
  g_handle_to_index.clear();
  for (auto sock : sockets) {
    uint32_t nproc = 0;
    if (get_procs(sock, &nproc, nullptr) != AMDSMI_STATUS_SUCCESS)
      continue;
    std::vector<amdsmi_processor_handle> procs(nproc);
    if (get_procs(sock, &nproc, procs.data()) != AMDSMI_STATUS_SUCCESS)
      continue;
    for (auto p : procs) {
      g_handle_to_index[p] = static_cast<uint32_t>(g_processors.size());
      g_processors.push_back(p);
    }
  }
}

void smi_shutdown() {
  std::lock_guard<std::mutex> lk(g_mutex);
  smi_shutdown_locked();
}

amdsmi_processor_handle handle_of(uint32_t index) {
  std::lock_guard<std::mutex> lk(g_mutex);
  if (!g_inited) throw std::runtime_error("amdsmi not initialized");
  if (index >= g_processors.size())
    throw std::runtime_error("device index out of range");
  return g_processors[index];
}

py::list get_devices() {
  using BdfFn = amdsmi_status_t (*)(amdsmi_processor_handle,
                                    amdsmi_bdf_t *);
  using UuidFn = amdsmi_status_t (*)(amdsmi_processor_handle,
                                     unsigned int *, char *);
  using AsicFn = amdsmi_status_t (*)(amdsmi_processor_handle,
                                     amdsmi_asic_info_t *);
  auto get_bdf = sym<BdfFn>("amdsmi_get_gpu_device_bdf");
  auto get_uuid = sym<UuidFn>("amdsmi_get_gpu_device_uuid");
  auto get_asic = sym<AsicFn>("amdsmi_get_gpu_asic_info");

  std::vector<amdsmi_processor_handle> procs;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (!g_inited) throw std::runtime_error("amdsmi not initialized");
    procs = g_processors;
  }
  py::list out;
  for (size_t i = 0; i < procs.size(); ++i) {
    py::dict d;
    d["index"] = static_cast<uint32_t>(i);
    amdsmi_bdf_t bdf{};
    if (get_bdf(procs[i], &bdf) == AMDSMI_STATUS_SUCCESS)
      d["bdf"] = bdf_str(bdf);
    else
      d["bdf"] = "";
    char uuid[AMDSMI_GPU_UUID_SIZE + 1] = {0};
    unsigned int len = AMDSMI_GPU_UUID_SIZE;
    if (get_uuid(procs[i], &len, uuid) == AMDSMI_STATUS_SUCCESS)
      d["uuid"] = std::string(uuid);
    else
      d["uuid"] = "";
    amdsmi_asic_info_t asic{};
    if (get_asic(procs[i], &asic) == AMDSMI_STATUS_SUCCESS) {
      d["name"] = std::string(asic.market_name);
      char devid[16];
      snprintf(devid, sizeof devid, "%04lx",
               static_cast<unsigned long>(asic.device_id));
      d["device_id"] = devid;
      d["num_compute_units"] = asic.num_of_compute_units;
    }
    out.append(d);
  }
  return out;
}

py::dict xgmi_info(uint32_t index) {
  using Fn = amdsmi_status_t (*)(amdsmi_processor_handle,
                                 amdsmi_xgmi_info_t *);
  amdsmi_xgmi_info_t info{};
  check(sym<Fn>("amdsmi_get_xgmi_info")(handle_of(index), &info),
        "amdsmi_get_xgmi_info");
  py::dict d;
  d["lanes"] = static_cast<unsigned>(info.xgmi_lanes);
  d["hive_id"] = info.xgmi_hive_id;
  d["node_id"] = info.xgmi_node_id;
  d["index"] = info.index;
  return d;
}

py::dict ecc_count(uint32_t index) {
  using Fn = amdsmi_status_t (*)(amdsmi_processor_handle,
                                 amdsmi_error_count_t *);
  amdsmi_error_count_t ec{};
  check(sym<Fn>("amdsmi_get_gpu_total_ecc_count")(handle_of(index), &ec),
        "amdsmi_get_gpu_total_ecc_count");
  py::dict d;
  d["correctable"] = ec.correctable_count;
  d["uncorrectable"] = ec.uncorrectable_count;
  d["deferred"] = ec.deferred_count;
  return d;
}

void event_init(uint32_t index) {
  using Fn = amdsmi_status_t (*)(amdsmi_processor_handle);
  check(sym<Fn>("amdsmi_init_gpu_event_notification")(handle_of(index)),
        "amdsmi_init_gpu_event_notification");
}

void event_mask(uint32_t index, uint64_t mask) {
  using Fn = amdsmi_status_t (*)(amdsmi_processor_handle, uint64_t);
  check(sym<Fn>("amdsmi_set_gpu_event_notification_mask")(
            handle_of(index), mask),
        "amdsmi_set_gpu_event_notification_mask");
}

void event_stop(uint32_t index) {
  using Fn = amdsmi_status_t (*)(amdsmi_processor_handle);
  sym<Fn>("amdsmi_stop_gpu_event_notification")(handle_of(index));
}

// Collect pending events: list of (device_index, event_type, message).
// Polling model matches the reference's WaitForEvent(eventSet, 5000)
// loop (generic_vgpu_device_plugin.go:406).
py::list get_events(int timeout_ms) {
  using Fn = amdsmi_status_t (*)(int, uint32_t *,
                                 amdsmi_evt_notification_data_t *);
  auto get = sym<Fn>("amdsmi_get_gpu_event_notification");
  uint32_t n = 64;
  std::vector<amdsmi_evt_notification_data_t> buf(n);
  amdsmi_status_t st;
  {
    py::gil_scoped_release release;
    st = get(timeout_ms, &n, buf.data());
  }
  py::list out;
  if (st != AMDSMI_STATUS_SUCCESS)
    return out;  // timeout / no data — empty list, like an empty poll
  std::map<amdsmi_processor_handle, uint32_t> h2i;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    h2i = g_handle_to_index;
  }
  for (uint32_t i = 0; i < n; ++i) {
    auto it = h2i.find(buf[i].processor_handle);
    int idx = it == h2i.end() ? -1 : static_cast<int>(it->second);
    out.append(py::make_tuple(idx, static_cast<int>(buf[i].event),
                              std::string(buf[i].message)));
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_amdsmi, m) {
  m.doc() = "lazy dlopen binding over libamd_smi.so";
  m.def("available", &available,
        "true when libamd_smi.so is loadable on this host");
  m.def("init", &smi_init);
  m.def("shutdown", &smi_shutdown);
  m.def("get_devices", &get_devices);
  m.def("xgmi_info", &xgmi_info, py::arg("index"));
  m.def("ecc_count", &ecc_count, py::arg("index"));
  m.def("event_init", &event_init, py::arg("index"));
  m.def("event_mask", &event_mask, py::arg("index"), py::arg("mask"));
  m.def("event_stop", &event_stop, py::arg("index"));
  m.def("get_events", &get_events, py::arg("timeout_ms"));
}
