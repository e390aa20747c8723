// cea_amd native AMD-SMI shim.
//
// Role parity: the reference's cgo C shim nvmlDeviceGetAverageUsage
// (/root/reference/pkg/gpu/nvidia/metrics/util.go:17-88), which averages
// ~100 NVML utilization samples (~16 s window) in C, plus the NVML event
// wait used by the health checker (health_checker.go:461).  AMD-SMI has no
// sample-buffer API, so the windowed average is produced by an in-shim
// background sampler thread over amdsmi_get_gpu_activity, and the event
// wait maps to amdsmi_get_gpu_event_notification (VM fault / thermal /
// reset / ring hang).
//
// Exposed as a plain C ABI consumed from Python via ctypes
// (cea_amd/amdsmi/shim.py).  Linked directly against libamd_smi.so.
//
// Build: see Makefile target `smi` (g++ -shared -fPIC, -lamd_smi).

#include <amd_smi/amdsmi.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <deque>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

namespace {

struct Sample {
  double t;      // seconds, monotonic
  double gfx;    // %
};

struct DeviceState {
  amdsmi_processor_handle handle{};
  std::string uuid;
  std::string name;
  std::string bdf;
  std::string serial;
  int render_minor = -1;
  int card_minor = -1;
  unsigned long long vram_total = 0;
  std::string compute_partition = "SPX";
  std::string memory_partition = "NPS1";
  unsigned partition_id = 0;
  int physical_index = 0;
  std::deque<Sample> samples;  // guarded by g_mutex
};

std::vector<DeviceState> g_devices;
std::mutex g_mutex;
std::thread g_sampler;
std::atomic<bool> g_sampler_run{false};
std::atomic<bool> g_inited{false};
thread_local char g_err[512];

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

int fail(const char* what, amdsmi_status_t st) {
  const char* s = nullptr;
  amdsmi_status_code_to_string(st, &s);
  snprintf(g_err, sizeof(g_err), "%s: %s (%d)", what, s ? s : "?", (int)st);
  return (int)st == 0 ? -1 : (int)st;
}

}  // namespace

extern "C" {

typedef struct {
  int index;
  char uuid[256];
  char name[256];
  char bdf[32];
  int render_minor;
  int card_minor;
  unsigned long long vram_total;
  char compute_partition[16];
  char memory_partition[16];
  unsigned partition_id;
  int physical_index;
} cea_device_info_t;

typedef struct {
  char uuid[256];
  int code;
  char message[256];
} cea_event_t;

const char* cea_smi_last_error() { return g_err; }

int cea_smi_init() {
  // serialize concurrent first-callers (ctypes callers may race init)
  static std::mutex init_mutex;
  std::lock_guard<std::mutex> init_lk(init_mutex);
  if (g_inited.load()) return 0;
  amdsmi_status_t st = amdsmi_init(AMDSMI_INIT_AMD_GPUS);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("amdsmi_init", st);

  uint32_t socket_count = 0;
  st = amdsmi_get_socket_handles(&socket_count, nullptr);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("get_socket_handles", st);
  std::vector<amdsmi_socket_handle> sockets(socket_count);
  st = amdsmi_get_socket_handles(&socket_count, sockets.data());
  if (st != AMDSMI_STATUS_SUCCESS) return fail("get_socket_handles", st);

  std::vector<DeviceState> devices;
  for (auto sock : sockets) {
    uint32_t n = 0;
    st = amdsmi_get_processor_handles(sock, &n, nullptr);
    if (st != AMDSMI_STATUS_SUCCESS) continue;
    std::vector<amdsmi_processor_handle> procs(n);
    st = amdsmi_get_processor_handles(sock, &n, procs.data());
    if (st != AMDSMI_STATUS_SUCCESS) continue;
    for (auto h : procs) {
      processor_type_t ptype;
      if (amdsmi_get_processor_type(h, &ptype) != AMDSMI_STATUS_SUCCESS ||
          ptype != AMDSMI_PROCESSOR_TYPE_AMD_GPU)
        continue;
      DeviceState d;
      d.handle = h;

      unsigned int len = 256;
      char uuid[256] = {0};
      if (amdsmi_get_gpu_device_uuid(h, &len, uuid) == AMDSMI_STATUS_SUCCESS)
        d.uuid = uuid;

      amdsmi_asic_info_t asic{};
      if (amdsmi_get_gpu_asic_info(h, &asic) == AMDSMI_STATUS_SUCCESS) {
        d.name = asic.market_name;
        d.serial = asic.asic_serial;
      }

      amdsmi_bdf_t bdf{};
      if (amdsmi_get_gpu_device_bdf(h, &bdf) == AMDSMI_STATUS_SUCCESS) {
        char buf[32];
        snprintf(buf, sizeof(buf), "%04lx:%02lx:%02lx.%lx",
                 (unsigned long)bdf.bdf.domain_number,
                 (unsigned long)bdf.bdf.bus_number,
                 (unsigned long)bdf.bdf.device_number,
                 (unsigned long)bdf.bdf.function_number);
        d.bdf = buf;
      }

      amdsmi_enumeration_info_t en{};
      if (amdsmi_get_gpu_enumeration_info(h, &en) == AMDSMI_STATUS_SUCCESS) {
        d.render_minor = (int)en.drm_render;
        d.card_minor = (int)en.drm_card;
      }

      uint64_t total = 0;
      if (amdsmi_get_gpu_memory_total(h, AMDSMI_MEM_TYPE_VRAM, &total) ==
          AMDSMI_STATUS_SUCCESS)
        d.vram_total = total;

      char part[16] = {0};
      if (amdsmi_get_gpu_compute_partition(h, part, sizeof(part)) ==
          AMDSMI_STATUS_SUCCESS && part[0])
        d.compute_partition = part;
      char mpart[16] = {0};
      if (amdsmi_get_gpu_memory_partition(h, mpart, sizeof(mpart)) ==
          AMDSMI_STATUS_SUCCESS && mpart[0])
        d.memory_partition = mpart;

      amdsmi_kfd_info_t kfd{};
      if (amdsmi_get_gpu_kfd_info(h, &kfd) == AMDSMI_STATUS_SUCCESS &&
          kfd.current_partition_id != 0xFFFFFFFFu)
        d.partition_id = kfd.current_partition_id;

      devices.push_back(std::move(d));
    }
  }

  // physical_index: partitions of one die share the ASIC serial (fallback:
  // BDF sans function).  This is the attribution key the health checker
  // uses to mark all partitions of a faulting die unhealthy — the analog of
  // the reference's MIG UUID->GI/CI matching (health_checker.go:426-445).
  std::vector<std::string> dies;
  for (auto& d : devices) {
    std::string key = !d.serial.empty() ? d.serial
                      : d.bdf.substr(0, d.bdf.find_last_of('.'));
    int idx = -1;
    for (size_t i = 0; i < dies.size(); i++)
      if (dies[i] == key) { idx = (int)i; break; }
    if (idx < 0) { dies.push_back(key); idx = (int)dies.size() - 1; }
    d.physical_index = idx;
  }

  {
    std::lock_guard<std::mutex> lk(g_mutex);
    g_devices = std::move(devices);
  }
  g_inited.store(true);
  return 0;
}

int cea_smi_shutdown() {
  if (!g_inited.load()) return 0;
  g_sampler_run.store(false);
  if (g_sampler.joinable()) g_sampler.join();
  g_inited.store(false);
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    g_devices.clear();
  }
  amdsmi_shut_down();
  return 0;
}

int cea_smi_device_count() {
  std::lock_guard<std::mutex> lk(g_mutex);
  return (int)g_devices.size();
}

int cea_smi_device_info(int idx, cea_device_info_t* out) {
  std::lock_guard<std::mutex> lk(g_mutex);
  if (idx < 0 || idx >= (int)g_devices.size()) {
    snprintf(g_err, sizeof(g_err), "device index %d out of range", idx);
    return -2;
  }
  const DeviceState& d = g_devices[idx];
  memset(out, 0, sizeof(*out));
  out->index = idx;
  snprintf(out->uuid, sizeof(out->uuid), "%s", d.uuid.c_str());
  snprintf(out->name, sizeof(out->name), "%s", d.name.c_str());
  snprintf(out->bdf, sizeof(out->bdf), "%s", d.bdf.c_str());
  out->render_minor = d.render_minor;
  out->card_minor = d.card_minor;
  out->vram_total = d.vram_total;
  snprintf(out->compute_partition, sizeof(out->compute_partition), "%s",
           d.compute_partition.c_str());
  snprintf(out->memory_partition, sizeof(out->memory_partition), "%s",
           d.memory_partition.c_str());
  out->partition_id = d.partition_id;
  out->physical_index = d.physical_index;
  return 0;
}

int cea_smi_memory_info(int idx, unsigned long long* total,
                        unsigned long long* used) {
  amdsmi_processor_handle h;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (idx < 0 || idx >= (int)g_devices.size()) return -2;
    h = g_devices[idx].handle;
  }
  uint64_t t = 0, u = 0;
  amdsmi_status_t st = amdsmi_get_gpu_memory_total(h, AMDSMI_MEM_TYPE_VRAM, &t);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("memory_total", st);
  st = amdsmi_get_gpu_memory_usage(h, AMDSMI_MEM_TYPE_VRAM, &u);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("memory_usage", st);
  *total = t;
  *used = u;
  return 0;
}

int cea_smi_gpu_activity(int idx, double* gfx, double* umc, double* mm) {
  amdsmi_processor_handle h;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (idx < 0 || idx >= (int)g_devices.size()) return -2;
    h = g_devices[idx].handle;
  }
  amdsmi_engine_usage_t eu{};
  amdsmi_status_t st = amdsmi_get_gpu_activity(h, &eu);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("gpu_activity", st);
  *gfx = eu.gfx_activity;
  *umc = eu.umc_activity;
  *mm = eu.mm_activity;
  return 0;
}

// --- windowed-average sampler ------------------------------------------
// Default cadence 160 ms -> ~100 samples per 16 s window, matching the
// reference's NVML sample buffer contents (util.go:34-36).

int cea_smi_start_sampler(int interval_ms) {
  if (!g_inited.load()) return -3;
  if (g_sampler_run.load()) return 0;
  if (interval_ms <= 0) interval_ms = 160;
  g_sampler_run.store(true);
  g_sampler = std::thread([interval_ms]() {
    while (g_sampler_run.load()) {
      {
        std::lock_guard<std::mutex> lk(g_mutex);
        double t = now_s();
        for (auto& d : g_devices) {
          amdsmi_engine_usage_t eu{};
          if (amdsmi_get_gpu_activity(d.handle, &eu) == AMDSMI_STATUS_SUCCESS) {
            d.samples.push_back({t, (double)eu.gfx_activity});
            while (d.samples.size() > 512) d.samples.pop_front();
          }
        }
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(interval_ms));
    }
  });
  return 0;
}

int cea_smi_stop_sampler() {
  g_sampler_run.store(false);
  if (g_sampler.joinable()) g_sampler.join();
  return 0;
}

int cea_smi_average_utilization(int idx, double window_s, double* avg,
                                int* count) {
  std::lock_guard<std::mutex> lk(g_mutex);
  if (idx < 0 || idx >= (int)g_devices.size()) return -2;
  const auto& samples = g_devices[idx].samples;
  double cutoff = now_s() - window_s;
  double sum = 0;
  int n = 0;
  for (auto it = samples.rbegin(); it != samples.rend(); ++it) {
    if (it->t < cutoff) break;
    sum += it->gfx;
    n++;
  }
  if (n == 0) {
    // No samples yet: fall back to one instantaneous reading, the same
    // graceful degradation the reference shim has for short windows.
    amdsmi_engine_usage_t eu{};
    if (amdsmi_get_gpu_activity(g_devices[idx].handle, &eu) !=
        AMDSMI_STATUS_SUCCESS) {
      snprintf(g_err, sizeof(g_err), "no samples and instant read failed");
      return -4;
    }
    *avg = eu.gfx_activity;
    *count = 1;
    return 0;
  }
  *avg = sum / n;
  *count = n;
  return 0;
}

int cea_smi_ecc_count(int idx, unsigned long long* correctable,
                      unsigned long long* uncorrectable,
                      unsigned long long* deferred) {
  amdsmi_processor_handle h;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (idx < 0 || idx >= (int)g_devices.size()) return -2;
    h = g_devices[idx].handle;
  }
  amdsmi_error_count_t ec{};
  amdsmi_status_t st = amdsmi_get_gpu_total_ecc_count(h, &ec);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("total_ecc_count", st);
  *correctable = ec.correctable_count;
  *uncorrectable = ec.uncorrectable_count;
  *deferred = ec.deferred_count;
  return 0;
}

// xGMI link error state: 0 = no errors, 1 = errors, 2 = multiple errors
// (amdsmi_xgmi_status_t).  The health checker's polling watchdog raises
// synthetic event 63 on nonzero (parity: Xid 63 class link errors).
int cea_smi_xgmi_error_status(int idx, int* status) {
  amdsmi_processor_handle h;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (idx < 0 || idx >= (int)g_devices.size()) return -2;
    h = g_devices[idx].handle;
  }
  amdsmi_xgmi_status_t st{};
  amdsmi_status_t rc = amdsmi_gpu_xgmi_error_status(h, &st);
  if (rc != AMDSMI_STATUS_SUCCESS) return fail("xgmi_error_status", rc);
  *status = (int)st;
  return 0;
}

int cea_smi_driver_version(char* buf, int len) {
  amdsmi_processor_handle h;
  {
    std::lock_guard<std::mutex> lk(g_mutex);
    if (g_devices.empty()) return -2;
    h = g_devices[0].handle;
  }
  amdsmi_driver_info_t info{};
  amdsmi_status_t st = amdsmi_get_gpu_driver_info(h, &info);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("driver_info", st);
  snprintf(buf, len, "%s", info.driver_version);
  return 0;
}

int cea_smi_lib_version(char* buf, int len) {
  amdsmi_version_t v{};
  amdsmi_status_t st = amdsmi_get_lib_version(&v);
  if (st != AMDSMI_STATUS_SUCCESS) return fail("lib_version", st);
  snprintf(buf, len, "%u.%u.%u", v.major, v.minor, v.release);
  return 0;
}

// --- event notification --------------------------------------------------

// Default notification mask built from THIS header's enum values, so the
// armed bits always agree with the amdsmi library the shim is compiled
// against (the driver installer builds the shim on the node, against the
// same ROCm it installs).  Hardcoding bit indices in Python silently armed
// the wrong events on amdsmi builds whose amdsmi_evt_notification_type_t
// numbers differ (e.g. trees where index 5 is RING_HANG instead of
// MIGRATE_START) — ADVICE r01.  If an enumerator is renamed/removed this
// fails loudly at compile time, which is the behavior we want.
unsigned long long cea_smi_default_event_mask(void) {
  return AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_VMFAULT) |
         AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_THERMAL_THROTTLE) |
         AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_GPU_PRE_RESET) |
         AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_GPU_POST_RESET) |
         AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_PAGE_FAULT_START) |
         AMDSMI_EVENT_MASK_FROM_INDEX(AMDSMI_EVT_NOTIF_PAGE_FAULT_END);
  // MIGRATE_START/END (5/6) and QUEUE_EVICTION/RESTORE (9/10) are benign
  // traffic and deliberately not armed.  On amdsmi builds that add a
  // RING_HANG enumerator, extend this expression — the compile-time name
  // reference keeps the gating tied to the actual header version.
}

int cea_smi_event_init(unsigned long long mask) {
  std::lock_guard<std::mutex> lk(g_mutex);
  int rc = 0;
  for (auto& d : g_devices) {
    amdsmi_status_t st = amdsmi_init_gpu_event_notification(d.handle);
    if (st != AMDSMI_STATUS_SUCCESS) { rc = fail("event_init", st); continue; }
    st = amdsmi_set_gpu_event_notification_mask(d.handle, mask);
    if (st != AMDSMI_STATUS_SUCCESS) rc = fail("event_mask", st);
  }
  return rc;
}

int cea_smi_wait_events(int timeout_ms, cea_event_t* out, int max_events,
                        int* num) {
  *num = 0;
  uint32_t n = (uint32_t)max_events;
  std::vector<amdsmi_evt_notification_data_t> data(max_events);
  amdsmi_status_t st =
      amdsmi_get_gpu_event_notification(timeout_ms, &n, data.data());
  if (st == AMDSMI_STATUS_NO_DATA || (st == AMDSMI_STATUS_SUCCESS && n == 0))
    return 0;
  if (st != AMDSMI_STATUS_SUCCESS) return fail("event_wait", st);
  std::lock_guard<std::mutex> lk(g_mutex);
  for (uint32_t i = 0; i < n && (int)i < max_events; i++) {
    memset(&out[i], 0, sizeof(out[i]));
    out[i].code = (int)data[i].event;
    snprintf(out[i].message, sizeof(out[i].message), "%s", data[i].message);
    for (auto& d : g_devices)
      if (d.handle == data[i].processor_handle) {
        snprintf(out[i].uuid, sizeof(out[i].uuid), "%s", d.uuid.c_str());
        break;
      }
  }
  *num = (int)n;
  return 0;
}

int cea_smi_event_stop() {
  std::lock_guard<std::mutex> lk(g_mutex);
  for (auto& d : g_devices) amdsmi_stop_gpu_event_notification(d.handle);
  return 0;
}

}  // extern "C"
