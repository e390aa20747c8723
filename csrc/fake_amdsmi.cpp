// Fake libamd_smi for C-boundary testing of csrc/amdsmi_shim.cpp.
//
// Loaded via LD_PRELOAD so the REAL shim .so (the one production runs)
// is exercised end-to-end on CPU — enumeration, die attribution, the
// event-notification path and ECC/xGMI polling — with this library
// standing in for the amdgpu driver stack.  This is the AMD analog of
// the reference's mock-at-the-NVML-boundary test strategy
// (/root/reference/pkg/gpu/nvidia/nvmlutil/nvml_mock.go) moved down one
// layer: the Python mock seam (cea_amd/amdsmi/mock.py) bypasses the C
// shim entirely, so shim logic (serial->physical_index attribution,
// event mask arming, message plumbing) was only reachable on real
// hardware before this ("mock at the shim's C boundary,
// not the Python seam").
//
// Control interface (all via environment, read at amdsmi_init time):
//   CEA_FAKE_SMI_DEVICES        enumerated GPU count (default 2)
//   CEA_FAKE_SMI_PARTITIONS     partitions per physical die (default 1;
//                               8 models CPX: devices i, i+1, ... share
//                               an ASIC serial and count partition_id up)
//   CEA_FAKE_SMI_PARTITION_MODE compute-partition string (default SPX,
//                               or CPX when PARTITIONS=8 etc.)
//   CEA_FAKE_SMI_DIR            control directory:
//     events       appended lines "idx code message..." become event
//                  notifications (consumed once, offset tracked)
//     ecc_<idx>    uncorrectable ECC count (read fresh per call)
//     xgmi_<idx>   xGMI error status 0/1/2 (read fresh per call)
//
// Build: `make fake-smi` -> tests/_build/libamd_smi.so
#include <amd_smi/amdsmi.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

namespace {

int env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v && *v ? atoi(v) : dflt;
}

std::string env_str(const char* name, const char* dflt) {
  const char* v = getenv(name);
  return v && *v ? v : dflt;
}

int g_devices = 2;
int g_partitions = 1;
std::string g_mode = "SPX";
std::string g_dir;
long g_event_offset = 0;
std::mutex g_mutex;
bool g_inited = false;

amdsmi_processor_handle handle_for(int idx) {
  return (amdsmi_processor_handle)(uintptr_t)(0x1000 + idx);
}

int index_for(amdsmi_processor_handle h) {
  long v = (long)(uintptr_t)h - 0x1000;
  return (v >= 0 && v < g_devices) ? (int)v : -1;
}

long file_int(const std::string& path, long dflt) {
  FILE* f = fopen(path.c_str(), "r");
  if (!f) return dflt;
  long v = dflt;
  if (fscanf(f, "%ld", &v) != 1) v = dflt;
  fclose(f);
  return v;
}

}  // namespace

extern "C" {

amdsmi_status_t amdsmi_init(uint64_t) {
  std::lock_guard<std::mutex> lk(g_mutex);
  g_devices = env_int("CEA_FAKE_SMI_DEVICES", 2);
  g_partitions = env_int("CEA_FAKE_SMI_PARTITIONS", 1);
  if (g_partitions < 1) g_partitions = 1;
  g_mode = env_str("CEA_FAKE_SMI_PARTITION_MODE",
                   g_partitions > 1 ? "CPX" : "SPX");
  g_dir = env_str("CEA_FAKE_SMI_DIR", "");
  g_event_offset = 0;
  g_inited = true;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_shut_down(void) { return AMDSMI_STATUS_SUCCESS; }

amdsmi_status_t amdsmi_get_socket_handles(uint32_t* count,
                                          amdsmi_socket_handle* handles) {
  if (handles && *count >= 1) handles[0] = (amdsmi_socket_handle)0x9000;
  *count = 1;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_processor_handles(amdsmi_socket_handle,
                                             uint32_t* count,
                                             amdsmi_processor_handle* out) {
  if (out) {
    uint32_t n = *count < (uint32_t)g_devices ? *count : (uint32_t)g_devices;
    for (uint32_t i = 0; i < n; i++) out[i] = handle_for((int)i);
  }
  *count = (uint32_t)g_devices;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_processor_type(amdsmi_processor_handle,
                                          processor_type_t* type) {
  *type = AMDSMI_PROCESSOR_TYPE_AMD_GPU;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_device_uuid(amdsmi_processor_handle h,
                                           unsigned int* len, char* uuid) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  snprintf(uuid, *len, "fake-uuid-%d", i);
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_asic_info(amdsmi_processor_handle h,
                                         amdsmi_asic_info_t* info) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  memset(info, 0, sizeof(*info));
  snprintf(info->market_name, sizeof(info->market_name),
           "Fake AMD Instinct MI355X");
  // partitions of one die share the serial — the attribution key the
  // shim's physical_index derivation groups by
  snprintf(info->asic_serial, sizeof(info->asic_serial), "FAKESERIAL%02d",
           i / g_partitions);
  info->num_of_compute_units = 256 / g_partitions;
  info->target_graphics_version = 90500;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_device_bdf(amdsmi_processor_handle h,
                                          amdsmi_bdf_t* bdf) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  memset(bdf, 0, sizeof(*bdf));
  bdf->bdf.domain_number = 0;
  bdf->bdf.bus_number = 0x10 + i / g_partitions;
  bdf->bdf.device_number = 0;
  bdf->bdf.function_number = i % g_partitions;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_enumeration_info(
    amdsmi_processor_handle h, amdsmi_enumeration_info_t* info) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  memset(info, 0, sizeof(*info));
  info->drm_render = 128 + i;
  info->drm_card = i;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_memory_total(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t,
                                            uint64_t* total) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  *total = (288ULL << 30) / (uint64_t)g_partitions;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_memory_usage(amdsmi_processor_handle h,
                                            amdsmi_memory_type_t,
                                            uint64_t* used) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  *used = 1ULL << 30;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_compute_partition(amdsmi_processor_handle h,
                                                 char* out, uint32_t len) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  snprintf(out, len, "%s", g_mode.c_str());
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_memory_partition(amdsmi_processor_handle h,
                                                char* out, uint32_t len) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  snprintf(out, len, "NPS1");
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_kfd_info(amdsmi_processor_handle h,
                                        amdsmi_kfd_info_t* info) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  memset(info, 0, sizeof(*info));
  info->kfd_id = 36000 + i;
  info->node_id = 2 + i;
  info->current_partition_id = (uint32_t)(i % g_partitions);
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_activity(amdsmi_processor_handle h,
                                        amdsmi_engine_usage_t* usage) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  memset(usage, 0, sizeof(*usage));
  usage->gfx_activity = 42;
  usage->umc_activity = 17;
  usage->mm_activity = 0;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_total_ecc_count(amdsmi_processor_handle h,
                                               amdsmi_error_count_t* ec) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  memset(ec, 0, sizeof(*ec));
  if (!g_dir.empty())
    ec->uncorrectable_count =
        (uint64_t)file_int(g_dir + "/ecc_" + std::to_string(i), 0);
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_gpu_xgmi_error_status(amdsmi_processor_handle h,
                                             amdsmi_xgmi_status_t* status) {
  int i = index_for(h);
  if (i < 0) return AMDSMI_STATUS_INVAL;
  long v = 0;
  if (!g_dir.empty()) v = file_int(g_dir + "/xgmi_" + std::to_string(i), 0);
  *status = (amdsmi_xgmi_status_t)v;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_gpu_driver_info(amdsmi_processor_handle h,
                                           amdsmi_driver_info_t* info) {
  if (index_for(h) < 0) return AMDSMI_STATUS_INVAL;
  memset(info, 0, sizeof(*info));
  snprintf(info->driver_version, sizeof(info->driver_version), "6.fake.0");
  snprintf(info->driver_name, sizeof(info->driver_name), "amdgpu");
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_get_lib_version(amdsmi_version_t* v) {
  memset(v, 0, sizeof(*v));
  v->major = 99;
  v->minor = 0;
  v->release = 0;
  return AMDSMI_STATUS_SUCCESS;
}

amdsmi_status_t amdsmi_status_code_to_string(amdsmi_status_t status,
                                             const char** str) {
  *str = status == AMDSMI_STATUS_SUCCESS ? "success" : "fake error";
  return AMDSMI_STATUS_SUCCESS;
}

// --- event notification ---------------------------------------------------

amdsmi_status_t amdsmi_init_gpu_event_notification(amdsmi_processor_handle h) {
  return index_for(h) >= 0 ? AMDSMI_STATUS_SUCCESS : AMDSMI_STATUS_INVAL;
}

amdsmi_status_t amdsmi_set_gpu_event_notification_mask(
    amdsmi_processor_handle h, uint64_t) {
  return index_for(h) >= 0 ? AMDSMI_STATUS_SUCCESS : AMDSMI_STATUS_INVAL;
}

amdsmi_status_t amdsmi_get_gpu_event_notification(
    int timeout_ms, uint32_t* num_elem, amdsmi_evt_notification_data_t* data) {
  uint32_t cap = *num_elem;
  *num_elem = 0;
  if (g_dir.empty()) return AMDSMI_STATUS_NO_DATA;
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::milliseconds(timeout_ms);
  std::string path = g_dir + "/events";
  while (true) {
    {
      std::lock_guard<std::mutex> lk(g_mutex);
      FILE* f = fopen(path.c_str(), "r");
      if (f) {
        fseek(f, g_event_offset, SEEK_SET);
        char line[512];
        uint32_t n = 0;
        while (n < cap && fgets(line, sizeof(line), f)) {
          size_t ln = strlen(line);
          if (ln == 0 || line[ln - 1] != '\n') {
            // partial write in flight; retry from the same offset
            fseek(f, -(long)ln, SEEK_CUR);
            break;
          }
          g_event_offset = ftell(f);
          int idx = -1, code = 0;
          char msg[256] = {0};
          if (sscanf(line, "%d %d %255[^\n]", &idx, &code, msg) >= 2 &&
              idx >= 0 && idx < g_devices) {
            memset(&data[n], 0, sizeof(data[n]));
            data[n].processor_handle = handle_for(idx);
            data[n].event = (amdsmi_evt_notification_type_t)code;
            snprintf(data[n].message, sizeof(data[n].message), "%s", msg);
            n++;
          }
        }
        fclose(f);
        if (n > 0) {
          *num_elem = n;
          return AMDSMI_STATUS_SUCCESS;
        }
      }
    }
    if (std::chrono::steady_clock::now() >= deadline)
      return AMDSMI_STATUS_NO_DATA;
    std::this_thread::sleep_for(std::chrono::milliseconds(20));
  }
}

amdsmi_status_t amdsmi_stop_gpu_event_notification(amdsmi_processor_handle h) {
  return index_for(h) >= 0 ? AMDSMI_STATUS_SUCCESS : AMDSMI_STATUS_INVAL;
}

}  // extern "C"
