// libaccelerator_amd.so — MI355X device discovery/metrics/isolation over
// amd-smi (see tf_accelerator.h for the ABI contract and the reference
// mapping). libamd_smi.so is dlopen'd lazily so this library loads on
// GPU-less CI machines, where the mock backend takes over
// (TF_ACCEL_MOCK=<ndev>, default 8 fake MI355X).

#include "tf_accelerator.h"

#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <mutex>
#include <vector>

// ---- minimal amdsmi ABI (mirrors /opt/rocm/include/amd_smi/amdsmi.h; we
// declare only what we call, resolved via dlsym so there is no link dep).
namespace smi {

typedef int status_t;  // amdsmi_status_t (0 == success)
typedef void* socket_handle;
typedef void* processor_handle;

struct engine_usage_t {
  uint32_t gfx_activity;
  uint32_t umc_activity;
  uint32_t mm_activity;
  uint32_t reserved[13];
};

static const int MAX_STRING = 256;

struct proc_info_t {
  char name[MAX_STRING];
  uint32_t pid;
  uint64_t mem;
  struct {
    uint64_t gfx;
    uint64_t enc;
    uint32_t reserved[12];
  } engine_usage;
  struct {
    uint64_t gtt_mem;
    uint64_t cpu_mem;
    uint64_t vram_mem;
    uint32_t reserved[10];
  } memory_usage;
  char container_name[MAX_STRING];
  uint32_t cu_occupancy;
  uint32_t evicted_time;
  uint32_t reserved[10];
};

typedef status_t (*fn_init)(uint64_t flags);
typedef status_t (*fn_shutdown)(void);
typedef status_t (*fn_get_sockets)(uint32_t*, socket_handle*);
typedef status_t (*fn_get_processors)(socket_handle, uint32_t*,
                                      processor_handle*);
typedef status_t (*fn_get_uuid)(processor_handle, unsigned int*, char*);
typedef status_t (*fn_get_mem_total)(processor_handle, int mem_type, uint64_t*);
typedef status_t (*fn_get_mem_usage)(processor_handle, int mem_type, uint64_t*);
typedef status_t (*fn_get_activity)(processor_handle, engine_usage_t*);
typedef status_t (*fn_get_proc_list)(processor_handle, uint32_t*, proc_info_t*);
typedef status_t (*fn_get_bdf_id)(processor_handle, uint64_t*);
typedef status_t (*fn_numa_affinity)(processor_handle, int32_t*);
typedef status_t (*fn_get_str)(processor_handle, char*, uint32_t);
typedef status_t (*fn_set_u32)(processor_handle, uint32_t);

struct Api {
  void* handle = nullptr;
  fn_init init = nullptr;
  fn_shutdown shutdown = nullptr;
  fn_get_sockets get_sockets = nullptr;
  fn_get_processors get_processors = nullptr;
  fn_get_uuid get_uuid = nullptr;
  fn_get_mem_total get_mem_total = nullptr;
  fn_get_mem_usage get_mem_usage = nullptr;
  fn_get_activity get_activity = nullptr;
  fn_get_proc_list get_proc_list = nullptr;
  fn_get_bdf_id get_bdf_id = nullptr;
  fn_numa_affinity numa_affinity = nullptr;
  // compute/memory partition APIs (optional — older amd-smi lacks them)
  fn_get_str get_compute_partition = nullptr;
  fn_set_u32 set_compute_partition = nullptr;
  fn_get_str get_memory_partition = nullptr;
  fn_set_u32 set_memory_partition = nullptr;

  bool load() {
    const char* names[] = {"libamd_smi.so", "libamd_smi.so.26"};
    for (const char* n : names) {
      handle = dlopen(n, RTLD_LAZY);
      if (handle) break;
    }
    if (!handle) return false;
#define R(field, sym)                                         \
  field = reinterpret_cast<decltype(field)>(dlsym(handle, sym)); \
  if (!(field)) return false;
    R(init, "amdsmi_init")
    R(shutdown, "amdsmi_shut_down")
    R(get_sockets, "amdsmi_get_socket_handles")
    R(get_processors, "amdsmi_get_processor_handles")
    R(get_uuid, "amdsmi_get_gpu_device_uuid")
    R(get_mem_total, "amdsmi_get_gpu_memory_total")
    R(get_mem_usage, "amdsmi_get_gpu_memory_usage")
    R(get_activity, "amdsmi_get_gpu_activity")
    R(get_proc_list, "amdsmi_get_gpu_process_list")
    R(get_bdf_id, "amdsmi_get_gpu_bdf_id")
#undef R
    numa_affinity = reinterpret_cast<fn_numa_affinity>(
        dlsym(handle, "amdsmi_get_gpu_topo_numa_affinity"));
    get_compute_partition = reinterpret_cast<fn_get_str>(
        dlsym(handle, "amdsmi_get_gpu_compute_partition"));
    set_compute_partition = reinterpret_cast<fn_set_u32>(
        dlsym(handle, "amdsmi_set_gpu_compute_partition"));
    get_memory_partition = reinterpret_cast<fn_get_str>(
        dlsym(handle, "amdsmi_get_gpu_memory_partition"));
    set_memory_partition = reinterpret_cast<fn_set_u32>(
        dlsym(handle, "amdsmi_set_gpu_memory_partition"));
    return true;
  }
};

static const uint64_t INIT_AMD_GPUS = 1ull << 1;
static const int MEM_TYPE_VRAM = 0;  // AMDSMI_MEM_TYPE_VRAM

}  // namespace smi

// ------------------------------------------------------------------ state

namespace {

struct State {
  bool initialized = false;
  bool mock = false;
  smi::Api api;
  std::vector<smi::processor_handle> procs;
  std::vector<TfAccelDevice> devices;
  tf_accel_log_fn log_cb = nullptr;
  std::mutex mu;
  // mock backend: per-device partition modes (defaults = whole-GPU)
  std::vector<std::string> mock_compute_mode;
  std::vector<std::string> mock_memory_mode;
};

State& S() {
  static State s;
  return s;
}

void logln(int level, const char* msg) {
  if (S().log_cb) S().log_cb(level, msg);
}

void fill_mock_devices(int n) {
  auto& s = S();
  s.devices.clear();
  for (int i = 0; i < n; ++i) {
    TfAccelDevice d{};
    snprintf(d.uuid, sizeof d.uuid, "GPU-mock-%02d", i);
    snprintf(d.name, sizeof d.name, "AMD Instinct MI355X (mock)");
    d.index = i;
    d.numa_node = i < n / 2 ? 0 : 1;
    d.vram_total_bytes = 288ull << 30;
    d.compute_units = 256;
    d.xcd_count = 8;
    d.bdf = 0xc0de00 + i;
    d.fp16_tflops = 2500.0;
    d.is_mock = 1;
    s.devices.push_back(d);
  }
  s.mock_compute_mode.assign(n, "SPX");
  s.mock_memory_mode.assign(n, "NPS1");
}

// amdsmi enum values (amdsmi.h): SPX=1 DPX=2 TPX=3 QPX=4 CPX=5;
// NPS1=1 NPS2=2 NPS4=3 NPS8=4.
int compute_mode_enum(const char* m) {
  static const char* names[] = {"", "SPX", "DPX", "TPX", "QPX", "CPX"};
  for (int i = 1; i <= 5; ++i)
    if (!strcasecmp(m, names[i])) return i;
  return 0;
}
int memory_mode_enum(const char* m) {
  static const char* names[] = {"", "NPS1", "NPS2", "NPS4", "NPS8"};
  for (int i = 1; i <= 4; ++i)
    if (!strcasecmp(m, names[i])) return i;
  return 0;
}

}  // namespace

extern "C" {

int tf_accel_init(void) {
  auto& s = S();
  std::lock_guard<std::mutex> l(s.mu);
  if (s.initialized) return TF_ACCEL_OK;
  const char* mock_env = getenv("TF_ACCEL_MOCK");
  if (!mock_env && s.api.load() && s.api.init(smi::INIT_AMD_GPUS) == 0) {
    uint32_t nsock = 0;
    if (s.api.get_sockets(&nsock, nullptr) == 0 && nsock > 0) {
      std::vector<smi::socket_handle> socks(nsock);
      s.api.get_sockets(&nsock, socks.data());
      int idx = 0;
      for (auto sock : socks) {
        uint32_t nproc = 0;
        if (s.api.get_processors(sock, &nproc, nullptr) != 0 || nproc == 0)
          continue;
        std::vector<smi::processor_handle> ph(nproc);
        s.api.get_processors(sock, &nproc, ph.data());
        for (auto p : ph) {
          TfAccelDevice d{};
          unsigned int ulen = sizeof d.uuid;
          if (s.api.get_uuid(p, &ulen, d.uuid) != 0)
            snprintf(d.uuid, sizeof d.uuid, "GPU-unknown-%d", idx);
          snprintf(d.name, sizeof d.name, "AMD Instinct MI355X");
          d.index = idx++;
          uint64_t vram = 0;
          s.api.get_mem_total(p, smi::MEM_TYPE_VRAM, &vram);
          d.vram_total_bytes = vram;
          d.compute_units = 256;
          d.xcd_count = 8;
          d.fp16_tflops = 2500.0;
          s.api.get_bdf_id(p, &d.bdf);
          int32_t numa = 0;
          if (s.api.numa_affinity && s.api.numa_affinity(p, &numa) == 0)
            d.numa_node = numa;
          d.is_mock = 0;
          s.procs.push_back(p);
          s.devices.push_back(d);
        }
      }
    }
    if (!s.devices.empty()) {
      s.initialized = true;
      logln(0, "amd-smi backend initialized");
      return TF_ACCEL_OK;
    }
    s.api.shutdown();
  }
  // mock fallback
  int n = mock_env ? atoi(mock_env) : 8;
  if (n <= 0) n = 8;
  fill_mock_devices(n);
  s.mock = true;
  s.initialized = true;
  logln(0, "mock backend initialized");
  return TF_ACCEL_OK;
}

int tf_accel_shutdown(void) {
  auto& s = S();
  std::lock_guard<std::mutex> l(s.mu);
  if (!s.initialized) return TF_ACCEL_OK;
  if (!s.mock && s.api.shutdown) s.api.shutdown();
  s.devices.clear();
  s.procs.clear();
  s.initialized = false;
  return TF_ACCEL_OK;
}

int tf_accel_device_count(int* count) {
  if (!S().initialized) return TF_ACCEL_ERR;
  *count = (int)S().devices.size();
  return TF_ACCEL_OK;
}

int tf_accel_get_devices(TfAccelDevice* out, int max_devices, int* count) {
  auto& s = S();
  if (!s.initialized) return TF_ACCEL_ERR;
  int n = (int)s.devices.size();
  if (n > max_devices) n = max_devices;
  memcpy(out, s.devices.data(), n * sizeof(TfAccelDevice));
  *count = n;
  return TF_ACCEL_OK;
}

int tf_accel_get_topology(int32_t* tiers, int n) {
  auto& s = S();
  if (!s.initialized || n != (int)s.devices.size()) return TF_ACCEL_ERR;
  // MI355X node: 7 xGMI links per GPU = a full mesh of 8 → every intra-node
  // pair is tier 0 (treat any intra-node set as equal cost).
  for (int i = 0; i < n; ++i)
    for (int j = 0; j < n; ++j)
      tiers[i * n + j] = i == j ? 0 : (n <= 8 ? 0 : 3);
  return TF_ACCEL_OK;
}

int tf_accel_get_metrics(int device, TfAccelMetrics* out) {
  auto& s = S();
  if (!s.initialized || device < 0 || device >= (int)s.devices.size())
    return TF_ACCEL_ERR;
  memset(out, 0, sizeof *out);
  out->vram_total_bytes = s.devices[device].vram_total_bytes;
  if (s.mock) {
    // Mock: utilization driven by a test hook file-less env knob.
    const char* u = getenv("TF_ACCEL_MOCK_UTIL");
    out->gfx_activity_percent = u ? atoi(u) : 0;
    return TF_ACCEL_OK;
  }
  smi::engine_usage_t eu{};
  if (s.api.get_activity(s.procs[device], &eu) == 0) {
    out->gfx_activity_percent = eu.gfx_activity;
    out->umc_activity_percent = eu.umc_activity;
  }
  uint64_t used = 0;
  if (s.api.get_mem_usage(s.procs[device], smi::MEM_TYPE_VRAM, &used) == 0)
    out->vram_used_bytes = used;
  return TF_ACCEL_OK;
}

int tf_accel_get_processes(int device, TfAccelProc* out, int max_procs,
                           int* count) {
  auto& s = S();
  *count = 0;
  if (!s.initialized || device < 0 || device >= (int)s.devices.size())
    return TF_ACCEL_ERR;
  if (s.mock) return TF_ACCEL_OK;
  uint32_t n = (uint32_t)max_procs;
  std::vector<smi::proc_info_t> list(max_procs);
  smi::status_t st = s.api.get_proc_list(s.procs[device], &n, list.data());
  if (st != 0 && n == 0) return TF_ACCEL_OK;  // no procs / unsupported
  if ((int)n > max_procs) n = max_procs;
  for (uint32_t i = 0; i < n; ++i) {
    out[i].pid = (int32_t)list[i].pid;
    out[i].vram_bytes = list[i].memory_usage.vram_mem ? list[i].memory_usage.vram_mem
                                                      : list[i].mem;
    out[i].gfx_busy_ns = list[i].engine_usage.gfx;
    out[i].cu_occupancy = list[i].cu_occupancy;
    strncpy(out[i].name, list[i].name, sizeof out[i].name - 1);
    out[i].name[sizeof out[i].name - 1] = 0;
  }
  *count = (int)n;
  return TF_ACCEL_OK;
}

// ---- isolation helpers -----------------------------------------------

static int mask_env_from_ranges(const char* ranges, char* out_env, int out_len) {
  // ROCr syntax: HSA_CU_MASK=<queue-list>:<cu-list>. Cover queues 0-15 so
  // every compute queue a workload creates is confined (validated on
  // MI355X: tests/test_gpu_accelerator.py partition test).
  int w = snprintf(out_env, out_len, "HSA_CU_MASK=0-15:%s", ranges);
  return (w > 0 && w < out_len) ? TF_ACCEL_OK : TF_ACCEL_ERR;
}

int tf_accel_compose_cu_mask_env(int device, const int32_t* xcds, int n_xcds,
                                 char* out_env, int out_len) {
  (void)device;
  char ranges[256] = {0};
  int off = 0;
  for (int i = 0; i < n_xcds; ++i) {
    int lo = xcds[i] * 32, hi = xcds[i] * 32 + 31;
    off += snprintf(ranges + off, sizeof ranges - off, "%s%d-%d",
                    i ? "," : "", lo, hi);
  }
  return mask_env_from_ranges(ranges, out_env, out_len);
}

int tf_accel_compose_percent_mask_env(int device, double percent,
                                      char* out_env, int out_len) {
  (void)device;
  int cus = (int)(256.0 * percent / 100.0 + 0.5);
  if (cus < 1) cus = 1;
  if (cus > 256) cus = 256;
  char ranges[64];
  snprintf(ranges, sizeof ranges, "0-%d", cus - 1);
  return mask_env_from_ranges(ranges, out_env, out_len);
}

int tf_accel_assign_partition(int device, const int32_t* xcds, int n_xcds) {
  auto& s = S();
  if (!s.initialized || device < 0 || device >= (int)s.devices.size())
    return TF_ACCEL_ERR;
  if (n_xcds < 1 || n_xcds > (int)s.devices[device].xcd_count)
    return TF_ACCEL_ERR;
  for (int i = 0; i < n_xcds; ++i)
    if (xcds[i] < 0 || xcds[i] >= (int)s.devices[device].xcd_count)
      return TF_ACCEL_ERR;
  return TF_ACCEL_OK;  // slot bookkeeping lives in the allocator
}

int tf_accel_remove_partition(int device, const int32_t* xcds, int n_xcds) {
  (void)xcds;
  (void)n_xcds;
  auto& s = S();
  if (!s.initialized || device < 0 || device >= (int)s.devices.size())
    return TF_ACCEL_ERR;
  return TF_ACCEL_OK;
}

int tf_accel_get_compute_partition(int device, char* out, int out_len) {
  auto& s = S();
  if (!s.initialized || device < 0 || device >= (int)s.devices.size() ||
      !out || out_len < 4)
    return TF_ACCEL_ERR;
  if (s.mock) {
    snprintf(out, out_len, "%s", s.mock_compute_mode[device].c_str());
    return TF_ACCEL_OK;
  }
  if (!s.api.get_compute_partition) return TF_ACCEL_NOT_SUPPORTED;
  return s.api.get_compute_partition(s.procs[device], out,
                                     (uint32_t)out_len) == 0
             ? TF_ACCEL_OK
             : TF_ACCEL_ERR;
}

int tf_accel_set_compute_partition(int device, const char* mode) {
  auto& s = S();
  int e = compute_mode_enum(mode ? mode : "");
  if (!s.initialized || device < 0 || device >= (int)s.devices.size() || !e)
    return TF_ACCEL_ERR;
  if (s.mock) {
    s.mock_compute_mode[device] = mode;
    return TF_ACCEL_OK;
  }
  if (!s.api.set_compute_partition) return TF_ACCEL_NOT_SUPPORTED;
  return s.api.set_compute_partition(s.procs[device], (uint32_t)e) == 0
             ? TF_ACCEL_OK
             : TF_ACCEL_ERR;
}

int tf_accel_get_memory_partition(int device, char* out, int out_len) {
  auto& s = S();
  if (!s.initialized || device < 0 || device >= (int)s.devices.size() ||
      !out || out_len < 4)
    return TF_ACCEL_ERR;
  if (s.mock) {
    snprintf(out, out_len, "%s", s.mock_memory_mode[device].c_str());
    return TF_ACCEL_OK;
  }
  if (!s.api.get_memory_partition) return TF_ACCEL_NOT_SUPPORTED;
  return s.api.get_memory_partition(s.procs[device], out,
                                    (uint32_t)out_len) == 0
             ? TF_ACCEL_OK
             : TF_ACCEL_ERR;
}

int tf_accel_set_memory_partition(int device, const char* mode) {
  auto& s = S();
  int e = memory_mode_enum(mode ? mode : "");
  if (!s.initialized || device < 0 || device >= (int)s.devices.size() || !e)
    return TF_ACCEL_ERR;
  if (s.mock) {
    s.mock_memory_mode[device] = mode;
    return TF_ACCEL_OK;
  }
  if (!s.api.set_memory_partition) return TF_ACCEL_NOT_SUPPORTED;
  return s.api.set_memory_partition(s.procs[device], (uint32_t)e) == 0
             ? TF_ACCEL_OK
             : TF_ACCEL_ERR;
}

int tf_accel_snapshot(int pid, const char* dest_dir) {
  (void)pid;
  (void)dest_dir;
  // Process-level GPU snapshot needs CRIU + ROCm plugin on the host; the
  // tiering engine handles VRAM dumps. Report honestly (the reference's
  // handlers return HTTP 501 for the same feature — handlers/worker.go:103).
  return TF_ACCEL_NOT_SUPPORTED;
}

int tf_accel_resume(int pid, const char* src_dir) {
  (void)pid;
  (void)src_dir;
  return TF_ACCEL_NOT_SUPPORTED;
}

int tf_accel_register_log_callback(tf_accel_log_fn fn) {
  S().log_cb = fn;
  return TF_ACCEL_OK;
}

}  // extern "C"
