// tf_accelerator.h — the vendor-neutral accelerator ABI of this stack.
//
// Capability parity with the reference's provider/accelerator.h:275-439
// (14-function C ABI: init/shutdown, device enumeration, topology matrix,
// partition assign/remove, hard limits, snapshot/resume, process info,
// device metrics, mount libs, log callback) — re-designed for MI355X:
// devices come from amd-smi, topology is the xGMI link matrix (full mesh
// intra-node ⇒ tier 0), partitions are XCD slabs realised as ROCr CU masks
// (HSA_CU_MASK env returned to the hypervisor), hard limits are CU mask +
// VRAM cap, process info is amdsmi_get_gpu_process_list.
//
// Loaded by the hypervisor via ctypes (tensor_fusion_amd/hypervisor/device.py).
// A CPU-only mock backend activates when TF_ACCEL_MOCK=<ndev> is set or no
// AMD GPU is present, mirroring the reference's provider/example stub.
#pragma once

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define TF_ACCEL_OK 0
#define TF_ACCEL_ERR 1
#define TF_ACCEL_NOT_SUPPORTED 2
#define TF_ACCEL_MAX_DEVICES 32
#define TF_ACCEL_UUID_LEN 64
#define TF_ACCEL_NAME_LEN 96

typedef struct TfAccelDevice {
  char uuid[TF_ACCEL_UUID_LEN];
  char name[TF_ACCEL_NAME_LEN];
  int32_t index;
  int32_t numa_node;
  uint64_t vram_total_bytes;
  uint32_t compute_units;
  uint32_t xcd_count;
  uint64_t bdf;           /* pcie bus-device-function id */
  double fp16_tflops;     /* dense MFMA peak for the model */
  uint32_t is_mock;
} TfAccelDevice;

typedef struct TfAccelMetrics {
  uint32_t gfx_activity_percent;
  uint32_t umc_activity_percent; /* memory controller */
  uint64_t vram_used_bytes;
  uint64_t vram_total_bytes;
  uint32_t power_w;
  uint32_t temp_c;
  uint32_t clock_mhz;
} TfAccelMetrics;

typedef struct TfAccelProc {
  int32_t pid;
  uint64_t vram_bytes;
  uint64_t gfx_busy_ns; /* cumulative engine time */
  uint32_t cu_occupancy;
  char name[TF_ACCEL_NAME_LEN];
} TfAccelProc;

/* lifecycle */
int tf_accel_init(void);
int tf_accel_shutdown(void);

/* enumeration */
int tf_accel_device_count(int* count);
int tf_accel_get_devices(TfAccelDevice* out, int max_devices, int* count);

/* topology: tier[i*n+j]: 0=xGMI direct, 1=same NUMA via host, 2=cross NUMA,
 * 3=unknown. MI355X nodes are an xGMI full mesh => all pairs 0. */
int tf_accel_get_topology(int32_t* tiers, int n);

/* monitoring */
int tf_accel_get_metrics(int device, TfAccelMetrics* out);
int tf_accel_get_processes(int device, TfAccelProc* out, int max_procs,
                           int* count);

/* isolation: returns the env assignment implementing the limit (CU mask) —
 * the hypervisor injects it into the worker (ROCr applies per process). */
int tf_accel_compose_cu_mask_env(int device, const int32_t* xcds, int n_xcds,
                                 char* out_env, int out_len);
int tf_accel_compose_percent_mask_env(int device, double percent,
                                      char* out_env, int out_len);

/* partition: validate an XCD-slab partition on the device (slot accounting
 * lives in the allocator; this checks device capability). */
/* AMD compute-partition modes (SPX/DPX/TPX/QPX/CPX) and memory modes
 * (NPS1/NPS2/NPS4/NPS8) — the MI355X device-global partitioning the
 * reference models per-vendor (partition_strategy.go). get returns the
 * current mode string; set switches it (requires an idle device; the
 * mock backend tracks modes in-process; returns TF_ACCEL_NOT_SUPPORTED
 * when the amd-smi build lacks the APIs). */
int tf_accel_get_compute_partition(int device, char* out, int out_len);
int tf_accel_set_compute_partition(int device, const char* mode);
int tf_accel_get_memory_partition(int device, char* out, int out_len);
int tf_accel_set_memory_partition(int device, const char* mode);

int tf_accel_assign_partition(int device, const int32_t* xcds, int n_xcds);
int tf_accel_remove_partition(int device, const int32_t* xcds, int n_xcds);

/* snapshot/resume of a workload process (CRIU-style; device-level VRAM dump
 * handled by the tiering engine). Returns TF_ACCEL_NOT_SUPPORTED when the
 * host lacks support — callers must handle. */
int tf_accel_snapshot(int pid, const char* dest_dir);
int tf_accel_resume(int pid, const char* src_dir);

/* log callback */
typedef void (*tf_accel_log_fn)(int level, const char* msg);
int tf_accel_register_log_callback(tf_accel_log_fn fn);

#ifdef __cplusplus
}
#endif
