// Prints the shm layout constants as JSON so tests/test_shm.py can verify
// the Python mirror (tensor_fusion_amd/hypervisor/shm.py) byte-for-byte.
// Built by setup.py (plain g++, no HIP needed) as tf_shm_layout_dump.
#include <cstddef>
#include <cstdio>

#include "limiter_shm.h"

int main() {
  std::printf(
      "{\"size\": %zu, \"magic\": %u, \"version\": %u, \"dev_off\": %zu, "
      "\"dev_stride\": %zu, \"device_count_off\": %zu, \"flags_off\": %zu, "
      "\"heartbeat_off\": %zu, \"hyp_heartbeat_off\": %zu, \"mutex_off\": %zu, "
      "\"pid_count_off\": %zu, \"pids_off\": %zu, "
      "\"e_uuid\": %zu, \"e_up_limit\": %zu, \"e_total_cus\": %zu, "
      "\"e_mem_limit\": %zu, \"e_mem_used\": %zu, \"e_rate\": %zu, "
      "\"e_capacity\": %zu, \"e_tokens\": %zu, \"e_last_update\": %zu, "
      "\"e_active\": %zu, \"e_launches\": %zu, \"e_block_ns\": %zu, "
      "\"e_alloc_bytes\": %zu}\n",
      sizeof(TfSharedState), TF_SHM_MAGIC, TF_SHM_VERSION,
      offsetof(TfSharedState, dev), sizeof(TfDeviceEntry),
      offsetof(TfSharedState, device_count), offsetof(TfSharedState, flags),
      offsetof(TfSharedState, heartbeat_ns),
      offsetof(TfSharedState, hyp_heartbeat_ns),
      offsetof(TfSharedState, mutex), offsetof(TfSharedState, pid_count),
      offsetof(TfSharedState, pids), offsetof(TfDeviceEntry, uuid),
      offsetof(TfDeviceEntry, up_limit_percent),
      offsetof(TfDeviceEntry, total_cus), offsetof(TfDeviceEntry, mem_limit_bytes),
      offsetof(TfDeviceEntry, pod_memory_used), offsetof(TfDeviceEntry, erl_refill_rate),
      offsetof(TfDeviceEntry, erl_capacity), offsetof(TfDeviceEntry, erl_tokens),
      offsetof(TfDeviceEntry, erl_last_update_ns), offsetof(TfDeviceEntry, active),
      offsetof(TfDeviceEntry, launch_count), offsetof(TfDeviceEntry, block_ns_total),
      offsetof(TfDeviceEntry, alloc_bytes_total));
  return 0;
}
