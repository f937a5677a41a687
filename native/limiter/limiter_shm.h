// Shared-memory protocol between the node hypervisor and the in-process
// HIP limiter (LD_PRELOAD) / remoting worker.
//
// Capability parity with the reference's soft-limiter shm
// (pkg/hypervisor/worker/state/soft_limiter_shm.go:140-356: V2 ERL token
// bucket, per-device entries, PID set under a shm mutex, heartbeats) —
// the layout here is our own: fixed 4 KiB page, explicit offsets, lock-free
// token consumption via 64-bit CAS on double bits, futex-free spin mutex for
// the PID set. One file per worker pod at
//   /run/tensor-fusion/shm/<namespace>/<pod>/shm
// created by the hypervisor (open-not-truncate: recreate must preserve a
// live worker's counters) and mapped read-write by the limiter.
//
// The Python mirror is tensor_fusion_amd/hypervisor/shm.py; offsets are
// cross-checked by native/limiter/shm_layout_dump.cpp + tests/test_shm.py.
#pragma once

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define TF_SHM_MAGIC 0x5446414Du /* "TFAM" */
#define TF_SHM_VERSION 2u
#define TF_SHM_SIZE 4096u
#define TF_MAX_DEVICES 16
#define TF_MAX_PIDS 64
#define TF_UUID_LEN 64

/* flags bits */
#define TF_FLAG_FREEZE (1u << 0)      /* worker must block all GPU ops      */
#define TF_FLAG_VRAM_PRESSURE (1u << 1) /* hypervisor signals tiering needed */

typedef struct TfDeviceEntry {
  char uuid[TF_UUID_LEN];        /* +0   device UUID, NUL-padded            */
  uint32_t up_limit_percent;     /* +64  target compute %% (ERL setpoint)   */
  uint32_t total_cus;            /* +68  256 on MI355X                      */
  uint64_t mem_limit_bytes;      /* +72  hard VRAM cap for this vGPU        */
  uint64_t pod_memory_used;      /* +80  written by limiter on alloc/free   */
  uint64_t erl_refill_rate;      /* +88  double bits: tokens/second         */
  uint64_t erl_capacity;         /* +96  double bits: bucket capacity       */
  uint64_t erl_tokens;           /* +104 double bits: current tokens (CAS)  */
  uint64_t erl_last_update_ns;   /* +112 CLOCK_MONOTONIC of last refill     */
  uint32_t active;               /* +120 entry in use                       */
  uint32_t launch_count;         /* +124 kernels launched (stats)           */
  uint64_t block_ns_total;       /* +128 cumulative ns throttled            */
  uint64_t alloc_bytes_total;    /* +136 cumulative hipMalloc bytes         */
  uint64_t vmm_bytes;            /* +144 remoting worker's VMM heap bytes
                                         (hipMemCreate bypasses the
                                         hipMalloc accounting; the worker
                                         reports here so caps and metrics
                                         see remote-vGPU VRAM)            */
  uint64_t pad1;                 /* +152 → sizeof == 160                    */
} TfDeviceEntry;

typedef struct TfSharedState {
  uint32_t magic;                    /* +0                                   */
  uint32_t version;                  /* +4                                   */
  TfDeviceEntry dev[TF_MAX_DEVICES]; /* +8 .. +8+16*160 = 2568               */
  uint32_t device_count;             /* +2568                                */
  uint32_t flags;                    /* +2572                                */
  uint64_t heartbeat_ns;             /* +2576 limiter → hypervisor           */
  uint64_t hyp_heartbeat_ns;         /* +2584 hypervisor → limiter           */
  uint32_t mutex;                    /* +2592 spin mutex guarding pid set    */
  uint32_t pid_count;                /* +2596                                */
  int32_t pids[TF_MAX_PIDS];         /* +2600 .. +2856                       */
  uint8_t pad[TF_SHM_SIZE - 2856];   /* fill to 4096                         */
} TfSharedState;

/* Offsets the Python side must agree on (asserted in shm_layout_dump). */
#define TF_OFF_DEV 8u
#define TF_DEV_STRIDE 160u
#define TF_OFF_DEVICE_COUNT 2568u
#define TF_OFF_FLAGS 2572u
#define TF_OFF_HEARTBEAT 2576u
#define TF_OFF_HYP_HEARTBEAT 2584u
#define TF_OFF_MUTEX 2592u
#define TF_OFF_PID_COUNT 2596u
#define TF_OFF_PIDS 2600u

#ifdef __cplusplus
} /* extern "C" */

#include <atomic>
#include <cstddef>
static_assert(sizeof(TfDeviceEntry) == TF_DEV_STRIDE, "device entry stride");
static_assert(sizeof(TfSharedState) == TF_SHM_SIZE, "shm page size");
static_assert(offsetof(TfSharedState, dev) == TF_OFF_DEV, "dev offset");
static_assert(offsetof(TfSharedState, device_count) == TF_OFF_DEVICE_COUNT,
              "device_count offset");
static_assert(offsetof(TfSharedState, mutex) == TF_OFF_MUTEX, "mutex offset");
static_assert(offsetof(TfSharedState, pids) == TF_OFF_PIDS, "pids offset");

// Helpers shared by limiter / worker / tests. All fields are accessed with
// C++11 atomics over the raw struct members (the file is mapped MAP_SHARED
// by unrelated processes; every member is naturally aligned).
namespace tfshm {

template <typename T>
static inline std::atomic<T>* at(T* p) {
  return reinterpret_cast<std::atomic<T>*>(p);
}

static inline double load_double(uint64_t* slot) {
  uint64_t bits = at(slot)->load(std::memory_order_acquire);
  double d;
  __builtin_memcpy(&d, &bits, 8);
  return d;
}

static inline void store_double(uint64_t* slot, double d) {
  uint64_t bits;
  __builtin_memcpy(&bits, &d, 8);
  at(slot)->store(bits, std::memory_order_release);
}

// Spin-lock over the shm mutex word (cross-process; holders are short).
static inline void lock(TfSharedState* s) {
  uint32_t expect = 0;
  while (!at(&s->mutex)->compare_exchange_weak(expect, 1,
                                               std::memory_order_acquire)) {
    expect = 0;
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
}

static inline void unlock(TfSharedState* s) {
  at(&s->mutex)->store(0, std::memory_order_release);
}

}  // namespace tfshm
#endif /* __cplusplus */
