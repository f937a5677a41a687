// libmockhip.so — a CPU-only mock of the HIP runtime entry points the
// limiter/remoting layers interpose. Same role as the reference's
// provider/example/device_mock/driver_mock.c (which itself mocks HIP/amdsmi
// signatures): lets CI exercise the LD_PRELOAD limiter, the accelerator ABI
// and the worker protocol on GPU-less machines.
//
// Semantics: device memory is plain malloc with byte accounting; kernel
// launches are no-ops with an optional per-launch busy-wait
// (MOCKHIP_LAUNCH_NS) so throttling tests see realistic pacing.

#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>

#include <atomic>

namespace {
std::atomic<uint64_t> g_allocated{0};
std::atomic<uint64_t> g_launches{0};
thread_local int g_device = 0;

uint64_t launch_ns() {
  static uint64_t v = [] {
    const char* e = getenv("MOCKHIP_LAUNCH_NS");
    return e ? strtoull(e, nullptr, 10) : 0ull;
  }();
  return v;
}

void spin_ns(uint64_t ns) {
  if (!ns) return;
  timespec a;
  clock_gettime(CLOCK_MONOTONIC, &a);
  uint64_t t0 = uint64_t(a.tv_sec) * 1000000000ull + a.tv_nsec;
  for (;;) {
    clock_gettime(CLOCK_MONOTONIC, &a);
    if (uint64_t(a.tv_sec) * 1000000000ull + a.tv_nsec - t0 >= ns) break;
  }
}
}  // namespace

extern "C" {

typedef int hipError_t;
typedef void* hipStream_t;
struct dim3u {
  unsigned x, y, z;
};

int hipSetDevice(int d) {
  g_device = d;
  return 0;
}
int hipGetDevice(int* d) {
  *d = g_device;
  return 0;
}
int hipGetDeviceCount(int* n) {
  const char* e = getenv("MOCKHIP_DEVICES");
  *n = e ? atoi(e) : 1;
  return 0;
}

hipError_t hipMalloc(void** p, size_t sz) {
  *p = malloc(sz ? sz : 1);
  if (!*p) return 2;
  g_allocated += sz;
  return 0;
}
hipError_t hipMallocAsync(void** p, size_t sz, hipStream_t) {
  return hipMalloc(p, sz);
}
hipError_t hipMallocManaged(void** p, size_t sz, unsigned) {
  return hipMalloc(p, sz);
}
hipError_t hipMallocPitch(void** p, size_t* pitch, size_t w, size_t h) {
  *pitch = w;
  return hipMalloc(p, w * h);
}
hipError_t hipFree(void* p) {
  free(p);
  return 0;
}
hipError_t hipFreeAsync(void* p, hipStream_t) { return hipFree(p); }

hipError_t hipMemGetInfo(size_t* free_b, size_t* total_b) {
  if (total_b) *total_b = 288ull << 30;
  if (free_b) *free_b = (288ull << 30) - g_allocated.load();
  return 0;
}

hipError_t hipLaunchKernel(const void*, dim3u, dim3u, void**, size_t,
                           hipStream_t) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipLaunchKernel_spt(const void* f, dim3u g, dim3u b, void** a,
                               size_t s, hipStream_t st) {
  return hipLaunchKernel(f, g, b, a, s, st);
}
hipError_t hipLaunchKernelExC(const void*, const void*, void**) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipExtLaunchKernel(const void*, dim3u, dim3u, void**, size_t,
                              hipStream_t, void*, void*, int) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipModuleLaunchKernel(void*, unsigned, unsigned, unsigned, unsigned,
                                 unsigned, unsigned, unsigned, hipStream_t,
                                 void**, void**) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipExtModuleLaunchKernel(void*, unsigned, unsigned, unsigned,
                                    unsigned, unsigned, unsigned, size_t,
                                    hipStream_t, void**, void**, void*, void*,
                                    unsigned) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipLaunchCooperativeKernel(const void*, dim3u, dim3u, void**,
                                      unsigned, hipStream_t) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipGraphLaunch(void*, hipStream_t) {
  ++g_launches;
  spin_ns(launch_ns());
  return 0;
}
hipError_t hipGraphLaunch_spt(void* g, hipStream_t s) {
  return hipGraphLaunch(g, s);
}

hipError_t hipMemcpy(void* dst, const void* src, size_t n, int) {
  memmove(dst, src, n);
  return 0;
}
hipError_t hipMemcpyAsync(void* dst, const void* src, size_t n, int,
                          hipStream_t) {
  memmove(dst, src, n);
  return 0;
}
hipError_t hipMemcpyWithStream(void* dst, const void* src, size_t n, int,
                               hipStream_t) {
  memmove(dst, src, n);
  return 0;
}
hipError_t hipMemset(void* dst, int v, size_t n) {
  memset(dst, v, n);
  return 0;
}
hipError_t hipMemsetAsync(void* dst, int v, size_t n, hipStream_t) {
  memset(dst, v, n);
  return 0;
}
hipError_t hipDeviceSynchronize() { return 0; }
hipError_t hipStreamSynchronize(hipStream_t) { return 0; }

// mock introspection for tests
unsigned long long mockhip_allocated() { return g_allocated.load(); }
unsigned long long mockhip_launches() { return g_launches.load(); }

}  // extern "C"
