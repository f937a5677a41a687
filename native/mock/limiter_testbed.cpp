// tf_limiter_testbed — exercises the LD_PRELOAD limiter against libmockhip
// on CPU-only machines (tests/test_limiter.py drives it). Prints one JSON
// line with the scenario result.
//
// Usage: tf_limiter_testbed <scenario>
//   alloc <n> <bytes>   n hipMallocs of <bytes>; prints admitted/denied
//   launch <n>          n hipLaunchKernels; prints elapsed seconds
//   stats               prints tf_limiter_stats of device 0 (if preloaded)
//   lat <n>             n launches with TF_LIMITER_TRACE; prints histogram totals

#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>

extern "C" {
typedef int hipError_t;
typedef void* hipStream_t;
struct dim3u {
  unsigned x, y, z;
};
hipError_t hipMalloc(void**, size_t);
hipError_t hipFree(void*);
hipError_t hipLaunchKernel(const void*, dim3u, dim3u, void**, size_t,
                           hipStream_t);
}

static double now_s() {
  timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec + ts.tv_nsec * 1e-9;
}

int main(int argc, char** argv) {
  if (argc < 2) return 2;
  const char* sc = argv[1];
  if (!strcmp(sc, "lat")) {
    int n = argc > 2 ? atoi(argv[2]) : 100;
    dim3u g{1, 1, 1}, b{64, 1, 1};
    for (int i = 0; i < n; ++i)
      hipLaunchKernel((void*)0x1, g, b, nullptr, 0, nullptr);
    typedef unsigned long long (*hist_fn)(int, unsigned long long*);
    hist_fn f = (hist_fn)dlsym(RTLD_DEFAULT, "tf_limiter_latency_hist");
    unsigned long long total = f ? f(0, nullptr) : 0;
    printf("{\"launch_samples\": %llu}\n", total);
    return 0;
  }
  if (!strcmp(sc, "alloc")) {
    int n = argc > 2 ? atoi(argv[2]) : 10;
    size_t bytes = argc > 3 ? strtoull(argv[3], nullptr, 10) : (1u << 20);
    int ok = 0, denied = 0;
    void* last = nullptr;
    for (int i = 0; i < n; ++i) {
      void* p = nullptr;
      if (hipMalloc(&p, bytes) == 0) {
        ++ok;
        last = p;
      } else {
        ++denied;
      }
    }
    // free one and retry: cap must admit again
    int readmitted = 0;
    if (denied && last) {
      hipFree(last);
      void* p = nullptr;
      if (hipMalloc(&p, bytes) == 0) readmitted = 1;
    }
    printf("{\"ok\": %d, \"denied\": %d, \"readmitted\": %d}\n", ok, denied,
           readmitted);
  } else if (!strcmp(sc, "launch")) {
    int n = argc > 2 ? atoi(argv[2]) : 100;
    dim3u g{1, 1, 1}, b{64, 1, 1};
    double t0 = now_s();
    for (int i = 0; i < n; ++i)
      hipLaunchKernel(nullptr, g, b, nullptr, 0, nullptr);
    double dt = now_s() - t0;
    printf("{\"launches\": %d, \"elapsed_s\": %.6f, \"rate\": %.1f}\n", n, dt,
           n / dt);
  } else if (!strcmp(sc, "stats")) {
    using stats_fn = int (*)(int, unsigned long long*, unsigned long long*,
                             unsigned*, unsigned long long*, double*);
    auto fn = (stats_fn)dlsym(RTLD_DEFAULT, "tf_limiter_stats");
    if (!fn) {
      printf("{\"error\": \"limiter not loaded\"}\n");
      return 0;
    }
    unsigned long long used = 0, lim = 0, blk = 0;
    unsigned launches = 0;
    double tok = 0;
    int r = fn(0, &used, &lim, &launches, &blk, &tok);
    printf(
        "{\"rc\": %d, \"mem_used\": %llu, \"mem_limit\": %llu, \"launches\": "
        "%u, \"block_ns\": %llu, \"tokens\": %.2f}\n",
        r, used, lim, launches, blk, tok);
  } else {
    return 2;
  }
  return 0;
}
