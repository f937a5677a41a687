// tf_bw_sweep — copy-kernel configuration sweep on a real MI355X.
// Finds the bandwidth-optimal (variant × grid) for the tiering copy path;
// the winner is what tier_kernels.hip ships with.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

typedef uint32_t u4 __attribute__((ext_vector_type(4)));

__global__ void k_plain(const u4* __restrict__ s, u4* __restrict__ d,
                        size_t n16) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t st = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += st) d[i] = s[i];
}

__global__ void k_nt(const u4* __restrict__ s, u4* __restrict__ d,
                     size_t n16) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t st = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += st) {
    u4 v = __builtin_nontemporal_load(&s[i]);
    __builtin_nontemporal_store(v, &d[i]);
  }
}

__global__ void k_nt_u4s(const u4* __restrict__ s, u4* __restrict__ d,
                         size_t n16) {  // 4x strided unroll
  size_t st = (size_t)gridDim.x * blockDim.x;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + 3 * st < n16; i += 4 * st) {
    u4 a = __builtin_nontemporal_load(&s[i]);
    u4 b = __builtin_nontemporal_load(&s[i + st]);
    u4 c = __builtin_nontemporal_load(&s[i + 2 * st]);
    u4 e = __builtin_nontemporal_load(&s[i + 3 * st]);
    __builtin_nontemporal_store(a, &d[i]);
    __builtin_nontemporal_store(b, &d[i + st]);
    __builtin_nontemporal_store(c, &d[i + 2 * st]);
    __builtin_nontemporal_store(e, &d[i + 3 * st]);
  }
  for (; i < n16; i += st) {
    u4 v = __builtin_nontemporal_load(&s[i]);
    __builtin_nontemporal_store(v, &d[i]);
  }
}

__global__ void k_nt_u4b(const u4* __restrict__ s, u4* __restrict__ d,
                         size_t n16) {  // 4x blocked unroll (64B/lane)
  size_t st = (size_t)gridDim.x * blockDim.x;
  size_t t = blockIdx.x * blockDim.x + threadIdx.x;
  size_t n4 = n16 / 4;
  for (size_t j = t; j < n4; j += st) {
    size_t i = 4 * j;
    u4 a = __builtin_nontemporal_load(&s[i]);
    u4 b = __builtin_nontemporal_load(&s[i + 1]);
    u4 c = __builtin_nontemporal_load(&s[i + 2]);
    u4 e = __builtin_nontemporal_load(&s[i + 3]);
    __builtin_nontemporal_store(a, &d[i]);
    __builtin_nontemporal_store(b, &d[i + 1]);
    __builtin_nontemporal_store(c, &d[i + 2]);
    __builtin_nontemporal_store(e, &d[i + 3]);
  }
}

__global__ void k_plain_u4b(const u4* __restrict__ s, u4* __restrict__ d,
                            size_t n16) {
  size_t st = (size_t)gridDim.x * blockDim.x;
  size_t t = blockIdx.x * blockDim.x + threadIdx.x;
  size_t n4 = n16 / 4;
  for (size_t j = t; j < n4; j += st) {
    size_t i = 4 * j;
    u4 a = s[i], b = s[i + 1], c = s[i + 2], e = s[i + 3];
    d[i] = a;
    d[i + 1] = b;
    d[i + 2] = c;
    d[i + 3] = e;
  }
}

#define CK(x)                          \
  do {                                 \
    hipError_t _e = (x);               \
    if (_e) {                          \
      printf("err %d @%d\n", _e, __LINE__); \
      return 1;                        \
    }                                  \
  } while (0)

int main() {
  size_t bytes = 4ull << 30;
  size_t n16 = bytes / 16;
  void *s, *d;
  CK(hipMalloc(&s, bytes));
  CK(hipMalloc(&d, bytes));
  CK(hipMemset(s, 1, bytes));
  hipEvent_t e0, e1;
  CK(hipEventCreate(&e0));
  CK(hipEventCreate(&e1));
  struct Var {
    const char* name;
    void (*fn)(const u4*, u4*, size_t);
  } vars[] = {
      {"plain", k_plain},       {"nt", k_nt},
      {"nt_u4strided", k_nt_u4s}, {"nt_u4blocked", k_nt_u4b},
      {"plain_u4blocked", k_plain_u4b},
  };
  int grids[] = {1024, 2048, 4096, 8192, 16384};
  for (auto& v : vars) {
    for (int g : grids) {
      // warmup
      hipLaunchKernelGGL(v.fn, dim3(g), dim3(256), 0, 0, (const u4*)s,
                         (u4*)d, n16);
      CK(hipDeviceSynchronize());
      CK(hipEventRecord(e0, 0));
      for (int r = 0; r < 5; ++r)
        hipLaunchKernelGGL(v.fn, dim3(g), dim3(256), 0, 0, (const u4*)s,
                           (u4*)d, n16);
      CK(hipEventRecord(e1, 0));
      CK(hipEventSynchronize(e1));
      float ms = 0;
      CK(hipEventElapsedTime(&ms, e0, e1));
      double tbs = 2.0 * bytes * 5 / (ms / 1e3) / 1e12;
      printf("%-16s grid=%-6d %.2f TB/s\n", v.name, g, tbs);
    }
  }
  printf("SWEEP_DONE\n");
  return 0;
}
