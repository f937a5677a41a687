// tf_vmm_probe — exercises the HIP VMM sequence the worker's VA-stable heap
// uses, printing every rc (gpurun diagnostic for the snapshot/migration
// allocator).
#include <dlfcn.h>
#include <stdio.h>
#include <stdint.h>
#include <string.h>

typedef int hipError_t;
struct Loc { int type; int id; };
struct Prop {
  int type; int requestedHandleType; Loc location; void* win32;
  struct { unsigned char c, g; unsigned short u; } allocFlags;
};
struct Acc { Loc location; int flags; };

int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  if (!h) { printf("no hip\n"); return 1; }
#define GET(n) auto n = (hipError_t(*)(...))dlsym(h, #n); if (!n) { printf("missing %s\n", #n); return 1; }
  GET(hipInit); GET(hipSetDevice);
  GET(hipMemGetAllocationGranularity); GET(hipMemAddressReserve);
  GET(hipMemCreate); GET(hipMemMap); GET(hipMemSetAccess);
  GET(hipMemUnmap); GET(hipMemRelease); GET(hipMemcpyAsync);
  GET(hipStreamSynchronize);
  hipSetDevice(0);
  Prop prop{}; prop.type = 1; prop.location = {1, 0};
  size_t gran = 0;
  printf("gran_rc=%d gran=%zu\n",
         (int)hipMemGetAllocationGranularity(&gran, &prop, 1), gran);
  void* base = nullptr;
  size_t heap = 8ull << 30;
  printf("reserve_rc=%d base=%p\n",
         (int)hipMemAddressReserve(&base, heap, 0, (void*)0x7a0000000000ull, 0),
         base);
  uint64_t off = 0;
  size_t sizes[4] = {4u << 20, 4u << 20, gran, gran};
  for (int i = 0; i < 4; ++i) {
    size_t len = (sizes[i] + gran - 1) & ~(gran - 1);
    void* handle = nullptr;
    int rc1 = hipMemCreate(&handle, len, &prop, 0);
    int rc2 = rc1 ? -1 : (int)hipMemMap((char*)base + off, len, 0, handle, 0);
    Acc acc{{1, 0}, 3};
    int rc3 = rc2 ? -1 : (int)hipMemSetAccess((char*)base + off, len, &acc, 1);
    // touch it
    char buf[16] = {42};
    int rc4 = rc3 ? -1 : (int)hipMemcpyAsync((char*)base + off, buf, 16, 1, nullptr);
    int rc5 = rc4 ? -1 : (int)hipStreamSynchronize(nullptr);
    printf("alloc[%d] len=%zu off=%llx create=%d map=%d access=%d h2d=%d sync=%d\n",
           i, len, (unsigned long long)off, rc1, rc2, rc3, rc4, rc5);
    off += len;
  }
  printf("PROBE_DONE\n");
  return 0;
}
