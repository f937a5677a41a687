// tf_vmm_probe — bisects the VMM failure seen in the worker: for each
// configuration (heap size × hostRegister × small-alloc size) runs
// reserve + create/map/setaccess sequence with properly-typed calls.
#include <dlfcn.h>
#include <stdio.h>
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

typedef int hipError_t;
struct Loc { int type; int id; };
struct Prop {
  int type; int requestedHandleType; Loc location; void* win32;
  struct { unsigned char c, g; unsigned short u; } allocFlags;
};
struct Acc { Loc location; int flags; };

typedef hipError_t (*fn_gran)(size_t*, const Prop*, int);
typedef hipError_t (*fn_reserve)(void**, size_t, size_t, void*, unsigned long long);
typedef hipError_t (*fn_addrfree)(void*, size_t);
typedef hipError_t (*fn_create)(void**, size_t, const Prop*, unsigned long long);
typedef hipError_t (*fn_release)(void*);
typedef hipError_t (*fn_map)(void*, size_t, size_t, void*, unsigned long long);
typedef hipError_t (*fn_unmap)(void*, size_t);
typedef hipError_t (*fn_setacc)(void*, size_t, const Acc*, size_t);
typedef hipError_t (*fn_setdev)(int);
typedef hipError_t (*fn_hostreg)(void*, size_t, unsigned);

int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  if (!h) { printf("no hip\n"); return 1; }
  auto Gran = (fn_gran)dlsym(h, "hipMemGetAllocationGranularity");
  auto Reserve = (fn_reserve)dlsym(h, "hipMemAddressReserve");
  auto AddrFree = (fn_addrfree)dlsym(h, "hipMemAddressFree");
  auto Create = (fn_create)dlsym(h, "hipMemCreate");
  auto Release = (fn_release)dlsym(h, "hipMemRelease");
  auto Map = (fn_map)dlsym(h, "hipMemMap");
  auto Unmap = (fn_unmap)dlsym(h, "hipMemUnmap");
  auto SetAcc = (fn_setacc)dlsym(h, "hipMemSetAccess");
  auto SetDev = (fn_setdev)dlsym(h, "hipSetDevice");
  auto HostReg = (fn_hostreg)dlsym(h, "hipHostRegister");
  SetDev(0);
  Prop prop{}; prop.type = 1; prop.location = {1, 0};
  size_t gran_min = 0, gran_rec = 0;
  int g0 = Gran(&gran_min, &prop, 0);
  int g1 = Gran(&gran_rec, &prop, 1);
  printf("gran min rc=%d %zu  rec rc=%d %zu\n", g0, gran_min, g1, gran_rec);

  void* arena = malloc(256u << 20);
  for (int hostreg = 0; hostreg < 2; ++hostreg) {
    if (hostreg) {
      int rr = HostReg(arena, 256u << 20, 0);
      printf("hostRegister rc=%d\n", rr);
    }
    for (int big = 0; big < 2; ++big) {
      size_t heap = big ? (192ull << 30) : (8ull << 30);
      void* base = nullptr;
      int rr = Reserve(&base, heap, 0, (void*)0x7b0000000000ull, 0);
      printf("cfg hostreg=%d heap=%zuGB reserve rc=%d base=%p\n",
             hostreg, heap >> 30, rr, base);
      if (rr) continue;
      uint64_t off = 0;
      size_t tests[3] = {4u << 20, 4u << 20, 4096};
      for (int i = 0; i < 3; ++i) {
        size_t len = (tests[i] + gran_rec - 1) & ~(gran_rec - 1);
        void* handle = nullptr;
        int rc1 = Create(&handle, len, &prop, 0);
        int rc2 = rc1 ? -1 : Map((char*)base + off, len, 0, handle, 0);
        Acc acc{{1, 0}, 3};
        int rc3 = rc2 ? -1 : SetAcc((char*)base + off, len, &acc, 1);
        printf("  alloc[%d] len=%zu create=%d map=%d access=%d\n",
               i, len, rc1, rc2, rc3);
        if (!rc2) Unmap((char*)base + off, len);
        if (!rc1) Release(handle);
        off += len;
      }
      AddrFree(base, heap);
    }
  }
  printf("PROBE_DONE\n");
  return 0;
}
