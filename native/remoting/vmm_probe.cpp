// tf_vmm_probe — SetAccess failure characterization (see worker VMM heap).
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

fn_create Create; fn_map Map; fn_setacc SetAcc; fn_unmap Unmap;
fn_release Release; fn_reserve Reserve; fn_addrfree AddrFree;
Prop prop{};

int try_alloc(void* va, size_t len, bool do_access, void** hout) {
  void* handle = nullptr;
  int rc1 = Create(&handle, len, &prop, 0);
  int rc2 = rc1 ? -1 : Map(va, len, 0, handle, 0);
  Acc acc{{1, 0}, 3};
  int rc3 = (!do_access || rc2) ? -2 : SetAcc(va, len, &acc, 1);
  printf("    map %p len=%zu create=%d map=%d access=%d\n", va, len, rc1,
         rc2, rc3);
  if (hout) *hout = handle;
  return rc3 > 0 ? rc3 : (rc2 > 0 ? rc2 : rc1);
}

int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  auto SetDev = (fn_setdev)dlsym(h, "hipSetDevice");
  auto Gran = (fn_gran)dlsym(h, "hipMemGetAllocationGranularity");
  Reserve = (fn_reserve)dlsym(h, "hipMemAddressReserve");
  AddrFree = (fn_addrfree)dlsym(h, "hipMemAddressFree");
  Create = (fn_create)dlsym(h, "hipMemCreate");
  Release = (fn_release)dlsym(h, "hipMemRelease");
  Map = (fn_map)dlsym(h, "hipMemMap");
  Unmap = (fn_unmap)dlsym(h, "hipMemUnmap");
  SetAcc = (fn_setacc)dlsym(h, "hipMemSetAccess");
  SetDev(0);
  prop.type = 1; prop.location = {1, 0};
  size_t gran = 0;
  Gran(&gran, &prop, 1);
  size_t heap = 8ull << 30;

  printf("A) holes between mappings (gap=gran):\n");
  {
    void* base = nullptr;
    Reserve(&base, heap, 0, nullptr, 0);
    uint64_t off = 0;
    size_t sizes[3] = {4u<<20, 4u<<20, 4096};
    for (int i = 0; i < 3; ++i) {
      try_alloc((char*)base + off, sizes[i], true, nullptr);
      off += sizes[i] + gran;  // hole
    }
    AddrFree(base, heap);
  }
  printf("B) small first (4K,4M,4M adjacent):\n");
  {
    void* base = nullptr;
    Reserve(&base, heap, 0, nullptr, 0);
    uint64_t off = 0;
    size_t sizes[3] = {4096, 4u<<20, 4u<<20};
    for (int i = 0; i < 3; ++i) {
      try_alloc((char*)base + off, sizes[i], true, nullptr);
      off += sizes[i];
    }
    AddrFree(base, heap);
  }
  printf("C) adjacent, SetAccess over the union each time:\n");
  {
    void* base = nullptr;
    Reserve(&base, heap, 0, nullptr, 0);
    uint64_t off = 0;
    size_t sizes[4] = {4u<<20, 4u<<20, 4096, 2u<<20};
    for (int i = 0; i < 4; ++i) {
      try_alloc((char*)base + off, sizes[i], false, nullptr);
      off += sizes[i];
      Acc acc{{1, 0}, 3};
      int rc = SetAcc(base, off, &acc, 1);
      printf("    setaccess union [base, +%llx) -> %d\n",
             (unsigned long long)off, rc);
    }
    AddrFree(base, heap);
  }
  printf("D) adjacent, access BEFORE... map-then-access per alloc, but re-assert prior range after failure:\n");
  {
    void* base = nullptr;
    Reserve(&base, heap, 0, nullptr, 0);
    uint64_t off = 0;
    size_t sizes[3] = {4u<<20, 4u<<20, 4096};
    for (int i = 0; i < 3; ++i) {
      int rc = try_alloc((char*)base + off, sizes[i], true, nullptr);
      if (rc > 0) {
        Acc acc{{1, 0}, 3};
        int rc2 = SetAcc((char*)base + off, sizes[i], &acc, 1);
        printf("    retry same range -> %d\n", rc2);
      }
      off += sizes[i];
    }
    AddrFree(base, heap);
  }
  printf("PROBE_DONE\n");
  return 0;
}
