// tf_vmm_host_probe — can ROCm 7.2 VMM back a device-visible VA range
// with HOST memory? Decides the tiering engine's mechanism:
//   yes → pointer-stable VMM tiering (demote = remap VA device→host)
//   no  → managed-memory tier (hipMemPrefetchAsync over SDMA) stays.
// Location types tried: 2 (hipMemLocationTypeHost, per hip_runtime_api.h)
// and 3 (hipMemLocationTypeHostNuma, id = numa node).
#include <dlfcn.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>

#include <initializer_list>

typedef int hipError_t;
struct Loc { int type; int id; };
struct Prop {
  int type; int requestedHandleType; Loc location; void* win32;
  struct { unsigned char c, g; unsigned short u; } allocFlags;
};
struct Acc { Loc location; int flags; };

int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  if (!h) { printf("PROBE no libamdhip64\n"); return 1; }
  auto SetDev = (hipError_t(*)(int))dlsym(h, "hipSetDevice");
  auto Gran = (hipError_t(*)(size_t*, const Prop*, int))dlsym(
      h, "hipMemGetAllocationGranularity");
  auto Reserve = (hipError_t(*)(void**, size_t, size_t, void*,
                                unsigned long long))dlsym(
      h, "hipMemAddressReserve");
  auto Create = (hipError_t(*)(void**, size_t, const Prop*,
                               unsigned long long))dlsym(h, "hipMemCreate");
  auto Map = (hipError_t(*)(void*, size_t, size_t, void*,
                            unsigned long long))dlsym(h, "hipMemMap");
  auto SetAcc = (hipError_t(*)(void*, size_t, const Acc*, size_t))dlsym(
      h, "hipMemSetAccess");
  auto Unmap = (hipError_t(*)(void*, size_t))dlsym(h, "hipMemUnmap");
  auto Release = (hipError_t(*)(void*))dlsym(h, "hipMemRelease");
  auto AddrFree = (hipError_t(*)(void*, size_t))dlsym(h, "hipMemAddressFree");
  auto Memset = (hipError_t(*)(void*, int, size_t))dlsym(h, "hipMemset");
  auto Memcpy = (hipError_t(*)(void*, const void*, size_t, int))dlsym(
      h, "hipMemcpy");
  auto DevSync = (hipError_t(*)())dlsym(h, "hipDeviceSynchronize");
  SetDev(0);

  const size_t len = 16ull << 20;
  for (int loctype : {2, 3, 1}) {
    Prop prop{};
    prop.type = 1;  // pinned
    prop.location = {loctype, 0};
    size_t gran = 0;
    int eg = Gran(&gran, &prop, 1);
    void* base = nullptr;
    int er = Reserve(&base, len, 0, nullptr, 0);
    void* handle = nullptr;
    int ec = Create(&handle, len, &prop, 0);
    int em = -99, ea = -99, ez = -99, ey = -99;
    if (ec == 0 && er == 0) {
      em = Map(base, len, 0, handle, 0);
      if (em == 0) {
        Acc acc{{1, 0}, 3};
        ea = SetAcc(base, len, &acc, 1);
        if (ea == 0) {
          ez = Memset(base, 0xAB, len);  // device writes the range
          DevSync();
          unsigned char buf[8] = {0};
          ey = Memcpy(buf, base, 8, 2 /*D2H path*/);
          printf("  loctype=%d first bytes %02x%02x\n", loctype, buf[0],
                 buf[1]);
        }
        Unmap(base, len);
      }
      Release(handle);
    }
    if (er == 0) AddrFree(base, len);
    printf("PROBE loctype=%d gran(e=%d)=%zu reserve=%d create=%d map=%d "
           "access=%d memset=%d readback=%d\n",
           loctype, eg, gran, er, ec, em, ea, ez, ey);
  }
  printf("PROBE_DONE\n");
  return 0;
}
