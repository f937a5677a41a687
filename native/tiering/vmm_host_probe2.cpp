// size-sweep host-located VMM handles + chunked mapping of one VA range
#include <dlfcn.h>
#include <stdio.h>
#include <stdint.h>
#include <initializer_list>
typedef int hipError_t;
struct Loc { int type; int id; };
struct Prop { int type; int rht; Loc location; void* w; struct { unsigned char c,g; unsigned short u; } f; };
struct Acc { Loc location; int flags; };
int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  auto SetDev=(hipError_t(*)(int))dlsym(h,"hipSetDevice");
  auto Reserve=(hipError_t(*)(void**,size_t,size_t,void*,unsigned long long))dlsym(h,"hipMemAddressReserve");
  auto Create=(hipError_t(*)(void**,size_t,const Prop*,unsigned long long))dlsym(h,"hipMemCreate");
  auto Map=(hipError_t(*)(void*,size_t,size_t,void*,unsigned long long))dlsym(h,"hipMemMap");
  auto SetAcc=(hipError_t(*)(void*,size_t,const Acc*,size_t))dlsym(h,"hipMemSetAccess");
  auto Unmap=(hipError_t(*)(void*,size_t))dlsym(h,"hipMemUnmap");
  auto Release=(hipError_t(*)(void*))dlsym(h,"hipMemRelease");
  auto AddrFree=(hipError_t(*)(void*,size_t))dlsym(h,"hipMemAddressFree");
  auto Memset=(hipError_t(*)(void*,int,size_t))dlsym(h,"hipMemset");
  auto DevSync=(hipError_t(*)())dlsym(h,"hipDeviceSynchronize");
  SetDev(0);
  // A) single host handle size sweep
  for (size_t mb : {16ul, 64ul, 128ul, 256ul, 512ul, 1024ul, 2048ul}) {
    Prop p{}; p.type=1; p.location={2,0};
    void* hd=nullptr;
    int rc = Create(&hd, mb<<20, &p, 0);
    printf("host create %zu MiB -> %d\n", mb, rc);
    if (!rc) Release(hd);
  }
  // B) chunked: one 1 GiB VA backed by 16x64MiB host handles, memset whole
  {
    size_t total = 1ull<<30, chunk = 64ull<<20;
    void* base=nullptr;
    int rr = Reserve(&base, total, 4096, nullptr, 0);
    int bad=0;
    for (size_t off=0; off<total && !bad; off+=chunk) {
      Prop p{}; p.type=1; p.location={2,0};
      void* hd=nullptr;
      if (Create(&hd, chunk, &p, 0)) { bad=1; break; }
      if (Map((char*)base+off, chunk, 0, hd, 0)) { bad=2; break; }
      Acc a{{1,0},3};
      if (SetAcc((char*)base+off, chunk, &a, 1)) { bad=3; break; }
    }
    int mz = bad?-1:Memset(base, 0x5A, total);
    DevSync();
    printf("chunked 1GiB host: reserve=%d bad=%d memset=%d byte=%02x\n", rr, bad, mz,
           bad?0:((unsigned char*)0,0));
    // readback via hipMemcpy
    auto Memcpy=(hipError_t(*)(void*,const void*,size_t,int))dlsym(h,"hipMemcpy");
    unsigned char b[4]={0};
    int mc = bad?-1:Memcpy(b, (char*)base+total-4, 4, 4);
    printf("chunked readback rc=%d byte=%02x\n", mc, b[0]);
  }
  // C) device handle big size (control)
  for (size_t mb : {1024ul, 4096ul}) {
    Prop p{}; p.type=1; p.location={1,0};
    void* hd=nullptr;
    int rc = Create(&hd, mb<<20, &p, 0);
    printf("device create %zu MiB -> %d\n", mb, rc);
    if (!rc) Release(hd);
  }
  printf("PROBE2_DONE\n");
  return 0;
}
