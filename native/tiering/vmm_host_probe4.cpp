#include <dlfcn.h>
#include <stdio.h>
#include <stdint.h>
#include <unistd.h>
#include <initializer_list>
typedef int hipError_t;
struct Loc { int type; int id; };
struct Prop { int type; int rht; Loc location; void* w; struct { unsigned char c,g; unsigned short u; } f; };
struct Acc { Loc location; int flags; };
long rss_kb() { long r=0; FILE* f=fopen("/proc/self/status","r"); char l[256];
  while (f && fgets(l,sizeof l,f)) if (sscanf(l,"VmRSS: %ld",&r)==1) break;
  if (f) fclose(f); return r; }
int main() {
  void* h = dlopen("libamdhip64.so", RTLD_LAZY | RTLD_GLOBAL);
  auto SetDev=(hipError_t(*)(int))dlsym(h,"hipSetDevice");
  auto Reserve=(hipError_t(*)(void**,size_t,size_t,void*,unsigned long long))dlsym(h,"hipMemAddressReserve");
  auto Create=(hipError_t(*)(void**,size_t,const Prop*,unsigned long long))dlsym(h,"hipMemCreate");
  auto Map=(hipError_t(*)(void*,size_t,size_t,void*,unsigned long long))dlsym(h,"hipMemMap");
  auto SetAcc=(hipError_t(*)(void*,size_t,const Acc*,size_t))dlsym(h,"hipMemSetAccess");
  auto Memset=(hipError_t(*)(void*,int,size_t))dlsym(h,"hipMemset");
  auto GetInfo=(hipError_t(*)(size_t*,size_t*))dlsym(h,"hipMemGetInfo");
  auto DevSync=(hipError_t(*)())dlsym(h,"hipDeviceSynchronize");
  SetDev(0);
  size_t f0,t0; GetInfo(&f0,&t0);
  long r0 = rss_kb();
  const size_t LEN=2ull<<30;
  for (int loct : {2,1}) {
    Prop p{}; p.type=1; p.location={loct,0};
    void* hd=nullptr; void* va=nullptr;
    int rc=Create(&hd,LEN,&p,0);
    int rr=Reserve(&va,LEN,4096,nullptr,0);
    int rm=Map(va,LEN,0,hd,0);
    Acc a{{1,0},3};
    int ra=SetAcc(va,LEN,&a,1);
    int rz=Memset(va,7,LEN); DevSync();
    size_t f1,t1; GetInfo(&f1,&t1);
    long r1=rss_kb();
    printf("loct=%d create=%d map=%d acc=%d memset=%d | hbm_free_drop=%zd MiB rss_grow=%ld MiB\n",
      loct, rc, rm, ra, rz, (ssize_t)(f0-f1)>>20, (r1-r0)>>10);
    f0=f1; r0=r1;
  }
  printf("PROBE4_DONE\n");
  return 0;
}
