// tf_vgpu_worker — the per-vGPU remote execution process.
//
// Owns one MI355X (or a fraction of one, under the same limiter/CU-mask
// regime as local workers) and executes the HIP command stream produced by
// libtfhip_client.so in a GPU-less client process. See protocol.h for the
// transport design and the <4% overhead argument.
//
// Bootstrap: listens on a Unix socket (TF_WORKER_SOCKET or argv). A client
// connects, sends the memfd of the shared segment via SCM_RIGHTS; the
// worker maps it, hipHostRegisters the arena (GPU DMAs directly to/from the
// shared pages) and enters the command loop. One client at a time (one
// worker per vGPU — the reference's model; the connection URL selects a
// worker, tensorfusionconnection_controller.go:136).
//
// Build: g++ (no HIP link — dlopens libamdhip64 so the binary also builds
// and starts in GPU-less CI, failing only on first real command).

#include <dlfcn.h>
#include <signal.h>
#include <errno.h>
#include <fcntl.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <map>
#include <string>
#include <unordered_map>
#include <vector>

#include "codeobj.h"
#include "limiter_shm.h"
#include "protocol.h"
#include "ring.h"

namespace {

// ------------------------------------------------------------- hip ABI
typedef int hipError_t;
typedef void* hipStream_t;
typedef void* hipEvent_t;
typedef void* hipModule_t;
typedef void* hipFunction_t;

#define HIP_LAUNCH_PARAM_BUFFER_POINTER ((void*)0x01)
#define HIP_LAUNCH_PARAM_BUFFER_SIZE ((void*)0x02)
#define HIP_LAUNCH_PARAM_END ((void*)0x03)

struct Hip {
  void* h = nullptr;
  hipError_t (*Init)(unsigned);
  hipError_t (*GetDeviceCount)(int*);
  hipError_t (*SetDevice)(int);
  hipError_t (*GetDevicePropertiesR0600)(void*, int);
  hipError_t (*DeviceGetAttribute)(int*, int, int);
  hipError_t (*Malloc)(void**, size_t);
  hipError_t (*Free)(void*);
  hipError_t (*HostRegister)(void*, size_t, unsigned);
  hipError_t (*HostUnregister)(void*);
  hipError_t (*MemcpyAsync)(void*, const void*, size_t, int, hipStream_t);
  hipError_t (*MemsetD8Async)(void*, unsigned char, size_t, hipStream_t);
  hipError_t (*StreamCreateWithPriority)(hipStream_t*, unsigned, int);
  hipError_t (*StreamDestroy)(hipStream_t);
  hipError_t (*StreamSynchronize)(hipStream_t);
  hipError_t (*StreamQuery)(hipStream_t);
  hipError_t (*EventCreateWithFlags)(hipEvent_t*, unsigned);
  hipError_t (*EventRecord)(hipEvent_t, hipStream_t);
  hipError_t (*EventSynchronize)(hipEvent_t);
  hipError_t (*EventQuery)(hipEvent_t);
  hipError_t (*EventElapsedTime)(float*, hipEvent_t, hipEvent_t);
  hipError_t (*EventDestroy)(hipEvent_t);
  hipError_t (*StreamWaitEvent)(hipStream_t, hipEvent_t, unsigned);
  hipError_t (*DeviceSynchronize)(void);
  hipError_t (*ModuleLoadData)(hipModule_t*, const void*);
  hipError_t (*ModuleGetFunction)(hipFunction_t*, hipModule_t, const char*);
  hipError_t (*ModuleLaunchKernel)(hipFunction_t, unsigned, unsigned, unsigned,
                                   unsigned, unsigned, unsigned, unsigned,
                                   hipStream_t, void**, void**);
  hipError_t (*MemGetInfo)(size_t*, size_t*);
  hipError_t (*DeviceCanAccessPeer)(int*, int, int);
  hipError_t (*GetLastError)(void);
  hipError_t (*StreamBeginCapture)(hipStream_t, int) = nullptr;
  hipError_t (*StreamEndCapture)(hipStream_t, void**) = nullptr;
  hipError_t (*GraphInstantiateWithFlags)(void**, void*,
                                          unsigned long long) = nullptr;
  hipError_t (*GraphLaunch)(void*, hipStream_t) = nullptr;
  hipError_t (*GraphDestroy)(void*) = nullptr;
  hipError_t (*GraphExecDestroy)(void*) = nullptr;
  hipError_t (*GraphGetNodes)(void*, void**, size_t*) = nullptr;
  hipError_t (*GraphNodeGetType)(void*, int*) = nullptr;
  hipError_t (*GraphKernelNodeGetParams)(void*, void*) = nullptr;

  // VMM surface (optional — absent on very old runtimes; worker falls back
  // to plain hipMalloc and snapshot/restore is disabled).
  hipError_t (*MemAddressReserve)(void**, size_t, size_t, void*,
                                  unsigned long long) = nullptr;
  hipError_t (*MemAddressFree)(void*, size_t) = nullptr;
  hipError_t (*MemCreate)(void**, size_t, const void*,
                          unsigned long long) = nullptr;
  hipError_t (*MemRelease)(void*) = nullptr;
  hipError_t (*MemMap)(void*, size_t, size_t, void*,
                       unsigned long long) = nullptr;
  hipError_t (*MemUnmap)(void*, size_t) = nullptr;
  hipError_t (*MemSetAccess)(void*, size_t, const void*, size_t) = nullptr;
  hipError_t (*MemGetAllocationGranularity)(size_t*, const void*,
                                            int) = nullptr;
  // stream-ordered allocator / mempool surface (optional)
  hipError_t (*MallocAsync)(void**, size_t, void*) = nullptr;
  hipError_t (*FreeAsync)(void*, void*) = nullptr;
  hipError_t (*DeviceGetDefaultMemPool)(void**, int) = nullptr;
  hipError_t (*MemPoolSetAttribute)(void*, int, void*) = nullptr;
  hipError_t (*MemPoolGetAttribute)(void*, int, void*) = nullptr;
  hipError_t (*MemPoolTrimTo)(void*, size_t) = nullptr;
  hipError_t (*ModuleGetGlobal)(void**, size_t*, void*, const char*) =
      nullptr;

  bool load() {
    const char* names[] = {"libamdhip64.so", "libamdhip64.so.7",
                           "/opt/rocm/lib/libamdhip64.so"};
    for (const char* n : names) {
      h = dlopen(n, RTLD_LAZY | RTLD_GLOBAL);
      if (h) break;
    }
    if (!h) return false;
// Resolve through the global scope FIRST so an LD_PRELOADed limiter
// (soft isolation on the worker pod) interposes the worker's own HIP
// calls; fall back to the libamdhip64 handle.
#define R(f, sym)                                              \
  f = reinterpret_cast<decltype(f)>(dlsym(RTLD_DEFAULT, sym)); \
  if (!(f)) f = reinterpret_cast<decltype(f)>(dlsym(h, sym));  \
  if (!(f)) {                                                  \
    fprintf(stderr, "[worker] missing %s\n", sym);             \
    return false;                                              \
  }
    R(GetDeviceCount, "hipGetDeviceCount")
    R(SetDevice, "hipSetDevice")
    R(GetDevicePropertiesR0600, "hipGetDevicePropertiesR0600")
    R(DeviceGetAttribute, "hipDeviceGetAttribute")
    R(Malloc, "hipMalloc")
    R(Free, "hipFree")
    R(HostRegister, "hipHostRegister")
    R(HostUnregister, "hipHostUnregister")
    R(MemcpyAsync, "hipMemcpyAsync")
    R(MemsetD8Async, "hipMemsetD8Async")
    R(StreamCreateWithPriority, "hipStreamCreateWithPriority")
    R(StreamDestroy, "hipStreamDestroy")
    R(StreamSynchronize, "hipStreamSynchronize")
    R(StreamQuery, "hipStreamQuery")
    R(EventCreateWithFlags, "hipEventCreateWithFlags")
    R(EventRecord, "hipEventRecord")
    R(EventSynchronize, "hipEventSynchronize")
    R(EventQuery, "hipEventQuery")
    R(EventElapsedTime, "hipEventElapsedTime")
    R(EventDestroy, "hipEventDestroy")
    R(StreamWaitEvent, "hipStreamWaitEvent")
    R(DeviceSynchronize, "hipDeviceSynchronize")
    R(ModuleLoadData, "hipModuleLoadData")
    R(ModuleGetFunction, "hipModuleGetFunction")
    R(ModuleLaunchKernel, "hipModuleLaunchKernel")
    R(MemGetInfo, "hipMemGetInfo")
    R(DeviceCanAccessPeer, "hipDeviceCanAccessPeer")
    R(GetLastError, "hipGetLastError")
#undef R
#define O(f, sym) f = reinterpret_cast<decltype(f)>(dlsym(h, sym));
    O(StreamBeginCapture, "hipStreamBeginCapture")
    O(StreamEndCapture, "hipStreamEndCapture")
    O(GraphInstantiateWithFlags, "hipGraphInstantiateWithFlags")
    O(GraphLaunch, "hipGraphLaunch")
    O(GraphDestroy, "hipGraphDestroy")
    O(GraphExecDestroy, "hipGraphExecDestroy")
    O(GraphGetNodes, "hipGraphGetNodes")
    O(GraphNodeGetType, "hipGraphNodeGetType")
    O(GraphKernelNodeGetParams, "hipGraphKernelNodeGetParams")
    O(MemAddressReserve, "hipMemAddressReserve")
    O(MemAddressFree, "hipMemAddressFree")
    O(MemCreate, "hipMemCreate")
    O(MemRelease, "hipMemRelease")
    O(MemMap, "hipMemMap")
    O(MemUnmap, "hipMemUnmap")
    O(MemSetAccess, "hipMemSetAccess")
    O(MemGetAllocationGranularity, "hipMemGetAllocationGranularity")
    O(MallocAsync, "hipMallocAsync")
    O(FreeAsync, "hipFreeAsync")
    O(DeviceGetDefaultMemPool, "hipDeviceGetDefaultMemPool")
    O(MemPoolSetAttribute, "hipMemPoolSetAttribute")
    O(MemPoolGetAttribute, "hipMemPoolGetAttribute")
    O(MemPoolTrimTo, "hipMemPoolTrimTo")
    O(ModuleGetGlobal, "hipModuleGetGlobal")
#undef O
    return true;
  }

  bool vmm_available() const {
    return MemAddressReserve && MemAddressFree && MemCreate && MemRelease &&
           MemMap && MemUnmap && MemSetAccess && MemGetAllocationGranularity;
  }
};

Hip hip;

// ------------------------------------------------- VA-stable VMM heap
// Device memory is suballocated from ONE VA reservation at a fixed base,
// mapped page-by-page with hipMemCreate/hipMemMap. Snapshot/restore and
// live migration rely on this: a resumed worker reserves the same base, so
// every device pointer the client holds stays valid — no pointer
// translation (the part the reference left closed-source and returned 501
// for).

struct hipMemLocation_ {
  int type;
  int id;
};
struct hipMemAllocationProp_ {
  int type;                 // hipMemAllocationTypePinned = 1 (device mem)
  int requestedHandleType;  // none
  hipMemLocation_ location;
  void* win32HandleMetaData;
  struct {
    unsigned char compressionType, gpuDirectRDMACapable;
    unsigned short usage;
  } allocFlags;
};
struct hipMemAccessDesc_ {
  hipMemLocation_ location;
  int flags;  // hipMemAccessFlagsProtReadWrite = 3
};

constexpr uint64_t VMM_BASE_HINT = 0x7a0000000000ull;

struct VmmRange {
  void* handle = nullptr;
  size_t bytes = 0;  // granularity-rounded mapped size
  size_t req_bytes = 0;  // what the client asked for (snapshot copies this)
};

struct Vmm {
  // VA range manager WITHOUT one giant reservation: each allocation gets
  // its own hipMemAddressReserve at an exact address inside [base,
  // base+heap). Rationale (tf_vmm_probe characterization): ROCm 7.2's
  // hipMemSetAccess intermittently rejects ranges deep inside a shared
  // reservation, but the first SetAccess of a fresh reservation always
  // succeeds — per-allocation reservations make every SetAccess exactly
  // that. VA stability across snapshot/restore is preserved because WE
  // choose the addresses.
  bool enabled = false;
  int device = 0;
  uint64_t base = 0;
  size_t heap_bytes = 0;
  // ---- VRAM accounting (closes the round-1 hole: hipMemCreate bypassed
  // the limiter's hipMalloc-based cap, so a remote vGPU could exceed its
  // VRAM limit). The worker enforces TF_VRAM_LIMIT_BYTES on its own heap
  // and reports mapped bytes into the limiter shm (vmm_bytes field).
  uint64_t mapped_bytes = 0;
  uint64_t vram_limit = 0;
  TfSharedState* shm = nullptr;
  size_t gran = 2u << 20;
  std::map<uint64_t, size_t> free_spans;          // va → len
  std::map<uint64_t, VmmRange> mapped;            // va → range

  bool init(int dev, uint64_t base_hint, size_t heap) {
    if (!hip.vmm_available()) return false;
    device = dev;
    hipMemAllocationProp_ prop{};
    prop.type = 1;
    prop.location = {1 /*device*/, dev};
    size_t g = 0;
    if (hip.MemGetAllocationGranularity(&g, &prop, 1 /*recommended*/) != 0 ||
        g == 0)
      return false;
    gran = g;
    // probe that the base region is reservable, then release it — real
    // reservations are per-allocation at exact addresses
    uint64_t want = base_hint ? base_hint : VMM_BASE_HINT;
    void* p = nullptr;
    if (hip.MemAddressReserve(&p, gran, 0, (void*)want, 0) != 0)
      return false;
    hip.MemAddressFree(p, gran);
    if ((uint64_t)p != want) return false;
    base = want;
    heap_bytes = heap;
    free_spans[base] = heap;
    const char* lim = getenv("TF_VRAM_LIMIT_BYTES");
    if (lim) vram_limit = strtoull(lim, nullptr, 10);
    const char* shm_path = getenv("TF_SHM_PATH");
    if (shm_path && *shm_path) {
      int fd = open(shm_path, O_RDWR);
      if (fd >= 0) {
        void* m = mmap(nullptr, TF_SHM_SIZE, PROT_READ | PROT_WRITE,
                       MAP_SHARED, fd, 0);
        close(fd);
        if (m != MAP_FAILED) {
          auto* st = reinterpret_cast<TfSharedState*>(m);
          if (st->magic == TF_SHM_MAGIC) {
            shm = st;
            uint64_t sl = tfshm::at(&st->dev[0].mem_limit_bytes)
                              ->load(std::memory_order_relaxed);
            if (sl && (!vram_limit || sl < vram_limit)) vram_limit = sl;
          } else {
            munmap(m, TF_SHM_SIZE);
          }
        }
      }
    }
    enabled = true;
    return true;
  }

  void report_usage() {
    if (!shm) return;
    tfshm::at(&shm->dev[0].vmm_bytes)
        ->store(mapped_bytes, std::memory_order_relaxed);
  }

  uint64_t round_up(uint64_t n) const { return (n + gran - 1) & ~(gran - 1); }

  hipError_t map_at(uint64_t va, size_t len, size_t req) {
    void* got = nullptr;
    hipError_t e = hip.MemAddressReserve(&got, len, 0, (void*)va, 0);
    if (e != 0 || (uint64_t)got != va) {
      fprintf(stderr, "[worker] vmm: reserve(%llx,%zu) -> %d (got %p)\n",
              (unsigned long long)va, len, e, got);
      if (e == 0) hip.MemAddressFree(got, len);
      return e ? e : 1;
    }
    hipMemAllocationProp_ prop{};
    prop.type = 1;
    prop.location = {1, device};
    void* handle = nullptr;
    e = hip.MemCreate(&handle, len, &prop, 0);
    if (e != 0) {
      fprintf(stderr, "[worker] vmm: hipMemCreate(%zu) -> %d\n", len, e);
      hip.MemAddressFree((void*)va, len);
      return e;
    }
    e = hip.MemMap((void*)va, len, 0, handle, 0);
    if (e != 0) {
      fprintf(stderr, "[worker] vmm: hipMemMap(%llx,%zu) -> %d\n",
              (unsigned long long)va, len, e);
      hip.MemRelease(handle);
      hip.MemAddressFree((void*)va, len);
      return e;
    }
    hipMemAccessDesc_ acc{{1, device}, 3 /*RW*/};
    e = hip.MemSetAccess((void*)va, len, &acc, 1);
    if (e != 0) {
      fprintf(stderr, "[worker] vmm: hipMemSetAccess(%llx,%zu) -> %d\n",
              (unsigned long long)va, len, e);
      hip.MemUnmap((void*)va, len);
      hip.MemRelease(handle);
      hip.MemAddressFree((void*)va, len);
      return e;
    }
    mapped[va] = VmmRange{handle, len, req};
    return 0;
  }

  void* alloc(size_t sz, hipError_t* err) {
    size_t len = round_up(sz ? sz : 1);
    if (vram_limit && mapped_bytes + len > vram_limit) {
      *err = 2;  // hipErrorOutOfMemory: the vGPU's VRAM cap
      return nullptr;
    }
    for (auto it = free_spans.begin(); it != free_spans.end(); ++it) {
      if (it->second < len) continue;
      uint64_t va = it->first;
      size_t span = it->second;
      free_spans.erase(it);
      if (span > len) free_spans[va + len] = span - len;
      hipError_t e = map_at(va, len, sz);
      if (e != 0) {
        free_spans[va] = len;  // give back (coalesce lazily)
        *err = e;
        return nullptr;
      }
      *err = 0;
      mapped_bytes += len;
      report_usage();
      return (void*)va;
    }
    *err = 2;  // hipErrorOutOfMemory (VA heap exhausted)
    return nullptr;
  }

  // Restore path: map at an exact VA recorded in the snapshot.
  hipError_t alloc_exact(uint64_t va, size_t req) {
    size_t len = round_up(req ? req : 1);
    // carve [va, va+len) out of the containing free span
    auto it = free_spans.upper_bound(va);
    if (it == free_spans.begin()) return 1;
    --it;
    uint64_t sb = it->first;
    size_t slen = it->second;
    if (va < sb || va + len > sb + slen) return 1;
    free_spans.erase(it);
    if (va > sb) free_spans[sb] = va - sb;
    if (sb + slen > va + len) free_spans[va + len] = sb + slen - (va + len);
    return map_at(va, len, req);
  }

  hipError_t free_(void* p) {
    auto it = mapped.find((uint64_t)p);
    if (it == mapped.end()) return 1;
    hip.MemUnmap(p, it->second.bytes);
    hip.MemRelease(it->second.handle);
    hip.MemAddressFree(p, it->second.bytes);
    uint64_t va = it->first;
    size_t len = it->second.bytes;
    mapped_bytes -= len < mapped_bytes ? len : mapped_bytes;
    report_usage();
    mapped.erase(it);
    // coalesce with neighbours
    auto nxt = free_spans.find(va + len);
    if (nxt != free_spans.end()) {
      len += nxt->second;
      free_spans.erase(nxt);
    }
    if (!free_spans.empty()) {
      auto prv = free_spans.upper_bound(va);
      if (prv != free_spans.begin()) {
        --prv;
        if (prv->first + prv->second == va) {
          prv->second += len;
          return 0;
        }
      }
    }
    free_spans[va] = len;
    return 0;
  }
};

struct PendingChunk {
  hipEvent_t ev;
  uint64_t end_off;  // arena_freed advances to this when ev completes
};

struct FuncRec {
  uint64_t image_id;
  std::string name;
};
struct StreamRec {
  uint32_t flags;
  int prio;
};

// Private pinned staging ring for INLINE H2D payloads: copying the bytes
// out of the cmd ring lets the worker pop the command immediately and keep
// the copy fully async (no StreamSynchronize in the hot path).
struct Staging {
  uint8_t* buf = nullptr;
  size_t cap = 8u << 20;
  uint64_t head = 0;             // bump offset (free-running)
  uint64_t freed = 0;            // advanced when chunk events complete
  std::vector<std::pair<hipEvent_t, uint64_t>> pending;  // (ev, end_off)

  bool init() {
    buf = (uint8_t*)aligned_alloc(4096, cap);
    if (!buf) return false;
    return hip.HostRegister(buf, cap, 0) == 0;
  }

  void retire(bool wait_all = false) {
    while (!pending.empty()) {
      auto& p = pending.front();
      hipError_t q = wait_all ? hip.EventSynchronize(p.first)
                              : hip.EventQuery(p.first);
      if (q != 0 && !wait_all) break;
      hip.EventDestroy(p.first);
      freed = p.second;
      pending.erase(pending.begin());
    }
  }

  // Returns a pointer able to hold n bytes (contiguous), or nullptr if the
  // payload is too large for staging at all.
  uint8_t* reserve(size_t n, uint64_t* end_off) {
    if (n > cap / 2) return nullptr;
    for (;;) {
      uint64_t off = head % cap;
      uint64_t avail_end = freed + cap;
      uint64_t want_end = (off + n <= cap) ? head + n
                                           : head + (cap - off) + n;
      if (want_end <= avail_end) {
        if (off + n > cap) head += cap - off;  // skip wrap gap
        uint8_t* p = buf + (head % cap);
        *end_off = head + n;
        head += n;
        return p;
      }
      retire(false);
      if (pending.empty()) retire(true);  // shouldn't happen
    }
  }

  void track(hipStream_t stream, uint64_t end_off) {
    hipEvent_t ev;
    if (hip.EventCreateWithFlags(&ev, 0x2) != 0) return;
    hip.EventRecord(ev, stream);
    pending.push_back({ev, end_off});
  }
};

struct Worker {
  tfrpc::Header* hdr = nullptr;
  int cli_fd = -1;  // client socket: EOF ⇒ client exited
  bool tcp_mode = false;  // cross-node session: D2H data rides in replies
  tfrpc::RingView cmd;  // consumer
  tfrpc::RingView cpl;  // producer
  uint8_t* arena = nullptr;
  std::vector<PendingChunk> pending;  // FIFO by arena order
  std::unordered_map<uint64_t, std::map<std::string, tfrpc::KernelSig>> sigs;
  std::unordered_map<uint64_t, hipModule_t> modules;  // image_id → module
  bool verbose = getenv("TF_WORKER_DEBUG") != nullptr;
  Staging staging;

  // ---- migratable state (snapshot/restore) ----
  Vmm vmm;
  int device = 0;
  std::map<uint64_t, size_t> plain_allocs;  // fallback when !vmm.enabled
  std::unordered_map<uint64_t, std::vector<uint8_t>> images;  // id → bytes
  std::unordered_map<uint64_t, uint64_t> module_image;  // handle → image_id
  std::unordered_map<uint64_t, FuncRec> funcs;    // func handle → origin
  std::unordered_map<uint64_t, StreamRec> streams;
  std::unordered_map<uint64_t, uint32_t> events;  // handle → flags
  std::unordered_map<uint64_t, uint64_t> tr;  // old handle → new (restore)
};

Worker W;

volatile sig_atomic_t g_snapshot_req = 0;

// Translate a handle from a pre-migration snapshot to its live value.
inline uint64_t xl(uint64_t h) {
  if (W.tr.empty() || h == 0) return h;
  auto it = W.tr.find(h);
  return it == W.tr.end() ? h : it->second;
}

void set_sticky(hipError_t e, const char* what = "") {
  if (e != 0) {
    if (W.verbose)
      fprintf(stderr, "[worker] %s failed: hipError %d\n", what, e);
    uint64_t expect = 0;
    tfrpc::at(&W.hdr->sticky_error)
        ->compare_exchange_strong(expect, (uint64_t)e);
  }
}

void reply(uint64_t seq, int32_t err, const void* body, uint32_t body_len) {
  size_t need = sizeof(tfrpc::CplHdr) + body_len;
  uint8_t* p;
  while (!(p = W.cpl.try_reserve(need))) {
    // client is slow draining completions; rare
    usleep(50);
  }
  auto* ch = reinterpret_cast<tfrpc::CplHdr*>(p);
  ch->seq = seq;
  ch->err = err;
  ch->body_len = body_len;
  if (body_len) memcpy(p + sizeof(tfrpc::CplHdr), body, body_len);
  W.cpl.commit();
  // wake a parked client
  if (tfrpc::at(&W.hdr->futex_cpl)->exchange(1) == 0)
    tfrpc::futex_wake(&W.hdr->futex_cpl);
}

void retire_pending(bool wait_all = false) {
  while (!W.pending.empty()) {
    auto& c = W.pending.front();
    hipError_t q = wait_all ? hip.EventSynchronize(c.ev) : hip.EventQuery(c.ev);
    if (q != 0 && !wait_all) break;  // not ready
    hip.EventDestroy(c.ev);
    tfrpc::at(&W.hdr->arena_freed)
        ->store(c.end_off, std::memory_order_release);
    W.pending.erase(W.pending.begin());
  }
}

void track_arena_use(uint64_t end_off, hipStream_t stream) {
  hipEvent_t ev;
  if (hip.EventCreateWithFlags(&ev, 0x2 /*DisableTiming*/) != 0) return;
  hip.EventRecord(ev, stream);
  W.pending.push_back({ev, end_off});
}

// ------------------------------------------------------------- handlers

void handle(tfrpc::CmdHdr* c, uint8_t* body) {
  using namespace tfrpc;
  switch (c->op) {
    case OP_HELLO: {
      int n = 0;
      hipError_t e = hip.GetDeviceCount(&n);
      struct {
        int32_t count;
      } r{n};
      reply(c->seq, e, &r, sizeof r);
      break;
    }
    case OP_SET_DEVICE: {
      int dev;
      memcpy(&dev, body, 4);
      hipError_t e = hip.SetDevice(dev);
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "async-op");
      break;
    }
    case OP_GET_PROPS: {
      int dev;
      memcpy(&dev, body, 4);
      static thread_local uint8_t props[2048];
      memset(props, 0, sizeof props);
      hipError_t e = hip.GetDevicePropertiesR0600(props, dev);
      reply(c->seq, e, props, 1472);
      break;
    }
    case OP_GET_ATTRIBUTE: {
      int dev, attr, val = 0;
      memcpy(&dev, body, 4);
      memcpy(&attr, body + 4, 4);
      hipError_t e = hip.DeviceGetAttribute(&val, attr, dev);
      reply(c->seq, e, &val, 4);
      break;
    }
    case OP_MALLOC: {
      uint64_t size;
      memcpy(&size, body, 8);
      void* p = nullptr;
      hipError_t e;
      if (W.vmm.enabled) {
        p = W.vmm.alloc(size, &e);
      } else {
        e = hip.Malloc(&p, size);
        if (e == 0) W.plain_allocs[(uint64_t)p] = size;
      }
      uint64_t r = (uint64_t)p;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_FREE: {
      uint64_t p;
      memcpy(&p, body, 8);
      hipError_t e;
      if (W.vmm.enabled) {
        e = W.vmm.free_((void*)p);
      } else {
        e = hip.Free((void*)p);
        W.plain_allocs.erase(p);
      }
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "hipFree");
      break;
    }
    case OP_MEMCPY_H2D: {
      auto* m = reinterpret_cast<MemcpyBody*>(body);
      m->stream = xl(m->stream);
      const void* src;
      if (c->flags & F_INLINE_DATA)
        src = body + sizeof(MemcpyBody);
      else
        src = W.arena + (m->arena_off % ARENA_BYTES);
      hipError_t e;
      if (c->flags & F_INLINE_DATA) {
        // inline payload lives in the cmd ring: stage it into the private
        // pinned ring so the command pops immediately and the DMA stays
        // fully async (a per-op StreamSynchronize here cost ~1% tok/s on
        // the decode bench).
        uint64_t end_off = 0;
        uint8_t* stage = W.staging.buf
                             ? W.staging.reserve(m->size, &end_off)
                             : nullptr;
        if (stage) {
          memcpy(stage, src, m->size);
          e = hip.MemcpyAsync((void*)m->dst, stage, m->size, 1 /*H2D*/,
                              (hipStream_t)m->stream);
          if (e == 0) W.staging.track((hipStream_t)m->stream, end_off);
        } else {
          e = hip.MemcpyAsync((void*)m->dst, src, m->size, 1 /*H2D*/,
                              (hipStream_t)m->stream);
          if (e == 0) e = hip.StreamSynchronize((hipStream_t)m->stream);
        }
      } else {
        e = hip.MemcpyAsync((void*)m->dst, src, m->size, 1,
                            (hipStream_t)m->stream);
        track_arena_use(m->arena_off + m->size, (hipStream_t)m->stream);
      }
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "async-op");
      break;
    }
    case OP_MEMCPY_D2H: {
      auto* m = reinterpret_cast<MemcpyBody*>(body);
      m->stream = xl(m->stream);
      if (W.tcp_mode) {
        // cross-node: data travels inside the reply body (the client has
        // no shared arena); client chunks large reads to fit the ring
        static std::vector<uint8_t> tmp;
        tmp.resize(m->size);
        hipError_t e = hip.MemcpyAsync(tmp.data(), (const void*)m->src,
                                       m->size, 2, (hipStream_t)m->stream);
        if (e == 0) e = hip.StreamSynchronize((hipStream_t)m->stream);
        reply(c->seq, e, tmp.data(), e == 0 ? (uint32_t)m->size : 0);
        break;
      }
      void* dst = W.arena + (m->arena_off % ARENA_BYTES);
      hipError_t e = hip.MemcpyAsync(dst, (const void*)m->src, m->size,
                                     2 /*D2H*/, (hipStream_t)m->stream);
      if (e == 0) e = hip.StreamSynchronize((hipStream_t)m->stream);
      reply(c->seq, e, nullptr, 0);  // data is in the arena now
      break;
    }
    case OP_MEMCPY_D2D: {
      auto* m = reinterpret_cast<MemcpyBody*>(body);
      m->stream = xl(m->stream);
      hipError_t e = hip.MemcpyAsync((void*)m->dst, (const void*)m->src,
                                     m->size, 3, (hipStream_t)m->stream);
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "async-op");
      break;
    }
    case OP_MEMSET: {
      auto* m = reinterpret_cast<MemcpyBody*>(body);
      m->stream = xl(m->stream);
      hipError_t e = hip.MemsetD8Async((void*)m->dst, (unsigned char)m->kind,
                                       m->size, (hipStream_t)m->stream);
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "async-op");
      break;
    }
    case OP_LAUNCH: {
      auto* l = reinterpret_cast<LaunchBody*>(body);
      l->func = xl(l->func);
      l->stream = xl(l->stream);
      void* kernarg = body + sizeof(LaunchBody);
      size_t sz = l->kernarg_size;
      hipError_t e;
      static int use_params = getenv("TF_WORKER_LAUNCH_PARAMS") ? 1 : 0;
      const tfrpc::KernelSig* sig = nullptr;
      if (use_params) {
        auto fit = W.funcs.find(l->func);
        if (fit != W.funcs.end()) {
          auto mit = W.sigs.find(fit->second.image_id);
          if (mit != W.sigs.end()) {
            auto sit = mit->second.find(fit->second.name);
            if (sit != mit->second.end()) sig = &sit->second;
          }
        }
      }
      if (sig && !sig->args.empty()) {
        // per-arg params path: lets ROCr pack kernargs at enqueue instead
        // of recording a kernarg staging copy as a second graph node
        void* params[64];
        size_t n = sig->args.size() < 64 ? sig->args.size() : 64;
        for (size_t i = 0; i < n; ++i)
          params[i] = (uint8_t*)kernarg + sig->args[i].offset;
        e = hip.ModuleLaunchKernel(
            (hipFunction_t)l->func, l->grid[0], l->grid[1], l->grid[2],
            l->block[0], l->block[1], l->block[2], l->shmem,
            (hipStream_t)l->stream, params, nullptr);
      } else {
        void* extra[] = {HIP_LAUNCH_PARAM_BUFFER_POINTER, kernarg,
                         HIP_LAUNCH_PARAM_BUFFER_SIZE, &sz,
                         HIP_LAUNCH_PARAM_END};
        e = hip.ModuleLaunchKernel(
            (hipFunction_t)l->func, l->grid[0], l->grid[1], l->grid[2],
            l->block[0], l->block[1], l->block[2], l->shmem,
            (hipStream_t)l->stream, nullptr, extra);
      }
      if (W.verbose)
        fprintf(stderr,
                "[worker] launch fn=%llx grid=%u,%u,%u block=%u,%u,%u "
                "kernarg=%u stream=%llx -> %d\n",
                (unsigned long long)l->func, l->grid[0], l->grid[1],
                l->grid[2], l->block[0], l->block[1], l->block[2],
                l->kernarg_size, (unsigned long long)l->stream, e);
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "ModuleLaunchKernel");
      break;
    }
    case OP_STREAM_CREATE: {
      uint32_t flags;
      int prio;
      memcpy(&flags, body, 4);
      memcpy(&prio, body + 4, 4);
      hipStream_t st = nullptr;
      hipError_t e = hip.StreamCreateWithPriority(&st, flags, prio);
      if (e == 0) W.streams[(uint64_t)st] = StreamRec{flags, prio};
      uint64_t r = (uint64_t)st;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_STREAM_DESTROY: {
      uint64_t st;
      memcpy(&st, body, 8);
      st = xl(st);
      W.streams.erase(st);
      set_sticky(hip.StreamDestroy((hipStream_t)st), "StreamDestroy");
      break;
    }
    case OP_STREAM_SYNC: {
      uint64_t st;
      memcpy(&st, body, 8);
      st = xl(st);
      hipError_t e = hip.StreamSynchronize((hipStream_t)st);
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_STREAM_QUERY: {
      uint64_t st;
      memcpy(&st, body, 8);
      st = xl(st);
      reply(c->seq, hip.StreamQuery((hipStream_t)st), nullptr, 0);
      break;
    }
    case OP_EVENT_CREATE: {
      uint32_t flags;
      memcpy(&flags, body, 4);
      hipEvent_t ev = nullptr;
      hipError_t e = hip.EventCreateWithFlags(&ev, flags);
      if (e == 0) W.events[(uint64_t)ev] = flags;
      uint64_t r = (uint64_t)ev;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_EVENT_RECORD: {
      uint64_t ev, st;
      memcpy(&ev, body, 8);
      memcpy(&st, body + 8, 8);
      ev = xl(ev);
      st = xl(st);
      set_sticky(hip.EventRecord((hipEvent_t)ev, (hipStream_t)st), "EventRecord");
      break;
    }
    case OP_EVENT_SYNC: {
      uint64_t ev;
      memcpy(&ev, body, 8);
      ev = xl(ev);
      reply(c->seq, hip.EventSynchronize((hipEvent_t)ev), nullptr, 0);
      break;
    }
    case OP_EVENT_QUERY: {
      uint64_t ev;
      memcpy(&ev, body, 8);
      ev = xl(ev);
      reply(c->seq, hip.EventQuery((hipEvent_t)ev), nullptr, 0);
      break;
    }
    case OP_EVENT_ELAPSED: {
      uint64_t e0, e1;
      memcpy(&e0, body, 8);
      memcpy(&e1, body + 8, 8);
      e0 = xl(e0);
      e1 = xl(e1);
      float ms = 0;
      hipError_t e = hip.EventElapsedTime(&ms, (hipEvent_t)e0, (hipEvent_t)e1);
      reply(c->seq, e, &ms, 4);
      break;
    }
    case OP_EVENT_DESTROY: {
      uint64_t ev;
      memcpy(&ev, body, 8);
      ev = xl(ev);
      W.events.erase(ev);
      set_sticky(hip.EventDestroy((hipEvent_t)ev), "EventDestroy");
      break;
    }
    case OP_DEVICE_SYNC: {
      reply(c->seq, hip.DeviceSynchronize(), nullptr, 0);
      break;
    }
    case OP_LOAD_MODULE: {
      struct B {
        uint64_t image_id;
        uint64_t size;
        uint64_t arena_off;
      } b;
      memcpy(&b, body, sizeof b);
      const void* img = W.arena + (b.arena_off % ARENA_BYTES);
      hipModule_t mod = nullptr;
      hipError_t e = 0;
      auto it = W.modules.find(b.image_id);
      if (it != W.modules.end()) {
        mod = it->second;
      } else {
        e = hip.ModuleLoadData(&mod, img);
        if (e == 0) {
          W.modules[b.image_id] = mod;
          W.module_image[(uint64_t)mod] = b.image_id;
          // snapshot needs the bytes after the arena chunk is recycled
          W.images[b.image_id].assign((const uint8_t*)img,
                                      (const uint8_t*)img + b.size);
          std::string err;
          auto& m = W.sigs[b.image_id];
          if (!tfrpc::parse_kernel_signatures(img, b.size, &m, &err) &&
              W.verbose)
            fprintf(stderr, "[worker] sig parse %llu: %s\n",
                    (unsigned long long)b.image_id, err.c_str());
        }
      }
      // arena chunk is consumed synchronously by ModuleLoadData
      tfrpc::at(&W.hdr->arena_freed)
          ->store(b.arena_off + b.size, std::memory_order_release);
      uint64_t r = (uint64_t)mod;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_GET_FUNCTION: {
      uint64_t image_id, mod;
      memcpy(&image_id, body, 8);
      memcpy(&mod, body + 8, 8);
      mod = xl(mod);
      const char* name = (const char*)(body + 16);
      hipFunction_t fn = nullptr;
      hipError_t e = hip.ModuleGetFunction(&fn, (hipModule_t)mod, name);
      if (e == 0) W.funcs[(uint64_t)fn] = FuncRec{image_id, name};
      // reply: {func u64, kernarg_size u32, explicit u32, nargs u32,
      //         {size,offset} x nargs}
      std::vector<uint8_t> out(20);
      tfrpc::KernelSig* sig = nullptr;
      auto mit = W.sigs.find(image_id);
      if (mit != W.sigs.end()) {
        auto sit = mit->second.find(name);
        if (sit != mit->second.end()) sig = &sit->second;
      }
      uint64_t f = (uint64_t)fn;
      uint32_t ks = sig ? sig->kernarg_segment_size : 0;
      uint32_t ex = sig ? sig->explicit_bytes : 0;
      uint32_t na = sig ? (uint32_t)sig->args.size() : 0;
      memcpy(out.data(), &f, 8);
      memcpy(out.data() + 8, &ks, 4);
      memcpy(out.data() + 12, &ex, 4);
      memcpy(out.data() + 16, &na, 4);
      if (sig)
        for (auto& a : sig->args) {
          uint8_t rec[8];
          memcpy(rec, &a.size, 4);
          memcpy(rec + 4, &a.offset, 4);
          out.insert(out.end(), rec, rec + 8);
        }
      if (e == 0 && sig == nullptr) e = 98;  // hipErrorInvalidDeviceFunction
      reply(c->seq, e, out.data(), (uint32_t)out.size());
      break;
    }
    case OP_MEM_GET_INFO: {
      size_t fr = 0, tot = 0;
      hipError_t e = hip.MemGetInfo(&fr, &tot);
      uint64_t r[2] = {fr, tot};
      reply(c->seq, e, r, 16);
      break;
    }
    case OP_CAN_ACCESS_PEER: {
      int dev, peer, v = 0;
      memcpy(&dev, body, 4);
      memcpy(&peer, body + 4, 4);
      hipError_t e = hip.DeviceCanAccessPeer(&v, dev, peer);
      reply(c->seq, e, &v, 4);
      break;
    }
    case OP_STREAM_WAIT_EVENT: {
      uint64_t st, ev;
      uint32_t flags;
      memcpy(&st, body, 8);
      memcpy(&ev, body + 8, 8);
      memcpy(&flags, body + 16, 4);
      st = xl(st);
      ev = xl(ev);
      set_sticky(hip.StreamWaitEvent((hipStream_t)st, (hipEvent_t)ev, flags), "StreamWaitEvent");
      break;
    }
    case OP_BEGIN_CAPTURE: {
      uint64_t st;
      uint32_t mode;
      memcpy(&st, body, 8);
      memcpy(&mode, body + 8, 4);
      st = xl(st);
      hipError_t e = hip.StreamBeginCapture
                         ? hip.StreamBeginCapture((hipStream_t)st, (int)mode)
                         : 801;
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_END_CAPTURE: {
      uint64_t st;
      memcpy(&st, body, 8);
      st = xl(st);
      void* graph = nullptr;
      hipError_t e = hip.StreamEndCapture
                         ? hip.StreamEndCapture((hipStream_t)st, &graph)
                         : 801;
      uint64_t r = (uint64_t)graph;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_GRAPH_INSTANTIATE: {
      uint64_t graph;
      uint64_t flags;
      memcpy(&graph, body, 8);
      memcpy(&flags, body + 8, 8);
      void* ge = nullptr;
      hipError_t e = hip.GraphInstantiateWithFlags
                         ? hip.GraphInstantiateWithFlags(&ge, (void*)graph,
                                                         flags)
                         : 801;
      uint64_t r = (uint64_t)ge;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_GRAPH_LAUNCH: {
      uint64_t ge, st;
      memcpy(&ge, body, 8);
      memcpy(&st, body + 8, 8);
      st = xl(st);
      set_sticky(hip.GraphLaunch ? hip.GraphLaunch((void*)ge, (hipStream_t)st)
                                 : 801,
                 "GraphLaunch");
      break;
    }
    case OP_GRAPH_DESTROY: {
      uint64_t g;
      memcpy(&g, body, 8);
      if (hip.GraphDestroy) set_sticky(hip.GraphDestroy((void*)g),
                                       "GraphDestroy");
      break;
    }
    case OP_GRAPH_EXEC_DESTROY: {
      uint64_t g;
      memcpy(&g, body, 8);
      if (hip.GraphExecDestroy) set_sticky(hip.GraphExecDestroy((void*)g),
                                           "GraphExecDestroy");
      break;
    }
    case OP_GRAPH_GET_NODES: {
      uint64_t graph, cap;
      memcpy(&graph, body, 8);
      memcpy(&cap, body + 8, 8);
      size_t count = 0;
      hipError_t e = hip.GraphGetNodes
                         ? hip.GraphGetNodes((void*)graph, nullptr, &count)
                         : 801;
      std::vector<uint8_t> outb(8);
      uint64_t cnt = count;
      memcpy(outb.data(), &cnt, 8);
      if (e == 0 && cap > 0 && count > 0) {
        size_t n = count < cap ? count : cap;
        std::vector<void*> nodes(n);
        size_t got = n;
        e = hip.GraphGetNodes((void*)graph, nodes.data(), &got);
        outb.resize(8 + got * 8);
        memcpy(outb.data() + 8, nodes.data(), got * 8);
      }
      reply(c->seq, e, outb.data(), (uint32_t)outb.size());
      break;
    }
    case OP_GRAPH_NODE_TYPES: {
      uint64_t graph;
      memcpy(&graph, body, 8);
      uint64_t out[17] = {0};
      hipError_t e = 801;
      if (hip.GraphGetNodes && hip.GraphNodeGetType) {
        size_t count = 0;
        e = hip.GraphGetNodes((void*)graph, nullptr, &count);
        if (e == 0 && count) {
          std::vector<void*> nodes(count);
          size_t got = count;
          e = hip.GraphGetNodes((void*)graph, nodes.data(), &got);
          out[0] = got;
          for (size_t i = 0; i < got && e == 0; ++i) {
            int t = 15;
            hip.GraphNodeGetType(nodes[i], &t);
            if (t < 0 || t > 15) t = 15;
            out[1 + t]++;
          }
        }
      }
      reply(c->seq, e, out, sizeof out);
      break;
    }
    case OP_GRAPH_KERNEL_HISTO: {
      struct KNP {  // hipKernelNodeParams (ROCm 7 layout)
        uint32_t blockDim[3];
        uint32_t _pad0;
        void** extra;
        void* func;
        uint32_t gridDim[3];
        uint32_t _pad1;
        void** kernelParams;
        unsigned int sharedMemBytes;
        uint32_t _pad2;
      };
      uint64_t graph;
      memcpy(&graph, body, 8);
      std::string text;
      hipError_t e = 801;
      if (hip.GraphGetNodes && hip.GraphNodeGetType &&
          hip.GraphKernelNodeGetParams) {
        size_t count = 0;
        e = hip.GraphGetNodes((void*)graph, nullptr, &count);
        std::vector<void*> nodes(count);
        if (e == 0 && count) {
          size_t got = count;
          e = hip.GraphGetNodes((void*)graph, nodes.data(), &got);
          std::map<std::string, int> histo;
          for (size_t i = 0; i < got; ++i) {
            int t = -1;
            hip.GraphNodeGetType(nodes[i], &t);
            if (t != 0) {
              histo["<node type " + std::to_string(t) + ">"]++;
              continue;
            }
            KNP p{};
            if (hip.GraphKernelNodeGetParams(nodes[i], &p) != 0) {
              histo["<params failed>"]++;
              continue;
            }
            auto fit = W.funcs.find((uint64_t)p.func);
            char buf[160];
            const char* nm = fit != W.funcs.end()
                                 ? fit->second.name.c_str()
                                 : "<unknown func>";
            snprintf(buf, sizeof buf, "%.100s g=%u,%u,%u b=%u,%u,%u", nm,
                     p.gridDim[0], p.gridDim[1], p.gridDim[2], p.blockDim[0],
                     p.blockDim[1], p.blockDim[2]);
            histo[buf]++;
          }
          std::vector<std::pair<int, std::string>> top;
          for (auto& [k, v] : histo) top.push_back({v, k});
          std::sort(top.rbegin(), top.rend());
          for (size_t i = 0; i < top.size() && i < 40; ++i)
            text += std::to_string(top[i].first) + "x " + top[i].second +
                    "\n";
          if (text.size() > 60000) text.resize(60000);
        }
      }
      reply(c->seq, e, text.data(), (uint32_t)text.size());
      break;
    }
    // ---- client-forwarded VMM surface (PyTorch expandable_segments /
    // vLLM-class allocators). VAs and handles live in THIS process;
    // ranges mapped here are ordinary device pointers for every other op.
    case OP_VMM_RESERVE: {
      struct B { uint64_t size, align, hint, flags; } b;
      memcpy(&b, body, sizeof b);
      void* p = nullptr;
      hipError_t e = hip.MemAddressReserve
          ? hip.MemAddressReserve(&p, b.size, b.align, (void*)b.hint,
                                  b.flags)
          : 801;
      uint64_t r = (uint64_t)p;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_VMM_ADDR_FREE: {
      struct B { uint64_t ptr, size; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e = hip.MemAddressFree
          ? hip.MemAddressFree((void*)b.ptr, b.size) : 801;
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "hipMemAddressFree");
      break;
    }
    case OP_VMM_CREATE: {
      struct B { uint64_t size, flags; uint8_t prop[48]; } b;
      memcpy(&b, body, sizeof b);
      void* h2 = nullptr;
      hipError_t e = hip.MemCreate
          ? hip.MemCreate(&h2, b.size, b.prop, b.flags) : 801;
      uint64_t r = (uint64_t)h2;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_VMM_RELEASE: {
      uint64_t h2;
      memcpy(&h2, body, 8);
      hipError_t e = hip.MemRelease ? hip.MemRelease((void*)h2) : 801;
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "hipMemRelease");
      break;
    }
    case OP_VMM_MAP: {
      struct B { uint64_t va, size, off, handle, flags; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e = hip.MemMap
          ? hip.MemMap((void*)b.va, b.size, b.off, (void*)b.handle,
                       b.flags)
          : 801;
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_VMM_UNMAP: {
      struct B { uint64_t va, size; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e = hip.MemUnmap
          ? hip.MemUnmap((void*)b.va, b.size) : 801;
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_VMM_SET_ACCESS: {
      struct B { uint64_t va, size; uint32_t count, pad; } b;
      memcpy(&b, body, sizeof b);
      // desc array follows: count * {int type, int id, int flags}
      hipError_t e = hip.MemSetAccess
          ? hip.MemSetAccess((void*)b.va, b.size, body + sizeof b,
                             b.count)
          : 801;
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_VMM_GRANULARITY: {
      struct B { uint64_t opt; uint8_t prop[48]; } b;
      memcpy(&b, body, sizeof b);
      size_t g2 = 0;
      hipError_t e = hip.MemGetAllocationGranularity
          ? hip.MemGetAllocationGranularity(&g2, b.prop, (int)b.opt)
          : 801;
      uint64_t r = g2;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_MALLOC_ASYNC: {
      struct B { uint64_t size, stream; } b;
      memcpy(&b, body, sizeof b);
      void* p = nullptr;
      hipError_t e;
      if (hip.MallocAsync) {
        e = hip.MallocAsync(&p, b.size, (hipStream_t)xl(b.stream));
      } else if (W.vmm.enabled) {
        p = W.vmm.alloc(b.size, &e);
      } else {
        e = hip.Malloc(&p, b.size);
      }
      uint64_t r = (uint64_t)p;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_FREE_ASYNC: {
      struct B { uint64_t ptr, stream; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e;
      if (hip.FreeAsync) {
        e = hip.FreeAsync((void*)b.ptr, (hipStream_t)xl(b.stream));
      } else if (W.vmm.enabled) {
        e = W.vmm.free_((void*)b.ptr);
      } else {
        e = hip.Free((void*)b.ptr);
      }
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "hipFreeAsync");
      break;
    }
    case OP_MEMPOOL_DEFAULT: {
      uint32_t dev;
      memcpy(&dev, body, 4);
      void* pool = nullptr;
      hipError_t e = hip.DeviceGetDefaultMemPool
          ? hip.DeviceGetDefaultMemPool(&pool, (int)dev) : 801;
      uint64_t r = (uint64_t)pool;
      reply(c->seq, e, &r, 8);
      break;
    }
    case OP_MEMPOOL_SET_ATTR: {
      struct B { uint64_t pool, attr, value; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e = hip.MemPoolSetAttribute
          ? hip.MemPoolSetAttribute((void*)b.pool, (int)b.attr, &b.value)
          : 801;
      reply(c->seq, e, nullptr, 0);
      break;
    }
    case OP_MEMPOOL_GET_ATTR: {
      struct B { uint64_t pool, attr; } b;
      memcpy(&b, body, sizeof b);
      uint64_t value = 0;
      hipError_t e = hip.MemPoolGetAttribute
          ? hip.MemPoolGetAttribute((void*)b.pool, (int)b.attr, &value)
          : 801;
      reply(c->seq, e, &value, 8);
      break;
    }
    case OP_MEMPOOL_TRIM: {
      struct B { uint64_t pool, keep; } b;
      memcpy(&b, body, sizeof b);
      hipError_t e = hip.MemPoolTrimTo
          ? hip.MemPoolTrimTo((void*)b.pool, b.keep) : 801;
      if (c->flags & F_WANT_REPLY) reply(c->seq, e, nullptr, 0);
      else set_sticky(e, "hipMemPoolTrimTo");
      break;
    }
    case OP_GET_GLOBAL: {
      uint64_t module;
      memcpy(&module, body, 8);
      const char* name = (const char*)body + 8;
      void* dptr = nullptr;
      size_t bytes = 0;
      hipError_t e = hip.ModuleGetGlobal
          ? hip.ModuleGetGlobal(&dptr, &bytes, (void*)xl(module), name)
          : 801;
      struct {
        uint64_t dptr, bytes;
      } r{(uint64_t)dptr, (uint64_t)bytes};
      reply(c->seq, e, &r, sizeof r);
      break;
    }
    case OP_SHUTDOWN:
      W.hdr->shutdown = 1;
      reply(c->seq, 0, nullptr, 0);
      break;
    default:
      if (c->flags & F_WANT_REPLY) reply(c->seq, 801 /*NotSupported*/, nullptr, 0);
      break;
  }
}


// --------------------------------------------------- snapshot / restore
// Device-level snapshot: quiesce, dump every allocation (VA + contents),
// module images, function/stream/event registries. Restore rebuilds the
// identical VA layout (VMM heap at the same base) and loads a handle
// translation table for pre-snapshot stream/event/module/function values.
// Triggered by SIGUSR1 with TF_WORKER_SNAPSHOT_PATH set; a worker started
// with TF_WORKER_RESTORE_PATH resumes serving the SAME client segment —
// live migration of a vGPU under a live client.

constexpr uint64_t SNAP_MAGIC = 0x50414e53465431ull;  // "TFSNAP1"

bool write_all(FILE* f, const void* p, size_t n) {
  return fwrite(p, 1, n, f) == n;
}
bool read_all(FILE* f, void* p, size_t n) {
  return fread(p, 1, n, f) == n;
}

int do_snapshot(const char* path) {
  if (!W.vmm.enabled) {
    fprintf(stderr, "[worker] snapshot requires the VMM heap (TF_WORKER_NO_VMM unset)\n");
    return -1;
  }
  hipError_t e = hip.DeviceSynchronize();
  if (e != 0) {
    fprintf(stderr, "[worker] snapshot: device sync failed %d\n", e);
    return -1;
  }
  FILE* f = fopen(path, "wb");
  if (!f) return -1;
  uint64_t magic = SNAP_MAGIC;
  write_all(f, &magic, 8);
  write_all(f, &W.vmm.base, 8);
  uint64_t heap = W.vmm.heap_bytes;
  write_all(f, &heap, 8);
  int32_t dev = W.device;
  write_all(f, &dev, 4);

  uint32_t n_allocs = (uint32_t)W.vmm.mapped.size();
  write_all(f, &n_allocs, 4);
  std::vector<uint8_t> staging;
  for (auto& [va, r] : W.vmm.mapped) {
    uint64_t v = va, req = r.req_bytes;
    write_all(f, &v, 8);
    write_all(f, &req, 8);
    staging.resize(r.req_bytes);
    e = hip.MemcpyAsync(staging.data(), (void*)va, r.req_bytes, 2 /*D2H*/,
                        nullptr);
    if (e == 0) e = hip.StreamSynchronize(nullptr);
    if (e != 0) {
      fclose(f);
      fprintf(stderr, "[worker] snapshot: D2H %llx failed %d\n",
              (unsigned long long)va, e);
      return -1;
    }
    write_all(f, staging.data(), r.req_bytes);
  }

  uint32_t n_mod = (uint32_t)W.module_image.size();
  write_all(f, &n_mod, 4);
  for (auto& [handle, image_id] : W.module_image) {
    auto& img = W.images[image_id];
    uint64_t h = handle, id = image_id, sz = img.size();
    write_all(f, &h, 8);
    write_all(f, &id, 8);
    write_all(f, &sz, 8);
    write_all(f, img.data(), img.size());
  }

  uint32_t n_fn = (uint32_t)W.funcs.size();
  write_all(f, &n_fn, 4);
  for (auto& [handle, rec] : W.funcs) {
    uint64_t h = handle, id = rec.image_id;
    uint32_t nl = (uint32_t)rec.name.size();
    write_all(f, &h, 8);
    write_all(f, &id, 8);
    write_all(f, &nl, 4);
    write_all(f, rec.name.data(), nl);
  }

  uint32_t n_st = (uint32_t)W.streams.size();
  write_all(f, &n_st, 4);
  for (auto& [handle, rec] : W.streams) {
    uint64_t h = handle;
    int32_t prio = rec.prio;
    write_all(f, &h, 8);
    write_all(f, &rec.flags, 4);
    write_all(f, &prio, 4);
  }

  uint32_t n_ev = (uint32_t)W.events.size();
  write_all(f, &n_ev, 4);
  for (auto& [handle, flags] : W.events) {
    uint64_t h = handle;
    write_all(f, &h, 8);
    write_all(f, &flags, 4);
  }
  bool ok = fflush(f) == 0;
  fclose(f);
  fprintf(stderr, "[worker] snapshot -> %s (%u allocs, %u modules, %u fns, "
          "%u streams, %u events)\n", path, n_allocs, n_mod, n_fn, n_st,
          n_ev);
  return ok ? 0 : -1;
}

int do_restore(const char* path) {
  FILE* f = fopen(path, "rb");
  if (!f) {
    fprintf(stderr, "[worker] restore: cannot open %s\n", path);
    return -1;
  }
  uint64_t magic = 0, base = 0, heap = 0;
  int32_t dev = 0;
  if (!read_all(f, &magic, 8) || magic != SNAP_MAGIC ||
      !read_all(f, &base, 8) || !read_all(f, &heap, 8) ||
      !read_all(f, &dev, 4)) {
    fclose(f);
    return -1;
  }
  if (!W.vmm.init(W.device, base, heap)) {
    fprintf(stderr, "[worker] restore: cannot re-reserve VA base %llx\n",
            (unsigned long long)base);
    fclose(f);
    return -1;
  }
  uint32_t n_allocs = 0;
  read_all(f, &n_allocs, 4);
  std::vector<uint8_t> staging;
  for (uint32_t i = 0; i < n_allocs; ++i) {
    uint64_t va = 0, req = 0;
    if (!read_all(f, &va, 8) || !read_all(f, &req, 8)) goto fail;
    if (W.vmm.alloc_exact(va, req) != 0) {
      fprintf(stderr, "[worker] restore: map %llx failed\n",
              (unsigned long long)va);
      goto fail;
    }
    staging.resize(req);
    if (!read_all(f, staging.data(), req)) goto fail;
    if (hip.MemcpyAsync((void*)va, staging.data(), req, 1, nullptr) != 0)
      goto fail;
  }
  hip.StreamSynchronize(nullptr);

  {
    uint32_t n_mod = 0;
    read_all(f, &n_mod, 4);
    for (uint32_t i = 0; i < n_mod; ++i) {
      uint64_t old_h = 0, id = 0, sz = 0;
      if (!read_all(f, &old_h, 8) || !read_all(f, &id, 8) ||
          !read_all(f, &sz, 8))
        goto fail;
      auto& img = W.images[id];
      img.resize(sz);
      if (!read_all(f, img.data(), sz)) goto fail;
      hipModule_t mod = nullptr;
      if (hip.ModuleLoadData(&mod, img.data()) != 0) goto fail;
      W.modules[id] = mod;
      W.module_image[(uint64_t)mod] = id;
      W.tr[old_h] = (uint64_t)mod;
      std::string err;
      tfrpc::parse_kernel_signatures(img.data(), sz, &W.sigs[id], &err);
    }
    uint32_t n_fn = 0;
    read_all(f, &n_fn, 4);
    for (uint32_t i = 0; i < n_fn; ++i) {
      uint64_t old_h = 0, id = 0;
      uint32_t nl = 0;
      if (!read_all(f, &old_h, 8) || !read_all(f, &id, 8) ||
          !read_all(f, &nl, 4))
        goto fail;
      std::string name(nl, 0);
      if (!read_all(f, name.data(), nl)) goto fail;
      hipFunction_t fn = nullptr;
      auto mit = W.modules.find(id);
      if (mit == W.modules.end() ||
          hip.ModuleGetFunction(&fn, mit->second, name.c_str()) != 0)
        goto fail;
      W.tr[old_h] = (uint64_t)fn;
      W.funcs[(uint64_t)fn] = FuncRec{id, name};
    }
    uint32_t n_st = 0;
    read_all(f, &n_st, 4);
    for (uint32_t i = 0; i < n_st; ++i) {
      uint64_t old_h = 0;
      uint32_t flags = 0;
      int32_t prio = 0;
      if (!read_all(f, &old_h, 8) || !read_all(f, &flags, 4) ||
          !read_all(f, &prio, 4))
        goto fail;
      hipStream_t st = nullptr;
      if (hip.StreamCreateWithPriority(&st, flags, prio) != 0) goto fail;
      W.tr[old_h] = (uint64_t)st;
      W.streams[(uint64_t)st] = StreamRec{flags, prio};
    }
    uint32_t n_ev = 0;
    read_all(f, &n_ev, 4);
    for (uint32_t i = 0; i < n_ev; ++i) {
      uint64_t old_h = 0;
      uint32_t flags = 0;
      if (!read_all(f, &old_h, 8) || !read_all(f, &flags, 4)) goto fail;
      hipEvent_t ev = nullptr;
      if (hip.EventCreateWithFlags(&ev, flags) != 0) goto fail;
      W.tr[old_h] = (uint64_t)ev;
      W.events[(uint64_t)ev] = flags;
    }
  }
  fclose(f);
  fprintf(stderr, "[worker] restored %u allocs from %s (VA base %llx)\n",
          n_allocs, path, (unsigned long long)base);
  return 0;
fail:
  fclose(f);
  fprintf(stderr, "[worker] restore: truncated/invalid snapshot %s\n", path);
  return -1;
}

void on_sigusr1(int) { g_snapshot_req = 1; }

int serve(tfrpc::Header* hdr) {
  W.hdr = hdr;
  W.cmd = tfrpc::RingView(&hdr->cmd, tfrpc::cmd_buf(hdr),
                          tfrpc::CMD_RING_BYTES);
  W.cpl = tfrpc::RingView(&hdr->cpl, tfrpc::cpl_buf(hdr),
                          tfrpc::CPL_RING_BYTES);
  W.arena = tfrpc::arena(hdr);
  // Register the shared arena once: GPU DMA goes straight to/from these
  // pages (the zero-copy staging path).
  hipError_t e = hip.HostRegister(W.arena, tfrpc::ARENA_BYTES, 0);
  if (e != 0)
    fprintf(stderr, "[worker] hipHostRegister(arena) failed: %d "
                    "(transfers fall back to pageable)\n", e);
  if (!W.staging.buf && !W.staging.init())
    fprintf(stderr, "[worker] staging init failed (inline H2D will sync)\n");
  tfrpc::at(&hdr->worker_ready)->store(1, std::memory_order_release);
  fprintf(stderr, "[worker] serving\n");
  while (!hdr->shutdown) {
    if (g_snapshot_req) {
      g_snapshot_req = 0;
      const char* sp = getenv("TF_WORKER_SNAPSHOT_PATH");
      if (sp && *sp) {
        retire_pending(true);
        if (do_snapshot(sp) == 0) {
          // migration handoff: mark not-ready and exit; the client's
          // reconnect thread re-attaches to whoever binds the socket next
          tfrpc::at(&hdr->worker_ready)->store(0, std::memory_order_release);
          fprintf(stderr, "[worker] exiting for migration\n");
          exit(0);
        }
      }
    }
    size_t len;
    uint8_t* p = W.cmd.try_next(&len);
    if (!p) {
      retire_pending();
      // idle: is the client still alive? (EOF ⇒ tear down; a MIGRATING
      // client keeps its socket open to the successor, so this only
      // fires when the client process is gone)
      if (W.cli_fd >= 0) {
        char b;
        ssize_t n = recv(W.cli_fd, &b, 1, MSG_DONTWAIT);
        if (n == 0) {
          fprintf(stderr, "[worker] client disconnected\n");
          break;
        }
      }
      W.cmd.wait_nonempty();
      continue;
    }
    auto* c = reinterpret_cast<tfrpc::CmdHdr*>(p);
    handle(c, p + sizeof(tfrpc::CmdHdr));
    W.cmd.pop();
    if (!W.pending.empty()) retire_pending();
  }
  retire_pending(true);
  fprintf(stderr, "[worker] shutdown\n");
  return 0;
}

// ---------------------------------------------------- TCP transport
// Cross-node session: the worker owns a PRIVATE segment; a reader thread
// writes incoming cmd frames into its cmd ring (+ arena payloads), and a
// writer thread drains the cpl ring back over the socket. serve() and
// every handler run unchanged.

bool read_full(int fd, void* p, size_t n) {
  uint8_t* b = (uint8_t*)p;
  while (n) {
    ssize_t r = recv(fd, b, n, 0);
    if (r <= 0) return false;
    b += r;
    n -= (size_t)r;
  }
  return true;
}

bool write_full(int fd, const void* p, size_t n) {
  const uint8_t* b = (const uint8_t*)p;
  while (n) {
    ssize_t r = send(fd, b, n, MSG_NOSIGNAL);
    if (r <= 0) return false;
    b += r;
    n -= (size_t)r;
  }
  return true;
}

struct TcpSession {
  int fd;
  tfrpc::Header* hdr;
  std::atomic<bool> dead{false};
};

void* tcp_reader_main(void* arg) {
  auto* s = (TcpSession*)arg;
  tfrpc::RingView cmd(&s->hdr->cmd, tfrpc::cmd_buf(s->hdr),
                      tfrpc::CMD_RING_BYTES);
  uint8_t* arena = tfrpc::arena(s->hdr);
  std::vector<uint8_t> rec;
  for (;;) {
    tfrpc::FrameHdr fh;
    if (!read_full(s->fd, &fh, sizeof fh)) break;
    if (fh.kind != 0 || fh.rec_len > tfrpc::CMD_RING_BYTES / 2) break;
    rec.resize(fh.rec_len);
    if (!read_full(s->fd, rec.data(), fh.rec_len)) break;
    if (fh.extra_len) {
      size_t off = fh.arena_off % tfrpc::ARENA_BYTES;
      size_t first = tfrpc::ARENA_BYTES - off;
      if (fh.extra_len <= first) {
        if (!read_full(s->fd, arena + off, fh.extra_len)) break;
      } else {
        if (!read_full(s->fd, arena + off, first)) break;
        if (!read_full(s->fd, arena, fh.extra_len - first)) break;
      }
    }
    uint8_t* p;
    while (!(p = cmd.try_reserve(fh.rec_len))) usleep(50);
    memcpy(p, rec.data(), fh.rec_len);
    cmd.commit();
    cmd.wake_consumer();
  }
  s->dead.store(true);
  // unblock serve(): mark shutdown so the loop exits
  s->hdr->shutdown = 1;
  tfrpc::at(&s->hdr->cmd.futex_nonempty)->store(1);
  tfrpc::futex_wake(&s->hdr->cmd.futex_nonempty);
  return nullptr;
}

void* tcp_writer_main(void* arg) {
  auto* s = (TcpSession*)arg;
  tfrpc::RingView cpl(&s->hdr->cpl, tfrpc::cpl_buf(s->hdr),
                      tfrpc::CPL_RING_BYTES);
  while (!s->dead.load()) {
    size_t len;
    uint8_t* p = cpl.try_next(&len);
    if (!p) {
      // park on the cpl futex the worker's reply() wakes
      tfrpc::at(&s->hdr->futex_cpl)->exchange(0);
      p = cpl.try_next(&len);
      if (!p) {
        tfrpc::futex_wait(&s->hdr->futex_cpl, 0, 100);
        continue;
      }
    }
    tfrpc::FrameHdr fh{1, (uint32_t)len, 0, 0, 0};
    if (!write_full(s->fd, &fh, sizeof fh) || !write_full(s->fd, p, len))
      break;
    cpl.pop();
  }
  s->dead.store(true);
  return nullptr;
}

int serve_tcp(int cli) {
  uint32_t magic = 0;
  if (!read_full(cli, &magic, 4) || magic != tfrpc::TCP_MAGIC) {
    close(cli);
    return -1;
  }
  write_full(cli, &magic, 4);
  void* seg = mmap(nullptr, tfrpc::SEG_BYTES, PROT_READ | PROT_WRITE,
                   MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
  if (seg == MAP_FAILED) {
    close(cli);
    return -1;
  }
  auto* hdr = reinterpret_cast<tfrpc::Header*>(seg);
  memset(hdr, 0, tfrpc::HDR_BYTES);
  hdr->magic = tfrpc::MAGIC;
  hdr->version = tfrpc::VERSION;
  hdr->total_bytes = tfrpc::SEG_BYTES;
  TcpSession sess{cli, hdr};
  pthread_t rt, wt;
  pthread_create(&rt, nullptr, tcp_reader_main, &sess);
  pthread_create(&wt, nullptr, tcp_writer_main, &sess);
  W.tcp_mode = true;
  W.cli_fd = -1;  // EOF detection is the reader's job in TCP mode
  fprintf(stderr, "[worker] tcp session started\n");
  serve(hdr);
  sess.dead.store(true);
  shutdown(cli, SHUT_RDWR);
  pthread_join(rt, nullptr);
  // wake a parked writer
  tfrpc::at(&hdr->futex_cpl)->store(1);
  tfrpc::futex_wake(&hdr->futex_cpl);
  pthread_join(wt, nullptr);
  close(cli);
  munmap(seg, tfrpc::SEG_BYTES);
  W = Worker{};
  return 0;
}

int recv_fd(int sock) {
  char buf[1];
  iovec iov{buf, 1};
  char ctrl[CMSG_SPACE(sizeof(int))];
  msghdr msg{};
  msg.msg_iov = &iov;
  msg.msg_iovlen = 1;
  msg.msg_control = ctrl;
  msg.msg_controllen = sizeof ctrl;
  if (recvmsg(sock, &msg, 0) <= 0) return -1;
  cmsghdr* c = CMSG_FIRSTHDR(&msg);
  if (!c || c->cmsg_type != SCM_RIGHTS) return -1;
  int fd;
  memcpy(&fd, CMSG_DATA(c), sizeof fd);
  return fd;
}

}  // namespace

int main(int argc, char** argv) {
  const char* sock_path = argc > 1 ? argv[1] : getenv("TF_WORKER_SOCKET");
  if (!sock_path && !getenv("TF_WORKER_TCP_PORT")) {
    fprintf(stderr,
            "usage: tf_vgpu_worker <socket-path>  (or TF_WORKER_TCP_PORT=N)\n");
    return 2;
  }
  if (!hip.load()) {
    fprintf(stderr, "[worker] cannot load libamdhip64\n");
    return 3;
  }
  int n = 0;
  if (hip.GetDeviceCount(&n) != 0 || n == 0) {
    fprintf(stderr, "[worker] no GPU visible\n");
    return 3;
  }
  hip.SetDevice(0);
  W.device = 0;
  signal(SIGUSR1, on_sigusr1);

  const char* restore_path = getenv("TF_WORKER_RESTORE_PATH");
  bool want_vmm = !getenv("TF_WORKER_NO_VMM");
  if (want_vmm && !restore_path) {
    // fresh start: VA-stable heap (restore re-reserves from the snapshot).
    // VA only — physical memory commits per mapping, so size it well past
    // HBM (fp32-init peaks of a 70B model exceed 192 GB before the bf16
    // cast frees it).
    size_t heap = 512ull << 30;
    const char* hb = getenv("TF_WORKER_VMM_BYTES");
    if (hb) heap = strtoull(hb, nullptr, 10);
    if (!W.vmm.init(0, VMM_BASE_HINT, heap))
      fprintf(stderr, "[worker] VMM unavailable — plain hipMalloc, "
                      "snapshot/migration disabled\n");
  }

  const char* tcp_port = getenv("TF_WORKER_TCP_PORT");
  if (tcp_port && *tcp_port) {
    // cross-node mode: accept GPU-over-IP sessions on TCP
    int srv = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
    sockaddr_in a{};
    a.sin_family = AF_INET;
    a.sin_addr.s_addr = INADDR_ANY;
    a.sin_port = htons((uint16_t)atoi(tcp_port));
    if (bind(srv, (sockaddr*)&a, sizeof a) != 0 || listen(srv, 4) != 0) {
      fprintf(stderr, "[worker] tcp bind :%s: %s\n", tcp_port,
              strerror(errno));
      return 2;
    }
    fprintf(stderr, "[worker] listening on tcp :%s (%d devices)\n",
            tcp_port, n);
    for (;;) {
      int cli = accept(srv, nullptr, nullptr);
      if (cli < 0) continue;
      int nd = 1;
      setsockopt(cli, IPPROTO_TCP, TCP_NODELAY, &nd, sizeof nd);
      serve_tcp(cli);
      if (getenv("TF_WORKER_ONESHOT")) break;
    }
    return 0;
  }

  unlink(sock_path);
  int srv = socket(AF_UNIX, SOCK_STREAM, 0);
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, sock_path, sizeof addr.sun_path - 1);
  if (bind(srv, (sockaddr*)&addr, sizeof addr) != 0 || listen(srv, 1) != 0) {
    fprintf(stderr, "[worker] bind %s: %s\n", sock_path, strerror(errno));
    return 2;
  }
  fprintf(stderr, "[worker] listening on %s (%d devices)\n", sock_path, n);
  for (;;) {
    int cli = accept(srv, nullptr, nullptr);
    if (cli < 0) continue;
    int segfd = recv_fd(cli);
    if (segfd < 0) {
      close(cli);
      continue;
    }
    void* seg = mmap(nullptr, tfrpc::SEG_BYTES, PROT_READ | PROT_WRITE,
                     MAP_SHARED, segfd, 0);
    close(segfd);
    if (seg == MAP_FAILED) {
      close(cli);
      continue;
    }
    auto* hdr = reinterpret_cast<tfrpc::Header*>(seg);
    if (hdr->magic != tfrpc::MAGIC || hdr->version != tfrpc::VERSION) {
      munmap(seg, tfrpc::SEG_BYTES);
      close(cli);
      continue;
    }
    W.cli_fd = cli;
    if (restore_path && *restore_path) {
      if (do_restore(restore_path) != 0) {
        fprintf(stderr, "[worker] restore failed — refusing to serve\n");
        munmap(seg, tfrpc::SEG_BYTES);
        close(cli);
        return 4;
      }
      restore_path = nullptr;  // one restore per process
    }
    serve(hdr);
    hip.HostUnregister(tfrpc::arena(hdr));
    munmap(seg, tfrpc::SEG_BYTES);
    close(cli);
    W = Worker{};  // reset per-client state (modules leak into HIP ctx; ok)
    if (getenv("TF_WORKER_ONESHOT")) break;
  }
  return 0;
}
