// libtfhip_limiter.so — soft-isolation limiter for MI355X vGPUs.
//
// LD_PRELOADed into workload processes. Interposes the HIP runtime entry
// points (hipMalloc/hipFree/hipMemcpy*/hipLaunchKernel*/hipModuleLaunch*/
// hipGraphLaunch/hipMemset*/hipExtStreamCreate*) via dlsym(RTLD_NEXT) and
// enforces, per vGPU:
//   - a VRAM hard cap (hipMalloc beyond the cap returns hipErrorOutOfMemory,
//     which PyTorch's caching allocator handles by trimming its cache), and
//   - an ERL (elastic rate limit) compute token bucket whose refill rate is
//     PID-controlled by the node hypervisor through shared memory
//     (native/limiter/limiter_shm.h). Consuming is a lock-free CAS on the
//     f64 token word; an empty bucket blocks the calling thread in short
//     sleeps, which throttles kernel-launch throughput and thereby GPU time.
//
// Capability parity with the reference's closed-source libcuda_limiter.so
// (contract: provider/limiter.h:71-106 + soft_limiter_shm.go) — this is a
// fresh HIP/CDNA4 implementation, not a port: throttling happens at launch
// granularity so RCCL collectives in flight are never split,
// and hard isolation uses ROCr CU masks (HSA_CU_MASK / CU-masked streams,
// 256 CUs => 0.39% granularity) instead of SM fractions.
//
// Config (env):
//   TF_SHM_PATH           shm file created by the hypervisor
//   TF_VRAM_LIMIT_BYTES   standalone mode: VRAM cap without a hypervisor
//   TF_UP_LIMIT_PERCENT   standalone mode: compute % target
//   TF_ERL_RATE / TF_ERL_CAPACITY  standalone ERL parameters (tokens/s)
//   TF_TOKENS_PER_LAUNCH / _PER_GRAPH / _PER_MEMCPY  op costs (default 1/8/1)
//   TF_LIMITER_DEBUG=1    stderr diagnostics
//
// Build: hipcc-free — plain g++ -shared -fPIC -O2 -ldl -pthread.

#include <dlfcn.h>
#include <errno.h>
#include <fcntl.h>
#include <pthread.h>
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <time.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <map>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "limiter_shm.h"

// ---------------------------------------------------------------- hip ABI
// Minimal ABI surface; we do not include hip headers (no HIP dependency —
// the limiter must load even in GPU-less CI).
typedef int hipError_t;
static const hipError_t hipSuccess = 0;
static const hipError_t hipErrorOutOfMemory = 2;
typedef void* hipStream_t;
struct dim3u {
  unsigned x, y, z;
};

// ------------------------------------------------------------------ state

namespace {

struct AllocMap {
  // ptr -> bytes for device allocations we admitted.
  std::mutex mu;
  std::unordered_map<const void*, size_t> m;
};

// One oversubscribed (managed-memory) allocation and its current tier.
struct ExpandedRange {
  size_t bytes = 0;
  bool device_resident = false;  // last prefetch target was HBM
  int device = 0;  // owning device at expand_alloc time — the tier thread
                   // never calls hipSetDevice, so its tls_device is always
                   // 0 and must not be used as a prefetch/advise target
  uint64_t last_use = 0;  // LRU epoch (launch count at last touch)
};

struct Limiter {
  TfSharedState* shm = nullptr;  // mapped page (file or private standalone)
  bool standalone = false;
  bool enabled = false;
  bool debug = false;
  double tokens_per_launch = 1.0;
  double tokens_per_graph = 8.0;
  double tokens_per_memcpy = 1.0;
  AllocMap allocs;
  std::atomic<long> blocked_threads{0};

  // ---- VRAM oversubscription (BASELINE config 4) ----
  // Over-cap hipMalloc falls back to HSA managed memory placed in host
  // DRAM (preferred-location CPU, accessed-by GPU): the workload keeps
  // running past the HBM cap; a tier thread promotes ranges into free
  // HBM headroom and demotes them on hypervisor VRAM-pressure signal.
  bool expand_enabled = false;
  uint64_t expand_limit = 0;  // 0 = unlimited host expansion
  std::mutex expand_mu;
  std::map<uint64_t, ExpandedRange> expanded;  // ordered: containment find
  std::atomic<uint64_t> expanded_bytes{0};
  std::atomic<uint64_t> promoted_bytes{0};
  std::atomic<int> tier_thread_running{0};
  // working-set + migration stats (SURVEY §2.4(c) tracking requirement)
  std::atomic<uint64_t> tier_epoch{1};
  std::atomic<uint64_t> demoted_total{0}, promoted_total{0};
  std::atomic<uint64_t> demote_ns{0}, promote_ns{0};

  // ---- per-call latency histograms (SURVEY §5.1 tracing) ----
  // log2-bucketed ns per op class, enabled with TF_LIMITER_TRACE=1:
  // class 0 = kernel launches, 1 = memcpy/memset, 2 = sync/other.
  bool trace = false;
  std::atomic<uint64_t> lat_hist[3][32] = {};

  Limiter();  // all init lives in the constructor: the instance is a Meyers
              // singleton so there is no static-init-order hazard between
              // the ELF constructor and this TU's dynamic initializers.
};

Limiter& G() {
  static Limiter inst;
  return inst;
}
#define g G()

uint64_t now_ns() {
  timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

void record_latency(int cls, uint64_t t0_ns) {
  uint64_t dt = now_ns() - t0_ns;
  int b = 0;
  while ((1ull << (b + 1)) <= dt && b < 31) ++b;
  g.lat_hist[cls][b].fetch_add(1, std::memory_order_relaxed);
}

#define TF_TRACE_CALL(cls, expr)                 \
  ({                                             \
    hipError_t _r;                               \
    if (g.trace) {                               \
      uint64_t _t0 = now_ns();                   \
      _r = (expr);                               \
      record_latency(cls, _t0);                  \
    } else {                                     \
      _r = (expr);                               \
    }                                            \
    _r;                                          \
  })

void dbg(const char* fmt, ...) {
  if (!g.debug) return;
  va_list ap;
  va_start(ap, fmt);
  fprintf(stderr, "[tf-limiter %d] ", getpid());
  vfprintf(stderr, fmt, ap);
  fprintf(stderr, "\n");
  va_end(ap);
}

double env_f(const char* k, double d) {
  const char* v = getenv(k);
  return v ? atof(v) : d;
}

// Thread-local current device (updated by interposed hipSetDevice).
thread_local int tls_device = 0;

void tier_note_launch();  // LRU epoch tick (defined with the tier engine)

void record_latency(int cls, uint64_t t0_ns);  // fwd (uses now_ns)

TfDeviceEntry* cur_dev() {
  if (!g.shm) return nullptr;
  int i = tls_device;
  if (i < 0 || i >= TF_MAX_DEVICES) i = 0;
  TfDeviceEntry* e = &g.shm->dev[i];
  if (!tfshm::at(&e->active)->load(std::memory_order_relaxed)) {
    // Single-device workers always govern through entry 0.
    e = &g.shm->dev[0];
    if (!tfshm::at(&e->active)->load(std::memory_order_relaxed)) return nullptr;
  }
  return e;
}

void heartbeat() {
  if (g.shm)
    tfshm::at(&g.shm->heartbeat_ns)->store(now_ns(), std::memory_order_relaxed);
}

// ------------------------------------------------------------ token bucket

// Refill + consume `cost` tokens; returns seconds to sleep (0 = admitted).
double try_consume(TfDeviceEntry* e, double cost) {
  uint64_t now = now_ns();
  double rate = tfshm::load_double(&e->erl_refill_rate);
  double cap = tfshm::load_double(&e->erl_capacity);
  if (rate <= 0) return 0.0;  // unlimited
  for (int spin = 0; spin < 64; ++spin) {
    uint64_t tok_bits = tfshm::at(&e->erl_tokens)->load(std::memory_order_acquire);
    double tok;
    __builtin_memcpy(&tok, &tok_bits, 8);
    uint64_t last = tfshm::at(&e->erl_last_update_ns)->load(std::memory_order_relaxed);
    double dt = last < now ? double(now - last) * 1e-9 : 0.0;
    double avail = tok + dt * rate;
    if (avail > cap) avail = cap;
    if (avail < cost) {
      double deficit = cost - avail;
      return deficit / rate;  // seconds until enough tokens
    }
    double next = avail - cost;
    uint64_t next_bits;
    __builtin_memcpy(&next_bits, &next, 8);
    if (tfshm::at(&e->erl_tokens)->compare_exchange_weak(
            tok_bits, next_bits, std::memory_order_acq_rel)) {
      tfshm::at(&e->erl_last_update_ns)->store(now, std::memory_order_relaxed);
      return 0.0;
    }
  }
  return 0.0;  // heavy contention: admit rather than livelock
}

void consume_blocking(double cost) {
  tier_note_launch();
  TfDeviceEntry* e = cur_dev();
  if (!e) return;
  heartbeat();
  // Freeze: hypervisor wants this worker fully stopped (auto-freeze /
  // snapshot preparation). Block before token accounting.
  while (g.shm && (tfshm::at(&g.shm->flags)->load(std::memory_order_relaxed) &
                   TF_FLAG_FREEZE)) {
    timespec ts{0, 2000000};  // 2 ms
    nanosleep(&ts, nullptr);
  }
  uint32_t up = tfshm::at(&e->up_limit_percent)->load(std::memory_order_relaxed);
  if (up >= 100) {  // unthrottled vGPU: count the launch, skip the bucket
    tfshm::at(&e->launch_count)->fetch_add(1, std::memory_order_relaxed);
    return;
  }
  uint64_t t0 = 0;
  double wait;
  while ((wait = try_consume(e, cost)) > 0) {
    if (!t0) {
      t0 = now_ns();
      g.blocked_threads.fetch_add(1, std::memory_order_relaxed);
    }
    if (wait > 0.05) wait = 0.05;  // re-check rate/freeze every <=50 ms
    timespec ts{time_t(wait), long((wait - time_t(wait)) * 1e9)};
    nanosleep(&ts, nullptr);
  }
  if (t0) {
    g.blocked_threads.fetch_sub(1, std::memory_order_relaxed);
    tfshm::at(&e->block_ns_total)
        ->fetch_add(now_ns() - t0, std::memory_order_relaxed);
  }
  tfshm::at(&e->launch_count)->fetch_add(1, std::memory_order_relaxed);
}

// ------------------------------------------------------------ vram account

bool admit_alloc(size_t bytes) {
  TfDeviceEntry* e = cur_dev();
  if (!e) return true;
  uint64_t limit = tfshm::at(&e->mem_limit_bytes)->load(std::memory_order_relaxed);
  if (limit == 0) return true;  // no cap
  uint64_t used = tfshm::at(&e->pod_memory_used)->load(std::memory_order_relaxed);
  if (used + bytes > limit) {
    dbg("deny alloc %zu (used %lu limit %lu)", bytes, used, limit);
    return false;
  }
  return true;
}

void record_alloc(const void* p, size_t bytes) {
  TfDeviceEntry* e = cur_dev();
  if (!e || !p) return;
  {
    std::lock_guard<std::mutex> l(g.allocs.mu);
    g.allocs.m[p] = bytes;
  }
  tfshm::at(&e->pod_memory_used)->fetch_add(bytes, std::memory_order_relaxed);
  tfshm::at(&e->alloc_bytes_total)->fetch_add(bytes, std::memory_order_relaxed);
}

void record_free(const void* p) {
  TfDeviceEntry* e = cur_dev();
  if (!e || !p) return;
  size_t bytes = 0;
  {
    std::lock_guard<std::mutex> l(g.allocs.mu);
    auto it = g.allocs.m.find(p);
    if (it == g.allocs.m.end()) return;
    bytes = it->second;
    g.allocs.m.erase(it);
  }
  tfshm::at(&e->pod_memory_used)->fetch_sub(bytes, std::memory_order_relaxed);
}

// ----------------------------------------------------- VRAM expansion

// Real-HIP entry points the expansion path needs (resolved lazily so the
// limiter still loads in GPU-less CI). resolve_real is defined below.
void* resolve_real(const char* name);

struct ExpandHip {
  hipError_t (*MallocManaged)(void**, size_t, unsigned) = nullptr;
  hipError_t (*MemAdvise)(const void*, size_t, int, int) = nullptr;
  hipError_t (*MemPrefetchAsync)(const void*, size_t, int, void*) = nullptr;
  hipError_t (*StreamSynchronize)(void*) = nullptr;
  bool ok = false;
  void init() {
    MallocManaged = (decltype(MallocManaged))resolve_real("hipMallocManaged");
    MemAdvise = (decltype(MemAdvise))resolve_real("hipMemAdvise");
    MemPrefetchAsync =
        (decltype(MemPrefetchAsync))resolve_real("hipMemPrefetchAsync");
    StreamSynchronize =
        (decltype(StreamSynchronize))resolve_real("hipStreamSynchronize");
    ok = MallocManaged && MemAdvise && MemPrefetchAsync;
  }
};
ExpandHip& ehip() {
  static ExpandHip e;
  return e;
}

static const int kHipCpuDeviceId = -1;

// ------------------------------------------------- tier mechanism note
//
// A VMM-based host tier (hipMemCreate location.type=2 + pointer-stable
// VA remap) was built and probed on MI355X: ROCm 7.2 ACCEPTS host-located
// handles but BACKS THEM WITH HBM (native/tiering/vmm_host_probe4.cpp:
// a 2 GiB "host" handle drops device free memory by 2 GiB, process RSS
// grows ~0) — so it frees no VRAM and was removed as a placebo. Managed
// memory (preferred-location CPU + hipMemPrefetchAsync over SDMA) is the
// genuine host tier on this stack; the engine below adds per-range LRU
// working-set tracking, touch hints from interposed memcpy/memset, and
// migration byte/latency stats on top of it.

hipError_t expand_alloc(void** p, size_t sz);  // defined below

hipError_t tier_alloc_overcap(void** p, size_t sz) {
  uint64_t lim = g.expand_limit;
  if (lim) {
    uint64_t cur = g.expanded_bytes.load(std::memory_order_relaxed);
    if (cur + sz > lim) return hipErrorOutOfMemory;
  }
  return expand_alloc(p, sz);
}
static const int kAdviseSetPreferredLocation = 3;
static const int kAdviseSetAccessedBy = 5;

// Allocate an over-cap range in host-DRAM-backed managed memory.
hipError_t expand_alloc(void** p, size_t sz) {
  ExpandHip& eh = ehip();
  if (!eh.MallocManaged) eh.init();
  if (!eh.ok) return hipErrorOutOfMemory;
  uint64_t lim = g.expand_limit;
  if (lim && g.expanded_bytes.load(std::memory_order_relaxed) + sz > lim)
    return hipErrorOutOfMemory;  // host expansion budget exhausted too
  hipError_t r = eh.MallocManaged(p, sz, 1 /*hipMemAttachGlobal*/);
  if (r != hipSuccess) return r;
  // host tier: pages prefer DRAM; GPU keeps direct access (no fault storm
  // on XNACK-disabled parts — fine-grain host access over PCIe).
  eh.MemAdvise(*p, sz, kAdviseSetPreferredLocation, kHipCpuDeviceId);
  eh.MemAdvise(*p, sz, kAdviseSetAccessedBy, tls_device);
  {
    std::lock_guard<std::mutex> l(g.expand_mu);
    g.expanded[(uint64_t)*p] = ExpandedRange{
        sz, false, tls_device,
        g.tier_epoch.load(std::memory_order_relaxed)};
  }
  g.expanded_bytes.fetch_add(sz, std::memory_order_relaxed);
  dbg("expand_alloc %zu B -> host tier (total expanded %lu)", sz,
      g.expanded_bytes.load());
  return hipSuccess;
}

// Returns bytes whose residency changed. to_device: promote the HOTTEST
// host ranges (descending last_use) into up to `budget` bytes of HBM;
// else demote the COLDEST device-resident ranges (ascending last_use)
// back to host DRAM — LRU in both directions, from the working-set
// signal collected by touch_managed() + the launch epoch.
uint64_t tier_migrate(bool to_device, uint64_t budget) {
  ExpandHip& eh = ehip();
  if (!eh.ok) return 0;
  uint64_t t0 = now_ns();
  uint64_t moved = 0;
  std::lock_guard<std::mutex> l(g.expand_mu);
  std::vector<std::pair<uint64_t, uint64_t>> order;  // (last_use, va)
  for (auto& [va, r] : g.expanded)
    if (r.device_resident != to_device) order.emplace_back(r.last_use, va);
  if (to_device)
    std::sort(order.rbegin(), order.rend());  // hottest first
  else
    std::sort(order.begin(), order.end());  // coldest first
  for (auto& [lu, va] : order) {
    ExpandedRange& r = g.expanded[va];
    const void* ptr = (const void*)va;
    if (to_device && moved + r.bytes > budget) continue;
    int dst = to_device ? r.device : kHipCpuDeviceId;
    eh.MemAdvise(ptr, r.bytes, kAdviseSetPreferredLocation, dst);
    if (eh.MemPrefetchAsync(ptr, r.bytes, dst, nullptr) == hipSuccess) {
      r.device_resident = to_device;
      moved += r.bytes;
      if (to_device)
        r.last_use = g.tier_epoch.load(std::memory_order_relaxed);
      if (!to_device && moved >= budget && budget) break;
    }
  }
  if (moved && eh.StreamSynchronize) eh.StreamSynchronize(nullptr);
  uint64_t dt = now_ns() - t0;
  if (to_device) {
    g.promoted_bytes.fetch_add(moved, std::memory_order_relaxed);
    g.promoted_total.fetch_add(moved, std::memory_order_relaxed);
    g.promote_ns.fetch_add(dt, std::memory_order_relaxed);
  } else {
    g.promoted_bytes.fetch_sub(
        moved > g.promoted_bytes.load() ? g.promoted_bytes.load() : moved,
        std::memory_order_relaxed);
    g.demoted_total.fetch_add(moved, std::memory_order_relaxed);
    g.demote_ns.fetch_add(dt, std::memory_order_relaxed);
  }
  return moved;
}

// LRU epoch: one tick per governed GPU op (launch/memcpy) — the
// working-set clock tier_migrate orders by.
void tier_note_launch() {
  if (g.expand_enabled)
    g.tier_epoch.fetch_add(1, std::memory_order_relaxed);
}

// Working-set touch: refresh the LRU stamp of the expanded range
// containing p (called from the memcpy/memset interposers — the only
// pointer-visible ops; kernel args are opaque).
void touch_managed(const void* p) {
  if (!g.expand_enabled || !p) return;
  std::lock_guard<std::mutex> l(g.expand_mu);
  if (g.expanded.empty()) return;
  auto it = g.expanded.upper_bound((uint64_t)p);
  if (it == g.expanded.begin()) return;
  --it;
  if ((uint64_t)p < it->first + it->second.bytes)
    it->second.last_use = g.tier_epoch.load(std::memory_order_relaxed);
}

void* tier_thread_main(void*) {
  // Hypervisor-driven loop: VRAM pressure flag → demote; otherwise promote
  // expanded ranges into free cap headroom (hot data earns HBM residency).
  while (true) {
    timespec ts{0, 250000000};  // 250 ms
    nanosleep(&ts, nullptr);
    if (!g.shm) continue;
    uint32_t flags = tfshm::at(&g.shm->flags)->load(std::memory_order_relaxed);
    if (flags & TF_FLAG_VRAM_PRESSURE) {
      tier_migrate(false, 0 /*all*/);
      continue;
    }
    TfDeviceEntry* e = cur_dev();
    if (!e) continue;
    uint64_t lim = tfshm::at(&e->mem_limit_bytes)->load(std::memory_order_relaxed);
    uint64_t used = tfshm::at(&e->pod_memory_used)->load(std::memory_order_relaxed);
    uint64_t promoted = g.promoted_bytes.load(std::memory_order_relaxed);
    if (lim > used + promoted) {
      uint64_t headroom = lim - used - promoted;
      if (headroom > (64u << 20)) tier_migrate(true, headroom);
    }
  }
  return nullptr;
}

void ensure_tier_thread() {
  int expect = 0;
  if (g.tier_thread_running.compare_exchange_strong(expect, 1)) {
    pthread_t t;
    pthread_create(&t, nullptr, tier_thread_main, nullptr);
    pthread_detach(t);
  }
}

// ----------------------------------------------------------------- init

Limiter::Limiter() {
  debug = getenv("TF_LIMITER_DEBUG") != nullptr;
  trace = getenv("TF_LIMITER_TRACE") != nullptr;
  expand_enabled = getenv("TF_VRAM_EXPAND") != nullptr &&
                   atoi(getenv("TF_VRAM_EXPAND")) != 0;
  expand_limit = (uint64_t)env_f("TF_VRAM_EXPAND_LIMIT_BYTES", 0);
  tokens_per_launch = env_f("TF_TOKENS_PER_LAUNCH", 1.0);
  tokens_per_graph = env_f("TF_TOKENS_PER_GRAPH", 8.0);
  tokens_per_memcpy = env_f("TF_TOKENS_PER_MEMCPY", 1.0);
  const char* shm_path = getenv("TF_SHM_PATH");
  if (shm_path && *shm_path) {
    int fd = open(shm_path, O_RDWR);
    if (fd >= 0) {
      void* p = mmap(nullptr, TF_SHM_SIZE, PROT_READ | PROT_WRITE, MAP_SHARED,
                     fd, 0);
      close(fd);
      if (p != MAP_FAILED) {
        auto* s = reinterpret_cast<TfSharedState*>(p);
        if (s->magic == TF_SHM_MAGIC) {
          shm = s;
        } else {
          munmap(p, TF_SHM_SIZE);
          fprintf(stderr, "[tf-limiter] bad magic in %s, ignoring\n", shm_path);
        }
      }
    } else {
      fprintf(stderr, "[tf-limiter] cannot open %s: %s\n", shm_path,
              strerror(errno));
    }
  }
  if (!shm && (getenv("TF_VRAM_LIMIT_BYTES") || getenv("TF_UP_LIMIT_PERCENT") ||
               getenv("TF_ERL_RATE"))) {
    // Private page: standalone mode (no hypervisor), limits straight from env.
    void* p = mmap(nullptr, TF_SHM_SIZE, PROT_READ | PROT_WRITE,
                   MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
    if (p != MAP_FAILED) {
      memset(p, 0, TF_SHM_SIZE);
      shm = reinterpret_cast<TfSharedState*>(p);
      shm->magic = TF_SHM_MAGIC;
      shm->version = TF_SHM_VERSION;
      standalone = true;
      TfDeviceEntry* e = &shm->dev[0];
      snprintf(e->uuid, TF_UUID_LEN, "standalone-0");
      e->up_limit_percent = uint32_t(env_f("TF_UP_LIMIT_PERCENT", 100));
      e->total_cus = 256;
      e->mem_limit_bytes = uint64_t(env_f("TF_VRAM_LIMIT_BYTES", 0));
      double rate = env_f("TF_ERL_RATE", 0.0);
      if (rate <= 0 && e->up_limit_percent < 100) {
        // No hypervisor PID loop: a fixed open-loop rate; standalone users
        // set TF_ERL_RATE for precise pacing.
        rate = 20.0 * e->up_limit_percent;
      }
      tfshm::store_double(&e->erl_refill_rate, rate);
      double cap = env_f("TF_ERL_CAPACITY", rate > 0 ? rate * 0.1 : 0.0);
      tfshm::store_double(&e->erl_capacity, cap);
      tfshm::store_double(&e->erl_tokens, cap);
      e->erl_last_update_ns = now_ns();
      e->active = 1;
      shm->device_count = 1;
    }
  }
  enabled = shm != nullptr;
  if (shm) {
    tfshm::lock(shm);
    if (shm->pid_count < TF_MAX_PIDS) {
      bool present = false;
      for (uint32_t i = 0; i < shm->pid_count; ++i)
        if (shm->pids[i] == getpid()) present = true;
      if (!present) shm->pids[shm->pid_count++] = getpid();
    }
    tfshm::unlock(shm);
    tfshm::at(&shm->heartbeat_ns)->store(now_ns(), std::memory_order_relaxed);
  }
  if (debug)
    fprintf(stderr, "[tf-limiter %d] init enabled=%d standalone=%d\n", getpid(),
            (int)enabled, (int)standalone);
}

__attribute__((constructor)) void tf_limiter_init() {
  G();  // force init at load; interposers also call G() so lazy init is safe
}

// --------------------------------------------------------------- dispatch

// Resolve the real HIP entry point. RTLD_NEXT covers the LD_PRELOAD case
// where the app links libamdhip64 directly; PyTorch instead dlopens its
// bundled libamdhip64.so into a local namespace, so fall back to grabbing
// the already-loaded instance by soname (RTLD_NOLOAD first — never load a
// second HIP runtime beside the app's).
void* resolve_real(const char* name) {
  void* p = dlsym(RTLD_NEXT, name);
  if (p) return p;
  static std::atomic<void*> hip_handle{nullptr};
  void* h = hip_handle.load(std::memory_order_acquire);
  if (!h) {
    static const char* sonames[] = {"libamdhip64.so", "libamdhip64.so.7",
                                    "libamdhip64.so.6"};
    for (const char* so : sonames) {
      h = dlopen(so, RTLD_LAZY | RTLD_NOLOAD);
      if (h) break;
    }
    if (!h) {
      for (const char* so : sonames) {
        h = dlopen(so, RTLD_LAZY);
        if (h) break;
      }
    }
    if (h) hip_handle.store(h, std::memory_order_release);
  }
  if (h) p = dlsym(h, name);
  return p;
}

template <typename Fn>
Fn real(const char* name, std::atomic<void*>* cache) {
  void* p = cache->load(std::memory_order_acquire);
  if (!p) {
    p = resolve_real(name);
    if (!p) {
      fprintf(stderr, "[tf-limiter] missing real symbol %s\n", name);
      abort();
    }
    cache->store(p, std::memory_order_release);
  }
  return reinterpret_cast<Fn>(p);
}

#define REAL(ret, name, ...)                          \
  static std::atomic<void*> real_##name{nullptr};     \
  using fn_##name = ret (*)(__VA_ARGS__);             \
  auto call_##name = real<fn_##name>(#name, &real_##name);

}  // namespace

// ------------------------------------------------------------- interposers

extern "C" {

// ----- device selection (track TLS current device)
int hipSetDevice(int dev) {
  REAL(int, hipSetDevice, int);
  int r = call_hipSetDevice(dev);
  if (r == hipSuccess) tls_device = dev;
  return r;
}

// ----- memory
hipError_t hipMalloc(void** p, size_t sz) {
  REAL(hipError_t, hipMalloc, void**, size_t);
  if (g.enabled && !admit_alloc(sz)) {
    if (g.expand_enabled) {
      // VRAM oversubscription: land the range in the host-DRAM tier
      // (VMM arena remap when available, managed memory otherwise)
      hipError_t r = tier_alloc_overcap(p, sz);
      if (r == hipSuccess) ensure_tier_thread();
      return r;
    }
    return hipErrorOutOfMemory;
  }
  hipError_t r = call_hipMalloc(p, sz);
  if (g.enabled && r == hipSuccess) record_alloc(*p, sz);
  return r;
}

hipError_t hipMallocAsync(void** p, size_t sz, hipStream_t s) {
  REAL(hipError_t, hipMallocAsync, void**, size_t, hipStream_t);
  if (g.enabled && !admit_alloc(sz)) return hipErrorOutOfMemory;
  hipError_t r = call_hipMallocAsync(p, sz, s);
  if (g.enabled && r == hipSuccess) record_alloc(*p, sz);
  return r;
}

hipError_t hipMallocManaged(void** p, size_t sz, unsigned flags) {
  REAL(hipError_t, hipMallocManaged, void**, size_t, unsigned);
  if (g.enabled && !admit_alloc(sz)) return hipErrorOutOfMemory;
  hipError_t r = call_hipMallocManaged(p, sz, flags);
  if (g.enabled && r == hipSuccess) record_alloc(*p, sz);
  return r;
}

hipError_t hipMallocPitch(void** p, size_t* pitch, size_t w, size_t h) {
  REAL(hipError_t, hipMallocPitch, void**, size_t*, size_t, size_t);
  if (g.enabled && !admit_alloc(w * h)) return hipErrorOutOfMemory;
  hipError_t r = call_hipMallocPitch(p, pitch, w, h);
  if (g.enabled && r == hipSuccess) record_alloc(*p, *pitch * h);
  return r;
}

hipError_t hipFree(void* p) {
  REAL(hipError_t, hipFree, void*);
  hipError_t r = call_hipFree(p);
  if (g.enabled && r == hipSuccess) {
    record_free(p);
    std::lock_guard<std::mutex> l(g.expand_mu);
    auto it = g.expanded.find((uint64_t)p);
    if (it != g.expanded.end()) {
      g.expanded_bytes.fetch_sub(it->second.bytes, std::memory_order_relaxed);
      if (it->second.device_resident)
        g.promoted_bytes.fetch_sub(it->second.bytes,
                                   std::memory_order_relaxed);
      g.expanded.erase(it);
    }
  }
  return r;
}

hipError_t hipFreeAsync(void* p, hipStream_t s) {
  REAL(hipError_t, hipFreeAsync, void*, hipStream_t);
  hipError_t r = call_hipFreeAsync(p, s);
  if (g.enabled && r == hipSuccess) record_free(p);
  return r;
}

hipError_t hipMemGetInfo(size_t* free_b, size_t* total_b) {
  REAL(hipError_t, hipMemGetInfo, size_t*, size_t*);
  hipError_t r = call_hipMemGetInfo(free_b, total_b);
  if (g.enabled && r == hipSuccess) {
    TfDeviceEntry* e = cur_dev();
    if (e) {
      uint64_t lim = tfshm::at(&e->mem_limit_bytes)->load(std::memory_order_relaxed);
      if (lim) {  // vGPU-scoped view (fake amd-smi reads this too)
        uint64_t used =
            tfshm::at(&e->pod_memory_used)->load(std::memory_order_relaxed);
        if (total_b) *total_b = lim;
        if (free_b) *free_b = used < lim ? lim - used : 0;
      }
    }
  }
  return r;
}

// ----- kernel launches (the ERL-governed hot path)
hipError_t hipLaunchKernel(const void* f, dim3u grid, dim3u block, void** args,
                           size_t shmem, hipStream_t stream) {
  REAL(hipError_t, hipLaunchKernel, const void*, dim3u, dim3u, void**, size_t,
       hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return TF_TRACE_CALL(0, call_hipLaunchKernel(f, grid, block, args, shmem,
                                               stream));
}

hipError_t hipLaunchKernel_spt(const void* f, dim3u grid, dim3u block,
                               void** args, size_t shmem, hipStream_t stream) {
  REAL(hipError_t, hipLaunchKernel_spt, const void*, dim3u, dim3u, void**,
       size_t, hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipLaunchKernel_spt(f, grid, block, args, shmem, stream);
}

hipError_t hipLaunchKernelExC(const void* cfg, const void* f, void** args) {
  REAL(hipError_t, hipLaunchKernelExC, const void*, const void*, void**);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipLaunchKernelExC(cfg, f, args);
}

hipError_t hipExtLaunchKernel(const void* f, dim3u grid, dim3u block,
                              void** args, size_t shmem, hipStream_t stream,
                              void* startEv, void* stopEv, int flags) {
  REAL(hipError_t, hipExtLaunchKernel, const void*, dim3u, dim3u, void**,
       size_t, hipStream_t, void*, void*, int);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipExtLaunchKernel(f, grid, block, args, shmem, stream, startEv,
                                 stopEv, flags);
}

hipError_t hipModuleLaunchKernel(void* func, unsigned gx, unsigned gy,
                                 unsigned gz, unsigned bx, unsigned by,
                                 unsigned bz, unsigned shmem, hipStream_t s,
                                 void** params, void** extra) {
  REAL(hipError_t, hipModuleLaunchKernel, void*, unsigned, unsigned, unsigned,
       unsigned, unsigned, unsigned, unsigned, hipStream_t, void**, void**);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipModuleLaunchKernel(func, gx, gy, gz, bx, by, bz, shmem, s,
                                    params, extra);
}

hipError_t hipExtModuleLaunchKernel(void* func, unsigned gwx, unsigned gwy,
                                    unsigned gwz, unsigned bx, unsigned by,
                                    unsigned bz, size_t shmem, hipStream_t s,
                                    void** params, void** extra, void* startEv,
                                    void* stopEv, unsigned flags) {
  REAL(hipError_t, hipExtModuleLaunchKernel, void*, unsigned, unsigned,
       unsigned, unsigned, unsigned, unsigned, size_t, hipStream_t, void**,
       void**, void*, void*, unsigned);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipExtModuleLaunchKernel(func, gwx, gwy, gwz, bx, by, bz, shmem,
                                       s, params, extra, startEv, stopEv,
                                       flags);
}

hipError_t hipLaunchCooperativeKernel(const void* f, dim3u grid, dim3u block,
                                      void** args, unsigned shmem,
                                      hipStream_t stream) {
  REAL(hipError_t, hipLaunchCooperativeKernel, const void*, dim3u, dim3u,
       void**, unsigned, hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_launch);
  return call_hipLaunchCooperativeKernel(f, grid, block, args, shmem, stream);
}

hipError_t hipGraphLaunch(void* graphExec, hipStream_t stream) {
  REAL(hipError_t, hipGraphLaunch, void*, hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_graph);
  return call_hipGraphLaunch(graphExec, stream);
}

hipError_t hipGraphLaunch_spt(void* graphExec, hipStream_t stream) {
  REAL(hipError_t, hipGraphLaunch_spt, void*, hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_graph);
  return call_hipGraphLaunch_spt(graphExec, stream);
}

// ----- memcpy / memset (memory-op tokens; VRAM cap already enforced)
hipError_t hipMemcpy(void* dst, const void* src, size_t n, int kind) {
  REAL(hipError_t, hipMemcpy, void*, const void*, size_t, int);
  if (g.enabled) {
    consume_blocking(g.tokens_per_memcpy);
    touch_managed(dst);
    touch_managed(src);
  }
  return TF_TRACE_CALL(1, call_hipMemcpy(dst, src, n, kind));
}

hipError_t hipMemcpyAsync(void* dst, const void* src, size_t n, int kind,
                          hipStream_t s) {
  REAL(hipError_t, hipMemcpyAsync, void*, const void*, size_t, int,
       hipStream_t);
  if (g.enabled) {
    consume_blocking(g.tokens_per_memcpy);
    touch_managed(dst);
    touch_managed(src);
  }
  return TF_TRACE_CALL(1, call_hipMemcpyAsync(dst, src, n, kind, s));
}

hipError_t hipMemcpyWithStream(void* dst, const void* src, size_t n, int kind,
                               hipStream_t s) {
  REAL(hipError_t, hipMemcpyWithStream, void*, const void*, size_t, int,
       hipStream_t);
  if (g.enabled) consume_blocking(g.tokens_per_memcpy);
  return call_hipMemcpyWithStream(dst, src, n, kind, s);
}

hipError_t hipMemsetAsync(void* dst, int v, size_t n, hipStream_t s) {
  REAL(hipError_t, hipMemsetAsync, void*, int, size_t, hipStream_t);
  if (g.enabled) {
    consume_blocking(g.tokens_per_memcpy);
    touch_managed(dst);
  }
  return call_hipMemsetAsync(dst, v, n, s);
}

hipError_t hipMemset(void* dst, int v, size_t n) {
  REAL(hipError_t, hipMemset, void*, int, size_t);
  if (g.enabled) consume_blocking(g.tokens_per_memcpy);
  return call_hipMemset(dst, v, n);
}

// ------------------------------------------------- limiter introspection ABI
// Small C ABI for the fake amd-smi shim and tests (reference limiter.h:71-106
// exposes an equivalent worker-facing surface).

int tf_limiter_enabled() { return g.enabled ? 1 : 0; }

int tf_limiter_stats(int dev, unsigned long long* mem_used,
                     unsigned long long* mem_limit, unsigned* launches,
                     unsigned long long* block_ns, double* tokens) {
  if (!g.shm || dev < 0 || dev >= TF_MAX_DEVICES) return -1;
  TfDeviceEntry* e = &g.shm->dev[dev];
  if (!e->active) return -1;
  if (mem_used) *mem_used = e->pod_memory_used;
  if (mem_limit) *mem_limit = e->mem_limit_bytes;
  if (launches) *launches = e->launch_count;
  if (block_ns) *block_ns = e->block_ns_total;
  if (tokens) *tokens = tfshm::load_double(&e->erl_tokens);
  return 0;
}

void tf_limiter_freeze(int on) {
  if (g.shm) {
    uint32_t f = g.shm->flags;
    g.shm->flags = on ? (f | TF_FLAG_FREEZE) : (f & ~TF_FLAG_FREEZE);
  }
}

// ---- VRAM tiering introspection / control (tests + fake amd-smi) ----

int tf_limiter_tier_stats(unsigned long long* expanded,
                          unsigned long long* promoted,
                          unsigned* n_ranges) {
  if (expanded) *expanded = g.expanded_bytes.load(std::memory_order_relaxed);
  if (promoted) *promoted = g.promoted_bytes.load(std::memory_order_relaxed);
  if (n_ranges) {
    std::lock_guard<std::mutex> l(g.expand_mu);
    *n_ranges = (unsigned)g.expanded.size();
  }
  return g.expand_enabled ? 1 : 0;
}

// Tier engine detail: out8 = {device_resident_bytes, host_resident_bytes,
// demoted_total, promoted_total, demote_ns, promote_ns, epoch, n_ranges}.
// Returns 1 when oversubscription is enabled.
int tf_limiter_tier_stats2(unsigned long long* out8) {
  if (out8) {
    uint64_t prom = g.promoted_bytes.load(std::memory_order_relaxed);
    uint64_t exp = g.expanded_bytes.load(std::memory_order_relaxed);
    out8[0] = prom;
    out8[1] = exp > prom ? exp - prom : 0;
    out8[2] = g.demoted_total.load(std::memory_order_relaxed);
    out8[3] = g.promoted_total.load(std::memory_order_relaxed);
    out8[4] = g.demote_ns.load(std::memory_order_relaxed);
    out8[5] = g.promote_ns.load(std::memory_order_relaxed);
    out8[6] = g.tier_epoch.load(std::memory_order_relaxed);
    std::lock_guard<std::mutex> l(g.expand_mu);
    out8[7] = g.expanded.size();
  }
  return g.expand_enabled ? 1 : 0;
}

// Working-set hint: mark the expanded range containing p as hot.
void tf_limiter_touch(const void* p) { touch_managed(p); }

// Per-range introspection (tests assert exact LRU behavior with this).
// Returns 1 if p is inside an expanded range, 0 otherwise.
int tf_limiter_tier_range(const void* p, unsigned long long* bytes,
                          int* device_resident,
                          unsigned long long* last_use) {
  std::lock_guard<std::mutex> l(g.expand_mu);
  if (g.expanded.empty()) return 0;
  auto it = g.expanded.upper_bound((uint64_t)p);
  if (it == g.expanded.begin()) return 0;
  --it;
  if ((uint64_t)p >= it->first + it->second.bytes) return 0;
  if (bytes) *bytes = it->second.bytes;
  if (device_resident) *device_resident = it->second.device_resident ? 1 : 0;
  if (last_use) *last_use = it->second.last_use;
  return 1;
}

// Force-demote every expanded range to host DRAM (pressure-trap path);
// returns bytes moved.
unsigned long long tf_limiter_demote_all() { return tier_migrate(false, 0); }

// Promote up to budget bytes into HBM; returns bytes moved.
unsigned long long tf_limiter_promote(unsigned long long budget) {
  return tier_migrate(true, budget);
}

// Per-call latency histogram readout: cls 0 launch, 1 memcpy, 2 other;
// out[32] = counts per log2(ns) bucket. Returns total samples.
unsigned long long tf_limiter_latency_hist(int cls,
                                           unsigned long long* out32) {
  if (cls < 0 || cls > 2) return 0;
  unsigned long long total = 0;
  for (int b = 0; b < 32; ++b) {
    unsigned long long v =
        g.lat_hist[cls][b].load(std::memory_order_relaxed);
    if (out32) out32[b] = v;
    total += v;
  }
  return total;
}

void tf_limiter_set_vram_pressure(int on) {
  if (g.shm) {
    uint32_t f = g.shm->flags;
    g.shm->flags =
        on ? (f | TF_FLAG_VRAM_PRESSURE) : (f & ~TF_FLAG_VRAM_PRESSURE);
  }
}

}  // extern "C"
