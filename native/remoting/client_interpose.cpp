// libtfhip_client.so — the GPU-over-IP client stub.
//
// LD_PRELOADed into a GPU-less app process (HIP_VISIBLE_DEVICES="");
// interposes the HIP runtime API surface PyTorch-class workloads use and
// forwards it over the shared-memory command ring to tf_vgpu_worker (see
// protocol.h). Async ops (kernel launches, async copies, event records)
// enqueue and return — the <4% overhead path; sync ops round-trip with
// futex parking. Capability parity with the reference's closed cuda_hook
// client.
//
// Connection: TF_WORKER_SOCKET names the worker's unix socket (same-node);
// the TensorFusionConnection URL native+ip+port+worker resolves to it via
// the operator /connection endpoint (python side, client/runtime.py).
//
// Any HIP entry point NOT interposed here falls through to the real (GPU
// -less) libamdhip64 and fails loudly — deliberately: silent local
// execution would falsify the remoting claim.

#include <dlfcn.h>
#include <pthread.h>
#include <errno.h>
#include <fcntl.h>
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <linux/errqueue.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/un.h>
#include <unistd.h>

#include <map>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "protocol.h"
#include "ring.h"

namespace {

using namespace tfrpc;

typedef int hipError_t;
static const hipError_t hipSuccess = 0;
static const hipError_t hipErrorInvalidValue = 1;
static const hipError_t hipErrorOutOfMemory = 2;
static const hipError_t hipErrorNotInitialized = 3;
static const hipError_t hipErrorInvalidDevice = 101;
static const hipError_t hipErrorNotReady = 600;
static const hipError_t hipErrorNotSupported = 801;

struct dim3u {
  unsigned x, y, z;
};

// ---------------------------------------------------------- client state

struct KernLayout {
  uint64_t func = 0;
  uint32_t explicit_bytes = 0;
  std::vector<std::pair<uint32_t, uint32_t>> args;  // (size, offset)
};

struct FatBin {
  const void* bundle = nullptr;  // in-process image (bundle or ELF)
  size_t size = 0;
  uint64_t image_id = 0;
  uint64_t module = 0;  // worker module handle (0 = not shipped)
};

struct Client {
  bool connected = false;
  bool debug = false;
  bool tcp_mode = false;    // cross-node session (TF_WORKER_TCP=host:port)
  int sock = -1;
  int seg_fd = -1;          // memfd of the segment, kept for re-handshake
  char sock_path[256] = {0};
  Header* hdr = nullptr;
  RingView cmd;  // producer
  RingView cpl;  // consumer
  uint8_t* arena_base = nullptr;
  std::mutex mu;          // serializes ring production + sync round trips
  uint64_t next_seq = 1;
  int device_count = 0;
  thread_local static int cur_device;
  hipError_t last_error = hipSuccess;

  // registrations
  std::mutex reg_mu;
  std::unordered_map<void*, FatBin*> fatbins;        // handle → fatbin
  std::unordered_map<const void*, std::pair<FatBin*, std::string>> stubs;
  // device globals: host shadow ptr → (fatbin, device name, size);
  // resolved worker addresses cached in dvar_addr
  std::unordered_map<const void*, std::pair<FatBin*, std::string>> dvars;
  std::unordered_map<const void*, std::pair<uint64_t, uint64_t>> dvar_addr;
  std::unordered_map<const void*, KernLayout*> launch_cache;
  std::unordered_map<uint64_t, KernLayout*> module_fn_layouts;  // worker fn
  uint64_t next_image_id = 1;

  // memory tracking
  std::mutex mem_mu;
  std::map<uint64_t, uint64_t> dev_ranges;   // base → size (worker ptrs)
  std::map<uint64_t, uint64_t> host_allocs;  // hipHostMalloc ranges

  // caches
  std::mutex cache_mu;
  std::unordered_map<int, std::vector<uint8_t>> props_cache;
  std::map<std::pair<int, int>, int> attr_cache;

  Client();
};

thread_local int Client::cur_device = 0;

Client& C() {
  static Client c;
  return c;
}

void cdbg(const char* fmt, ...) {
  if (!C().debug) return;
  va_list ap;
  va_start(ap, fmt);
  fprintf(stderr, "[tf-client %d] ", getpid());
  vfprintf(stderr, fmt, ap);
  fprintf(stderr, "\n");
  va_end(ap);
}

// hipGraph capture state (client-side mirror; worker owns the real capture)
std::mutex g_capture_mu;
std::map<void*, unsigned long long> g_capturing;  // stream → capture id
unsigned long long g_capture_seq = 1;

bool stream_capturing(void* stream, unsigned long long* id) {
  std::lock_guard<std::mutex> l(g_capture_mu);
  auto it = g_capturing.find(stream);
  if (it == g_capturing.end()) return false;
  if (id) *id = it->second;
  return true;
}

bool send_fd(int sock, int fd) {
  char buf[1] = {0};
  iovec iov{buf, 1};
  char ctrl[CMSG_SPACE(sizeof(int))] = {};
  msghdr msg{};
  msg.msg_iov = &iov;
  msg.msg_iovlen = 1;
  msg.msg_control = ctrl;
  msg.msg_controllen = sizeof ctrl;
  cmsghdr* c = CMSG_FIRSTHDR(&msg);
  c->cmsg_level = SOL_SOCKET;
  c->cmsg_type = SCM_RIGHTS;
  c->cmsg_len = CMSG_LEN(sizeof(int));
  memcpy(CMSG_DATA(c), &fd, sizeof fd);
  return sendmsg(sock, &msg, 0) == 1;
}

// ------------------------------------------------ TCP transport pumps
// Cross-node mode mirrors the local rings over a socket: the cmd pump
// consumes the cmd ring (as the worker would) and frames records +
// referenced arena payloads onto the wire; the cpl pump turns incoming
// frames back into cpl-ring records.

bool tcp_read_full(int fd, void* p, size_t n) {
  uint8_t* b = (uint8_t*)p;
  while (n) {
    ssize_t r = recv(fd, b, n, 0);
    if (r <= 0) return false;
    b += r;
    n -= (size_t)r;
  }
  return true;
}

bool tcp_write_full(int fd, const void* p, size_t n) {
  const uint8_t* b = (const uint8_t*)p;
  while (n) {
    ssize_t r = send(fd, b, n, MSG_NOSIGNAL);
    if (r <= 0) return false;
    b += r;
    n -= (size_t)r;
  }
  return true;
}

// One frame = one sendmsg (header + record + up to two arena slices —
// the payload goes straight from the hipHostRegister'd arena to the
// socket, no staging copy). With MSG_ZEROCOPY (TF_TCP_ZEROCOPY=1 and a
// payload past the threshold) the kernel pins those arena pages for DMA
// instead of copying into skbs; completion notifications arrive on the
// error queue and gate arena-chunk reuse (ZcPending below).
struct ZcState {
  bool enabled = false;       // SO_ZEROCOPY accepted on this socket
  uint32_t next_seq = 0;      // kernel numbers zerocopy sends 0,1,2,...
  uint32_t completed = 0;     // highest seq + 1 fully acked
  struct Pending {
    uint32_t seq;
    uint64_t arena_end;       // arena_freed advances here on completion
  };
  std::vector<Pending> pending;
};

void zc_drain(int fd, ZcState& zc, Header* hdr) {
  // non-blocking errqueue read: SO_EE_ORIGIN_ZEROCOPY ranges
  while (true) {
    char ctrl[128];
    msghdr msg{};
    msg.msg_control = ctrl;
    msg.msg_controllen = sizeof ctrl;
    ssize_t r = recvmsg(fd, &msg, MSG_ERRQUEUE | MSG_DONTWAIT);
    if (r < 0) break;
    for (cmsghdr* cm = CMSG_FIRSTHDR(&msg); cm;
         cm = CMSG_NXTHDR(&msg, cm)) {
      if ((cm->cmsg_level == SOL_IP && cm->cmsg_type == IP_RECVERR) ||
          (cm->cmsg_level == SOL_IPV6 && cm->cmsg_type == 25)) {
        auto* ee = reinterpret_cast<sock_extended_err*>(CMSG_DATA(cm));
        if (ee->ee_origin == SO_EE_ORIGIN_ZEROCOPY) {
          uint32_t hi = ee->ee_data;  // inclusive range [ee_info, ee_data]
          if (hi + 1 > zc.completed) zc.completed = hi + 1;
        }
      }
    }
  }
  // advance arena_freed for fully-acked zerocopy payloads (in order)
  size_t done = 0;
  for (auto& p : zc.pending) {
    if (p.seq < zc.completed) {
      uint64_t cur = at(&hdr->arena_freed)->load();
      if (p.arena_end > cur) at(&hdr->arena_freed)->store(p.arena_end);
      ++done;
    } else {
      break;
    }
  }
  if (done) zc.pending.erase(zc.pending.begin(), zc.pending.begin() + done);
}

// Returns the number of MSG_ZEROCOPY sendmsg calls performed (each gets
// its own kernel completion seq), or -1 on error. A partial send retries
// with the SAME flags, so every successful call is counted.
int tcp_send_frame(int fd, FrameHdr* fh, const uint8_t* rec, size_t len,
                   const uint8_t* pay1, size_t n1, const uint8_t* pay2,
                   size_t n2, bool zerocopy) {
  iovec iov[4];
  int cnt = 0;
  iov[cnt++] = {fh, sizeof *fh};
  iov[cnt++] = {const_cast<uint8_t*>(rec), len};
  if (n1) iov[cnt++] = {const_cast<uint8_t*>(pay1), n1};
  if (n2) iov[cnt++] = {const_cast<uint8_t*>(pay2), n2};
  msghdr msg{};
  msg.msg_iov = iov;
  msg.msg_iovlen = cnt;
  size_t total = sizeof *fh + len + n1 + n2;
  int flags = MSG_NOSIGNAL | (zerocopy ? MSG_ZEROCOPY : 0);
  int zc_sends = 0;
  while (total) {
    ssize_t r = sendmsg(fd, &msg, flags);
    if (r < 0 && zerocopy && (errno == ENOBUFS || errno == EINVAL))
      return -1;  // caller retries without zerocopy
    if (r <= 0) return -1;
    if (zerocopy) ++zc_sends;
    total -= (size_t)r;
    // partial send: advance the iovec
    size_t adv = (size_t)r;
    while (adv && msg.msg_iovlen) {
      if (adv >= msg.msg_iov[0].iov_len) {
        adv -= msg.msg_iov[0].iov_len;
        ++msg.msg_iov;
        --msg.msg_iovlen;
      } else {
        msg.msg_iov[0].iov_base = (uint8_t*)msg.msg_iov[0].iov_base + adv;
        msg.msg_iov[0].iov_len -= adv;
        adv = 0;
      }
    }
  }
  return zc_sends;
}

void* tcp_cmd_pump(void* arg) {
  Client* c = (Client*)arg;
  RingView cmd(&c->hdr->cmd, cmd_buf(c->hdr), CMD_RING_BYTES);
  uint8_t* arena = c->arena_base;
  ZcState zc;
  const char* zc_env = getenv("TF_TCP_ZEROCOPY");
  if (!zc_env || atoi(zc_env) != 0) {
    int one = 1;
    zc.enabled = setsockopt(c->sock, SOL_SOCKET, SO_ZEROCOPY, &one,
                            sizeof one) == 0;
  }
  const size_t kZcThreshold = 64u << 10;  // pinned-DMA pays past ~64 KiB
  for (;;) {
    if (zc.enabled && !zc.pending.empty()) zc_drain(c->sock, zc, c->hdr);
    size_t len;
    uint8_t* p = cmd.try_next(&len);
    if (!p) {
      cmd.wait_nonempty();
      continue;
    }
    auto* h = reinterpret_cast<CmdHdr*>(p);
    uint64_t arena_off = 0;
    uint32_t extra = 0;
    if (h->op == OP_MEMCPY_H2D && !(h->flags & F_INLINE_DATA)) {
      auto* m = reinterpret_cast<MemcpyBody*>(p + sizeof(CmdHdr));
      arena_off = m->arena_off;
      extra = (uint32_t)m->size;
    } else if (h->op == OP_LOAD_MODULE) {
      struct B {
        uint64_t image_id, size, arena_off;
      };
      auto* b = reinterpret_cast<B*>(p + sizeof(CmdHdr));
      arena_off = b->arena_off;
      extra = (uint32_t)b->size;
    }
    FrameHdr fh{0, (uint32_t)len, arena_off, extra, 0};
    const uint8_t *pay1 = nullptr, *pay2 = nullptr;
    size_t n1 = 0, n2 = 0;
    if (extra) {
      size_t off = arena_off % ARENA_BYTES;
      size_t first = ARENA_BYTES - off;
      pay1 = arena + off;
      n1 = extra <= first ? extra : first;
      if (extra > first) {
        pay2 = arena;
        n2 = extra - first;
      }
    }
    bool use_zc = zc.enabled && extra >= kZcThreshold;
    int zc_sends = tcp_send_frame(c->sock, &fh, p, len, pay1, n1, pay2,
                                  n2, use_zc);
    bool ok = zc_sends >= 0;
    if (!ok && use_zc) {  // kernel refused zerocopy: plain path once
      zc.enabled = false;
      use_zc = false;
      zc_sends = tcp_send_frame(c->sock, &fh, p, len, pay1, n1, pay2, n2,
                                false);
      ok = zc_sends >= 0;
    }
    if (ok && use_zc) zc.next_seq += (uint32_t)zc_sends;
    if (ok && extra) {
      uint64_t end = arena_off + ((extra + 63) & ~uint64_t(63));
      if (use_zc && zc_sends > 0) {
        // arena pages stay pinned by the kernel until the errqueue ack
        // of the LAST send of this frame: defer the freed advance
        zc.pending.push_back({zc.next_seq - 1, end});
      } else {
        uint64_t cur = at(&c->hdr->arena_freed)->load();
        if (end > cur) at(&c->hdr->arena_freed)->store(end);
      }
    }
    cmd.pop();
    if (!ok) {
      fprintf(stderr, "[tf-client] tcp send failed\n");
      return nullptr;
    }
  }
}

void* tcp_cpl_pump(void* arg) {
  Client* c = (Client*)arg;
  RingView cpl(&c->hdr->cpl, cpl_buf(c->hdr), CPL_RING_BYTES);
  std::vector<uint8_t> rec;
  for (;;) {
    FrameHdr fh;
    if (!tcp_read_full(c->sock, &fh, sizeof fh)) break;
    if (fh.kind != 1 || fh.rec_len > CPL_RING_BYTES / 2) break;
    rec.resize(fh.rec_len);
    if (!tcp_read_full(c->sock, rec.data(), fh.rec_len)) break;
    uint8_t* p;
    while (!(p = cpl.try_reserve(fh.rec_len))) usleep(50);
    memcpy(p, rec.data(), fh.rec_len);
    cpl.commit();
    if (at(&c->hdr->futex_cpl)->exchange(1) == 0)
      futex_wake(&c->hdr->futex_cpl);
  }
  fprintf(stderr, "[tf-client] tcp session closed\n");
  return nullptr;
}

void* reconnect_main(void* arg) {
  Client* c = reinterpret_cast<Client*>(arg);
  for (;;) {
    char b;
    ssize_t n = recv(c->sock, &b, 1, 0);
    if (n > 0) continue;          // workers never send on this socket
    if (n < 0 && (errno == EINTR || errno == EAGAIN)) continue;
    // EOF: worker gone (migration or crash)
    fprintf(stderr, "[tf-client] worker disconnected; waiting for successor on %s\n",
            c->sock_path);
    close(c->sock);
    for (;;) {
      int s2 = socket(AF_UNIX, SOCK_STREAM, 0);
      sockaddr_un addr{};
      addr.sun_family = AF_UNIX;
      strncpy(addr.sun_path, c->sock_path, sizeof addr.sun_path - 1);
      if (connect(s2, (sockaddr*)&addr, sizeof addr) == 0 &&
          send_fd(s2, c->seg_fd)) {
        c->sock = s2;
        fprintf(stderr, "[tf-client] re-attached to worker\n");
        break;
      }
      close(s2);
      usleep(200000);
    }
  }
  return nullptr;
}

Client::Client() {
  debug = getenv("TF_CLIENT_DEBUG") != nullptr;
  const char* tcp = getenv("TF_WORKER_TCP");  // host:port → cross-node
  const char* sock_path = getenv("TF_WORKER_SOCKET");
  if (tcp && *tcp) {
    char host[128] = {0};
    int port = 0;
    const char* colon = strrchr(tcp, ':');
    if (!colon || sscanf(colon + 1, "%d", &port) != 1) return;
    size_t hl = (size_t)(colon - tcp);
    if (hl >= sizeof host) return;
    memcpy(host, tcp, hl);
    void* seg = mmap(nullptr, SEG_BYTES, PROT_READ | PROT_WRITE,
                     MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
    if (seg == MAP_FAILED) return;
    memset(seg, 0, HDR_BYTES);
    hdr = reinterpret_cast<Header*>(seg);
    hdr->magic = MAGIC;
    hdr->version = VERSION;
    hdr->total_bytes = SEG_BYTES;
    sock = socket(AF_INET, SOCK_STREAM, 0);
    sockaddr_in a{};
    a.sin_family = AF_INET;
    a.sin_port = htons((uint16_t)port);
    if (inet_pton(AF_INET, host, &a.sin_addr) != 1) return;
    bool ok = false;
    for (int i = 0; i < 100; ++i) {
      if (connect(sock, (sockaddr*)&a, sizeof a) == 0) {
        ok = true;
        break;
      }
      usleep(100000);
    }
    if (!ok) {
      fprintf(stderr, "[tf-client] cannot reach worker at %s\n", tcp);
      return;
    }
    int nd = 1;
    setsockopt(sock, IPPROTO_TCP, TCP_NODELAY, &nd, sizeof nd);
    uint32_t magic = TCP_MAGIC;
    if (!tcp_write_full(sock, &magic, 4) ||
        !tcp_read_full(sock, &magic, 4) || magic != TCP_MAGIC) {
      fprintf(stderr, "[tf-client] tcp handshake failed\n");
      return;
    }
    cmd = RingView(&hdr->cmd, cmd_buf(hdr), CMD_RING_BYTES);
    cpl = RingView(&hdr->cpl, cpl_buf(hdr), CPL_RING_BYTES);
    arena_base = arena(hdr);
    at(&hdr->worker_ready)->store(1, std::memory_order_release);
    tcp_mode = true;
    connected = true;
    pthread_t t1, t2;
    pthread_create(&t1, nullptr, tcp_cmd_pump, this);
    pthread_detach(t1);
    pthread_create(&t2, nullptr, tcp_cpl_pump, this);
    pthread_detach(t2);
    if (debug)
      fprintf(stderr, "[tf-client %d] connected via tcp %s\n", getpid(),
              tcp);
    return;
  }
  if (!sock_path || !*sock_path) return;

  int fd = memfd_create("tfrpc-seg", 0);
  if (fd < 0) return;
  if (ftruncate(fd, SEG_BYTES) != 0) {
    close(fd);
    return;
  }
  void* seg = mmap(nullptr, SEG_BYTES, PROT_READ | PROT_WRITE, MAP_SHARED, fd,
                   0);
  if (seg == MAP_FAILED) {
    close(fd);
    return;
  }
  memset(seg, 0, HDR_BYTES);
  hdr = reinterpret_cast<Header*>(seg);
  hdr->magic = MAGIC;
  hdr->version = VERSION;
  hdr->total_bytes = SEG_BYTES;

  sock = socket(AF_UNIX, SOCK_STREAM, 0);
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, sock_path, sizeof addr.sun_path - 1);
  // the worker may still be starting: retry briefly
  bool ok = false;
  for (int i = 0; i < 100; ++i) {
    if (connect(sock, (sockaddr*)&addr, sizeof addr) == 0) {
      ok = true;
      break;
    }
    usleep(100000);
  }
  if (!ok || !send_fd(sock, fd)) {
    fprintf(stderr, "[tf-client] cannot reach worker at %s\n", sock_path);
    close(fd);
    return;
  }
  seg_fd = fd;  // kept open: live migration re-sends it to the new worker
  strncpy(this->sock_path, sock_path, sizeof this->sock_path - 1);
  cmd = RingView(&hdr->cmd, cmd_buf(hdr), CMD_RING_BYTES);
  cpl = RingView(&hdr->cpl, cpl_buf(hdr), CPL_RING_BYTES);
  arena_base = arena(hdr);
  for (int i = 0; i < 3000; ++i) {  // wait worker_ready (30 s)
    if (at(&hdr->worker_ready)->load(std::memory_order_acquire)) {
      connected = true;
      break;
    }
    usleep(10000);
  }
  if (!connected) {
    fprintf(stderr, "[tf-client] worker never became ready\n");
    return;
  }
  // NOTE: must not call cdbg()/C() here — this runs inside the function-local
  // static's guarded construction; re-entering C() throws recursive_init_error.
  if (debug)
    fprintf(stderr, "[tf-client %d] connected via %s\n", getpid(), sock_path);
  // watcher: when the worker exits (live migration), re-handshake with its
  // successor on the same socket path — the shared segment carries all
  // in-flight protocol state, so the app never notices beyond a pause.
  pthread_t t;
  pthread_create(&t, nullptr, reconnect_main, this);
  pthread_detach(t);
}

// ------------------------------------------------------------ transport

// mu must be held.
uint8_t* reserve_locked(Client& c, size_t len) {
  uint8_t* p;
  while (!(p = c.cmd.try_reserve(len))) {
    // ring full: the worker is behind; give it a moment
    c.cmd.wake_consumer();
    usleep(20);
  }
  return p;
}

void send_async(uint32_t op, uint32_t flags, const void* body, size_t blen,
                const void* tail = nullptr, size_t tlen = 0) {
  Client& c = C();
  std::lock_guard<std::mutex> l(c.mu);
  uint8_t* p = reserve_locked(c, sizeof(CmdHdr) + blen + tlen);
  auto* h = reinterpret_cast<CmdHdr*>(p);
  h->op = op;
  h->flags = flags;
  h->seq = c.next_seq++;
  h->body_len = (uint32_t)(blen + tlen);
  if (blen) memcpy(p + sizeof(CmdHdr), body, blen);
  if (tlen) memcpy(p + sizeof(CmdHdr) + blen, tail, tlen);
  c.cmd.commit();
  c.cmd.wake_consumer();
}

// Round trip. Returns err; reply payload copied into out (up to out_cap).
hipError_t send_sync(uint32_t op, const void* body, size_t blen, void* out,
                     size_t out_cap, size_t* out_len = nullptr,
                     const void* tail = nullptr, size_t tlen = 0) {
  Client& c = C();
  if (!c.connected) return hipErrorNotInitialized;
  std::unique_lock<std::mutex> l(c.mu);
  uint64_t seq = c.next_seq++;
  {
    uint8_t* p = reserve_locked(c, sizeof(CmdHdr) + blen + tlen);
    auto* h = reinterpret_cast<CmdHdr*>(p);
    h->op = op;
    h->flags = F_WANT_REPLY;
    h->seq = seq;
    h->body_len = (uint32_t)(blen + tlen);
    if (blen) memcpy(p + sizeof(CmdHdr), body, blen);
    if (tlen) memcpy(p + sizeof(CmdHdr) + blen, tail, tlen);
    c.cmd.commit();
    c.cmd.wake_consumer();
  }
  // wait for OUR completion (single outstanding sync per client: mu held)
  for (;;) {
    size_t len;
    uint8_t* p = c.cpl.try_next(&len);
    if (!p) {
      // spin then park on the cpl futex
      for (int i = 0; i < 2000; ++i) {
        p = c.cpl.try_next(&len);
        if (p) break;
#if defined(__x86_64__)
        __builtin_ia32_pause();
#endif
      }
      if (!p) {
        at(&c.hdr->futex_cpl)->exchange(0, std::memory_order_acq_rel);
        p = c.cpl.try_next(&len);
        if (!p) {
          futex_wait(&c.hdr->futex_cpl, 0, 200);
          continue;
        }
      }
    }
    auto* r = reinterpret_cast<CplHdr*>(p);
    hipError_t err = r->err;
    if (r->seq != seq) {
      // stale completion (shouldn't happen); drop it
      c.cpl.pop();
      continue;
    }
    size_t n = r->body_len;
    if (out && n) memcpy(out, p + sizeof(CplHdr), n < out_cap ? n : out_cap);
    if (out_len) *out_len = n;
    c.cpl.pop();
    return err;
  }
}

// ------------------------------------------------------------ arena

// Allocate `len` bytes in the arena; blocks while full. Returns the
// free-running offset. mu NOT required.
uint64_t arena_alloc(size_t len, uint8_t** ptr_out) {
  Client& c = C();
  len = (len + 63) & ~size_t(63);
  std::lock_guard<std::mutex> l(c.mu);
  for (;;) {
    uint64_t head = at(&c.hdr->arena_head)->load(std::memory_order_relaxed);
    uint64_t freed = at(&c.hdr->arena_freed)->load(std::memory_order_acquire);
    uint64_t off = head % ARENA_BYTES;
    uint64_t to_end = ARENA_BYTES - off;
    uint64_t eff_head = head, eff_len = len;
    if (to_end < len) {  // skip to start (no wraparound chunks)
      eff_head = head + to_end;
      eff_len = len + to_end;
    }
    if (eff_head + len - freed <= ARENA_BYTES) {
      at(&c.hdr->arena_head)
          ->store(eff_head + len, std::memory_order_release);
      *ptr_out = c.arena_base + (eff_head % ARENA_BYTES);
      return eff_head;
    }
    usleep(50);  // arena full: worker still consuming
  }
}

// ----------------------------------------------------------- ptr classes

bool is_device_ptr(const void* p) {
  Client& c = C();
  std::lock_guard<std::mutex> l(c.mem_mu);
  auto it = c.dev_ranges.upper_bound((uint64_t)p);
  if (it == c.dev_ranges.begin()) return false;
  --it;
  return (uint64_t)p < it->first + it->second;
}

// kind: 0=H2H 1=H2D 2=D2H 3=D2D 4=Default
int resolve_kind(void* dst, const void* src, int kind) {
  if (kind != 4) return kind;
  bool d_dev = is_device_ptr(dst), s_dev = is_device_ptr(src);
  return d_dev ? (s_dev ? 3 : 1) : (s_dev ? 2 : 0);
}

// ------------------------------------------------------------- kernels

uint64_t ship_image(FatBin* fb) {
  Client& c = C();
  if (fb->module) return fb->module;
  // push bytes through the arena in one chunk (grow-safe: chunked)
  size_t remaining = fb->size;
  const uint8_t* src = (const uint8_t*)fb->bundle;
  if (fb->size <= ARENA_BYTES / 2) {
    uint8_t* dst;
    uint64_t off = arena_alloc(fb->size, &dst);
    memcpy(dst, src, fb->size);
    struct {
      uint64_t image_id, size, arena_off;
    } b{fb->image_id, fb->size, off};
    uint64_t mod = 0;
    hipError_t e = send_sync(OP_LOAD_MODULE, &b, sizeof b, &mod, 8);
    if (e != hipSuccess || !mod) {
      fprintf(stderr, "[tf-client] module load failed (err %d, %zu bytes)\n",
              e, fb->size);
      return 0;
    }
    fb->module = mod;
    return mod;
  }
  // very large images: stage into a worker-side device... not needed yet.
  fprintf(stderr, "[tf-client] image %zu B exceeds arena staging\n",
          remaining);
  return 0;
}

KernLayout* fetch_layout(FatBin* fb, const char* name,
                         hipError_t* err_out = nullptr) {
  Client& c = C();
  uint64_t mod = ship_image(fb);
  if (!mod) {
    if (err_out) *err_out = hipErrorInvalidValue;
    return nullptr;
  }
  struct {
    uint64_t image_id, module;
  } b{fb->image_id, mod};
  std::vector<uint8_t> out(20 + 8 * 256);
  size_t out_len = 0;
  size_t nlen = strlen(name) + 1;
  hipError_t e = send_sync(OP_GET_FUNCTION, &b, sizeof b, out.data(),
                           out.size(), &out_len, name, nlen);
  if (e != hipSuccess || out_len < 20) {
    // 500 = hipErrorNotFound: NORMAL control flow — Tensile probes its
    // lazily-loaded code objects for each kernel. Stay quiet.
    if (e != 500)
      fprintf(stderr, "[tf-client] GET_FUNCTION %s failed: %d\n", name, e);
    if (err_out) *err_out = e ? e : hipErrorInvalidValue;
    return nullptr;
  }
  auto* kl = new KernLayout();
  uint32_t ks, na;
  memcpy(&kl->func, out.data(), 8);
  memcpy(&ks, out.data() + 8, 4);
  memcpy(&kl->explicit_bytes, out.data() + 12, 4);
  memcpy(&na, out.data() + 16, 4);
  for (uint32_t i = 0; i < na && 20 + 8 * i + 8 <= out_len; ++i) {
    uint32_t sz, off;
    memcpy(&sz, out.data() + 20 + 8 * i, 4);
    memcpy(&off, out.data() + 24 + 8 * i, 4);
    kl->args.emplace_back(sz, off);
  }
  return kl;
}

hipError_t launch_packed(uint64_t func, dim3u grid, dim3u block,
                         const void* kernarg, uint32_t kernarg_size,
                         size_t shmem, void* stream) {
  LaunchBody lb{};
  lb.func = func;
  lb.grid[0] = grid.x;
  lb.grid[1] = grid.y;
  lb.grid[2] = grid.z;
  lb.block[0] = block.x;
  lb.block[1] = block.y;
  lb.block[2] = block.z;
  lb.shmem = (uint32_t)shmem;
  lb.stream = (uint64_t)stream;
  lb.kernarg_size = kernarg_size;
  send_async(OP_LAUNCH, 0, &lb, sizeof lb, kernarg, kernarg_size);
  return hipSuccess;
}

}  // namespace

// =======================================================================
// interposed HIP API
// =======================================================================

extern "C" {

// ------------------------------------------------------- init / devices

hipError_t hipInit(unsigned) {
  return C().connected ? hipSuccess : hipErrorNotInitialized;
}

hipError_t hipGetDeviceCount(int* n) {
  Client& c = C();
  if (!c.connected) {
    *n = 0;
    return hipSuccess;
  }
  if (!c.device_count) {
    int32_t cnt = 0;
    if (send_sync(OP_HELLO, nullptr, 0, &cnt, 4) == hipSuccess)
      c.device_count = cnt;
  }
  *n = c.device_count;
  return hipSuccess;
}

hipError_t hipGetDevice(int* d) {
  *d = Client::cur_device;
  return hipSuccess;
}

hipError_t hipSetDevice(int d) {
  Client& c = C();
  int n = 0;
  hipGetDeviceCount(&n);
  if (d < 0 || d >= n) return hipErrorInvalidDevice;
  if (Client::cur_device != d) {
    Client::cur_device = d;
    send_async(OP_SET_DEVICE, 0, &d, 4);
  }
  return hipSuccess;
}

hipError_t hipGetDevicePropertiesR0600(void* props, int dev) {
  Client& c = C();
  {
    std::lock_guard<std::mutex> l(c.cache_mu);
    auto it = c.props_cache.find(dev);
    if (it != c.props_cache.end()) {
      memcpy(props, it->second.data(), it->second.size());
      return hipSuccess;
    }
  }
  std::vector<uint8_t> buf(1472);
  hipError_t e = send_sync(OP_GET_PROPS, &dev, 4, buf.data(), buf.size());
  if (e == hipSuccess) {
    memcpy(props, buf.data(), buf.size());
    std::lock_guard<std::mutex> l(c.cache_mu);
    c.props_cache[dev] = std::move(buf);
  }
  return e;
}

hipError_t hipGetDeviceProperties(void* props, int dev) {
  return hipGetDevicePropertiesR0600(props, dev);
}

hipError_t hipDeviceGetAttribute(int* value, int attr, int dev) {
  Client& c = C();
  {
    std::lock_guard<std::mutex> l(c.cache_mu);
    auto it = c.attr_cache.find({dev, attr});
    if (it != c.attr_cache.end()) {
      *value = it->second;
      return hipSuccess;
    }
  }
  struct {
    int dev, attr;
  } b{dev, attr};
  int v = 0;
  hipError_t e = send_sync(OP_GET_ATTRIBUTE, &b, sizeof b, &v, 4);
  if (e == hipSuccess) {
    std::lock_guard<std::mutex> l(c.cache_mu);
    c.attr_cache[{dev, attr}] = v;
    *value = v;
  }
  return e;
}

hipError_t hipDeviceCanAccessPeer(int* can, int dev, int peer) {
  struct {
    int dev, peer;
  } b{dev, peer};
  return send_sync(OP_CAN_ACCESS_PEER, &b, sizeof b, can, 4);
}

hipError_t hipDeviceSynchronize(void) {
  hipError_t e = send_sync(OP_DEVICE_SYNC, nullptr, 0, nullptr, 0);
  if (e == hipSuccess) {
    // surface async launch/copy failures at sync points (CUDA semantics)
    e = (hipError_t)at(&C().hdr->sticky_error)->exchange(0);
  }
  return e;
}

hipError_t hipDriverGetVersion(int* v) {
  *v = 70200000;
  return hipSuccess;
}
hipError_t hipRuntimeGetVersion(int* v) {
  *v = 70200000;
  return hipSuccess;
}
hipError_t hipGetDeviceFlags(unsigned* f) {
  *f = 0;
  return hipSuccess;
}
hipError_t hipSetDeviceFlags(unsigned) { return hipSuccess; }
hipError_t hipDeviceGetStreamPriorityRange(int* lo, int* hi) {
  if (lo) *lo = 0;
  if (hi) *hi = -2;  // ROCm convention: higher priority = more negative
  return hipSuccess;
}

hipError_t hipGetLastError(void) {
  Client& c = C();
  hipError_t sticky = (hipError_t)at(&c.hdr->sticky_error)->exchange(0);
  hipError_t e = c.last_error ? c.last_error : sticky;
  c.last_error = hipSuccess;
  return e;
}

hipError_t hipPeekAtLastError(void) {
  Client& c = C();
  return c.last_error ? c.last_error
                      : (hipError_t)at(&c.hdr->sticky_error)->load();
}

// ------------------------------------------------------------- memory

hipError_t hipMalloc(void** p, size_t sz) {
  Client& c = C();
  if (sz == 0) {
    *p = nullptr;
    return hipSuccess;
  }
  uint64_t ptr = 0;
  hipError_t e = send_sync(OP_MALLOC, &sz, sizeof sz, &ptr, 8);
  if (e == hipSuccess && ptr) {
    *p = (void*)ptr;
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges[ptr] = sz;
  } else if (e == hipSuccess) {
    e = hipErrorOutOfMemory;
  }
  return e;
}

hipError_t hipMallocAsync(void** p, size_t sz, void*) {
  return hipMalloc(p, sz);
}

hipError_t hipFree(void* p) {
  Client& c = C();
  if (!p) return hipSuccess;
  {
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges.erase((uint64_t)p);
  }
  uint64_t v = (uint64_t)p;
  // hipFree has implicit-sync semantics; keep it synchronous
  return send_sync(OP_FREE, &v, 8, nullptr, 0);
}

hipError_t hipFreeAsync(void* p, void*) {
  Client& c = C();
  if (!p) return hipSuccess;
  {
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges.erase((uint64_t)p);
  }
  uint64_t v = (uint64_t)p;
  send_async(OP_FREE, 0, &v, 8);
  return hipSuccess;
}

hipError_t hipMemGetInfo(size_t* free_b, size_t* total_b) {
  uint64_t r[2] = {0, 0};
  hipError_t e = send_sync(OP_MEM_GET_INFO, nullptr, 0, r, 16);
  if (free_b) *free_b = r[0];
  if (total_b) *total_b = r[1];
  return e;
}

hipError_t hipHostMalloc(void** p, size_t sz, unsigned) {
  if (posix_memalign(p, 4096, sz ? sz : 1) != 0) return hipErrorOutOfMemory;
  Client& c = C();
  std::lock_guard<std::mutex> l(c.mem_mu);
  c.host_allocs[(uint64_t)*p] = sz;
  return hipSuccess;
}
hipError_t hipMallocHost(void** p, size_t sz) {
  return hipHostMalloc(p, sz, 0);
}
hipError_t hipHostFree(void* p) {
  Client& c = C();
  {
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.host_allocs.erase((uint64_t)p);
  }
  free(p);
  return hipSuccess;
}
hipError_t hipFreeHost(void* p) { return hipHostFree(p); }
hipError_t hipHostRegister(void*, size_t, unsigned) { return hipSuccess; }
hipError_t hipHostUnregister(void*) { return hipSuccess; }

// hipPointerGetAttributes: torch uses it to classify pointers.
struct hipPointerAttribute_small {
  int type;  // 0 unregistered 1 host 2 device 3 managed
  int device;
  void* devicePointer;
  void* hostPointer;
  int isManaged;
  unsigned allocationFlags;
};

hipError_t hipPointerGetAttributes(hipPointerAttribute_small* attr,
                                   const void* ptr) {
  memset(attr, 0, sizeof *attr);
  if (is_device_ptr(ptr)) {
    attr->type = 2;
    attr->device = Client::cur_device;
    attr->devicePointer = const_cast<void*>(ptr);
    return hipSuccess;
  }
  Client& c = C();
  std::lock_guard<std::mutex> l(c.mem_mu);
  auto it = c.host_allocs.upper_bound((uint64_t)ptr);
  if (it != c.host_allocs.begin()) {
    --it;
    if ((uint64_t)ptr < it->first + it->second) {
      attr->type = 1;
      attr->device = -1;
      attr->hostPointer = const_cast<void*>(ptr);
      return hipSuccess;
    }
  }
  attr->type = 0;
  attr->device = -1;
  return hipErrorInvalidValue;
}

// ------------------------------------------------------------- memcpy

static hipError_t do_memcpy(void* dst, const void* src, size_t n, int kind,
                            void* stream, bool sync) {
  Client& c = C();
  if (n == 0) return hipSuccess;
  kind = resolve_kind(dst, src, kind);
  switch (kind) {
    case 0:  // H2H
      memmove(dst, src, n);
      return hipSuccess;
    case 1: {  // H2D
      const size_t INLINE_MAX = 32 << 10;
      if (n <= INLINE_MAX) {
        MemcpyBody b{};
        b.dst = (uint64_t)dst;
        b.size = n;
        b.stream = (uint64_t)stream;
        b.kind = 1;
        send_async(OP_MEMCPY_H2D, F_INLINE_DATA, &b, sizeof b, src, n);
        return hipSuccess;
      }
      size_t done = 0;
      while (done < n) {
        size_t chunk = n - done;
        if (chunk > ARENA_BYTES / 4) chunk = ARENA_BYTES / 4;
        uint8_t* stage;
        uint64_t off = arena_alloc(chunk, &stage);
        memcpy(stage, (const uint8_t*)src + done, chunk);
        MemcpyBody b{};
        b.dst = (uint64_t)dst + done;
        b.size = chunk;
        b.arena_off = off;
        b.stream = (uint64_t)stream;
        b.kind = 1;
        send_async(OP_MEMCPY_H2D, 0, &b, sizeof b);
        done += chunk;
      }
      if (sync)
        return send_sync(OP_STREAM_SYNC, &stream, 8, nullptr, 0);
      return hipSuccess;
    }
    case 2: {  // D2H: inherently needs the data — always a round trip
      if (c.tcp_mode) {
        // cross-node: data rides inside reply bodies; chunk to fit the
        // completion ring
        size_t done = 0;
        while (done < n) {
          size_t chunk = n - done;
          if (chunk > CPL_RING_BYTES / 4) chunk = CPL_RING_BYTES / 4;
          MemcpyBody b{};
          b.src = (uint64_t)src + done;
          b.size = chunk;
          b.stream = (uint64_t)stream;
          b.kind = 2;
          size_t got = 0;
          hipError_t e = send_sync(OP_MEMCPY_D2H, &b, sizeof b,
                                   (uint8_t*)dst + done, chunk, &got);
          if (e != hipSuccess) return e;
          if (got != chunk) return hipErrorInvalidValue;
          done += chunk;
        }
        return hipSuccess;
      }
      size_t done = 0;
      while (done < n) {
        size_t chunk = n - done;
        if (chunk > ARENA_BYTES / 4) chunk = ARENA_BYTES / 4;
        uint8_t* stage;
        uint64_t off = arena_alloc(chunk, &stage);
        MemcpyBody b{};
        b.src = (uint64_t)src + done;
        b.size = chunk;
        b.arena_off = off;
        b.stream = (uint64_t)stream;
        b.kind = 2;
        hipError_t e = send_sync(OP_MEMCPY_D2H, &b, sizeof b, nullptr, 0);
        if (e != hipSuccess) return e;
        memcpy((uint8_t*)dst + done, stage, chunk);
        // release the staging chunk
        at(&c.hdr->arena_freed)->store(off + ((chunk + 63) & ~size_t(63)),
                                       std::memory_order_release);
        done += chunk;
      }
      return hipSuccess;
    }
    case 3: {  // D2D
      MemcpyBody b{};
      b.dst = (uint64_t)dst;
      b.src = (uint64_t)src;
      b.size = n;
      b.stream = (uint64_t)stream;
      b.kind = 3;
      if (sync) {
        b.sync = 1;
        return send_sync(OP_MEMCPY_D2D, &b, sizeof b, nullptr, 0);
      }
      send_async(OP_MEMCPY_D2D, 0, &b, sizeof b);
      return hipSuccess;
    }
  }
  return hipErrorInvalidValue;
}

hipError_t hipMemcpy(void* dst, const void* src, size_t n, int kind) {
  hipError_t e = do_memcpy(dst, src, n, kind, nullptr, true);
  if (e == hipSuccess && (kind == 1 || kind == 4)) {
    void* s0 = nullptr;
    e = send_sync(OP_STREAM_SYNC, &s0, 8, nullptr, 0);
  }
  return e;
}

hipError_t hipMemcpyAsync(void* dst, const void* src, size_t n, int kind,
                          void* stream) {
  return do_memcpy(dst, src, n, kind, stream, false);
}

hipError_t hipMemcpyWithStream(void* dst, const void* src, size_t n, int kind,
                               void* stream) {
  hipError_t e = do_memcpy(dst, src, n, kind, stream, true);
  if (e == hipSuccess)
    e = send_sync(OP_STREAM_SYNC, &stream, 8, nullptr, 0);
  return e;
}

hipError_t hipMemcpyHtoD(void* dst, void* src, size_t n) {
  return hipMemcpy(dst, src, n, 1);
}
hipError_t hipMemcpyDtoH(void* dst, void* src, size_t n) {
  return hipMemcpy(dst, src, n, 2);
}
hipError_t hipMemcpyDtoD(void* dst, void* src, size_t n) {
  return hipMemcpy(dst, src, n, 3);
}
hipError_t hipMemcpyHtoDAsync(void* dst, void* src, size_t n, void* s) {
  return hipMemcpyAsync(dst, src, n, 1, s);
}
hipError_t hipMemcpyDtoHAsync(void* dst, void* src, size_t n, void* s) {
  return hipMemcpyAsync(dst, src, n, 2, s);
}

static hipError_t do_memset(void* dst, int value, size_t count, void* stream,
                            bool sync) {
  MemcpyBody b{};
  b.dst = (uint64_t)dst;
  b.size = count;
  b.stream = (uint64_t)stream;
  b.kind = (uint32_t)(unsigned char)value;
  if (sync) {
    b.sync = 1;
    return send_sync(OP_MEMSET, &b, sizeof b, nullptr, 0);
  }
  send_async(OP_MEMSET, 0, &b, sizeof b);
  return hipSuccess;
}

hipError_t hipMemset(void* dst, int v, size_t n) {
  hipError_t e = do_memset(dst, v, n, nullptr, true);
  return e;
}
hipError_t hipMemsetAsync(void* dst, int v, size_t n, void* s) {
  return do_memset(dst, v, n, s, false);
}
hipError_t hipMemsetD8(void* dst, unsigned char v, size_t n) {
  return hipMemset(dst, v, n);
}
hipError_t hipMemsetD8Async(void* dst, unsigned char v, size_t n, void* s) {
  return hipMemsetAsync(dst, v, n, s);
}
hipError_t hipMemsetD32(void* dst, int v, size_t n) {
  // byte-pattern only when all bytes equal; torch uses 0
  unsigned char b0 = (unsigned)v & 0xff;
  if (v == (int)(b0 * 0x01010101u)) return hipMemset(dst, b0, n * 4);
  return hipErrorNotSupported;
}
hipError_t hipMemsetD32Async(void* dst, int v, size_t n, void* s) {
  unsigned char b0 = (unsigned)v & 0xff;
  if (v == (int)(b0 * 0x01010101u)) return hipMemsetAsync(dst, b0, n * 4, s);
  return hipErrorNotSupported;
}

// ------------------------------------------------------------- streams

hipError_t hipStreamCreateWithPriority(void** stream, unsigned flags,
                                       int prio) {
  struct {
    uint32_t flags;
    int prio;
  } b{flags, prio};
  uint64_t st = 0;
  hipError_t e = send_sync(OP_STREAM_CREATE, &b, sizeof b, &st, 8);
  if (e == hipSuccess) *stream = (void*)st;
  return e;
}
hipError_t hipStreamCreateWithFlags(void** s, unsigned f) {
  return hipStreamCreateWithPriority(s, f, 0);
}
hipError_t hipStreamCreate(void** s) {
  return hipStreamCreateWithPriority(s, 0, 0);
}
hipError_t hipStreamDestroy(void* s) {
  uint64_t v = (uint64_t)s;
  send_async(OP_STREAM_DESTROY, 0, &v, 8);
  return hipSuccess;
}
hipError_t hipStreamSynchronize(void* s) {
  hipError_t e = send_sync(OP_STREAM_SYNC, &s, 8, nullptr, 0);
  if (e == hipSuccess)
    e = (hipError_t)at(&C().hdr->sticky_error)->exchange(0);
  return e;
}
hipError_t hipStreamQuery(void* s) {
  return send_sync(OP_STREAM_QUERY, &s, 8, nullptr, 0);
}
hipError_t hipStreamWaitEvent(void* s, void* ev, unsigned flags) {
  struct {
    uint64_t s, ev;
    uint32_t flags;
  } b{(uint64_t)s, (uint64_t)ev, flags};
  send_async(OP_STREAM_WAIT_EVENT, 0, &b, sizeof b);
  return hipSuccess;
}
hipError_t hipStreamIsCapturing(void* stream, int* status) {
  if (status)
    *status = stream_capturing(stream, nullptr) ? 1 : 0;
  return hipSuccess;
}
hipError_t hipStreamGetCaptureInfo(void* stream, int* status,
                                   unsigned long long* id) {
  if (status) *status = stream_capturing(stream, id) ? 1 : 0;
  if (id) *id = 0;
  return hipSuccess;
}
hipError_t hipStreamGetCaptureInfo_v2(void* stream, int* status,
                                      unsigned long long* id, void** g,
                                      const void*** deps, size_t* ndeps) {
  if (status) *status = stream_capturing(stream, id) ? 1 : 0;
  if (g) *g = nullptr;
  if (deps) *deps = nullptr;
  if (ndeps) *ndeps = 0;
  return hipSuccess;
}
hipError_t hipStreamGetPriority(void*, int* p) {
  if (p) *p = 0;
  return hipSuccess;
}
hipError_t hipStreamGetFlags(void*, unsigned* f) {
  if (f) *f = 0;
  return hipSuccess;
}

// ------------------------------------------------------------- events

hipError_t hipEventCreateWithFlags(void** ev, unsigned flags) {
  uint64_t v = 0;
  hipError_t e = send_sync(OP_EVENT_CREATE, &flags, 4, &v, 8);
  if (e == hipSuccess) *ev = (void*)v;
  return e;
}
hipError_t hipEventCreate(void** ev) { return hipEventCreateWithFlags(ev, 0); }
hipError_t hipEventRecord(void* ev, void* s) {
  struct {
    uint64_t ev, s;
  } b{(uint64_t)ev, (uint64_t)s};
  send_async(OP_EVENT_RECORD, 0, &b, sizeof b);
  return hipSuccess;
}
hipError_t hipEventSynchronize(void* ev) {
  uint64_t v = (uint64_t)ev;
  return send_sync(OP_EVENT_SYNC, &v, 8, nullptr, 0);
}
hipError_t hipEventQuery(void* ev) {
  uint64_t v = (uint64_t)ev;
  return send_sync(OP_EVENT_QUERY, &v, 8, nullptr, 0);
}
hipError_t hipEventElapsedTime(float* ms, void* e0, void* e1) {
  struct {
    uint64_t a, b;
  } b{(uint64_t)e0, (uint64_t)e1};
  return send_sync(OP_EVENT_ELAPSED, &b, sizeof b, ms, 4);
}
hipError_t hipEventDestroy(void* ev) {
  uint64_t v = (uint64_t)ev;
  send_async(OP_EVENT_DESTROY, 0, &v, 8);
  return hipSuccess;
}

// ------------------------------------------------- fatbin registration

struct FatBinWrapper {
  uint32_t magic;
  uint32_t version;
  const void* binary;
  void* unused;
};

void** __hipRegisterFatBinary(const void* data) {
  Client& c = C();
  auto* w = reinterpret_cast<const FatBinWrapper*>(data);
  auto* fb = new FatBin();
  fb->bundle = w ? w->binary : data;
  std::lock_guard<std::mutex> l(c.reg_mu);
  fb->image_id = c.next_image_id++;
  // size: for CCOB, total size = header field; for plain bundle, compute
  // from entries; defer to ship time (we size it there).
  auto** handle = new void*[1];
  handle[0] = fb;
  c.fatbins[handle] = fb;
  return handle;
}

void __hipUnregisterFatBinary(void** handle) {
  Client& c = C();
  std::lock_guard<std::mutex> l(c.reg_mu);
  c.fatbins.erase(handle);
}

void __hipRegisterFunction(void** handle, const void* hostFun, char*,
                           const char* deviceName, unsigned, void*, void*,
                           void*, void*, void*) {
  Client& c = C();
  std::lock_guard<std::mutex> l(c.reg_mu);
  auto it = c.fatbins.find(handle);
  if (it == c.fatbins.end()) return;
  c.stubs[hostFun] = {it->second, deviceName};
}

void __hipRegisterVar(void** handle, void* var, char*, char* deviceVar,
                      int, size_t, int, int) {
  Client& c = C();
  std::lock_guard<std::mutex> l(c.reg_mu);
  auto it = c.fatbins.find(handle);
  if (it == c.fatbins.end()) return;
  c.dvars[var] = {it->second, deviceVar};
}
void __hipRegisterManagedVar(void**, void**, void*, const char*, size_t,
                             unsigned) {}
void __hipRegisterSurface(void**, void*, char*, char*, int, int) {}
void __hipRegisterTexture(void**, void*, char*, char*, int, int, int) {}

// ------------------------------------------------------------- launch

static size_t bundle_size(const void* image);

hipError_t hipLaunchKernel(const void* func, dim3u grid, dim3u block,
                           void** args, size_t shmem, void* stream) {
  Client& c = C();
  KernLayout* kl = nullptr;
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    auto it = c.launch_cache.find(func);
    if (it != c.launch_cache.end()) kl = it->second;
  }
  if (!kl) {
    FatBin* fb = nullptr;
    std::string name;
    {
      std::lock_guard<std::mutex> l(c.reg_mu);
      auto it = c.stubs.find(func);
      if (it == c.stubs.end()) {
        fprintf(stderr, "[tf-client] launch of unregistered stub %p\n", func);
        return hipErrorInvalidValue;
      }
      fb = it->second.first;
      name = it->second.second;
      if (!fb->size) fb->size = bundle_size(fb->bundle);
    }
    kl = fetch_layout(fb, name.c_str());
    if (!kl) return hipErrorInvalidValue;
    std::lock_guard<std::mutex> l(c.reg_mu);
    c.launch_cache[func] = kl;
  }
  // pack explicit args
  uint8_t buf[4096];
  uint32_t total = kl->explicit_bytes;
  if (total > sizeof buf) return hipErrorInvalidValue;
  memset(buf, 0, total);
  for (size_t i = 0; i < kl->args.size(); ++i) {
    auto [sz, off] = kl->args[i];
    if (off + sz <= total && args && args[i]) memcpy(buf + off, args[i], sz);
  }
  return launch_packed(kl->func, grid, block, buf, total, shmem, stream);
}

hipError_t hipLaunchKernel_spt(const void* f, dim3u g, dim3u b, void** a,
                               size_t s, void* st) {
  return hipLaunchKernel(f, g, b, a, s, st);
}

hipError_t hipExtLaunchKernel(const void* f, dim3u g, dim3u b, void** a,
                              size_t shmem, void* stream, void* startEv,
                              void* stopEv, int) {
  if (startEv) hipEventRecord(startEv, stream);
  hipError_t e = hipLaunchKernel(f, g, b, a, shmem, stream);
  if (stopEv) hipEventRecord(stopEv, stream);
  return e;
}

// ------------------------------------------------------------- modules

hipError_t hipModuleLoadData(void** module, const void* image) {
  Client& c = C();
  auto* fb = new FatBin();
  fb->bundle = image;
  fb->size = bundle_size(image);
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    fb->image_id = c.next_image_id++;
  }
  uint64_t mod = ship_image(fb);
  if (!mod) return hipErrorInvalidValue;
  *module = new std::pair<FatBin*, uint64_t>(fb, mod);
  return hipSuccess;
}

hipError_t hipModuleLoad(void** module, const char* path) {
  FILE* f = fopen(path, "rb");
  if (!f) return hipErrorInvalidValue;
  fseek(f, 0, SEEK_END);
  long sz = ftell(f);
  fseek(f, 0, SEEK_SET);
  char* buf = (char*)malloc(sz);
  if (fread(buf, 1, sz, f) != (size_t)sz) {
    fclose(f);
    free(buf);
    return hipErrorInvalidValue;
  }
  fclose(f);
  return hipModuleLoadData(module, buf);  // buf intentionally kept
}

hipError_t hipModuleLoadDataEx(void** module, const void* image, unsigned,
                               void**, void**) {
  return hipModuleLoadData(module, image);
}

hipError_t hipModuleGetFunction(void** fn, void* module, const char* name) {
  auto* pr = reinterpret_cast<std::pair<FatBin*, uint64_t>*>(module);
  if (!pr) return hipErrorInvalidValue;
  hipError_t err = hipSuccess;
  KernLayout* kl = fetch_layout(pr->first, name, &err);
  if (!kl) return err ? err : hipErrorInvalidValue;
  Client& c = C();
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    c.module_fn_layouts[kl->func] = kl;
  }
  *fn = (void*)kl->func;
  return hipSuccess;
}

hipError_t hipModuleUnload(void*) { return hipSuccess; }

static hipError_t module_launch(void* fn, unsigned gx, unsigned gy,
                                unsigned gz, unsigned bx, unsigned by,
                                unsigned bz, unsigned shmem, void* stream,
                                void** params, void** extra) {
  Client& c = C();
  const void* kernarg = nullptr;
  size_t ksize = 0;
  uint8_t buf[4096];
  if (extra) {
    for (int i = 0; extra[i] && extra[i] != (void*)0x03; i += 2) {
      if (extra[i] == (void*)0x01) kernarg = extra[i + 1];
      if (extra[i] == (void*)0x02) ksize = *(size_t*)extra[i + 1];
    }
  }
  if (!kernarg && params) {
    KernLayout* kl = nullptr;
    {
      std::lock_guard<std::mutex> l(c.reg_mu);
      auto it = c.module_fn_layouts.find((uint64_t)fn);
      if (it != c.module_fn_layouts.end()) kl = it->second;
    }
    if (!kl) return hipErrorInvalidValue;
    if (kl->explicit_bytes > sizeof buf) return hipErrorInvalidValue;
    memset(buf, 0, kl->explicit_bytes);
    for (size_t i = 0; i < kl->args.size(); ++i) {
      auto [sz, off] = kl->args[i];
      if (params[i]) memcpy(buf + off, params[i], sz);
    }
    kernarg = buf;
    ksize = kl->explicit_bytes;
  }
  if (!kernarg && !params) ksize = 0;
  dim3u grid{gx, gy, gz}, block{bx, by, bz};
  return launch_packed((uint64_t)fn, grid, block, kernarg, (uint32_t)ksize,
                       shmem, stream);
}

hipError_t hipModuleLaunchKernel(void* fn, unsigned gx, unsigned gy,
                                 unsigned gz, unsigned bx, unsigned by,
                                 unsigned bz, unsigned shmem, void* stream,
                                 void** params, void** extra) {
  return module_launch(fn, gx, gy, gz, bx, by, bz, shmem, stream, params,
                       extra);
}

hipError_t hipExtModuleLaunchKernel(void* fn, unsigned gwx, unsigned gwy,
                                    unsigned gwz, unsigned bx, unsigned by,
                                    unsigned bz, size_t shmem, void* stream,
                                    void** params, void** extra, void* startEv,
                                    void* stopEv, unsigned) {
  if (startEv) hipEventRecord(startEv, stream);
  // ext variant takes GLOBAL work sizes
  hipError_t e = module_launch(fn, gwx / (bx ? bx : 1), gwy / (by ? by : 1),
                               gwz / (bz ? bz : 1), bx, by, bz,
                               (unsigned)shmem, stream, params, extra);
  if (stopEv) hipEventRecord(stopEv, stream);
  return e;
}

// occupancy heuristics (no round trip; used for launch-config only)
hipError_t hipOccupancyMaxActiveBlocksPerMultiprocessor(int* n, const void*,
                                                        int blk, size_t) {
  *n = blk >= 512 ? 2 : 4;
  return hipSuccess;
}
hipError_t hipModuleOccupancyMaxActiveBlocksPerMultiprocessor(int* n, void*,
                                                              int blk,
                                                              size_t) {
  *n = blk >= 512 ? 2 : 4;
  return hipSuccess;
}

hipError_t hipKernelNameRefByPtr(const void*, void*) { return hipErrorNotSupported; }

// --------------------------------------------------------- bundle size

// Compute the in-memory size of a code object / offload bundle so it can
// be shipped: ELF (section headers), plain bundle (max entry end) or CCOB
// (totalSize field).
static size_t bundle_size(const void* image) {
  const uint8_t* p = (const uint8_t*)image;
  if (memcmp(p, "\x7f" "ELF", 4) == 0) {
    // 64-bit ELF: e_shoff + e_shnum * e_shentsize
    uint64_t shoff;
    uint16_t shentsize, shnum;
    memcpy(&shoff, p + 0x28, 8);
    memcpy(&shentsize, p + 0x3a, 2);
    memcpy(&shnum, p + 0x3c, 2);
    size_t end = shoff + (size_t)shentsize * shnum;
    // sections may precede headers; scan section ends too
    for (uint16_t i = 0; i < shnum; ++i) {
      const uint8_t* sh = p + shoff + (size_t)i * shentsize;
      uint64_t off, size;
      uint32_t type;
      memcpy(&type, sh + 0x4, 4);
      memcpy(&off, sh + 0x18, 8);
      memcpy(&size, sh + 0x20, 8);
      if (type != 8 /*NOBITS*/ && off + size > end) end = off + size;
    }
    return end;
  }
  if (memcmp(p, "CCOB", 4) == 0) {
    // Compressed offload bundle (verified against this toolchain's
    // emissions): v2 = magic(4) ver(u16) method(u16) TotalFileSize(u32)
    // UncompressedSize(u32) hash(u64); v3 = same with u64 sizes.
    uint16_t version;
    memcpy(&version, p + 4, 2);
    if (version == 2) {
      uint32_t total;
      memcpy(&total, p + 8, 4);
      return (size_t)total;
    }
    if (version >= 3) {
      uint64_t total;
      memcpy(&total, p + 8, 8);
      return (size_t)total;
    }
    fprintf(stderr, "[tf-client] unsupported CCOB v%u\n", version);
    return 0;
  }
  static const char BMAGIC[] = "__CLANG_OFFLOAD_BUNDLE__";
  if (memcmp(p, BMAGIC, 24) == 0) {
    uint64_t n;
    memcpy(&n, p + 24, 8);
    size_t off = 32, end = 32;
    for (uint64_t i = 0; i < n; ++i) {
      uint64_t eoff, esz, tlen;
      memcpy(&eoff, p + off, 8);
      memcpy(&esz, p + off + 8, 8);
      memcpy(&tlen, p + off + 16, 8);
      off += 24 + tlen;
      if (eoff + esz > end) end = eoff + esz;
    }
    return end;
  }
  return 0;
}

// ------------------------------------------------------------- misc

hipError_t hipProfilerStart(void) { return hipSuccess; }
hipError_t hipProfilerStop(void) { return hipSuccess; }
hipError_t hipDeviceSetCacheConfig(int) { return hipSuccess; }
hipError_t hipDeviceSetSharedMemConfig(int) { return hipSuccess; }
hipError_t hipDeviceGetLimit(size_t* v, int) {
  if (v) *v = 0;
  return hipSuccess;
}
hipError_t hipCtxGetCurrent(void** ctx) {
  if (ctx) *ctx = (void*)0x1;
  return hipSuccess;
}
hipError_t hipDevicePrimaryCtxRetain(void** ctx, int) {
  if (ctx) *ctx = (void*)0x1;
  return hipSuccess;
}

int tf_client_connected(void) { return C().connected ? 1 : 0; }

// --------------------------------------------------- dlopen redirection
// PyTorch's driver-API shim dlopens libamdhip64.so and dlsyms entry points
// directly, which would bypass LD_PRELOAD and hit the GPU-less local
// runtime. Redirect any dlopen of the HIP runtime to THIS library so those
// dlsyms resolve to the interposers above.

void* dlopen(const char* filename, int flags) {
  using fn_t = void* (*)(const char*, int);
  static fn_t real_dlopen = nullptr;
  if (!real_dlopen) {
    // glibc ≥2.34 moved dlopen into libc (version GLIBC_2.34); older lives
    // in libdl (GLIBC_2.2.5). dlvsym avoids recursing into any interposer.
    real_dlopen = (fn_t)dlvsym(RTLD_NEXT, "dlopen", "GLIBC_2.34");
    if (!real_dlopen)
      real_dlopen = (fn_t)dlvsym(RTLD_NEXT, "dlopen", "GLIBC_2.2.5");
    if (!real_dlopen) real_dlopen = (fn_t)dlsym(RTLD_NEXT, "dlopen");
  }
  if (filename && strstr(filename, "libamdhip64")) {
    Dl_info info;
    if (dladdr((void*)&tf_client_connected, &info) && info.dli_fname) {
      void* h = real_dlopen(info.dli_fname, flags);
      if (h) return h;
    }
  }
  return real_dlopen(filename, flags);
}

// ------------------------------------------- driver-API context symbols

hipError_t hipDevicePrimaryCtxGetState(int dev, unsigned* flags,
                                       int* active) {
  if (dev < 0 || dev >= C().device_count) return hipErrorInvalidDevice;
  if (flags) *flags = 0;
  if (active) *active = 1;  // the remote worker's context is always live
  return hipSuccess;
}

hipError_t hipDevicePrimaryCtxSetFlags(int, unsigned) { return hipSuccess; }
hipError_t hipDevicePrimaryCtxRelease(int) { return hipSuccess; }

hipError_t hipDeviceGet(int* dev, int ordinal) {
  if (!dev) return hipErrorInvalidValue;
  if (ordinal < 0 || ordinal >= C().device_count)
    return hipErrorInvalidDevice;
  *dev = ordinal;
  return hipSuccess;
}

// ------------------------------------------------- call configuration
// <<<>>> launches compile to __hipPushCallConfiguration + stub +
// __hipPopCallConfiguration + hipLaunchKernel. These MUST be interposed:
// the real runtime's implementations fail (or return garbage) in a
// GPU-less process, which silently corrupts every launch config.

struct CallConfig {
  dim3u grid, block;
  size_t shmem;
  void* stream;
};

static thread_local std::vector<CallConfig> tls_call_stack;

hipError_t __hipPushCallConfiguration(dim3u grid, dim3u block, size_t shmem,
                                      void* stream) {
  tls_call_stack.push_back({grid, block, shmem, stream});
  return hipSuccess;
}

hipError_t __hipPopCallConfiguration(dim3u* grid, dim3u* block, size_t* shmem,
                                     void** stream) {
  if (tls_call_stack.empty()) return hipErrorInvalidValue;
  CallConfig c = tls_call_stack.back();
  tls_call_stack.pop_back();
  if (grid) *grid = c.grid;
  if (block) *block = c.block;
  if (shmem) *shmem = c.shmem;
  if (stream) *stream = c.stream;
  return hipSuccess;
}

// ------------------------------------------------- small local shims

const char* hipGetErrorName(hipError_t e) {
  switch (e) {
    case 0: return "hipSuccess";
    case 1: return "hipErrorInvalidValue";
    case 2: return "hipErrorOutOfMemory";
    case 3: return "hipErrorNotInitialized";
    case 98: return "hipErrorInvalidDeviceFunction";
    case 100: return "hipErrorNoDevice";
    case 101: return "hipErrorInvalidDevice";
    case 600: return "hipErrorNotReady";
    case 801: return "hipErrorNotSupported";
    default: return "hipErrorUnknown(remote)";
  }
}

const char* hipGetErrorString(hipError_t e) { return hipGetErrorName(e); }

hipError_t hipDeviceGetPCIBusId(char* id, int len, int dev) {
  if (!id || len < 13) return hipErrorInvalidValue;
  snprintf(id, len, "0000:%02x:00.0", dev & 0xff);
  return hipSuccess;
}

hipError_t hipGetStreamDeviceId(void*) { return Client::cur_device; }

hipError_t hipStreamGetDevice(void*, int* device) {
  // aotriton gates flash attention on this: falling through to the
  // GPU-less local runtime made SDPA silently use the math path remotely
  if (!device) return hipErrorInvalidValue;
  *device = Client::cur_device;
  return hipSuccess;
}

struct hipFuncAttributes_small {
  int binaryVersion, cacheModeCA;
  size_t constSizeBytes, localSizeBytes;
  int maxDynamicSharedSizeBytes, maxThreadsPerBlock, numRegs;
  int preferredShmemCarveout, ptxVersion;
  size_t sharedSizeBytes;
};

hipError_t hipFuncGetAttributes(hipFuncAttributes_small* a, const void*) {
  if (!a) return hipErrorInvalidValue;
  memset(a, 0, sizeof *a);
  a->maxThreadsPerBlock = 1024;
  a->maxDynamicSharedSizeBytes = 160 * 1024 - 16;  // CDNA4 LDS
  a->numRegs = 64;
  a->binaryVersion = 950;
  return hipSuccess;
}

hipError_t hipOccupancyMaxPotentialBlockSize(int* minGrid, int* blockSize,
                                             const void*, size_t, int) {
  if (minGrid) *minGrid = 2048;  // 256 CUs × 8 blocks
  if (blockSize) *blockSize = 256;
  return hipSuccess;
}

hipError_t hipThreadExchangeStreamCaptureMode(int* mode) {
  if (mode) *mode = 0;
  return hipSuccess;
}

hipError_t hipDeviceEnablePeerAccess(int, unsigned) {
  return hipErrorNotSupported;  // one vGPU per worker; RCCL rides xGMI
}

hipError_t hipStreamCreateWithFlags(void**, unsigned);  // fwd decl (above)

hipError_t hipExtStreamCreateWithCUMask(void** stream, unsigned,
                                        const unsigned*) {
  // CU masking of a remote vGPU is enforced worker-side by the hypervisor
  // (HSA_CU_MASK on the worker process); the client gets a normal stream.
  return hipStreamCreateWithFlags(stream, 0);
}

hipError_t hipExtStreamGetCUMask(void*, unsigned n, unsigned* mask) {
  if (!mask) return hipErrorInvalidValue;
  for (unsigned i = 0; i < n; ++i) mask[i] = 0xffffffffu;
  return hipSuccess;
}

// not-yet-supported surfaces: fail loudly rather than run locally
// ---- client-side VMM surface (forwarded): what PyTorch's
// expandable_segments allocator and vLLM-class servers drive. VAs and
// physical handles are WORKER-side values; a mapped range registers in
// dev_ranges so memcpy/memset route it like any device pointer.

extern "C" hipError_t hipMemAddressReserve(void** ptr, size_t size,
                                           size_t align, void* hint,
                                           unsigned long long flags) {
  struct {
    uint64_t size, align, hint, flags;
  } b{size, align, (uint64_t)hint, flags};
  uint64_t out = 0;
  hipError_t e = send_sync(OP_VMM_RESERVE, &b, sizeof b, &out, 8);
  if (e == hipSuccess) *ptr = (void*)out;
  return e;
}

extern "C" hipError_t hipMemAddressFree(void* ptr, size_t size) {
  struct {
    uint64_t ptr, size;
  } b{(uint64_t)ptr, size};
  send_async(OP_VMM_ADDR_FREE, 0, &b, sizeof b);
  return hipSuccess;
}

extern "C" hipError_t hipMemCreate(void* handle, size_t size,
                                   const void* prop,
                                   unsigned long long flags) {
  struct {
    uint64_t size, flags;
    uint8_t prop[48];
  } b{};
  b.size = size;
  b.flags = flags;
  memcpy(b.prop, prop, 48);  // hipMemAllocationProp fits in 48 bytes
  uint64_t out = 0;
  hipError_t e = send_sync(OP_VMM_CREATE, &b, sizeof b, &out, 8);
  if (e == hipSuccess) *(uint64_t*)handle = out;
  return e;
}

extern "C" hipError_t hipMemRelease(void* handle) {
  uint64_t h = (uint64_t)handle;
  send_async(OP_VMM_RELEASE, 0, &h, 8);
  return hipSuccess;
}

extern "C" hipError_t hipMemMap(void* va, size_t size, size_t offset,
                                void* handle, unsigned long long flags) {
  struct {
    uint64_t va, size, off, handle, flags;
  } b{(uint64_t)va, size, offset, (uint64_t)handle, flags};
  hipError_t e = send_sync(OP_VMM_MAP, &b, sizeof b, nullptr, 0);
  if (e == hipSuccess) {
    Client& c = C();
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges[(uint64_t)va] = size;
  }
  return e;
}

extern "C" hipError_t hipMemUnmap(void* va, size_t size) {
  struct {
    uint64_t va, size;
  } b{(uint64_t)va, size};
  hipError_t e = send_sync(OP_VMM_UNMAP, &b, sizeof b, nullptr, 0);
  if (e == hipSuccess) {
    Client& c = C();
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges.erase((uint64_t)va);
  }
  return e;
}

extern "C" hipError_t hipMemSetAccess(void* va, size_t size,
                                      const void* desc, size_t count) {
  if (count > 8) return hipErrorNotSupported;
  struct {
    uint64_t va, size;
    uint32_t count, pad;
  } hdr{(uint64_t)va, size, (uint32_t)count, 0};
  uint8_t body[sizeof hdr + 8 * 12];
  memcpy(body, &hdr, sizeof hdr);
  memcpy(body + sizeof hdr, desc, count * 12);  // {loc{type,id},flags}
  return send_sync(OP_VMM_SET_ACCESS, body, sizeof hdr + count * 12,
                   nullptr, 0);
}

extern "C" hipError_t hipMemGetAllocationGranularity(size_t* g,
                                                     const void* prop,
                                                     int opt) {
  struct {
    uint64_t opt;
    uint8_t prop[48];
  } b{};
  b.opt = (uint64_t)opt;
  memcpy(b.prop, prop, 48);
  uint64_t out = 0;
  hipError_t e = send_sync(OP_VMM_GRANULARITY, &b, sizeof b, &out, 8);
  if (e == hipSuccess) *g = (size_t)out;
  return e;
}

// ---- stream-ordered allocator + mempool surface (hipMallocAsync
// backend of the PyTorch caching allocator)

extern "C" hipError_t hipDeviceGetDefaultMemPool(void** pool, int dev) {
  uint32_t d = (uint32_t)dev;
  uint64_t out = 0;
  hipError_t e = send_sync(OP_MEMPOOL_DEFAULT, &d, 4, &out, 8);
  if (e == hipSuccess) *pool = (void*)out;
  return e;
}

extern "C" hipError_t hipMemPoolSetAttribute(void* pool, int attr,
                                             void* value) {
  struct {
    uint64_t pool, attr, value;
  } b{(uint64_t)pool, (uint64_t)attr, 0};
  // attrs are u64 or int-sized thresholds; copy 8 bytes conservatively
  memcpy(&b.value, value, 8);
  return send_sync(OP_MEMPOOL_SET_ATTR, &b, sizeof b, nullptr, 0);
}

extern "C" hipError_t hipMemPoolGetAttribute(void* pool, int attr,
                                             void* value) {
  struct {
    uint64_t pool, attr;
  } b{(uint64_t)pool, (uint64_t)attr};
  uint64_t out = 0;
  hipError_t e = send_sync(OP_MEMPOOL_GET_ATTR, &b, sizeof b, &out, 8);
  if (e == hipSuccess) memcpy(value, &out, 8);
  return e;
}

extern "C" hipError_t hipMemPoolTrimTo(void* pool, size_t keep) {
  struct {
    uint64_t pool, keep;
  } b{(uint64_t)pool, keep};
  send_async(OP_MEMPOOL_TRIM, 0, &b, sizeof b);
  return hipSuccess;
}

// ---- device globals (__device__ vars): __hipRegisterVar recorded the
// host shadow → (fatbin, name); resolution ships the module (if not
// already) and asks the worker for hipModuleGetGlobal. The range is
// registered in dev_ranges so plain memcpys route to the device.

bool resolve_symbol(const void* symbol, uint64_t* dptr, uint64_t* bytes) {
  Client& c = C();
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    auto hit = c.dvar_addr.find(symbol);
    if (hit != c.dvar_addr.end()) {
      *dptr = hit->second.first;
      *bytes = hit->second.second;
      return true;
    }
  }
  FatBin* fb = nullptr;
  std::string name;
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    auto it = c.dvars.find(symbol);
    if (it == c.dvars.end()) return false;
    fb = it->second.first;
    name = it->second.second;
  }
  uint64_t mod = ship_image(fb);
  if (!mod) return false;
  std::vector<uint8_t> body(8 + name.size() + 1);
  memcpy(body.data(), &mod, 8);
  memcpy(body.data() + 8, name.c_str(), name.size() + 1);
  struct {
    uint64_t dptr, bytes;
  } r{0, 0};
  hipError_t e = send_sync(OP_GET_GLOBAL, body.data(), body.size(), &r,
                           sizeof r);
  if (e != hipSuccess || !r.dptr) return false;
  {
    std::lock_guard<std::mutex> l(c.reg_mu);
    c.dvar_addr[symbol] = {r.dptr, r.bytes};
  }
  {
    std::lock_guard<std::mutex> l(c.mem_mu);
    c.dev_ranges[r.dptr] = r.bytes;
  }
  *dptr = r.dptr;
  *bytes = r.bytes;
  return true;
}

extern "C" hipError_t hipGetSymbolAddress(void** devPtr,
                                          const void* symbol) {
  uint64_t d = 0, b = 0;
  if (!resolve_symbol(symbol, &d, &b)) return 1 /*hipErrorInvalidValue*/;
  *devPtr = (void*)d;
  return hipSuccess;
}

extern "C" hipError_t hipGetSymbolSize(size_t* size, const void* symbol) {
  uint64_t d = 0, b = 0;
  if (!resolve_symbol(symbol, &d, &b)) return 1;
  *size = (size_t)b;
  return hipSuccess;
}

extern "C" hipError_t hipMemcpyToSymbol(const void* symbol, const void* src,
                                        size_t count, size_t offset,
                                        int kind) {
  uint64_t d = 0, b = 0;
  if (!resolve_symbol(symbol, &d, &b)) return 1;
  if (offset + count > b && b) return 1;
  return hipMemcpy((void*)(d + offset), src, count,
                   kind == 0 ? 1 /*default for ToSymbol is H2D*/ : kind);
}

extern "C" hipError_t hipMemcpyFromSymbol(void* dst, const void* symbol,
                                          size_t count, size_t offset,
                                          int kind) {
  uint64_t d = 0, b = 0;
  if (!resolve_symbol(symbol, &d, &b)) return 1;
  if (offset + count > b && b) return 1;
  return hipMemcpy(dst, (const void*)(d + offset), count,
                   kind == 0 ? 2 /*default FromSymbol is D2H*/ : kind);
}

#define TF_NOTSUP(name, ...)                         \
  hipError_t name(__VA_ARGS__) {                     \
    fprintf(stderr, "[tf-client] %s: not supported over remoting yet\n", \
            #name);                                  \
    return hipErrorNotSupported;                     \
  }

TF_NOTSUP(hipIpcGetMemHandle, void*, void*)
TF_NOTSUP(hipIpcOpenMemHandle, void**, const void*, unsigned)
TF_NOTSUP(hipIpcCloseMemHandle, void*)
TF_NOTSUP(hipMemExportToShareableHandle, void*, void*, int, unsigned long long)
TF_NOTSUP(hipMemImportFromShareableHandle, void*, void*, int)
TF_NOTSUP(hipMemPoolSetAccess, void*, const void*, size_t)
TF_NOTSUP(hipMemcpyPeerAsync, void*, int, const void*, int, size_t, void*)
TF_NOTSUP(hipStreamWriteValue32, void*, void*, unsigned, unsigned)
TF_NOTSUP(hipGraphNodeGetDependencies, void*, void**, size_t*)
TF_NOTSUP(hipGraphDebugDotPrint, void*, const char*, unsigned)
#undef TF_NOTSUP

// ------------------------------------------------------------ hipGraphs
// Capture runs worker-side on the real stream (every stream op already
// flows through the ring in order); the client tracks capture state
// locally so torch's per-op hipStreamIsCapturing checks stay free.

hipError_t hipStreamBeginCapture(void* stream, int mode) {
  struct {
    uint64_t st;
    uint32_t mode;
  } b{(uint64_t)stream, (uint32_t)mode};
  hipError_t e = send_sync(OP_BEGIN_CAPTURE, &b, 12, nullptr, 0);
  if (e == hipSuccess) {
    std::lock_guard<std::mutex> l(g_capture_mu);
    g_capturing[stream] = g_capture_seq++;
  }
  return e;
}

hipError_t hipStreamEndCapture(void* stream, void** graph) {
  uint64_t st = (uint64_t)stream, g = 0;
  hipError_t e = send_sync(OP_END_CAPTURE, &st, 8, &g, 8);
  {
    std::lock_guard<std::mutex> l(g_capture_mu);
    g_capturing.erase(stream);
  }
  if (graph) *graph = (void*)g;
  return e;
}

hipError_t hipGraphInstantiateWithFlags(void** graphExec, void* graph,
                                        unsigned long long flags) {
  struct {
    uint64_t graph, flags;
  } b{(uint64_t)graph, flags};
  uint64_t ge = 0;
  hipError_t e = send_sync(OP_GRAPH_INSTANTIATE, &b, 16, &ge, 8);
  if (graphExec) *graphExec = (void*)ge;
  return e;
}

hipError_t hipGraphInstantiate(void** graphExec, void* graph, void*, char*,
                               size_t) {
  return hipGraphInstantiateWithFlags(graphExec, graph, 0);
}

hipError_t hipGraphLaunch(void* graphExec, void* stream) {
  struct {
    uint64_t ge, st;
  } b{(uint64_t)graphExec, (uint64_t)stream};
  send_async(OP_GRAPH_LAUNCH, 0, &b, 16);
  return hipSuccess;
}

hipError_t hipGraphDestroy(void* graph) {
  uint64_t g = (uint64_t)graph;
  send_async(OP_GRAPH_DESTROY, 0, &g, 8);
  return hipSuccess;
}

hipError_t hipGraphGetNodes(void* graph, void** nodes, size_t* count) {
  if (!count) return hipErrorInvalidValue;
  struct {
    uint64_t graph, cap;
  } b{(uint64_t)graph, nodes ? (uint64_t)*count : 0};
  std::vector<uint8_t> out(8 + 8 * 4096);
  size_t out_len = 0;
  hipError_t e = send_sync(OP_GRAPH_GET_NODES, &b, 16, out.data(),
                           out.size(), &out_len);
  if (e != hipSuccess || out_len < 8) return e ? e : hipErrorInvalidValue;
  uint64_t cnt;
  memcpy(&cnt, out.data(), 8);
  if (nodes) {
    size_t have = (out_len - 8) / 8;
    size_t n = have < *count ? have : *count;
    memcpy(nodes, out.data() + 8, n * 8);
    *count = n;
  } else {
    *count = (size_t)cnt;
  }
  return hipSuccess;
}

int tf_graph_kernel_histo(void* graph, char* out, size_t cap) {
  std::vector<uint8_t> buf(64 << 10);
  size_t out_len = 0;
  uint64_t g = (uint64_t)graph;
  hipError_t e = send_sync(OP_GRAPH_KERNEL_HISTO, &g, 8, buf.data(),
                           buf.size(), &out_len);
  if (e != hipSuccess) return -1;
  size_t n = out_len < cap - 1 ? out_len : cap - 1;
  memcpy(out, buf.data(), n);
  out[n] = 0;
  return 0;
}

int tf_graph_node_types(void* graph, unsigned long long* hist17) {
  std::vector<uint8_t> out(17 * 8);
  size_t out_len = 0;
  uint64_t g = (uint64_t)graph;
  hipError_t e = send_sync(OP_GRAPH_NODE_TYPES, &g, 8, out.data(),
                           out.size(), &out_len);
  if (e != hipSuccess || out_len < 17 * 8) return -1;
  memcpy(hist17, out.data(), 17 * 8);
  return 0;
}

hipError_t hipGraphExecDestroy(void* graphExec) {
  uint64_t g = (uint64_t)graphExec;
  send_async(OP_GRAPH_EXEC_DESTROY, 0, &g, 8);
  return hipSuccess;
}

}  // extern "C"
