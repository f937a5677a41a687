// GPU-over-IP remoting protocol — client HIP interposer ⇄ per-vGPU worker.
//
// Capability parity with the reference's closed GPU-over-IP plane (SURVEY
// §2.4: cuda_hook/vgpu.rs forwarding CUDA over TCP/RDMA with <4% loss).
// MI355X-native redesign for the dominant same-node case:
//
//   client app (GPU-less: HIP_VISIBLE_DEVICES="")            vGPU worker
//   ┌──────────────────────────┐  shared-memory segment  ┌──────────────┐
//   │ libtfhip_client.so       │ ┌─────────────────────┐ │ tf_vgpu_worker│
//   │  interposes hip* ───────►│ │ cmd ring (SPSC)     │ │  executes on  │
//   │  async ops: write+return │ │ cpl ring (SPSC)     │ │  real HIP,    │
//   │  sync ops: futex wait ◄──│ │ data arena (pinned) │ │  hipHostRegist│
//   │                          │ └─────────────────────┘ │  -ers arena   │
//   └──────────────────────────┘   doorbells = futex      └──────────────┘
//
// Why this hits <4%: kernel launches and async copies are FIRE-AND-FORGET
// (enqueue ≈100ns, no syscall in steady state — the worker spins briefly
// before futex-parking), so the client only pays a round trip where HIP
// itself is synchronous (hipStreamSynchronize, D2H reads, mallocs). Bulk
// H2D/D2H stages through the arena, which the worker hipHostRegisters once:
// the GPU DMAs straight out of/into the shared mapping — zero copies beyond
// the client's own memcpy in/out of its private buffers.
//
// Handles: device pointers, streams, events, modules and functions are the
// WORKER's real values, opaque to the client — no translation tables. The
// only client-side mappings are kernel stubs (__hipRegisterFunction) →
// (fatbin id, name) and kernarg layouts fetched at first use.
//
// Cross-node transport (TCP/RDMA) carries the same command stream; the
// same-node fast path is the benchmarked one.
#pragma once

#include <stdint.h>

namespace tfrpc {

constexpr uint32_t MAGIC = 0x54465250;  // "TFRP"
constexpr uint32_t VERSION = 1;

// ---- segment layout ----------------------------------------------------
// [Header | cmd ring buf | cpl ring buf | arena]
constexpr size_t CMD_RING_BYTES = 4u << 20;
constexpr size_t CPL_RING_BYTES = 1u << 20;
constexpr size_t ARENA_BYTES = 256u << 20;

struct Ring {
  // SPSC byte ring: producer bumps head, consumer bumps tail. Records are
  // 8-byte aligned, length-prefixed; a zero length-word at end-of-buffer
  // means wrap. head/tail are free-running counters mod 2^64.
  alignas(64) volatile uint64_t head;   // producer writes
  alignas(64) volatile uint64_t tail;   // consumer writes
  alignas(64) volatile uint32_t futex_nonempty;  // consumer parks on this
  uint32_t _pad;
};

struct Header {
  uint32_t magic;
  uint32_t version;
  uint64_t total_bytes;
  Ring cmd;  // client → worker
  Ring cpl;  // worker → client
  // Arena: bump ring. Client allocates [head, head+len) when it fits in
  // (freed + ARENA_BYTES - head); worker advances freed when the GPU is
  // done with a chunk. Offsets are free-running; real offset = v % ARENA.
  alignas(64) volatile uint64_t arena_head;   // client
  alignas(64) volatile uint64_t arena_freed;  // worker
  alignas(64) volatile uint32_t futex_cpl;    // client parks on this
  uint32_t _pad2;
  volatile uint32_t worker_ready;
  volatile uint32_t shutdown;
  volatile uint64_t sticky_error;  // first async hipError_t, sticky
  uint8_t reserved[960];
};

constexpr size_t HDR_BYTES = 4096;
static_assert(sizeof(Header) <= HDR_BYTES, "header fits");

constexpr size_t SEG_BYTES = HDR_BYTES + CMD_RING_BYTES + CPL_RING_BYTES +
                             ARENA_BYTES;

inline uint8_t* cmd_buf(Header* h) {
  return reinterpret_cast<uint8_t*>(h) + HDR_BYTES;
}
inline uint8_t* cpl_buf(Header* h) {
  return reinterpret_cast<uint8_t*>(h) + HDR_BYTES + CMD_RING_BYTES;
}
inline uint8_t* arena(Header* h) {
  return reinterpret_cast<uint8_t*>(h) + HDR_BYTES + CMD_RING_BYTES +
         CPL_RING_BYTES;
}

// ---- commands ----------------------------------------------------------

enum Op : uint32_t {
  OP_NOP = 0,
  OP_HELLO,              // → reply {device_count}
  OP_SET_DEVICE,         // {dev}
  OP_GET_PROPS,          // {dev} → reply {hipDeviceProp_tR0600 blob}
  OP_GET_ATTRIBUTE,      // {dev, attr} → reply {value}
  OP_MALLOC,             // {size} → reply {ptr, err}
  OP_FREE,               // {ptr} async
  OP_MEMCPY_H2D,         // {dst, size, arena_off|inline} async
  OP_MEMCPY_D2H,         // {src, size, arena_off} → reply when data ready
  OP_MEMCPY_D2D,         // {dst, src, size, stream} async
  OP_MEMSET,             // {dst, value, size, stream} async
  OP_LAUNCH,             // {func, grid[3], block[3], shmem, stream,
                         //  kernarg_size, kernarg bytes} async
  OP_STREAM_CREATE,      // {flags, priority} → reply {stream}
  OP_STREAM_DESTROY,     // {stream} async
  OP_STREAM_SYNC,        // {stream} → reply {err}
  OP_STREAM_QUERY,       // {stream} → reply {err}
  OP_EVENT_CREATE,       // {flags} → reply {event}
  OP_EVENT_RECORD,       // {event, stream} async
  OP_EVENT_SYNC,         // {event} → reply {err}
  OP_EVENT_QUERY,        // {event} → reply {err}
  OP_EVENT_ELAPSED,      // {ev0, ev1} → reply {ms_bits}
  OP_EVENT_DESTROY,      // {event} async
  OP_DEVICE_SYNC,        // → reply {err}
  OP_LOAD_MODULE,        // {image_id, size, arena_off} → reply {module}
  OP_GET_FUNCTION,       // {module, name[]} → reply {func, nargs,
                         //  kernarg_size, {size,offset}×nargs}
  OP_MEM_GET_INFO,       // → reply {free, total}
  OP_CAN_ACCESS_PEER,    // {dev, peer} → reply {int}
  OP_STREAM_WAIT_EVENT,  // {stream, event, flags} async
  OP_SHUTDOWN,
  // hipGraphs: capture happens worker-side on the real stream; the
  // client only tracks "is capturing" locally for torch's hot checks.
  OP_BEGIN_CAPTURE,      // {stream, mode} → reply {err}
  OP_END_CAPTURE,        // {stream} → reply {graph}
  OP_GRAPH_INSTANTIATE,  // {graph, flags} → reply {graphExec}
  OP_GRAPH_LAUNCH,       // {graphExec, stream} async
  OP_GRAPH_DESTROY,      // {graph} async
  OP_GRAPH_EXEC_DESTROY, // {graphExec} async
  OP_GRAPH_GET_NODES,    // {graph, cap} → reply {count, nodes[...]}
  OP_GRAPH_NODE_TYPES,   // {graph} → reply {count, type_histogram[16]}
  OP_GRAPH_KERNEL_HISTO, // {graph} → reply {text histogram of node kernels}

  // client-side VMM surface (PyTorch expandable_segments, vLLM-class
  // allocators): forwarded 1:1; handles/VAs are worker-side values
  OP_VMM_RESERVE,        // {size, align, addr_hint, flags} → {ptr}
  OP_VMM_ADDR_FREE,      // {ptr, size} async
  OP_VMM_CREATE,         // {size, flags, prop[48]} → {handle}
  OP_VMM_RELEASE,        // {handle} async
  OP_VMM_MAP,            // {va, size, offset, handle, flags} → {err}
  OP_VMM_UNMAP,          // {va, size} → {err}
  OP_VMM_SET_ACCESS,     // {va, size, count, desc[count*12]} → {err}
  OP_VMM_GRANULARITY,    // {opt, prop[48]} → {granularity}

  // stream-ordered allocator + mempool surface (hipMallocAsync backend)
  OP_MALLOC_ASYNC,       // {size, stream} → {ptr}
  OP_FREE_ASYNC,         // {ptr, stream} async
  OP_MEMPOOL_DEFAULT,    // {dev} → {pool}
  OP_MEMPOOL_SET_ATTR,   // {pool, attr, value} → {err}
  OP_MEMPOOL_GET_ATTR,   // {pool, attr} → {value}
  OP_MEMPOOL_TRIM,       // {pool, keep} async
  OP_GET_GLOBAL,         // {module, name[]} → reply {dptr, size}
};

constexpr uint32_t F_WANT_REPLY = 1u << 0;
constexpr uint32_t F_INLINE_DATA = 1u << 1;  // payload carries the bytes

struct CmdHdr {
  uint32_t op;
  uint32_t flags;
  uint64_t seq;       // reply correlation (client-assigned)
  uint32_t body_len;  // bytes following this header
  uint32_t _pad;
};

struct CplHdr {
  uint64_t seq;
  int32_t err;        // hipError_t
  uint32_t body_len;  // reply payload following
};

// Fixed bodies (variable tails documented per-op above).
struct LaunchBody {
  uint64_t func;
  uint32_t grid[3];
  uint32_t block[3];
  uint32_t shmem;
  uint32_t _pad;
  uint64_t stream;
  uint32_t kernarg_size;  // bytes following this struct
  uint32_t _pad2;
};

struct MemcpyBody {
  uint64_t dst;
  uint64_t src;        // device ptr or unused
  uint64_t size;
  uint64_t arena_off;  // free-running arena offset (H2D/D2H staging)
  uint64_t stream;
  uint32_t kind;
  uint32_t sync;       // 1 = synchronous semantics requested
};

// ---- TCP transport framing (cross-node GPU-over-IP) -------------------
// Same command stream over a socket: each side runs a pump that mirrors
// its ring to the wire. Frames:
//   [FrameHdr][record bytes][extra bytes]
// kind 0 = cmd record (CmdHdr+body). extra = arena payload the command
//          references (non-inline H2D staging, module images), written
//          into the receiver's arena at `arena_off % ARENA_BYTES`.
// kind 1 = cpl record (CplHdr+body). D2H data rides INSIDE the reply body
//          in TCP sessions (the worker detects the session type), so
//          extra_len is 0 for completions.
struct FrameHdr {
  uint32_t kind;       // 0 cmd, 1 cpl
  uint32_t rec_len;    // bytes of the record that follows
  uint64_t arena_off;  // where extra bytes land (kind 0 only)
  uint32_t extra_len;
  uint32_t _pad;
};

constexpr uint32_t TCP_MAGIC = 0x54465443;  // "TFTC" handshake word

}  // namespace tfrpc
