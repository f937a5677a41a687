// libtftier.so — hand-written CDNA4 (gfx950) kernels for VRAM
// oversubscription / hot-cold tiering on MI355X (SURVEY §2.4(c), §7 stage 7).
//
// The tiering engine moves vGPU pages between 288 GB HBM3E and host DRAM:
//   - eviction: gather scattered cold device pages into a contiguous device
//     staging arena (these kernels, ~6 TB/s class), then one large SDMA
//     hipMemcpyAsync staging→pinned-host (PCIe Gen5, ~63 GB/s) overlapped on
//     its own stream;
//   - restore: the reverse (H2D SDMA, then scatter to fresh pages);
//   - compaction: defragment the vGPU arena with direct page moves.
// Gathering first keeps the PCIe DMA a single contiguous transfer (SDMA
// engines hate scatter lists) while the HBM-side scatter/gather runs at
// memory speed.
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//   - 16 B/lane vectorized uint4 access (G13), grid-strided,
//     256-thread workgroups, grids sized ≫256 workgroups to fill 8 XCDs;
//   - `nt` (non-temporal, aux=2) loads on evicted pages: they are read once
//     and must not displace the workload's L2/LLC working set
//     (microarch row nt-weights: streamed-once data wants nt);
//   - no inter-workgroup communication — every page chunk is independent, no
//     agent-scope fencing is needed (G16 does not apply).
//
// C ABI (ctypes from tensor_fusion_amd/ops/tiering.py); all funcs take an
// explicit hipStream_t so Python passes torch.cuda.current_stream().

#include <hip/hip_runtime.h>

#include <cstdint>

#define TF_CHECK(x)                                   \
  do {                                                \
    hipError_t _e = (x);                              \
    if (_e != hipSuccess) return (int)_e;             \
  } while (0)

namespace {

// 16-byte native vector (HIP's uint4 is a class type the nontemporal
// builtins reject; the clang ext_vector maps straight to dwordx4 ops).
typedef uint32_t u4 __attribute__((ext_vector_type(4)));

// Plain vectorized copy, grid-strided. Swept on hardware
// (native/tiering/bw_sweep.hip, gpurun_out/bw_sweep.log): the SIMPLE
// 16 B/lane loop at grid≈1024 (4 waves/CU) is the fastest variant
// (5.38 TB/s for a 4 GiB copy); 4×-unrolled variants LOSE (4.6–5.2),
// and 64 B/lane blocked non-temporal collapses to 1.1 TB/s.
__global__ void copy16_kernel(const u4* __restrict__ src, u4* __restrict__ dst,
                              size_t n16) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) dst[i] = src[i];
}

// Non-temporal variant for one-shot streams (eviction): bypass L1 and mark
// lines as last-use in L2 so tiering traffic does not evict workload data.
__global__ void copy16_nt_kernel(const u4* __restrict__ src,
                                 u4* __restrict__ dst, size_t n16) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) {
    u4 v = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(v, &dst[i]);
  }
}

// Gather pages: dst[i*page] = src_pages[idx[i]] for i in [0, npages).
// One y-slice of blocks per page; x covers the page body.
__global__ void gather_pages_kernel(const uint8_t* __restrict__ base,
                                    uint8_t* __restrict__ dst,
                                    const uint32_t* __restrict__ idx,
                                    size_t page_bytes, uint32_t npages,
                                    int nontemporal) {
  uint32_t page = blockIdx.y;
  if (page >= npages) return;
  const u4* src = reinterpret_cast<const u4*>(base + (size_t)idx[page] * page_bytes);
  u4* out = reinterpret_cast<u4*>(dst + (size_t)page * page_bytes);
  size_t n16 = page_bytes / 16;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  if (nontemporal) {
    for (; i < n16; i += stride) {
      u4 v = __builtin_nontemporal_load(&src[i]);
      __builtin_nontemporal_store(v, &out[i]);
    }
  } else {
    for (; i < n16; i += stride) out[i] = src[i];
  }
}

// Scatter pages: src contiguous staging → dst_pages[idx[i]].
__global__ void scatter_pages_kernel(const uint8_t* __restrict__ staging,
                                     uint8_t* __restrict__ base,
                                     const uint32_t* __restrict__ idx,
                                     size_t page_bytes, uint32_t npages) {
  uint32_t page = blockIdx.y;
  if (page >= npages) return;
  const u4* src = reinterpret_cast<const u4*>(staging + (size_t)page * page_bytes);
  u4* out = reinterpret_cast<u4*>(base + (size_t)idx[page] * page_bytes);
  size_t n16 = page_bytes / 16;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) out[i] = src[i];
}

inline int copy_grid(size_t n16, int block = 256) {
  // hardware-swept optimum: 1024 workgroups (4 per CU) saturates HBM;
  // bigger grids only add scheduling overhead (bw_sweep.hip)
  size_t want = (n16 + block - 1) / block;
  size_t cap = 1024;
  return (int)(want < cap ? (want ? want : 1) : cap);
}

}  // namespace

extern "C" {

// All sizes in bytes; pointers must be 16-byte aligned, sizes multiples of 16.
int tf_tier_copy(const void* src, void* dst, size_t bytes, void* stream,
                 int nontemporal) {
  size_t n16 = bytes / 16;
  dim3 block(256);
  dim3 grid(copy_grid(n16));
  if (nontemporal)
    hipLaunchKernelGGL(copy16_nt_kernel, grid, block, 0, (hipStream_t)stream,
                       (const u4*)src, (u4*)dst, n16);
  else
    hipLaunchKernelGGL(copy16_kernel, grid, block, 0, (hipStream_t)stream,
                       (const u4*)src, (u4*)dst, n16);
  return (int)hipGetLastError();
}

// base + idx[i]*page_bytes → staging (contiguous). idx is a DEVICE pointer.
int tf_tier_gather(const void* base, void* staging, const uint32_t* idx_dev,
                   size_t page_bytes, uint32_t npages, void* stream,
                   int nontemporal) {
  dim3 block(256);
  dim3 grid(copy_grid(page_bytes / 16, 256), npages);
  // target ≈2048 workgroups total: few big pages get wide x-grids, many
  // small pages narrow ones (256 CUs × 8 XCDs need ≫256 WGs in flight)
  unsigned cap = npages ? (2048u + npages - 1) / npages : 1;
  if (cap < 1) cap = 1;
  if (grid.x > cap) grid.x = cap;
  hipLaunchKernelGGL(gather_pages_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)base, (uint8_t*)staging, idx_dev,
                     page_bytes, npages, nontemporal);
  return (int)hipGetLastError();
}

int tf_tier_scatter(const void* staging, void* base, const uint32_t* idx_dev,
                    size_t page_bytes, uint32_t npages, void* stream) {
  dim3 block(256);
  dim3 grid(copy_grid(page_bytes / 16, 256), npages);
  unsigned cap = npages ? (2048u + npages - 1) / npages : 1;
  if (cap < 1) cap = 1;
  if (grid.x > cap) grid.x = cap;
  hipLaunchKernelGGL(scatter_pages_kernel, grid, block, 0, (hipStream_t)stream,
                     (const uint8_t*)staging, (uint8_t*)base, idx_dev,
                     page_bytes, npages);
  return (int)hipGetLastError();
}

int tf_tier_synchronize(void* stream) {
  TF_CHECK(hipStreamSynchronize((hipStream_t)stream));
  return 0;
}

int tf_tier_device_count(int* n) { return (int)hipGetDeviceCount(n); }

}  // extern "C"
