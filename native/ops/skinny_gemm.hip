// Skinny-GEMM (decode GEMV) on gfx950 matrix cores.
//
//   y[M,N] = x[M,K] @ W[N,K]^T        bf16 in, fp32 accumulate, bf16 out
//   M ≤ 16 (decode batch), K ∈ {4096, 14336, ...}, N up to 128256.
//
// Shape analysis (guide §3): at M ≤ 16 the GEMM is pure weight streaming
// (2·M flops per loaded W element); one v_mfma_f32_16x16x32_bf16 consumes a
// 16(col)×32(k) B tile = 1 KiB of W per wave per ~5-cycle instruction, so
// the matrix pipe is never the bound — HBM is. The kernel therefore
// optimizes the W access pattern: each wave owns one 16-column tile and
// streams its 16 rows sequentially in 64 B/row chunks (4 k-groups × 16 B)
// with non-temporal hints (weights are read once per token).
//
// Fragment mapping (cdna4: 2×K extension of the classic CDNA layout,
// validated by tests/test_gpu_fused.py numerics vs fp32 torch):
//   A (x):  lane l, elem j → A[i = l&15][k = (l>>4)*8 + j]
//   B (Wᵀ): lane l, elem j → B[k = (l>>4)*8 + j][n = l&15]  = W[n][k]
//   C/D:    lane l, reg r  → D[row = (l>>4)*4 + r][col = l&15]
//
// Launch: one wave per 16-column tile, 4 waves per workgroup,
// grid.x = ceil(N / 64). x is small (≤16×K×2 B ≤ 448 KiB) and shared by
// every workgroup — its loads hit L2 after the first tile.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

namespace {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef uint32_t u4 __attribute__((ext_vector_type(4)));

union B16x8 {
  u4 raw;
  bf16x8 v;
};

template <int UNROLL>
__global__ void skinny_gemm_kernel(const __hip_bfloat16* __restrict__ x,
                                   const __hip_bfloat16* __restrict__ w,
                                   __hip_bfloat16* __restrict__ y,
                                   int M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = (blockIdx.x * 4 + wave) * 16;
  if (n0 >= N) return;

  const int col = lane & 15;      // A row i / B col n / D col
  const int kgrp = lane >> 4;     // 0..3 → k-subgroup of 8
  const int n = n0 + col;
  const bool ncol_ok = n < N;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};

  // row pointer for this lane's W stream (8 bf16 = 16 B per step)
  const __hip_bfloat16* wrow = w + (size_t)(ncol_ok ? n : 0) * K;
  const __hip_bfloat16* xrow = x + (size_t)(col < M ? col : 0) * K;
  const bool arow_ok = col < M;

  // UNROLL-deep K loop: independent (a,b) load pairs in flight per lane
  // before their MFMAs — a single load→mfma chain left the memory queues
  // underfed (3.2 TB/s vs 4.4 at depth 8; flat 8→16, see
  // profiles/skinny_gemm_mfma_r01.md; hipBLASLt streams the shape at 5.7)
  // NOTE x loads are CACHED (not nt): every workgroup re-reads the same
  // small x, so last-use-marking it in L2 would re-fetch it from DRAM
  // N/64 times. W is streamed once → nt.
  int k0 = 0;
  for (; k0 + 32 * UNROLL <= K; k0 += 32 * UNROLL) {
    B16x8 a[UNROLL], b[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      int k = k0 + u * 32 + kgrp * 8;
      a[u].raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                         : u4{0, 0, 0, 0};
      b[u].raw = ncol_ok
                     ? __builtin_nontemporal_load(
                           reinterpret_cast<const u4*>(wrow + k))
                     : u4{0, 0, 0, 0};
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u].v, b[u].v, acc, 0,
                                                    0, 0);
  }
  for (; k0 < K; k0 += 32) {
    int k = k0 + kgrp * 8;
    B16x8 a, b;
    a.raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                    : u4{0, 0, 0, 0};
    b.raw = ncol_ok ? __builtin_nontemporal_load(
                          reinterpret_cast<const u4*>(wrow + k))
                    : u4{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = kgrp * 4 + r;
    if (row < M && ncol_ok)
      y[(size_t)row * N + n] = __float2bfloat16(acc[r]);
  }
}

// ---- v2: LDS-DMA weight streaming (guide T3 minimum 2-phase) ----------
// Each wave stages its 16 W rows through a double-buffered LDS tile with
// global_load_lds: the 64 lanes of one instruction read 1 KiB of ONE row
// contiguously (fully coalesced DRAM), and the MFMA B fragments are
// ds_read from LDS. Row stride padded by 4 elements so the 16 fragment
// rows land on distinct bank groups.

// KT (K elements per tile) is a template knob: 512 → 1 KiB rows staged
// with 16 B/lane glds but 132 KiB LDS caps occupancy at 1 WG/CU;
// 128 → 256 B rows (4 B/lane glds) at 33.8 KiB LDS → 4 WG/CU, letting
// other workgroups cover each one's __syncthreads() drains.
constexpr int ROW_PAD = 4;         // 8 B pad → fragment rows on distinct banks

template <int KT>
__global__ void __launch_bounds__(256)
skinny_gemm_lds_kernel(const __hip_bfloat16* __restrict__ x,
                       const __hip_bfloat16* __restrict__ w,
                       __hip_bfloat16* __restrict__ y,
                       int M, int N, int K) {
  constexpr int ROW_ELEMS = KT + ROW_PAD;
  __shared__ __hip_bfloat16 tile[4][2][16][ROW_ELEMS];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = (blockIdx.x * 4 + wave) * 16;
  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int n = n0 + col;
  const bool ncol_ok = n < N && n0 < N;
  const bool arow_ok = col < M;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const __hip_bfloat16* xrow = x + (size_t)(arow_ok ? col : 0) * K;

  auto stage = [&](int buf, int kt) {
    // 16 glds: instruction r streams W[n0+r][kt..kt+KT) → tile row r
#pragma unroll 4
    for (int r = 0; r < 16; ++r) {
      int row = n0 + r < N ? n0 + r : N - 1;
      const void* src = w + (size_t)row * K + kt + lane * (KT / 64);
      auto* gsrc = (const __attribute__((address_space(1))) void*)src;
      auto* ldst =
          (__attribute__((address_space(3))) void*)&tile[wave][buf][r][0];
      if constexpr (KT == 512)
        __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);
      else
        __builtin_amdgcn_global_load_lds(gsrc, ldst, 4, 0, 0);
    }
  };

  const int ntiles = K / KT;
  int cur = 0;
  stage(cur, 0);
  __syncthreads();  // vmcnt(0)+lgkmcnt(0)+barrier: tile 0 resident
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) stage(cur ^ 1, (t + 1) * KT);
#pragma unroll
    for (int k0 = 0; k0 < KT; k0 += 32) {
      int kk = k0 + kgrp * 8;
      B16x8 a, b;
      a.raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + t * KT + kk)
                      : u4{0, 0, 0, 0};
      b.raw = *reinterpret_cast<const u4*>(&tile[wave][cur][col][kk]);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
    }
    __syncthreads();  // drains the prefetch glds; next tile resident
    cur ^= 1;
  }
  // K remainder (K % KT) via direct loads
  for (int k0 = ntiles * KT; k0 < K; k0 += 32) {
    int k = k0 + kgrp * 8;
    B16x8 a, b;
    a.raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                    : u4{0, 0, 0, 0};
    b.raw = ncol_ok ? __builtin_nontemporal_load(
                          reinterpret_cast<const u4*>(w + (size_t)n * K + k))
                    : u4{0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = kgrp * 4 + r;
    if (row < M && ncol_ok)
      y[(size_t)row * N + n] = __float2bfloat16(acc[r]);
  }
}

// ---- v3: pre-shuffled weight layout (the hipBLASLt-closing move) -----
// The round-1 ablation pinned the direct kernel at 4.4 TB/s: ~8000
// concurrent 16-row K-strided W streams are a DRAM-unfriendly pattern no
// unroll depth fixes. The framework owns the weight layout, so shuffle W
// ONCE at model load into the exact per-lane MFMA fragment order:
//
//   P[nt][kt][lane][j] = W[nt*16 + (lane&15)][kt*32 + (lane>>4)*8 + j]
//
// (ops/fused.py pack_skinny_weight — a single torch permute). Each wave
// then reads ONE fully-sequential stream: per k-step the 64 lanes fetch
// 1 KiB contiguous, step after step, tile after tile. DRAM sees pure
// sequential bursts. No LDS, no layout math in the hot loop — the
// pointer just increments.

template <int UNROLL>
__global__ void skinny_gemm_packed_kernel(
    const __hip_bfloat16* __restrict__ x,
    const __hip_bfloat16* __restrict__ wp,  // packed [N/16][K/32][64][8]
    __hip_bfloat16* __restrict__ y, int M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nt = blockIdx.x * 4 + wave;
  const int n0 = nt * 16;
  if (n0 >= N) return;

  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int n = n0 + col;
  const bool arow_ok = col < M;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int ktiles = K / 32;
  // this wave's sequential W stream: 512 elems (1 KiB) per k-tile
  const __hip_bfloat16* wstream =
      wp + ((size_t)nt * ktiles) * 512 + lane * 8;
  const __hip_bfloat16* xrow = x + (size_t)(arow_ok ? col : 0) * K;

  int t = 0;
  for (; t + UNROLL <= ktiles; t += UNROLL) {
    B16x8 a[UNROLL], b[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      int k = (t + u) * 32 + kgrp * 8;
      a[u].raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                         : u4{0, 0, 0, 0};
      b[u].raw = __builtin_nontemporal_load(
          reinterpret_cast<const u4*>(wstream + (size_t)(t + u) * 512));
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u].v, b[u].v, acc, 0,
                                                    0, 0);
  }
  for (; t < ktiles; ++t) {
    int k = t * 32 + kgrp * 8;
    B16x8 a, b;
    a.raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                    : u4{0, 0, 0, 0};
    b.raw = __builtin_nontemporal_load(
        reinterpret_cast<const u4*>(wstream + (size_t)t * 512));
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = kgrp * 4 + r;
    if (row < M && n < N) y[(size_t)row * N + n] = __float2bfloat16(acc[r]);
  }
}

// ---- v3b: wave-split-K for underfilled N ------------------------------
// At N=4096 the packed kernel launches only 64 workgroups (1 wave-group
// per CU on a quarter of the chip) and the per-CU memory parallelism
// caps the stream at ~1.8 TB/s. This variant keeps the same grid but
// packs KSPLIT wave-groups per workgroup, each streaming a disjoint
// K-slice of the SAME four column tiles; the ks==0 group reduces the
// partials through LDS and writes y. 4x the outstanding loads per CU,
// one kernel, no workspace.

template <int UNROLL, int KSPLIT>
__global__ void __launch_bounds__(256 * KSPLIT)
skinny_gemm_packed_ws_kernel(const __hip_bfloat16* __restrict__ x,
                             const __hip_bfloat16* __restrict__ wp,
                             __hip_bfloat16* __restrict__ y, int M, int N,
                             int K) {
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;  // wave: [0, 4*KSPLIT)
  const int tile = w & 3;
  const int ks = w >> 2;  // K-slice of this wave
  const int nt = blockIdx.x * 4 + tile;
  const int n0 = nt * 16;
  __shared__ float red[4][KSPLIT][64][4];

  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int n = n0 + col;
  const bool arow_ok = col < M;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int ktiles = K / 32;
  const int per = (ktiles + KSPLIT - 1) / KSPLIT;
  const int t_lo = ks * per;
  const int t_hi = t_lo + per < ktiles ? t_lo + per : ktiles;

  if (n0 < N) {
    const __hip_bfloat16* wstream =
        wp + ((size_t)nt * ktiles) * 512 + lane * 8;
    const __hip_bfloat16* xrow = x + (size_t)(arow_ok ? col : 0) * K;
    int t = t_lo;
    for (; t + UNROLL <= t_hi; t += UNROLL) {
      B16x8 a[UNROLL], b[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        int k = (t + u) * 32 + kgrp * 8;
        a[u].raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                           : u4{0, 0, 0, 0};
        b[u].raw = __builtin_nontemporal_load(
            reinterpret_cast<const u4*>(wstream + (size_t)(t + u) * 512));
      }
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u].v, b[u].v, acc,
                                                      0, 0, 0);
    }
    for (; t < t_hi; ++t) {
      int k = t * 32 + kgrp * 8;
      B16x8 a, b;
      a.raw = arow_ok ? *reinterpret_cast<const u4*>(xrow + k)
                      : u4{0, 0, 0, 0};
      b.raw = __builtin_nontemporal_load(
          reinterpret_cast<const u4*>(wstream + (size_t)t * 512));
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
    }
  }

  if (ks > 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) red[tile][ks][lane][r] = acc[r];
  }
  __syncthreads();
  if (ks == 0 && n0 < N) {
#pragma unroll
    for (int s = 1; s < KSPLIT; ++s)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[r] += red[tile][s][lane][r];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = kgrp * 4 + r;
      if (row < M && n < N)
        y[(size_t)row * N + n] = __float2bfloat16(acc[r]);
    }
  }
}

}  // namespace

extern "C" {

// x[M,K] bf16; wp = packed weights (pack_skinny_weight: [N/16][K/32][64][8]);
// y[M,N] bf16. Requires K % 32 == 0, N % 16 == 0, M <= 16.
int tf_skinny_gemm_packed(const void* x, const void* wp, void* y, int M,
                          int N, int K, void* stream) {
  if (M < 1 || M > 16 || (K & 31) || (N & 15)) return 1;
  dim3 block(256);
  dim3 grid((N + 63) / 64);
  // underfilled grids (< 3/4 of the 256 CUs): wave-split-K packs 4
  // K-slice wave-groups per workgroup for 4x the in-flight loads/CU
  static int splitk = [] {
    const char* v = getenv("TF_SKINNY_SPLITK");
    return v ? atoi(v) : -1;  // -1 = auto
  }();
  bool want_ws = splitk == 4 || (splitk == -1 && grid.x < 192);
  if (want_ws && K / 32 >= 8) {
    hipLaunchKernelGGL((skinny_gemm_packed_ws_kernel<8, 4>), grid,
                       dim3(1024), 0, (hipStream_t)stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)wp,
                       (__hip_bfloat16*)y, M, N, K);
    return (int)hipGetLastError();
  }
  static int unroll = [] {
    const char* v = getenv("TF_SKINNY_UNROLL");
    int u = v ? atoi(v) : 8;
    return (u == 4 || u == 12 || u == 16) ? u : 8;
  }();
  switch (unroll) {
    case 4:
      hipLaunchKernelGGL(skinny_gemm_packed_kernel<4>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)wp, (__hip_bfloat16*)y, M,
                         N, K);
      break;
    case 12:
      hipLaunchKernelGGL(skinny_gemm_packed_kernel<12>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)wp, (__hip_bfloat16*)y, M,
                         N, K);
      break;
    case 16:
      hipLaunchKernelGGL(skinny_gemm_packed_kernel<16>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)wp, (__hip_bfloat16*)y, M,
                         N, K);
      break;
    default:
      hipLaunchKernelGGL(skinny_gemm_packed_kernel<8>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)wp, (__hip_bfloat16*)y, M,
                         N, K);
  }
  return (int)hipGetLastError();
}

// x[M,K], w[N,K] row-major bf16; y[M,N] bf16. K % 32 == 0, M <= 16.
int tf_skinny_gemm(const void* x, const void* w, void* y, int M, int N,
                   int K, void* stream) {
  if (M < 1 || M > 16 || (K & 31)) return 1;
  dim3 block(256);  // 4 waves × 16 cols each
  dim3 grid((N + 63) / 64);
  static int use_lds = [] {
    const char* v = getenv("TF_SKINNY_LDS");  // 1 → KT=512, 2 → KT=128
    return v ? atoi(v) : 0;
  }();
  if (use_lds == 2 && K >= 128) {
    hipLaunchKernelGGL(skinny_gemm_lds_kernel<128>, grid, block, 0,
                       (hipStream_t)stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (__hip_bfloat16*)y, M, N,
                       K);
  } else if (use_lds && K >= 512) {
    hipLaunchKernelGGL(skinny_gemm_lds_kernel<512>, grid, block, 0,
                       (hipStream_t)stream, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)w, (__hip_bfloat16*)y, M, N,
                       K);
  } else {
    static int unroll = [] {
      const char* v = getenv("TF_SKINNY_UNROLL");  // W-loads in flight/lane
      int u = v ? atoi(v) : 8;
      return (u == 12 || u == 16) ? u : 8;
    }();
    if (unroll == 16)
      hipLaunchKernelGGL(skinny_gemm_kernel<16>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)w, (__hip_bfloat16*)y, M, N,
                         K);
    else if (unroll == 12)
      hipLaunchKernelGGL(skinny_gemm_kernel<12>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)w, (__hip_bfloat16*)y, M, N,
                         K);
    else
      hipLaunchKernelGGL(skinny_gemm_kernel<8>, grid, block, 0,
                         (hipStream_t)stream, (const __hip_bfloat16*)x,
                         (const __hip_bfloat16*)w, (__hip_bfloat16*)y, M, N,
                         K);
  }
  return (int)hipGetLastError();
}

}  // extern "C"
