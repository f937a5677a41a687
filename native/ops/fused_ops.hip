// libtfops.so — fused gfx950 kernels for the serving workload's hot
// normalization path (guide G: fuse elementwise/normalisation work into
// one kernel; eager PyTorch spends ~5 kernels per RMSNorm: cast, pow,
// mean, rsqrt, two muls — ~15% of decode kernel time, see
// profiles/llama8b_decode_b8_kernel_stats_r01.md).
//
//   tf_rmsnorm:      out = x * rsqrt(mean(x², dim)+eps) * w      (bf16)
//   tf_add_rmsnorm:  res += x; out = rmsnorm(res) * w            (bf16)
//
// One 256-lane workgroup per row; squares accumulate in fp32 through a
// wave + LDS reduction (matches torch's fp32 internal math); 16-byte
// vectorized loads (8 bf16/lane/step). Rows are independent — no
// agent-scope fencing.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

constexpr int BLOCK = 256;

typedef uint32_t u4 __attribute__((ext_vector_type(4)));

union V8 {
  u4 v;
  __hip_bfloat16 h[8];
};

__device__ float block_sum(float v) {
  __shared__ float lds[BLOCK / 64];
  // wave-level reduce (64-wide)
  for (int off = 32; off; off >>= 1)
    v += __shfl_down(v, off, 64);
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) lds[wave] = v;
  __syncthreads();
  if (threadIdx.x < BLOCK / 64)
    v = lds[threadIdx.x];
  else
    v = 0.0f;
  if (threadIdx.x < 64)
    for (int off = BLOCK / 128; off; off >>= 1)
      v += __shfl_down(v, off, 64);
  __shared__ float total;
  if (threadIdx.x == 0) total = v;
  __syncthreads();
  return total;
}

__global__ void rmsnorm_kernel(const u4* __restrict__ x,
                               const u4* __restrict__ w,
                               u4* __restrict__ out, int n8, float eps) {
  const u4* xr = x + (size_t)blockIdx.x * n8;
  u4* orow = out + (size_t)blockIdx.x * n8;
  float acc = 0.0f;
  for (int i = threadIdx.x; i < n8; i += BLOCK) {
    V8 v{xr[i]};
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = __bfloat162float(v.h[k]);
      acc += f * f;
    }
  }
  float total = block_sum(acc);
  float inv = rsqrtf(total / (8.0f * n8) + eps);
  for (int i = threadIdx.x; i < n8; i += BLOCK) {
    V8 v{xr[i]}, g{w[i]}, o;
#pragma unroll
    for (int k = 0; k < 8; ++k)
      o.h[k] = __float2bfloat16(__bfloat162float(v.h[k]) * inv *
                                __bfloat162float(g.h[k]));
    orow[i] = o.v;
  }
}

__global__ void add_rmsnorm_kernel(const u4* __restrict__ x,
                                   u4* __restrict__ res,
                                   const u4* __restrict__ w,
                                   u4* __restrict__ out, int n8, float eps) {
  const u4* xr = x + (size_t)blockIdx.x * n8;
  u4* rr = res + (size_t)blockIdx.x * n8;
  u4* orow = out + (size_t)blockIdx.x * n8;
  float acc = 0.0f;
  for (int i = threadIdx.x; i < n8; i += BLOCK) {
    V8 a{xr[i]}, b{rr[i]}, s;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = __bfloat162float(a.h[k]) + __bfloat162float(b.h[k]);
      s.h[k] = __float2bfloat16(f);
      acc += f * f;
    }
    rr[i] = s.v;  // residual stream updated in place
  }
  float total = block_sum(acc);
  float inv = rsqrtf(total / (8.0f * n8) + eps);
  for (int i = threadIdx.x; i < n8; i += BLOCK) {
    V8 s{rr[i]}, g{w[i]}, o;
#pragma unroll
    for (int k = 0; k < 8; ++k)
      o.h[k] = __float2bfloat16(__bfloat162float(s.h[k]) * inv *
                                __bfloat162float(g.h[k]));
    orow[i] = o.v;
  }
}

// ---- fused RoPE + KV-cache write --------------------------------------
// The decode census (profiles/pmc_ktrace_r02.md) showed ~4.2 ms/step of
// rope/index elementwise soup — ~260 4.7 us kernels per step, as much
// time as all GEMMs. This kernel replaces apply_rope(q) + apply_rope(k)
// + the two cache index-writes of one attention layer with ONE launch:
//
//   qout[b,h,t,d]              = rope(qin[b,t,h*D+d], pos[t])
//   kcache[b,hk,pos[t],d]      = rope(kin[b,t,hk*D+d], pos[t])
//   vcache[b,hk,pos[t],d]      = vin[b,t,hk*D+d]
//
// Rope pairing is interleaved (x1=x[...,::2], x2=x[...,1::2]) against
// cos/sin[pos, j] fp32 tables — numerically identical math to the
// reference implementation (fp32 multiply, bf16 store).

__global__ void rope_qkv_cache_kernel(
    const u4* __restrict__ qin,   // [B,T,*] bf16, row stride qs elems
    const u4* __restrict__ kin,   // (may be slices of ONE merged qkv
    const u4* __restrict__ vin,   //  GEMM output — strides in elems)
    const float* __restrict__ cosT,  // [S, D/2]
    const float* __restrict__ sinT,
    const long long* __restrict__ pos,  // [T]
    u4* __restrict__ qout,        // [B,Hq,T,D]
    u4* __restrict__ kcache,      // [B,Hk,S,D]
    u4* __restrict__ vcache,
    int B, int T, int Hq, int Hk, int D, int S, int qs, int ks, int vs) {
  // grid.x = B*T, grid.y = Hq + 2*Hk; 64 threads each handling 8 elems
  // (4 rope pairs) — D % 8 == 0, D <= 512
  const int bt = blockIdx.x;
  const int b = bt / T, t = bt % T;
  int h = blockIdx.y;
  const int i8 = threadIdx.x;         // 8-element group index
  if (i8 * 8 >= D) return;
  const long long p = pos[t];
  const int half = D / 2;

  auto rope8 = [&](V8 in, int d0) {
    V8 o;
#pragma unroll
    for (int k = 0; k < 8; k += 2) {
      int j = (d0 + k) >> 1;  // pair index
      float c = cosT[(size_t)p * half + j];
      float sn = sinT[(size_t)p * half + j];
      float x1 = __bfloat162float(in.h[k]);
      float x2 = __bfloat162float(in.h[k + 1]);
      o.h[k] = __float2bfloat16(x1 * c - x2 * sn);
      o.h[k + 1] = __float2bfloat16(x2 * c + x1 * sn);
    }
    return o;
  };

  const int d0 = i8 * 8;
  if (h < Hq) {  // q head → rope → qout[b,h,t,:]
    V8 in{qin[((size_t)(b * T + t) * qs + h * D + d0) / 8]};
    V8 o = rope8(in, d0);
    qout[((size_t)((b * Hq + h) * T + t) * D + d0) / 8] = o.v;
  } else if (h < Hq + Hk) {  // k head → rope → kcache[b,hk,p,:]
    int hk = h - Hq;
    V8 in{kin[((size_t)(b * T + t) * ks + hk * D + d0) / 8]};
    V8 o = rope8(in, d0);
    kcache[((size_t)((b * Hk + hk) * S + p) * D + d0) / 8] = o.v;
  } else {  // v head → vcache
    int hk = h - Hq - Hk;
    vcache[((size_t)((b * Hk + hk) * S + p) * D + d0) / 8] =
        vin[((size_t)(b * T + t) * vs + hk * D + d0) / 8];
  }
}

__global__ void silu_mul_gu_kernel(const u4* __restrict__ gu,
                                   u4* __restrict__ out, int n8,
                                   int row8) {
  // grid.x = rows; each row: first n8 groups = gate, next n8 = up
  const u4* row = gu + (size_t)blockIdx.x * row8;
  u4* orow = out + (size_t)blockIdx.x * n8;
  for (int i = threadIdx.x; i < n8; i += blockDim.x) {
    V8 a{row[i]}, b{row[n8 + i]}, o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float x = __bfloat162float(a.h[k]);
      float sg = x / (1.0f + __expf(-x));
      o.h[k] = __float2bfloat16(sg * __bfloat162float(b.h[k]));
    }
    orow[i] = o.v;
  }
}

__global__ void silu_mul_kernel(const u4* __restrict__ g,
                                const u4* __restrict__ u,
                                u4* __restrict__ out, size_t n8) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n8) return;
  V8 a{g[i]}, b{u[i]}, o;
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    float x = __bfloat162float(a.h[k]);
    float s = x / (1.0f + __expf(-x));
    o.h[k] = __float2bfloat16(s * __bfloat162float(b.h[k]));
  }
  out[i] = o.v;
}

}  // namespace

extern "C" {

// out[r, i] = silu(gu[r, i]) * gu[r, I+i] — the gate+up halves of ONE
// merged GEMM output (bf16; I % 8 == 0).
int tf_silu_mul_gu(const void* gu, void* out, int rows, int inter,
                   void* stream);

// out = silu(g) * u, bf16, numel % 8 == 0 (one kernel instead of two
// eager ones on the MLP hot path).
int tf_silu_mul_gu(const void* gu, void* out, int rows, int inter,
                   void* stream) {
  if (inter % 8) return 1;
  hipLaunchKernelGGL(silu_mul_gu_kernel, dim3(rows), dim3(256), 0,
                     (hipStream_t)stream, (const u4*)gu, (u4*)out,
                     inter / 8, inter / 4);
  return (int)hipGetLastError();
}

int tf_silu_mul(const void* g, const void* u, void* out,
                long long numel, void* stream) {
  if (numel % 8) return 1;
  size_t n8 = (size_t)numel / 8;
  int block = 256;
  size_t grid = (n8 + block - 1) / block;
  hipLaunchKernelGGL(silu_mul_kernel, dim3((unsigned)grid), dim3(block), 0,
                     (hipStream_t)stream, (const u4*)g, (const u4*)u,
                     (u4*)out, n8);
  return (int)hipGetLastError();
}

// One launch per attention layer: rope(q) → qout, rope(k)/copy(v) →
// caches at pos. All tensors bf16 except cos/sin (fp32) and pos (i64).
int tf_rope_qkv_cache(const void* qin, const void* kin, const void* vin,
                      const void* cos_t, const void* sin_t,
                      const void* pos, void* qout, void* kcache,
                      void* vcache, int B, int T, int Hq, int Hk, int D,
                      int S, int q_stride, int k_stride, int v_stride,
                      void* stream) {
  if (D % 8 || D > 512 || (q_stride % 8) || (k_stride % 8) ||
      (v_stride % 8))
    return 1;
  dim3 grid(B * T, Hq + 2 * Hk);
  dim3 block((D + 7) / 8);
  hipLaunchKernelGGL(rope_qkv_cache_kernel, grid, block, 0,
                     (hipStream_t)stream, (const u4*)qin, (const u4*)kin,
                     (const u4*)vin, (const float*)cos_t,
                     (const float*)sin_t, (const long long*)pos, (u4*)qout,
                     (u4*)kcache, (u4*)vcache, B, T, Hq, Hk, D, S,
                     q_stride, k_stride, v_stride);
  return (int)hipGetLastError();
}

// x,w,out bf16; dim % 8 == 0. One workgroup per row.
int tf_rmsnorm(const void* x, const void* w, void* out, int rows, int dim,
               float eps, void* stream) {
  if (dim % 8) return 1;
  hipLaunchKernelGGL(rmsnorm_kernel, dim3(rows), dim3(BLOCK), 0,
                     (hipStream_t)stream, (const u4*)x, (const u4*)w,
                     (u4*)out, dim / 8, eps);
  return (int)hipGetLastError();
}

int tf_add_rmsnorm(const void* x, void* res, const void* w, void* out,
                   int rows, int dim, float eps, void* stream) {
  if (dim % 8) return 1;
  hipLaunchKernelGGL(add_rmsnorm_kernel, dim3(rows), dim3(BLOCK), 0,
                     (hipStream_t)stream, (const u4*)x, (u4*)res,
                     (const u4*)w, (u4*)out, dim / 8, eps);
  return (int)hipGetLastError();
}

}  // extern "C"
