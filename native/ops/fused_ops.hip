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

}  // namespace

extern "C" {

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
