// Multi-tensor (foreach-style) kernels: one launch covers a whole parameter
// list instead of one kernel per tensor.
//
// Round-1 profiling (profiles/albert_kernels.md) showed 2323 single-tensor
// grad-accumulation adds at 33 us each (6.2% of step) and FusedAdamW
// launching once per parameter. The tensor table
// travels in kernarg space (a by-value struct, ~2.5 KB < the 4 KB limit);
// each thread binary-searches its tensor from the cumulative-offset table
// once per grid-stride chunk -- ~6 compares against a memory-bound body.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#include "multi_tensor.h"

typedef unsigned short ushort_t;

__device__ __forceinline__ int mt_find(const long long* cum, int n, long long i) {
  int lo = 0, hi = n - 1;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (i < cum[mid + 1]) hi = mid;
    else lo = mid + 1;
  }
  return lo;
}

__device__ __forceinline__ float mt_load_g(const void* g, long long j, bool bf16) {
  if (bf16) return __uint_as_float(((unsigned int)((const ushort_t*)g)[j]) << 16);
  return ((const float*)g)[j];
}

extern "C" __global__ void multi_tensor_adamw(
    MTAdamArgs args, float lr, float beta1, float beta2, float eps,
    float weight_decay, float bias_corr1, float bias_corr2) {
  const long long total = args.cum[args.n];
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int t = mt_find(args.cum, args.n, i);
    long long j = i - args.cum[t];
    float g = mt_load_g(args.g[t], j, (args.g_bf16_mask >> t) & 1);
    float m = args.m[t][j] = beta1 * args.m[t][j] + (1.f - beta1) * g;
    float v = args.v[t][j] = beta2 * args.v[t][j] + (1.f - beta2) * g * g;
    float denom = sqrtf(v / bias_corr2) + eps;
    float p = args.p[t][j];
    p -= lr * (m / bias_corr1 / denom + weight_decay * p);
    args.p[t][j] = p;
    if (args.mirror[t] != nullptr)
      args.mirror[t][j] = __bfloat16_as_ushort(__float2bfloat16(p));
  }
}

extern "C" __global__ void multi_tensor_accumulate(MTAccArgs args, float alpha) {
  const long long total = args.cum[args.n];
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int t = mt_find(args.cum, args.n, i);
    long long j = i - args.cum[t];
    args.acc[t][j] += alpha * mt_load_g(args.x[t], j, (args.x_bf16_mask >> t) & 1);
  }
}

extern "C" void launch_multi_tensor_adamw(const MTAdamArgs* args, float lr, float beta1,
                                          float beta2, float eps, float weight_decay,
                                          float bias_corr1, float bias_corr2, void* stream) {
  long long total = args->cum[args->n];
  long long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(multi_tensor_adamw, dim3((int)blocks), dim3(256), 0, (hipStream_t)stream,
                     *args, lr, beta1, beta2, eps, weight_decay, bias_corr1, bias_corr2);
}

extern "C" void launch_multi_tensor_accumulate(const MTAccArgs* args, float alpha, void* stream) {
  long long total = args->cum[args->n];
  long long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(multi_tensor_accumulate, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, *args, alpha);
}
