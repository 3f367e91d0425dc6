// Shared host/device declarations for the multi-tensor kernels.
#pragma once
#include <cstdint>

#define MT_CHUNK 48

typedef unsigned short mt_ushort;

struct MTAdamArgs {
  float* p[MT_CHUNK];
  const void* g[MT_CHUNK];
  float* m[MT_CHUNK];
  float* v[MT_CHUNK];
  mt_ushort* mirror[MT_CHUNK];
  long long cum[MT_CHUNK + 1];
  int n;
  unsigned long long g_bf16_mask;
};

struct MTAccArgs {
  float* acc[MT_CHUNK];
  const void* x[MT_CHUNK];
  long long cum[MT_CHUNK + 1];
  int n;
  unsigned long long x_bf16_mask;
};

extern "C" void launch_multi_tensor_adamw(const MTAdamArgs*, float, float, float, float,
                                          float, float, float, void*);
extern "C" void launch_multi_tensor_accumulate(const MTAccArgs*, float, void*);
