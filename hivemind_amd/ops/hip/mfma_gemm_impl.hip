// Separately-compiled TU for the MFMA GEMM kernels (host launchers below).
#include <hip/hip_runtime.h>
#include <cstdint>
typedef unsigned short ushort_t;
__device__ __forceinline__ float bf16_to_f32(ushort_t u) {
  unsigned int w = ((unsigned int)u) << 16;
  return __uint_as_float(w);
}
__device__ __forceinline__ ushort_t f32_to_bf16(float f) {
  unsigned int w = __float_as_uint(f);
  unsigned int rounding_bias = 0x7FFF + ((w >> 16) & 1);
  return (ushort_t)((w + rounding_bias) >> 16);
}
__device__ __forceinline__ float fast_tanh_g(float x) {
  x = fminf(fmaxf(x, -15.f), 15.f);
  float e = __expf(2.f * x);
  return (e - 1.f) / (e + 1.f);
}
__device__ __forceinline__ float gelu_tanh(float x) {
  const float c = 0.7978845608028654f;
  float inner = c * (x + 0.044715f * x * x * x);
  return 0.5f * x * (1.f + fast_tanh_g(inner));
}
// Hand-written CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ W[N,K]^T (+ fused bias+GELU
// epilogue). Structure follows the measured gfx950 ladder in
// cdna_hip_programming.md §5 ("m97"): 128x128 output tile, 4 waves computing
// 64x64 each as 4x4 fragments of v_mfma_f32_16x16x32_bf16, A/B staged through
// LDS with __builtin_amdgcn_global_load_lds width 16 (the +67% lever, common
// mistake #1), XCD-aware blockIdx swizzle for L2 locality (technique T1).
//
// W is the torch nn.Linear weight, stored [N, K] row-major -- exactly the
// "B^T input" layout the ladder uses, so both A and W stage with coalesced
// contiguous rows.
//
// Constraints (checked in the binding; the python wrapper pads M):
//   M % 128 == 0, N % 128 == 0, K % 32 == 0.


typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;   // MFMA A/B operand
typedef __attribute__((ext_vector_type(4))) float f32x4_acc;     // MFMA C/D accumulator

// epilogue selector: 0 = none (plain GEMM), 1 = bias + gelu_tanh
template <int EPILOGUE>
__global__ __launch_bounds__(256) void gemm_bt_bf16_t(
    const ushort_t* __restrict__ A,   // [M, K] bf16 row-major
    const ushort_t* __restrict__ W,   // [N, K] bf16 row-major (linear weight)
    const ushort_t* __restrict__ bias,  // [N] bf16 (EPILOGUE==1) or nullptr
    ushort_t* __restrict__ C,         // [M, N] bf16 row-major
    ushort_t* __restrict__ pre_act,   // [M, N] saved x+bias for backward; may be null
    int M, int N, int K) {
  constexpr int BM = 128, BN = 128, BK = 32;
  __shared__ ushort_t lds_a[BM * BK];  // [128][32] row-major, matches gload_lds linear writes
  __shared__ ushort_t lds_b[BN * BK];

  // XCD-aware swizzle (bijective form, guide ERRATA #11): consecutive
  // workgroups share W panels; keep them on one XCD's L2.
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    constexpr int NXCD = 8;
    int q = nwg / NXCD, r = nwg % NXCD;
    int xcd = wg % NXCD, idx = wg / NXCD;
    wg = (xcd < r) ? (xcd * (q + 1) + idx) : (r * (q + 1) + (xcd - r) * q + idx);
  }
  const int tiles_n = N / BN;
  const int tile_m = wg / tiles_n, tile_n = wg % tiles_n;
  const int m0 = tile_m * BM, n0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;  // wave's 64x64 sub-tile

  f32x4_acc acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // staging: thread t loads 16 B of row (t/4), segment (t%4); one issue covers
  // 64 rows (256 threads x 8 bf16), so each 128x32 tile takes two issues.
  // LDS dest is linear in thread order (wave-uniform base + lane*16 --
  // the gload_lds contract, §5 caveat).
  const int row_of_t = tid >> 2, seg_of_t = tid & 3;
  const long long a_src_base = (long long)(m0 + row_of_t) * K + seg_of_t * 8;
  const long long b_src_base = (long long)(n0 + row_of_t) * K + seg_of_t * 8;
  const long long half_rows_stride = 64LL * K;  // second issue: rows 64..127
  constexpr int HALF_LDS = 64 * BK;             // 2048 bf16

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src_base + k0),
                                     (__attribute__((address_space(3))) void*)(lds_a + tid * 8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src_base + half_rows_stride + k0),
                                     (__attribute__((address_space(3))) void*)(lds_a + HALF_LDS + tid * 8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(W + b_src_base + k0),
                                     (__attribute__((address_space(3))) void*)(lds_b + tid * 8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(W + b_src_base + half_rows_stride + k0),
                                     (__attribute__((address_space(3))) void*)(lds_b + HALF_LDS + tid * 8), 16, 0, 0);
    __syncthreads();

    // fragment loads: lane l reads row (16-block + l&15), k = (l>>4)*8 .. +8
    const int fr = lane & 15, fq = lane >> 4;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      bf16x8_frag a_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_a[(wr + mi * 16 + fr) * BK + fq * 8]);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        bf16x8_frag b_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_b[(wc + ni * 16 + fr) * BK + fq * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
  }

  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg (guide §3)
  const int fr = lane & 15, fq = lane >> 4;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int row = m0 + wr + mi * 16 + fq * 4 + reg;
        int col = n0 + wc + ni * 16 + fr;
        float v = acc[mi][ni][reg];
        long long out_idx = (long long)row * N + col;
        if (EPILOGUE == 1) {
          v += bf16_to_f32(bias[col]);
          if (pre_act != nullptr) pre_act[out_idx] = f32_to_bf16(v);
          v = gelu_tanh(v);
        }
        C[out_idx] = f32_to_bf16(v);
      }
    }
  }
}


// Layout probe: lane l takes its 8 A/B operand values directly from
// a_vals[l*8+j]; the resulting C (written with the verified C/D mapping)
// reveals which (row, k) each (lane, j) slot corresponds to.
extern "C" __global__ void mfma_probe_16x16x32_bf16(const ushort_t* __restrict__ a_vals,
                                                    const ushort_t* __restrict__ b_vals,
                                                    float* __restrict__ c_out) {
  int lane = threadIdx.x & 63;
  bf16x8_frag a_frag, b_frag;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a_frag[j] = (short)a_vals[lane * 8 + j];
    b_frag[j] = (short)b_vals[lane * 8 + j];
  }
  f32x4_acc acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    int row = (lane >> 4) * 4 + reg;
    int col = lane & 15;
    c_out[row * 16 + col] = acc[reg];
  }
}


// staging probe: run the exact two-issue gload_lds staging pattern for one
// 128x32 tile, then copy LDS back out for host-side verification.
extern "C" __global__ __launch_bounds__(256) void stage_probe_bf16(const ushort_t* __restrict__ A,
                                                                   ushort_t* __restrict__ out, int K) {
  __shared__ ushort_t lds_a[128 * 32];
  int tid = threadIdx.x;
  const int row_of_t = tid >> 2, seg_of_t = tid & 3;
  const long long a_src = (long long)row_of_t * K + seg_of_t * 8;
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src),
                                   (__attribute__((address_space(3))) void*)(lds_a + tid * 8), 16, 0, 0);
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src + 64LL * K),
                                   (__attribute__((address_space(3))) void*)(lds_a + 2048 + tid * 8), 16, 0, 0);
  __syncthreads();
  for (int i = tid; i < 4096; i += 256) out[i] = lds_a[i];
}


extern "C" void launch_gemm_bt_bf16(const void* A, const void* W, const void* bias,
                                    void* C, void* pre_act, int M, int N, int K,
                                    int epilogue, void* stream) {
  int grid = (M / 128) * (N / 128);
  if (epilogue == 1) {
    hipLaunchKernelGGL(gemm_bt_bf16_t<1>, dim3(grid), dim3(256), 0, (hipStream_t)stream,
                       (const ushort_t*)A, (const ushort_t*)W, (const ushort_t*)bias,
                       (ushort_t*)C, (ushort_t*)pre_act, M, N, K);
  } else {
    hipLaunchKernelGGL(gemm_bt_bf16_t<0>, dim3(grid), dim3(256), 0, (hipStream_t)stream,
                       (const ushort_t*)A, (const ushort_t*)W, nullptr,
                       (ushort_t*)C, nullptr, M, N, K);
  }
}

extern "C" void launch_mfma_probe(const void* a_vals, const void* b_vals, void* c_out, void* stream) {
  hipLaunchKernelGGL(mfma_probe_16x16x32_bf16, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     (const ushort_t*)a_vals, (const ushort_t*)b_vals, (float*)c_out);
}

extern "C" void launch_stage_probe(const void* A, void* out, int K, void* stream) {
  hipLaunchKernelGGL(stage_probe_bf16, dim3(1), dim3(256), 0, (hipStream_t)stream,
                     (const ushort_t*)A, (ushort_t*)out, K);
}
