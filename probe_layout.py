import sys, torch, os
sys.path.insert(0, "/root/repo")
os.environ["PYTORCH_ROCM_ARCH"] = "gfx950"
from torch.utils.cpp_extension import load_inline

COMMON = r'''
#include <hip/hip_runtime.h>
typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8_frag;
typedef __attribute__((ext_vector_type(4))) float f32x4_acc;
__device__ __forceinline__ float bf16_to_f32(ushort_t u) { unsigned int w = ((unsigned int)u) << 16; return __uint_as_float(w); }
__device__ __forceinline__ ushort_t f32_to_bf16(float f) { unsigned int w = __float_as_uint(f); unsigned int rb = 0x7FFF + ((w >> 16) & 1); return (ushort_t)((w + rb) >> 16); }
'''

VARIANTS = {
 "dswrite_stage": r'''
__global__ __launch_bounds__(256) void gemm_var(const ushort_t* A, const ushort_t* W, ushort_t* C, int M, int N, int K) {
  __shared__ ushort_t lds_a[128*32];
  __shared__ ushort_t lds_b[128*32];
  int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  int tiles_n = N / 128;
  int m0 = (blockIdx.x / tiles_n) * 128, n0 = (blockIdx.x % tiles_n) * 128;
  int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  f32x4_acc acc[4][4];
  for (int i = 0; i < 4; ++i) for (int j = 0; j < 4; ++j) acc[i][j] = {0.f,0.f,0.f,0.f};
  for (int k0 = 0; k0 < K; k0 += 32) {
    __syncthreads();
    // plain per-thread LDS writes: thread t covers 16 elements
    for (int i = tid; i < 128*32/8; i += 256) {
      int r = i / 4, seg = i % 4;
      for (int j = 0; j < 8; ++j) {
        lds_a[r*32 + seg*8 + j] = A[(long long)(m0 + r) * K + k0 + seg*8 + j];
        lds_b[r*32 + seg*8 + j] = W[(long long)(n0 + r) * K + k0 + seg*8 + j];
      }
    }
    __syncthreads();
    int fr = lane & 15, fq = lane >> 4;
    for (int mi = 0; mi < 4; ++mi) {
      bf16x8_frag a_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_a[(wr + mi*16 + fr)*32 + fq*8]);
      for (int ni = 0; ni < 4; ++ni) {
        bf16x8_frag b_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_b[(wc + ni*16 + fr)*32 + fq*8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
  }
  int fr = lane & 15, fq = lane >> 4;
  for (int mi = 0; mi < 4; ++mi) for (int ni = 0; ni < 4; ++ni) for (int reg = 0; reg < 4; ++reg) {
    int row = m0 + wr + mi*16 + fq*4 + reg, col = n0 + wc + ni*16 + fr;
    C[(long long)row * N + col] = f32_to_bf16(acc[mi][ni][reg]);
  }
}
''',
 "gload_stage": r'''
__global__ __launch_bounds__(256) void gemm_var(const ushort_t* A, const ushort_t* W, ushort_t* C, int M, int N, int K) {
  __shared__ ushort_t lds_a[128*32];
  __shared__ ushort_t lds_b[128*32];
  int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  int tiles_n = N / 128;
  int m0 = (blockIdx.x / tiles_n) * 128, n0 = (blockIdx.x % tiles_n) * 128;
  int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  f32x4_acc acc[4][4];
  for (int i = 0; i < 4; ++i) for (int j = 0; j < 4; ++j) acc[i][j] = {0.f,0.f,0.f,0.f};
  int row_of_t = tid >> 2, seg_of_t = tid & 3;
  long long a_src = (long long)(m0 + row_of_t) * K + seg_of_t * 8;
  long long b_src = (long long)(n0 + row_of_t) * K + seg_of_t * 8;
  for (int k0 = 0; k0 < K; k0 += 32) {
    __syncthreads();
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src + k0), (__attribute__((address_space(3))) void*)(lds_a + tid*8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(A + a_src + 64LL*K + k0), (__attribute__((address_space(3))) void*)(lds_a + 2048 + tid*8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(W + b_src + k0), (__attribute__((address_space(3))) void*)(lds_b + tid*8), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)(W + b_src + 64LL*K + k0), (__attribute__((address_space(3))) void*)(lds_b + 2048 + tid*8), 16, 0, 0);
    __syncthreads();
    int fr = lane & 15, fq = lane >> 4;
    for (int mi = 0; mi < 4; ++mi) {
      bf16x8_frag a_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_a[(wr + mi*16 + fr)*32 + fq*8]);
      for (int ni = 0; ni < 4; ++ni) {
        bf16x8_frag b_frag = *reinterpret_cast<const bf16x8_frag*>(&lds_b[(wc + ni*16 + fr)*32 + fq*8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
  }
  int fr = lane & 15, fq = lane >> 4;
  for (int mi = 0; mi < 4; ++mi) for (int ni = 0; ni < 4; ++ni) for (int reg = 0; reg < 4; ++reg) {
    int row = m0 + wr + mi*16 + fq*4 + reg, col = n0 + wc + ni*16 + fr;
    C[(long long)row * N + col] = f32_to_bf16(acc[mi][ni][reg]);
  }
}
''',
}

WRAPPER = r'''
#include <torch/extension.h>
torch::Tensor gemm_run(torch::Tensor A, torch::Tensor W) {
  int M = A.size(0), K = A.size(1), N = W.size(0);
  auto C = torch::zeros({M, N}, A.options());
  int grid = (M/128) * (N/128);
  hipLaunchKernelGGL(gemm_var, dim3(grid), dim3(256), 0, 0,
    (const ushort_t*)A.data_ptr(), (const ushort_t*)W.data_ptr(), (ushort_t*)C.data_ptr(), M, N, K);
  return C;
}
'''

torch.manual_seed(0)
M, N, K = 256, 256, 64
x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
ref = x.float() @ w.float().t()
for name, body in VARIANTS.items():
    mod = load_inline(name=f"gemmvar_{name}", cpp_sources="torch::Tensor gemm_run(torch::Tensor A, torch::Tensor W);", cuda_sources=COMMON + body + WRAPPER,
                      functions=["gemm_run"], with_cuda=True, verbose=False,
                      extra_cuda_cflags=["-O3", "--offload-arch=gfx950"])
    out = mod.gemm_run(x, w)
    err = (out.float() - ref).abs().max().item()
    print(f"{name}: max err {err:.4f}", flush=True)
