// Memory-bound fused kernels for the training hot path, written directly for
// CDNA4 / gfx950 (MI355X): 64-wide wavefronts, vectorized bf16 loads
// (ushort2/short4 reinterpret -- hipcc does not auto-vectorize bf16, guide
// G13), grid-stride loops capped near CU count * waves (guide G11).
//
// Kernels (reference hot spots):
//   K2  apply_delta            tensor += alpha * delta
//   K1  weighted_accumulate    acc += w * x
//   K3  fp16 codec             clamp + cast both directions
//   K7  blockwise int8 codec   absmax per 4096 block, linear int8
//   K13 fused Adam             fp32 master step + bf16 param mirror
//   K9  bias+GELU fwd/bwd      (tanh approximation, as the reference's gelu_fast)
//   --  LayerNorm fwd/bwd      fused residual-add option
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define DEVINL __device__ __forceinline__

typedef unsigned short ushort_t;

DEVINL float bf16_to_f32(ushort_t u) {
  unsigned int w = ((unsigned int)u) << 16;
  return __uint_as_float(w);
}
DEVINL ushort_t f32_to_bf16(float f) {
  unsigned int w = __float_as_uint(f);
  // round-to-nearest-even
  unsigned int rounding_bias = 0x7FFF + ((w >> 16) & 1);
  return (ushort_t)((w + rounding_bias) >> 16);
}

// ---------------------------------------------------------------------------
// K2: tensor += alpha * delta   (bf16 tensor, fp32/bf16 delta)
// ---------------------------------------------------------------------------
extern "C" __global__ void apply_delta_bf16(ushort_t* __restrict__ tensor,
                                            const ushort_t* __restrict__ delta,
                                            float alpha, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  // vectorize: 2 bf16 per lane per iteration (4-byte coalesced)
  long long n2 = n / 2;
  const uint32_t* d2 = reinterpret_cast<const uint32_t*>(delta);
  uint32_t* t2 = reinterpret_cast<uint32_t*>(tensor);
  for (long long i = i0; i < n2; i += stride) {
    uint32_t tv = t2[i], dv = d2[i];
    float a0 = bf16_to_f32((ushort_t)(tv & 0xffff)) + alpha * bf16_to_f32((ushort_t)(dv & 0xffff));
    float a1 = bf16_to_f32((ushort_t)(tv >> 16)) + alpha * bf16_to_f32((ushort_t)(dv >> 16));
    t2[i] = (uint32_t)f32_to_bf16(a0) | ((uint32_t)f32_to_bf16(a1) << 16);
  }
  if (i0 == 0 && (n & 1)) {
    long long last = n - 1;
    tensor[last] = f32_to_bf16(bf16_to_f32(tensor[last]) + alpha * bf16_to_f32(delta[last]));
  }
}

extern "C" __global__ void apply_delta_f32(float* __restrict__ tensor,
                                           const float* __restrict__ delta,
                                           float alpha, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) tensor[i] += alpha * delta[i];
}

// ---------------------------------------------------------------------------
// K1: acc += w * x (fp32 accumulator, bf16 or fp32 input)
// ---------------------------------------------------------------------------
extern "C" __global__ void weighted_accumulate_f32(float* __restrict__ acc,
                                                   const float* __restrict__ x,
                                                   float w, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) acc[i] = fmaf(w, x[i], acc[i]);
}

extern "C" __global__ void weighted_accumulate_bf16_f32(float* __restrict__ acc,
                                                        const ushort_t* __restrict__ x,
                                                        float w, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) acc[i] = fmaf(w, bf16_to_f32(x[i]), acc[i]);
}

// ---------------------------------------------------------------------------
// K3: fp16 codec with clamp (reference floating.py:14-41 semantics)
// ---------------------------------------------------------------------------
extern "C" __global__ void compress_fp16(const float* __restrict__ in,
                                         __half* __restrict__ out, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) {
    float v = in[i];
    v = fminf(fmaxf(v, -65504.f), 65504.f);
    out[i] = __float2half(v);
  }
}

extern "C" __global__ void decompress_fp16(const __half* __restrict__ in,
                                           float* __restrict__ out, long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) out[i] = __half2float(in[i]);
}

// ---------------------------------------------------------------------------
// K7: blockwise int8: absmax per BLK elements, q = round(x * 127 / absmax)
// one workgroup (256 threads) handles one 4096-block: two-stage reduction in LDS
// ---------------------------------------------------------------------------
#define QBLK 4096
extern "C" __global__ void quantize_blockwise_int8(const float* __restrict__ in,
                                                   int8_t* __restrict__ q,
                                                   float* __restrict__ absmax,
                                                   long long n) {
  __shared__ float red[256 / 64];  // one partial max per wave
  long long block = blockIdx.x;
  long long base = block * QBLK;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  float local_max = 0.f;
  for (int i = tid; i < QBLK; i += blockDim.x) {
    long long idx = base + i;
    float v = (idx < n) ? in[idx] : 0.f;
    local_max = fmaxf(local_max, fabsf(v));
  }
  // wave reduce (64 lanes)
  for (int off = 32; off > 0; off >>= 1)
    local_max = fmaxf(local_max, __shfl_down(local_max, off));
  if (lane == 0) red[wave] = local_max;
  __syncthreads();
  if (tid == 0) {
    float m = red[0];
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) m = fmaxf(m, red[w]);
    red[0] = m;
    absmax[block] = m;
  }
  __syncthreads();
  float scale = red[0] > 0.f ? 127.f / red[0] : 0.f;
  for (int i = tid; i < QBLK; i += blockDim.x) {
    long long idx = base + i;
    if (idx < n) {
      float v = in[idx] * scale;
      v = fminf(fmaxf(v, -127.f), 127.f);
      q[idx] = (int8_t)__float2int_rn(v);
    }
  }
}

extern "C" __global__ void dequantize_blockwise_int8(const int8_t* __restrict__ q,
                                                     const float* __restrict__ absmax,
                                                     float* __restrict__ out,
                                                     long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) {
    float scale = absmax[i / QBLK] / 127.f;
    out[i] = (float)q[i] * scale;
  }
}

// ---------------------------------------------------------------------------
// K13: fused Adam step on fp32 master params with optional bf16 mirror
// p -= lr * ( m_hat / (sqrt(v_hat) + eps) + wd * p )   (AdamW-style decoupled wd)
// ---------------------------------------------------------------------------
extern "C" __global__ void fused_adamw_f32(float* __restrict__ param,
                                           const float* __restrict__ grad,
                                           float* __restrict__ exp_avg,
                                           float* __restrict__ exp_avg_sq,
                                           ushort_t* __restrict__ param_bf16_mirror,  // may be null
                                           float lr, float beta1, float beta2,
                                           float eps, float weight_decay,
                                           float bias_corr1, float bias_corr2,
                                           long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) {
    float g = grad[i];
    float m = exp_avg[i] = fmaf(beta1, exp_avg[i], (1.f - beta1) * g);
    float v = exp_avg_sq[i] = fmaf(beta2, exp_avg_sq[i], (1.f - beta2) * g * g);
    float m_hat = m / bias_corr1;
    float v_hat = v / bias_corr2;
    float p = param[i];
    p -= lr * (m_hat / (sqrtf(v_hat) + eps) + weight_decay * p);
    param[i] = p;
    if (param_bf16_mirror != nullptr) param_bf16_mirror[i] = f32_to_bf16(p);
  }
}

// grads arriving in bf16 (model grads) -- same step, bf16 grad load
extern "C" __global__ void fused_adamw_bf16grad(float* __restrict__ param,
                                                const ushort_t* __restrict__ grad,
                                                float* __restrict__ exp_avg,
                                                float* __restrict__ exp_avg_sq,
                                                ushort_t* __restrict__ param_bf16_mirror,
                                                float lr, float beta1, float beta2,
                                                float eps, float weight_decay,
                                                float bias_corr1, float bias_corr2,
                                                long long n) {
  long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n; i += stride) {
    float g = bf16_to_f32(grad[i]);
    float m = exp_avg[i] = fmaf(beta1, exp_avg[i], (1.f - beta1) * g);
    float v = exp_avg_sq[i] = fmaf(beta2, exp_avg_sq[i], (1.f - beta2) * g * g);
    float m_hat = m / bias_corr1;
    float v_hat = v / bias_corr2;
    float p = param[i];
    p -= lr * (m_hat / (sqrtf(v_hat) + eps) + weight_decay * p);
    param[i] = p;
    if (param_bf16_mirror != nullptr) param_bf16_mirror[i] = f32_to_bf16(p);
  }
}

// ---------------------------------------------------------------------------
// K9a: bias + GELU (tanh approx) forward, bf16 activations
//   out = gelu(x + bias);  x: [rows, cols], bias: [cols]
// ---------------------------------------------------------------------------
// libm tanhf costs ~10x a VALU op; tanh(x) = (e^2x - 1)/(e^2x + 1) with the
// hardware __expf is bf16-exact in practice. Clamp keeps e^2x finite.
DEVINL float fast_tanh(float x) {
  x = fminf(fmaxf(x, -15.f), 15.f);
  float e = __expf(2.f * x);
  return (e - 1.f) / (e + 1.f);
}
DEVINL float gelu_tanh(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float inner = c * (x + 0.044715f * x * x * x);
  return 0.5f * x * (1.f + fast_tanh(inner));
}
DEVINL float gelu_tanh_grad(float x) {
  const float c = 0.7978845608028654f;
  float x2 = x * x;
  float inner = c * (x + 0.044715f * x * x2);
  float t = fast_tanh(inner);
  float sech2 = 1.f - t * t;
  return 0.5f * (1.f + t) + 0.5f * x * sech2 * c * (1.f + 3.f * 0.044715f * x2);
}

extern "C" __global__ void bias_gelu_fwd_bf16(const ushort_t* __restrict__ x,
                                              const ushort_t* __restrict__ bias,
                                              ushort_t* __restrict__ out,
                                              ushort_t* __restrict__ pre_act,  // saved for bwd (x+bias); may be null
                                              long long rows, long long cols) {
  const uint32_t* x2 = reinterpret_cast<const uint32_t*>(x);
  const uint32_t* b2 = reinterpret_cast<const uint32_t*>(bias);
  uint32_t* out2 = reinterpret_cast<uint32_t*>(out);
  uint32_t* pre2 = reinterpret_cast<uint32_t*>(pre_act);
  long long cols2 = cols >> 1;
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    long long base = row * cols2;
    for (long long c = threadIdx.x; c < cols2; c += blockDim.x) {
      uint32_t xv = x2[base + c], bv = b2[c];
      float v0 = bf16_to_f32((ushort_t)(xv & 0xffff)) + bf16_to_f32((ushort_t)(bv & 0xffff));
      float v1 = bf16_to_f32((ushort_t)(xv >> 16)) + bf16_to_f32((ushort_t)(bv >> 16));
      if (pre_act != nullptr)
        pre2[base + c] = (uint32_t)f32_to_bf16(v0) | ((uint32_t)f32_to_bf16(v1) << 16);
      out2[base + c] = (uint32_t)f32_to_bf16(gelu_tanh(v0)) | ((uint32_t)f32_to_bf16(gelu_tanh(v1)) << 16);
    }
  }
}

// K9b: backward: dx = dy * gelu'(pre_act). Each thread owns a fixed set of
// column-pairs (c = tid + p*blockDim, identical in every row), so dbias
// accumulates in REGISTERS -- zero atomics inside the row loop; one global
// atomicAdd per owned column at the end (spread over `cols` addresses).
// PAIRS is compile-time: runtime-indexed register arrays spill to scratch
// (cdna_hip_programming.md rule #20). Dispatch in bindings covers
// cols <= 2 * PAIRS_MAX * 256.
// FROM_XB: `pre` is the raw GEMM output x and the bias is added here (the
// x tensor is saved by autograd for the GEMM backward anyway, so the forward
// pass never writes a separate pre_act -- 384 MB less traffic per call at
// ALBERT-base batch 128). FROM_XB=false: `pre` is a materialized pre_act
// (the MFMA fused-epilogue path, where x never exists in memory).
template <int PAIRS, bool FROM_XB>
__global__ void bias_gelu_bwd_bf16_t(const ushort_t* __restrict__ dy,
                                     const ushort_t* __restrict__ pre,
                                     const ushort_t* __restrict__ bias,  // null unless FROM_XB
                                     ushort_t* __restrict__ dx,
                                     float* __restrict__ partial,  // [gridDim, cols] per-block dbias
                                     long long rows, long long cols) {
  const uint32_t* dy2 = reinterpret_cast<const uint32_t*>(dy);
  const uint32_t* pre2 = reinterpret_cast<const uint32_t*>(pre);
  const uint32_t* b2 = reinterpret_cast<const uint32_t*>(bias);
  uint32_t* dx2 = reinterpret_cast<uint32_t*>(dx);
  long long cols2 = cols >> 1;

  float acc0[PAIRS], acc1[PAIRS];
  float bia0[PAIRS], bia1[PAIRS];  // per-thread bias columns, loaded once
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    acc0[p] = acc1[p] = 0.f;
    bia0[p] = bia1[p] = 0.f;
    if (FROM_XB) {
      long long c = threadIdx.x + (long long)p * blockDim.x;
      if (c < cols2) {
        uint32_t bv = b2[c];
        bia0[p] = bf16_to_f32((ushort_t)(bv & 0xffff));
        bia1[p] = bf16_to_f32((ushort_t)(bv >> 16));
      }
    }
  }

  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    long long base = row * cols2;
#pragma unroll
    for (int p = 0; p < PAIRS; ++p) {
      long long c = threadIdx.x + (long long)p * blockDim.x;
      if (c >= cols2) break;
      uint32_t dyv = dy2[base + c], pv = pre2[base + c];
      float p0 = bf16_to_f32((ushort_t)(pv & 0xffff));
      float p1 = bf16_to_f32((ushort_t)(pv >> 16));
      if (FROM_XB) { p0 += bia0[p]; p1 += bia1[p]; }
      float g0 = bf16_to_f32((ushort_t)(dyv & 0xffff)) * gelu_tanh_grad(p0);
      float g1 = bf16_to_f32((ushort_t)(dyv >> 16)) * gelu_tanh_grad(p1);
      dx2[base + c] = (uint32_t)f32_to_bf16(g0) | ((uint32_t)f32_to_bf16(g1) << 16);
      acc0[p] += g0;
      acc1[p] += g1;
    }
  }
  // each column is owned by exactly one thread in this block: store the
  // block's partial sums contiguously; a tiny second kernel reduces over
  // blocks. Global float atomics here would serialize gridDim-deep per
  // address (measured ~200 us of pure atomic tail at gridDim=2048).
  float* prow = partial + (long long)blockIdx.x * cols;
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    long long c = threadIdx.x + (long long)p * blockDim.x;
    if (c >= cols2) break;
    prow[2 * c] = acc0[p];
    prow[2 * c + 1] = acc1[p];
  }
}

// ---------------------------------------------------------------------------
// LayerNorm forward: one wave per row, bf16 in/out, fp32 stats. The row's
// values are STASHED IN REGISTERS between the stats pass and the normalize
// pass (PAIRS column-pairs per lane, compile-time -- rule #20), so the input
// is read from HBM exactly once. Optionally fuses a residual add:
// h = x + residual; y = ln(h). cols <= 2 * PAIRS * 64.
// ---------------------------------------------------------------------------
template <int PAIRS>
__global__ void layernorm_fwd_bf16_t(const ushort_t* __restrict__ x,
                                     const ushort_t* __restrict__ residual,  // may be null
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     ushort_t* __restrict__ y,
                                     ushort_t* __restrict__ h_out,  // saved normalized input source (x+res); may be null
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     float eps, long long rows, int cols) {
  // blockDim.x = 256 -> 4 waves; each wave owns one row
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  long long row = (long long)blockIdx.x * 4 + wave;
  if (row >= rows) return;
  const uint32_t* xr2 = reinterpret_cast<const uint32_t*>(x + row * cols);
  const uint32_t* rr2 = residual ? reinterpret_cast<const uint32_t*>(residual + row * cols) : nullptr;
  uint32_t* hr2 = h_out ? reinterpret_cast<uint32_t*>(h_out + row * cols) : nullptr;
  uint32_t* yr2 = reinterpret_cast<uint32_t*>(y + row * cols);
  int cols2 = cols >> 1;

  float v0[PAIRS], v1[PAIRS];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    int c = lane + p * 64;
    v0[p] = v1[p] = 0.f;
    if (c < cols2) {
      uint32_t xv = xr2[c];
      float a = bf16_to_f32((ushort_t)(xv & 0xffff));
      float b = bf16_to_f32((ushort_t)(xv >> 16));
      if (rr2) {
        uint32_t rv = rr2[c];
        a += bf16_to_f32((ushort_t)(rv & 0xffff));
        b += bf16_to_f32((ushort_t)(rv >> 16));
      }
      if (hr2) hr2[c] = (uint32_t)f32_to_bf16(a) | ((uint32_t)f32_to_bf16(b) << 16);
      v0[p] = a; v1[p] = b;
      sum += a + b;
      sumsq += a * a + b * b;
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off);
    sumsq += __shfl_down(sumsq, off);
  }
  sum = __shfl(sum, 0);
  sumsq = __shfl(sumsq, 0);
  float mean = sum / cols;
  float var = sumsq / cols - mean * mean;
  float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    int c = lane + p * 64;
    if (c < cols2) {
      float n0 = fmaf((v0[p] - mean) * rstd, gamma[2 * c], beta[2 * c]);
      float n1 = fmaf((v1[p] - mean) * rstd, gamma[2 * c + 1], beta[2 * c + 1]);
      yr2[c] = (uint32_t)f32_to_bf16(n0) | ((uint32_t)f32_to_bf16(n1) << 16);
    }
  }
}

// LayerNorm backward: wave-per-row dx; dgamma/dbeta first accumulated in LDS
// per block, then ONE global atomicAdd per column per block. Each block
// grid-strides over rows (4 waves x many rows) so the number of global
// atomics is cols x gridDim, independent of row count. Dynamic LDS:
// 2 * cols * sizeof(float).
// LayerNorm backward: wave-per-row dx; each lane owns fixed column-pairs
// (c = lane + p*64, identical in every row), so dgamma/dbeta accumulate in
// registers across all rows the wave visits; one global atomicAdd per owned
// column at the end. PAIRS compile-time (rule #20). cols <= 2 * PAIRS * 64.
template <int PAIRS>
__global__ void layernorm_bwd_bf16_t(const ushort_t* __restrict__ dy,
                                     const ushort_t* __restrict__ h,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     ushort_t* __restrict__ dx,
                                     float* __restrict__ partial,  // [gridDim, 2*cols]: dgamma then dbeta
                                     long long rows, int cols) {
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  int waves_per_block = blockDim.x >> 6;
  int cols2 = cols >> 1;

  float accg0[PAIRS], accg1[PAIRS], accb0[PAIRS], accb1[PAIRS];
  float gam0[PAIRS], gam1[PAIRS];  // each lane's gamma columns, loaded once
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    accg0[p] = accg1[p] = accb0[p] = accb1[p] = 0.f;
    int c = lane + p * 64;
    gam0[p] = (c < cols2) ? gamma[2 * c] : 0.f;
    gam1[p] = (c < cols2) ? gamma[2 * c + 1] : 0.f;
  }

  for (long long row = (long long)blockIdx.x * waves_per_block + wave; row < rows;
       row += (long long)gridDim.x * waves_per_block) {
    const uint32_t* dyr2 = reinterpret_cast<const uint32_t*>(dy + row * cols);
    const uint32_t* hr2 = reinterpret_cast<const uint32_t*>(h + row * cols);
    uint32_t* dxr2 = reinterpret_cast<uint32_t*>(dx + row * cols);
    float mu = mean[row], rs = rstd[row];

    float c1 = 0.f, c2 = 0.f;
#pragma unroll
    for (int p = 0; p < PAIRS; ++p) {
      int c = lane + p * 64;
      if (c >= cols2) break;
      uint32_t dyv = dyr2[c], hv = hr2[c];
      float dy0 = bf16_to_f32((ushort_t)(dyv & 0xffff));
      float dy1 = bf16_to_f32((ushort_t)(dyv >> 16));
      float xh0 = (bf16_to_f32((ushort_t)(hv & 0xffff)) - mu) * rs;
      float xh1 = (bf16_to_f32((ushort_t)(hv >> 16)) - mu) * rs;
      float dg0 = dy0 * gam0[p], dg1 = dy1 * gam1[p];
      c1 += dg0 * xh0 + dg1 * xh1;
      c2 += dg0 + dg1;
      accg0[p] += dy0 * xh0;
      accg1[p] += dy1 * xh1;
      accb0[p] += dy0;
      accb1[p] += dy1;
    }
    for (int off = 32; off > 0; off >>= 1) {
      c1 += __shfl_down(c1, off);
      c2 += __shfl_down(c2, off);
    }
    c1 = __shfl(c1, 0) / cols;
    c2 = __shfl(c2, 0) / cols;
#pragma unroll
    for (int p = 0; p < PAIRS; ++p) {
      int c = lane + p * 64;
      if (c >= cols2) break;
      uint32_t dyv = dyr2[c], hv = hr2[c];
      float xh0 = (bf16_to_f32((ushort_t)(hv & 0xffff)) - mu) * rs;
      float xh1 = (bf16_to_f32((ushort_t)(hv >> 16)) - mu) * rs;
      float dg0 = bf16_to_f32((ushort_t)(dyv & 0xffff)) * gam0[p];
      float dg1 = bf16_to_f32((ushort_t)(dyv >> 16)) * gam1[p];
      float o0 = (dg0 - c2 - xh0 * c1) * rs;
      float o1 = (dg1 - c2 - xh1 * c1) * rs;
      dxr2[c] = (uint32_t)f32_to_bf16(o0) | ((uint32_t)f32_to_bf16(o1) << 16);
    }
  }
  // columns are owned per WAVE here (4 waves share each column): merge the
  // waves' register accumulators through LDS (depth-4 LDS atomics), then
  // store ONE per-block partial row; a second kernel reduces over blocks.
  // Global atomics would be 4*gridDim-deep per address (~8192 at gridDim
  // 2048) and dominated this kernel's runtime.
  __shared__ float smem[2 * PAIRS * 128];  // [dgamma(cols) | dbeta(cols)], cols <= PAIRS*128
  for (int c = threadIdx.x; c < 2 * cols; c += blockDim.x) smem[c] = 0.f;
  __syncthreads();
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    int c = lane + p * 64;
    if (c >= cols2) break;
    atomicAdd(&smem[2 * c], accg0[p]);
    atomicAdd(&smem[2 * c + 1], accg1[p]);
    atomicAdd(&smem[cols + 2 * c], accb0[p]);
    atomicAdd(&smem[cols + 2 * c + 1], accb1[p]);
  }
  __syncthreads();
  float* prow = partial + (long long)blockIdx.x * (2 * cols);
  for (int c = threadIdx.x; c < 2 * cols; c += blockDim.x) prow[c] = smem[c];
}

// ---------------------------------------------------------------------------
// Fused masked-LM cross-entropy over a large vocab (reference: the ALBERT
// example's MLM loss; torch's F.cross_entropy materializes a [N, V] log-softmax
// for backward -- 3.9 GB at N=65536, V=30000). This version saves only the
// per-row logsumexp (fp32 [N]) and recomputes softmax in backward: two
// streaming passes over the bf16 logits, no giant activation.
//
// fwd: one block (256 threads) per row; online max/sum-exp merge in registers,
//      then wave shuffles + LDS. Rows with label < 0 (ignore_index) contribute
//      nothing. Emits sum of losses + count of valid rows via global atomics.
// bwd: dlogits = (softmax - onehot(label)) * (*upstream) / max(*valid, 1),
//      zero for ignored rows. upstream/valid stay on device (no host sync).
// ---------------------------------------------------------------------------
extern "C" __global__ void cross_entropy_fwd_bf16(const ushort_t* __restrict__ logits,
                                                  const long long* __restrict__ labels,
                                                  float* __restrict__ lse,
                                                  float* __restrict__ loss_sum,
                                                  int* __restrict__ valid_count,
                                                  int n_rows, int n_cols) {
  int row = blockIdx.x;
  if (row >= n_rows) return;
  const ushort_t* lrow = logits + (long long)row * n_cols;
  const uint32_t* l2 = reinterpret_cast<const uint32_t*>(lrow);
  int tid = threadIdx.x;
  int n2 = n_cols >> 1;
  float m = -__builtin_inff(), s = 0.f;
  for (int i = tid; i < n2; i += blockDim.x) {
    uint32_t v = l2[i];
    float a = bf16_to_f32((ushort_t)(v & 0xffff));
    float b = bf16_to_f32((ushort_t)(v >> 16));
    float mn = fmaxf(m, fmaxf(a, b));
    s = s * __expf(m - mn) + __expf(a - mn) + __expf(b - mn);
    m = mn;
  }
  if ((n_cols & 1) && tid == 0) {
    float a = bf16_to_f32(lrow[n_cols - 1]);
    float mn = fmaxf(m, a);
    s = s * __expf(m - mn) + __expf(a - mn);
    m = mn;
  }
  // merge partial (m, s) across the wave, then across waves via LDS
  for (int off = 32; off > 0; off >>= 1) {
    float mo = __shfl_down(m, off), so = __shfl_down(s, off);
    float mn = fmaxf(m, mo);
    s = s * __expf(m - mn) + so * __expf(mo - mn);
    m = mn;
  }
  __shared__ float sm[4], ss[4];
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0) { sm[wave] = m; ss[wave] = s; }
  __syncthreads();
  if (tid == 0) {
    float M = sm[0], S = ss[0];
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) {
      float mn = fmaxf(M, sm[w]);
      S = S * __expf(M - mn) + ss[w] * __expf(sm[w] - mn);
      M = mn;
    }
    float l = __logf(S) + M;
    lse[row] = l;
    long long lab = labels[row];
    if (lab >= 0 && lab < n_cols) {
      atomicAdd(loss_sum, l - bf16_to_f32(lrow[lab]));
      atomicAdd(valid_count, 1);
    }
  }
}

extern "C" __global__ void cross_entropy_bwd_bf16(const ushort_t* __restrict__ logits,
                                                  const long long* __restrict__ labels,
                                                  const float* __restrict__ lse,
                                                  const float* __restrict__ upstream,
                                                  const int* __restrict__ valid_count,
                                                  ushort_t* __restrict__ dlogits,
                                                  int n_rows, int n_cols) {
  int row = blockIdx.x;
  if (row >= n_rows) return;
  const ushort_t* lrow = logits + (long long)row * n_cols;
  const uint32_t* l2 = reinterpret_cast<const uint32_t*>(lrow);
  uint32_t* d2 = reinterpret_cast<uint32_t*>(dlogits + (long long)row * n_cols);
  int tid = threadIdx.x;
  int n2 = n_cols >> 1;
  long long lab = labels[row];
  float l = lse[row];
  int nvalid = *valid_count;
  float gs = (lab >= 0) ? (*upstream) / (float)(nvalid > 0 ? nvalid : 1) : 0.f;
  for (int i = tid; i < n2; i += blockDim.x) {
    uint32_t v = l2[i];
    int c0 = 2 * i, c1 = 2 * i + 1;
    float p0 = __expf(bf16_to_f32((ushort_t)(v & 0xffff)) - l);
    float p1 = __expf(bf16_to_f32((ushort_t)(v >> 16)) - l);
    float g0 = (p0 - (c0 == lab ? 1.f : 0.f)) * gs;
    float g1 = (p1 - (c1 == lab ? 1.f : 0.f)) * gs;
    d2[i] = (uint32_t)f32_to_bf16(g0) | ((uint32_t)f32_to_bf16(g1) << 16);
  }
  if ((n_cols & 1) && tid == 0) {
    int c = n_cols - 1;
    float p = __expf(bf16_to_f32(lrow[c]) - l);
    dlogits[(long long)row * n_cols + c] = f32_to_bf16((p - (c == lab ? 1.f : 0.f)) * gs);
  }
}

