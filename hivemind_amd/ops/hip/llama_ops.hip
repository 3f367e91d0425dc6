// Llama-family fused kernels for gfx950: RMSNorm, SwiGLU, RoPE.
// Conventions follow elementwise.hip: paired bf16 loads (4 B/lane),
// block-per-row with cross-wave LDS reduction for the norm statistics,
// per-thread register accumulation + one endgame atomic per column for
// weight gradients (no atomics on the hot path).
// RoPE uses HOST-precomputed cos/sin tables (on-device trig turns a
// memory-bound op VALU-bound -- cdna_hip_programming.md Appendix B).
#pragma once

// ---------------------------------------------------------------------------
// RMSNorm forward: y = x * rstd * gamma, rstd = 1/sqrt(mean(x^2) + eps).
// One 256-thread block per row (grid-strided): supports cols up to 16k.
// ---------------------------------------------------------------------------
extern "C" __global__ void rmsnorm_fwd_bf16(const ushort_t* __restrict__ x,
                                            const float* __restrict__ gamma,
                                            ushort_t* __restrict__ y,
                                            float* __restrict__ rstd_out,
                                            float eps, long long rows, int cols) {
  __shared__ float wave_sums[4];
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int cols2 = cols >> 1;
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const uint32_t* xr2 = reinterpret_cast<const uint32_t*>(x + row * cols);
    uint32_t* yr2 = reinterpret_cast<uint32_t*>(y + row * cols);
    float sumsq = 0.f;
    for (int c = tid; c < cols2; c += blockDim.x) {
      uint32_t xv = xr2[c];
      float v0 = bf16_to_f32((ushort_t)(xv & 0xffff));
      float v1 = bf16_to_f32((ushort_t)(xv >> 16));
      sumsq += v0 * v0 + v1 * v1;
    }
    for (int off = 32; off > 0; off >>= 1) sumsq += __shfl_down(sumsq, off);
    if (lane == 0) wave_sums[wave] = sumsq;
    __syncthreads();
    float total = wave_sums[0] + wave_sums[1] + wave_sums[2] + wave_sums[3];
    float rstd = rsqrtf(total / cols + eps);
    if (tid == 0) rstd_out[row] = rstd;
    for (int c = tid; c < cols2; c += blockDim.x) {
      uint32_t xv = xr2[c];
      float v0 = bf16_to_f32((ushort_t)(xv & 0xffff)) * rstd * gamma[2 * c];
      float v1 = bf16_to_f32((ushort_t)(xv >> 16)) * rstd * gamma[2 * c + 1];
      yr2[c] = (uint32_t)f32_to_bf16(v0) | ((uint32_t)f32_to_bf16(v1) << 16);
    }
    __syncthreads();
  }
}

// RMSNorm backward:
//   dx     = gamma*dy*rstd - x * (rstd^3 / N) * sum(dy * gamma * x)
//   dgamma = sum_rows dy * x * rstd           (register-accumulated)
// PAIRS compile-time; block-per-row; supports cols <= 2 * PAIRS * 256.
template <int PAIRS>
__global__ void rmsnorm_bwd_bf16_t(const ushort_t* __restrict__ dy,
                                   const ushort_t* __restrict__ x,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ rstd,
                                   ushort_t* __restrict__ dx,
                                   float* __restrict__ partial,  // [gridDim, cols] per-block dgamma
                                   long long rows, int cols) {
  __shared__ float wave_sums[4];
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int cols2 = cols >> 1;

  float accg0[PAIRS], accg1[PAIRS];
  float gam0[PAIRS], gam1[PAIRS];  // per-thread gamma columns, loaded once
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    accg0[p] = accg1[p] = 0.f;
    int c = tid + p * blockDim.x;
    gam0[p] = (c < cols2) ? gamma[2 * c] : 0.f;
    gam1[p] = (c < cols2) ? gamma[2 * c + 1] : 0.f;
  }

  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const uint32_t* dyr2 = reinterpret_cast<const uint32_t*>(dy + row * cols);
    const uint32_t* xr2 = reinterpret_cast<const uint32_t*>(x + row * cols);
    uint32_t* dxr2 = reinterpret_cast<uint32_t*>(dx + row * cols);
    float rs = rstd[row];

    float dot = 0.f;  // sum(dy * gamma * x)
#pragma unroll
    for (int p = 0; p < PAIRS; ++p) {
      int c = tid + p * blockDim.x;
      if (c >= cols2) break;
      uint32_t dyv = dyr2[c], xv = xr2[c];
      float dy0 = bf16_to_f32((ushort_t)(dyv & 0xffff));
      float dy1 = bf16_to_f32((ushort_t)(dyv >> 16));
      float x0 = bf16_to_f32((ushort_t)(xv & 0xffff));
      float x1 = bf16_to_f32((ushort_t)(xv >> 16));
      dot += dy0 * gam0[p] * x0 + dy1 * gam1[p] * x1;
      accg0[p] += dy0 * x0 * rs;
      accg1[p] += dy1 * x1 * rs;
    }
    for (int off = 32; off > 0; off >>= 1) dot += __shfl_down(dot, off);
    if (lane == 0) wave_sums[wave] = dot;
    __syncthreads();
    float total_dot = wave_sums[0] + wave_sums[1] + wave_sums[2] + wave_sums[3];
    float k = total_dot * rs * rs * rs / cols;
#pragma unroll
    for (int p = 0; p < PAIRS; ++p) {
      int c = tid + p * blockDim.x;
      if (c >= cols2) break;
      uint32_t dyv = dyr2[c], xv = xr2[c];
      float dy0 = bf16_to_f32((ushort_t)(dyv & 0xffff));
      float dy1 = bf16_to_f32((ushort_t)(dyv >> 16));
      float x0 = bf16_to_f32((ushort_t)(xv & 0xffff));
      float x1 = bf16_to_f32((ushort_t)(xv >> 16));
      float o0 = gam0[p] * dy0 * rs - x0 * k;
      float o1 = gam1[p] * dy1 * rs - x1 * k;
      dxr2[c] = (uint32_t)f32_to_bf16(o0) | ((uint32_t)f32_to_bf16(o1) << 16);
    }
    __syncthreads();
  }
  // one owner thread per column in this block: store the block partial row;
  // reduce_block_partials_f32 sums over blocks (contended global fp32 atomics
  // serialize gridDim-deep per address and dominated this kernel).
  float* prow = partial + (long long)blockIdx.x * cols;
#pragma unroll
  for (int p = 0; p < PAIRS; ++p) {
    int c = tid + p * blockDim.x;
    if (c >= cols2) break;
    prow[2 * c] = accg0[p];
    prow[2 * c + 1] = accg1[p];
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: out = silu(gate) * up   (elementwise over [rows, cols] pairs)
// backward: dgate = dy * up * silu'(gate);  dup = dy * silu(gate)
// ---------------------------------------------------------------------------
DEVINL float silu_f(float x) { return x / (1.f + __expf(-x)); }
DEVINL float silu_grad_f(float x) {
  float s = 1.f / (1.f + __expf(-x));
  return s * (1.f + x * (1.f - s));
}

extern "C" __global__ void swiglu_fwd_bf16(const ushort_t* __restrict__ gate,
                                           const ushort_t* __restrict__ up,
                                           ushort_t* __restrict__ out, long long n2) {
  const uint32_t* g2 = reinterpret_cast<const uint32_t*>(gate);
  const uint32_t* u2 = reinterpret_cast<const uint32_t*>(up);
  uint32_t* o2 = reinterpret_cast<uint32_t*>(out);
  long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n2; i += stride) {
    uint32_t gv = g2[i], uv = u2[i];
    float o0 = silu_f(bf16_to_f32((ushort_t)(gv & 0xffff))) * bf16_to_f32((ushort_t)(uv & 0xffff));
    float o1 = silu_f(bf16_to_f32((ushort_t)(gv >> 16))) * bf16_to_f32((ushort_t)(uv >> 16));
    o2[i] = (uint32_t)f32_to_bf16(o0) | ((uint32_t)f32_to_bf16(o1) << 16);
  }
}

extern "C" __global__ void swiglu_bwd_bf16(const ushort_t* __restrict__ dy,
                                           const ushort_t* __restrict__ gate,
                                           const ushort_t* __restrict__ up,
                                           ushort_t* __restrict__ dgate,
                                           ushort_t* __restrict__ dup, long long n2) {
  const uint32_t* dy2 = reinterpret_cast<const uint32_t*>(dy);
  const uint32_t* g2 = reinterpret_cast<const uint32_t*>(gate);
  const uint32_t* u2 = reinterpret_cast<const uint32_t*>(up);
  uint32_t* dg2 = reinterpret_cast<uint32_t*>(dgate);
  uint32_t* du2 = reinterpret_cast<uint32_t*>(dup);
  long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n2; i += stride) {
    uint32_t dyv = dy2[i], gv = g2[i], uv = u2[i];
    float d0 = bf16_to_f32((ushort_t)(dyv & 0xffff)), d1 = bf16_to_f32((ushort_t)(dyv >> 16));
    float g0 = bf16_to_f32((ushort_t)(gv & 0xffff)), g1 = bf16_to_f32((ushort_t)(gv >> 16));
    float u0 = bf16_to_f32((ushort_t)(uv & 0xffff)), u1 = bf16_to_f32((ushort_t)(uv >> 16));
    float dgate0 = d0 * u0 * silu_grad_f(g0), dgate1 = d1 * u1 * silu_grad_f(g1);
    float dup0 = d0 * silu_f(g0), dup1 = d1 * silu_f(g1);
    dg2[i] = (uint32_t)f32_to_bf16(dgate0) | ((uint32_t)f32_to_bf16(dgate1) << 16);
    du2[i] = (uint32_t)f32_to_bf16(dup0) | ((uint32_t)f32_to_bf16(dup1) << 16);
  }
}

// ---------------------------------------------------------------------------
// RoPE (Llama rotate-half convention): x shape [tokens, heads, head_dim],
// cos/sin tables [tokens, head_dim/2] fp32 precomputed on host.
//   out[..., :half]  = x1 * cos - x2 * sin
//   out[..., half:]  = x2 * cos + x1 * sin
// backward = rotation by -theta (swap the sin signs). direction=+1 fwd, -1 bwd.
// ---------------------------------------------------------------------------
extern "C" __global__ void rope_bf16(const ushort_t* __restrict__ x,
                                     const float* __restrict__ cos_table,
                                     const float* __restrict__ sin_table,
                                     ushort_t* __restrict__ out,
                                     float direction,
                                     long long tokens, int seq_len, int heads, int head_dim) {
  const int half = head_dim >> 1;
  long long total = tokens * heads * half;
  long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < total; i += stride) {
    long long token = i / (heads * half);
    int rem = (int)(i % (heads * half));
    int head = rem / half;
    int d = rem % half;
    long long base = (token * heads + head) * head_dim;
    long long pos = token % seq_len;  // x is [batch, seq, heads, head_dim] row-major
    float c = cos_table[pos * half + d];
    float s = sin_table[pos * half + d] * direction;
    float x1 = bf16_to_f32(x[base + d]);
    float x2 = bf16_to_f32(x[base + half + d]);
    out[base + d] = f32_to_bf16(x1 * c - x2 * s);
    out[base + half + d] = f32_to_bf16(x2 * c + x1 * s);
  }
}
