// PyTorch bindings for the hand-written CDNA4 kernels. Compiled by hipcc via
// torch.utils.cpp_extension with PYTORCH_ROCM_ARCH=gfx950 (see setup.py).
// All launches go onto the current PyTorch HIP stream so they compose with
// autograd and the RCCL side-stream logic in averaging/rccl.py.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "elementwise.hip"
#include "llama_ops.hip"

extern "C" void launch_gemm_bt_bf16(const void*, const void*, const void*, void*, void*, int, int, int, int, void*);
extern "C" void launch_mfma_probe(const void*, const void*, void*, void*);
extern "C" void launch_stage_probe(const void*, void*, int, void*);
extern "C" void launch_flash_fwd(const void*, const void*, const void*, void*, void*,
                                 int, int, int, int, int, float, int,
                                 const long long*, const long long*, void*);
extern "C" void launch_flash_delta(const void*, const void*, void*, long long, int, void*);
extern "C" void launch_flash_bwd_dkdv(const void*, const void*, const void*, const void*,
                                      const void*, const void*, const void*, const void*,
                                      void*, void*, int, int, int, int, float, int,
                                      const long long*, const long long*, const long long*, void*);
extern "C" void launch_flash_bwd_dq(const void*, const void*, const void*, const void*,
                                    const void*, const void*, const void*, void*,
                                    int, int, int, int, float, int,
                                    const long long*, const long long*, const long long*, void*);
extern "C" void launch_transpose_bhsd(const void*, void*, long long, int, int, int,
                                      long long, long long, long long, void*);

#include "multi_tensor.h"

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on the GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace {

inline hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

inline int grid_1d(long long n, int block = 256, int cap = 2048) {
  long long blocks = (n + block - 1) / block;
  if (blocks > cap) blocks = cap;   // grid-stride handles the rest (guide G11)
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace

// ---------------------------------------------------------------- averaging

void apply_delta_(torch::Tensor tensor, torch::Tensor delta, double alpha) {
  CHECK_GPU(tensor); CHECK_GPU(delta); CHECK_CONTIG(tensor); CHECK_CONTIG(delta);
  TORCH_CHECK(tensor.numel() == delta.numel(), "size mismatch");
  long long n = tensor.numel();
  if (tensor.scalar_type() == torch::kBFloat16 && delta.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(apply_delta_bf16, dim3(grid_1d(n / 2 + 1)), dim3(256), 0, current_stream(),
                       (unsigned short*)tensor.data_ptr(), (const unsigned short*)delta.data_ptr(),
                       (float)alpha, n);
  } else if (tensor.scalar_type() == torch::kFloat32 && delta.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(apply_delta_f32, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                       tensor.data_ptr<float>(), delta.data_ptr<float>(), (float)alpha, n);
  } else {
    TORCH_CHECK(false, "apply_delta_: unsupported dtype combination");
  }
}

void weighted_accumulate_(torch::Tensor acc, torch::Tensor x, double w) {
  CHECK_GPU(acc); CHECK_GPU(x); CHECK_CONTIG(acc); CHECK_CONTIG(x);
  TORCH_CHECK(acc.scalar_type() == torch::kFloat32, "accumulator must be fp32");
  long long n = acc.numel();
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(weighted_accumulate_f32, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                       acc.data_ptr<float>(), x.data_ptr<float>(), (float)w, n);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(weighted_accumulate_bf16_f32, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                       acc.data_ptr<float>(), (const unsigned short*)x.data_ptr(), (float)w, n);
  } else {
    TORCH_CHECK(false, "weighted_accumulate_: unsupported input dtype");
  }
}

// ------------------------------------------------------------------- codecs

torch::Tensor compress_fp16_gpu(torch::Tensor input) {
  CHECK_GPU(input); CHECK_CONTIG(input);
  auto in32 = input.scalar_type() == torch::kFloat32 ? input : input.to(torch::kFloat32);
  auto out = torch::empty_like(in32, in32.options().dtype(torch::kFloat16));
  long long n = in32.numel();
  hipLaunchKernelGGL(compress_fp16, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                     in32.data_ptr<float>(), (__half*)out.data_ptr(), n);
  return out;
}

torch::Tensor decompress_fp16_gpu(torch::Tensor input) {
  CHECK_GPU(input); CHECK_CONTIG(input);
  auto out = torch::empty_like(input, input.options().dtype(torch::kFloat32));
  long long n = input.numel();
  hipLaunchKernelGGL(decompress_fp16, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                     (const __half*)input.data_ptr(), out.data_ptr<float>(), n);
  return out;
}

std::vector<torch::Tensor> quantize_blockwise_gpu(torch::Tensor input) {
  CHECK_GPU(input); CHECK_CONTIG(input);
  auto in32 = input.scalar_type() == torch::kFloat32 ? input : input.to(torch::kFloat32);
  long long n = in32.numel();
  long long num_blocks = (n + 4095) / 4096;
  auto q = torch::empty({n}, in32.options().dtype(torch::kInt8));
  auto absmax = torch::empty({num_blocks}, in32.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(quantize_blockwise_int8, dim3((int)num_blocks), dim3(256), 0, current_stream(),
                     in32.data_ptr<float>(), q.data_ptr<int8_t>(), absmax.data_ptr<float>(), n);
  return {q, absmax};
}

torch::Tensor dequantize_blockwise_gpu(torch::Tensor q, torch::Tensor absmax) {
  CHECK_GPU(q); CHECK_CONTIG(q); CHECK_CONTIG(absmax);
  long long n = q.numel();
  auto out = torch::empty({n}, q.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(dequantize_blockwise_int8, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                     q.data_ptr<int8_t>(), absmax.data_ptr<float>(), out.data_ptr<float>(), n);
  return out;
}

// ---------------------------------------------------------------- optimizer

void fused_adamw_(torch::Tensor param, torch::Tensor grad, torch::Tensor exp_avg,
                  torch::Tensor exp_avg_sq, c10::optional<torch::Tensor> bf16_mirror,
                  double lr, double beta1, double beta2, double eps,
                  double weight_decay, long step) {
  CHECK_GPU(param); CHECK_CONTIG(param); CHECK_CONTIG(grad);
  CHECK_CONTIG(exp_avg); CHECK_CONTIG(exp_avg_sq);
  TORCH_CHECK(param.scalar_type() == torch::kFloat32, "master params must be fp32");
  long long n = param.numel();
  float bias_corr1 = 1.0f - powf((float)beta1, (float)step);
  float bias_corr2 = 1.0f - powf((float)beta2, (float)step);
  unsigned short* mirror_ptr = nullptr;
  if (bf16_mirror.has_value()) {
    CHECK_CONTIG(bf16_mirror.value());
    TORCH_CHECK(bf16_mirror->scalar_type() == torch::kBFloat16, "mirror must be bf16");
    mirror_ptr = (unsigned short*)bf16_mirror->data_ptr();
  }
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(fused_adamw_f32, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                       param.data_ptr<float>(), grad.data_ptr<float>(), exp_avg.data_ptr<float>(),
                       exp_avg_sq.data_ptr<float>(), mirror_ptr,
                       (float)lr, (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                       bias_corr1, bias_corr2, n);
  } else if (grad.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(fused_adamw_bf16grad, dim3(grid_1d(n)), dim3(256), 0, current_stream(),
                       param.data_ptr<float>(), (const unsigned short*)grad.data_ptr(),
                       exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(), mirror_ptr,
                       (float)lr, (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                       bias_corr1, bias_corr2, n);
  } else {
    TORCH_CHECK(false, "fused_adamw_: unsupported grad dtype");
  }
}

// ------------------------------------------------------------ cross-entropy

std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits, torch::Tensor labels) {
  CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_CONTIG(labels);
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16, "cross_entropy expects bf16 logits");
  TORCH_CHECK(labels.scalar_type() == torch::kInt64, "cross_entropy expects int64 labels");
  TORCH_CHECK(logits.dim() == 2 && labels.dim() == 1 && labels.size(0) == logits.size(0),
              "cross_entropy: logits [N, V], labels [N]");
  int n_rows = (int)logits.size(0), n_cols = (int)logits.size(1);
  auto lse = torch::empty({n_rows}, logits.options().dtype(torch::kFloat32));
  auto loss_sum = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  auto valid = torch::zeros({}, logits.options().dtype(torch::kInt32));
  hipLaunchKernelGGL(cross_entropy_fwd_bf16, dim3(n_rows), dim3(256), 0, current_stream(),
                     (const unsigned short*)logits.data_ptr(), (const long long*)labels.data_ptr(),
                     lse.data_ptr<float>(), loss_sum.data_ptr<float>(), valid.data_ptr<int>(),
                     n_rows, n_cols);
  return {loss_sum, valid, lse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                                torch::Tensor upstream, torch::Tensor valid) {
  CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_CONTIG(labels); CHECK_CONTIG(lse);
  TORCH_CHECK(upstream.scalar_type() == torch::kFloat32 && upstream.numel() == 1,
              "cross_entropy_bwd: upstream must be a single fp32 value on device");
  int n_rows = (int)logits.size(0), n_cols = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  hipLaunchKernelGGL(cross_entropy_bwd_bf16, dim3(n_rows), dim3(256), 0, current_stream(),
                     (const unsigned short*)logits.data_ptr(), (const long long*)labels.data_ptr(),
                     lse.data_ptr<float>(), upstream.data_ptr<float>(), valid.data_ptr<int>(),
                     (unsigned short*)dlogits.data_ptr(), n_rows, n_cols);
  return dlogits;
}

// --------------------------------------------------------------- activations

std::vector<torch::Tensor> bias_gelu_fwd(torch::Tensor x, torch::Tensor bias, bool save_pre_act) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(bias);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && bias.scalar_type() == torch::kBFloat16,
              "bias_gelu expects bf16");
  long long cols = x.size(-1);
  TORCH_CHECK(cols % 2 == 0 && cols <= 2 * 8 * 256, "bias_gelu: cols must be even and <= 4096");
  long long rows = x.numel() / cols;
  auto out = torch::empty_like(x);
  torch::Tensor pre_act;
  unsigned short* pre_ptr = nullptr;
  if (save_pre_act) {
    pre_act = torch::empty_like(x);
    pre_ptr = (unsigned short*)pre_act.data_ptr();
  }
  hipLaunchKernelGGL(bias_gelu_fwd_bf16, dim3((int)std::min<long long>(rows, 2048)), dim3(256), 0, current_stream(),
                     (const unsigned short*)x.data_ptr(), (const unsigned short*)bias.data_ptr(),
                     (unsigned short*)out.data_ptr(), pre_ptr, rows, cols);
  if (save_pre_act) return {out, pre_act};
  return {out};
}

// shared driver: pre_or_x is either a materialized pre-activation (MFMA
// fused-epilogue path) or the raw GEMM output with `bias` added in-kernel.
static std::vector<torch::Tensor> bias_gelu_bwd_impl(torch::Tensor dy, torch::Tensor pre_or_x,
                                                     const unsigned short* bias_ptr) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(pre_or_x);
  long long cols = dy.size(-1);
  TORCH_CHECK(cols % 2 == 0, "bias_gelu_bwd requires an even column count");
  long long rows = dy.numel() / cols;
  auto dx = torch::empty_like(dy);
  TORCH_CHECK(cols <= 2 * 8 * 256, "bias_gelu_bwd: cols must be <= 4096");
  int blocks = (int)std::min<long long>(rows, 2048);
  auto partial = torch::empty({blocks, cols}, dy.options().dtype(torch::kFloat32));
  int pairs = (int)((cols / 2 + 255) / 256);
  auto launch = [&](auto kernel) {
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, current_stream(),
                       (const unsigned short*)dy.data_ptr(), (const unsigned short*)pre_or_x.data_ptr(),
                       bias_ptr, (unsigned short*)dx.data_ptr(), partial.data_ptr<float>(), rows, cols);
  };
  if (bias_ptr) {
    switch (pairs) {
      case 1: launch(bias_gelu_bwd_bf16_t<1, true>); break;
      case 2: launch(bias_gelu_bwd_bf16_t<2, true>); break;
      case 3: launch(bias_gelu_bwd_bf16_t<3, true>); break;
      case 4: launch(bias_gelu_bwd_bf16_t<4, true>); break;
      case 5: case 6: launch(bias_gelu_bwd_bf16_t<6, true>); break;
      default: launch(bias_gelu_bwd_bf16_t<8, true>); break;
    }
  } else {
    switch (pairs) {
      case 1: launch(bias_gelu_bwd_bf16_t<1, false>); break;
      case 2: launch(bias_gelu_bwd_bf16_t<2, false>); break;
      case 3: launch(bias_gelu_bwd_bf16_t<3, false>); break;
      case 4: launch(bias_gelu_bwd_bf16_t<4, false>); break;
      case 5: case 6: launch(bias_gelu_bwd_bf16_t<6, false>); break;
      default: launch(bias_gelu_bwd_bf16_t<8, false>); break;
    }
  }
  // column-reduce the per-block partials with ATen's tuned reducer (a naive
  // one-thread-per-column kernel here measured 535 us -- latency-bound)
  auto dbias = partial.sum(0);
  return {dx, dbias};
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor pre_act) {
  return bias_gelu_bwd_impl(dy, pre_act, nullptr);
}

std::vector<torch::Tensor> bias_gelu_bwd_xb(torch::Tensor dy, torch::Tensor x, torch::Tensor bias) {
  CHECK_CONTIG(bias);
  TORCH_CHECK(bias.scalar_type() == torch::kBFloat16 && bias.numel() == dy.size(-1));
  return bias_gelu_bwd_impl(dy, x, (const unsigned short*)bias.data_ptr());
}

// ----------------------------------------------------------------- layernorm

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, c10::optional<torch::Tensor> residual,
                                         torch::Tensor gamma, torch::Tensor beta, double eps) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(gamma); CHECK_CONTIG(beta);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "layernorm expects bf16 input");
  TORCH_CHECK(gamma.scalar_type() == torch::kFloat32, "gamma/beta must be fp32");
  int cols = (int)x.size(-1);
  TORCH_CHECK(cols % 2 == 0, "layernorm requires an even column count");
  long long rows = x.numel() / cols;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  const unsigned short* res_ptr = nullptr;
  torch::Tensor h;
  unsigned short* h_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_CONTIG(residual.value());
    res_ptr = (const unsigned short*)residual->data_ptr();
    h = torch::empty_like(x);
    h_ptr = (unsigned short*)h.data_ptr();
  }
  int blocks = (int)((rows + 3) / 4);
  int pairs_fwd = (cols / 2 + 63) / 64;
  TORCH_CHECK(pairs_fwd <= 16, "layernorm_fwd: cols must be <= 2048");
  auto launch_fwd = [&](auto kernel) {
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, current_stream(),
                       (const unsigned short*)x.data_ptr(), res_ptr,
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       (unsigned short*)y.data_ptr(), h_ptr,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (float)eps, rows, cols);
  };
  switch (pairs_fwd) {
    case 1: launch_fwd(layernorm_fwd_bf16_t<1>); break;
    case 2: launch_fwd(layernorm_fwd_bf16_t<2>); break;
    case 3: case 4: launch_fwd(layernorm_fwd_bf16_t<4>); break;
    case 5: case 6: launch_fwd(layernorm_fwd_bf16_t<6>); break;
    case 7: case 8: launch_fwd(layernorm_fwd_bf16_t<8>); break;
    default: launch_fwd(layernorm_fwd_bf16_t<16>); break;
  }
  if (residual.has_value()) return {y, mean, rstd, h};
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor h, torch::Tensor gamma,
                                         torch::Tensor mean, torch::Tensor rstd) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(h);
  int cols = (int)dy.size(-1);
  long long rows = dy.numel() / cols;
  auto dx = torch::empty_like(dy);
  TORCH_CHECK(cols % 2 == 0 && cols <= 2 * 64 * 16, "layernorm_bwd: cols must be even and <= 2048");
  int blocks = (int)std::min<long long>((rows + 3) / 4, 2048);
  auto partial = torch::empty({blocks, 2 * (long long)cols}, dy.options().dtype(torch::kFloat32));
  int pairs = (cols / 2 + 63) / 64;
  auto launch = [&](auto kernel) {
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, current_stream(),
                       (const unsigned short*)dy.data_ptr(), (const unsigned short*)h.data_ptr(),
                       gamma.data_ptr<float>(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (unsigned short*)dx.data_ptr(), partial.data_ptr<float>(),
                       rows, cols);
  };
  switch (pairs) {
    case 1: launch(layernorm_bwd_bf16_t<1>); break;
    case 2: launch(layernorm_bwd_bf16_t<2>); break;
    case 3: case 4: launch(layernorm_bwd_bf16_t<4>); break;
    case 5: case 6: launch(layernorm_bwd_bf16_t<6>); break;
    case 7: case 8: launch(layernorm_bwd_bf16_t<8>); break;
    case 9: case 10: case 11: case 12: launch(layernorm_bwd_bf16_t<12>); break;
    default: launch(layernorm_bwd_bf16_t<16>); break;
  }
  auto fused = partial.sum(0);
  auto dgamma = fused.narrow(0, 0, cols);
  auto dbeta = fused.narrow(0, cols, cols);
  return {dx, dgamma, dbeta};
}

// ------------------------------------------------------------ llama kernels

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor gamma, double eps) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(gamma);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && gamma.scalar_type() == torch::kFloat32);
  int cols = (int)x.size(-1);
  TORCH_CHECK(cols % 2 == 0, "rmsnorm: cols must be even");
  long long rows = x.numel() / cols;
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  int blocks = (int)std::min<long long>(rows, 4096);
  hipLaunchKernelGGL(rmsnorm_fwd_bf16, dim3(blocks), dim3(256), 0, current_stream(),
                     (const unsigned short*)x.data_ptr(), gamma.data_ptr<float>(),
                     (unsigned short*)y.data_ptr(), rstd.data_ptr<float>(), (float)eps, rows, cols);
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor gamma, torch::Tensor rstd) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(x);
  int cols = (int)dy.size(-1);
  TORCH_CHECK(cols % 2 == 0 && cols <= 2 * 16 * 256, "rmsnorm_bwd: cols must be even and <= 8192");
  long long rows = dy.numel() / cols;
  auto dx = torch::empty_like(dy);
  int blocks = (int)std::min<long long>(rows, 2048);
  auto partial = torch::empty({blocks, (long long)cols}, dy.options().dtype(torch::kFloat32));
  int pairs = (cols / 2 + 255) / 256;
  auto launch = [&](auto kernel) {
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, current_stream(),
                       (const unsigned short*)dy.data_ptr(), (const unsigned short*)x.data_ptr(),
                       gamma.data_ptr<float>(), rstd.data_ptr<float>(),
                       (unsigned short*)dx.data_ptr(), partial.data_ptr<float>(), rows, cols);
  };
  switch (pairs) {
    case 1: launch(rmsnorm_bwd_bf16_t<1>); break;
    case 2: launch(rmsnorm_bwd_bf16_t<2>); break;
    case 3: case 4: launch(rmsnorm_bwd_bf16_t<4>); break;
    case 5: case 6: case 7: case 8: launch(rmsnorm_bwd_bf16_t<8>); break;
    default: launch(rmsnorm_bwd_bf16_t<16>); break;
  }
  auto dgamma = partial.sum(0);
  return {dx, dgamma};
}

torch::Tensor swiglu_fwd(torch::Tensor gate, torch::Tensor up) {
  CHECK_GPU(gate); CHECK_CONTIG(gate); CHECK_CONTIG(up);
  TORCH_CHECK(gate.numel() == up.numel() && gate.numel() % 2 == 0);
  auto out = torch::empty_like(gate);
  long long n2 = gate.numel() / 2;
  hipLaunchKernelGGL(swiglu_fwd_bf16, dim3(grid_1d(n2)), dim3(256), 0, current_stream(),
                     (const unsigned short*)gate.data_ptr(), (const unsigned short*)up.data_ptr(),
                     (unsigned short*)out.data_ptr(), n2);
  return out;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dy, torch::Tensor gate, torch::Tensor up) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(gate); CHECK_CONTIG(up);
  auto dgate = torch::empty_like(gate);
  auto dup = torch::empty_like(up);
  long long n2 = gate.numel() / 2;
  hipLaunchKernelGGL(swiglu_bwd_bf16, dim3(grid_1d(n2)), dim3(256), 0, current_stream(),
                     (const unsigned short*)dy.data_ptr(), (const unsigned short*)gate.data_ptr(),
                     (const unsigned short*)up.data_ptr(), (unsigned short*)dgate.data_ptr(),
                     (unsigned short*)dup.data_ptr(), n2);
  return {dgate, dup};
}

torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cos_table, torch::Tensor sin_table, double direction) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(cos_table); CHECK_CONTIG(sin_table);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  int head_dim = (int)x.size(-1);
  int heads = (int)x.size(-2);
  long long tokens = x.numel() / ((long long)heads * head_dim);
  TORCH_CHECK(head_dim % 2 == 0);
  auto out = torch::empty_like(x);
  int seq_len = (int)cos_table.size(0);
  long long total = tokens * heads * (head_dim / 2);
  hipLaunchKernelGGL(rope_bf16, dim3(grid_1d(total)), dim3(256), 0, current_stream(),
                     (const unsigned short*)x.data_ptr(), cos_table.data_ptr<float>(),
                     sin_table.data_ptr<float>(), (unsigned short*)out.data_ptr(),
                     (float)direction, tokens, seq_len, heads, head_dim);
  return out;
}

// ------------------------------------------------------- hand-written GEMM

std::vector<torch::Tensor> mfma_linear_bf16(torch::Tensor x, torch::Tensor weight,
                                            c10::optional<torch::Tensor> bias, bool gelu,
                                            bool save_pre_act) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(weight);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && weight.scalar_type() == torch::kBFloat16);
  long long K = x.size(-1);
  long long M = x.numel() / K;
  long long N = weight.size(0);
  TORCH_CHECK(weight.size(1) == K, "weight must be [N, K]");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 32 == 0,
              "mfma_linear: M%128, N%128, K%32 must be 0 (wrapper pads M)");
  auto out_sizes = x.sizes().vec();
  out_sizes.back() = N;
  auto out = torch::empty(out_sizes, x.options());
  torch::Tensor pre_act;
  unsigned short* pre_ptr = nullptr;
  const unsigned short* bias_ptr = nullptr;
  if (gelu) {
    TORCH_CHECK(bias.has_value() && bias->is_contiguous() && bias->scalar_type() == torch::kBFloat16);
    bias_ptr = (const unsigned short*)bias->data_ptr();
    if (save_pre_act) {
      pre_act = torch::empty(out_sizes, x.options());
      pre_ptr = (unsigned short*)pre_act.data_ptr();
    }
  }
  launch_gemm_bt_bf16(x.data_ptr(), weight.data_ptr(), (const void*)bias_ptr,
                      out.data_ptr(), (void*)pre_ptr, (int)M, (int)N, (int)K,
                      gelu ? 1 : 0, (void*)current_stream());
  if (save_pre_act && gelu) return {out, pre_act};
  return {out};
}

torch::Tensor mfma_probe(torch::Tensor a_vals, torch::Tensor b_vals) {
  CHECK_GPU(a_vals); CHECK_CONTIG(a_vals); CHECK_CONTIG(b_vals);
  auto c = torch::zeros({16, 16}, a_vals.options().dtype(torch::kFloat32));
  launch_mfma_probe(a_vals.data_ptr(), b_vals.data_ptr(), c.data_ptr(), (void*)current_stream());
  return c;
}

torch::Tensor stage_probe(torch::Tensor A) {
  CHECK_GPU(A); CHECK_CONTIG(A);
  int K = (int)A.size(1);
  auto out = torch::zeros({128 * 32}, A.options());
  launch_stage_probe(A.data_ptr(), out.data_ptr(), K, (void*)current_stream());
  return out;
}

// ----------------------------------------------------------- multi-tensor

void multi_adamw_(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                  std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                  std::vector<c10::optional<torch::Tensor>> mirrors,
                  double lr, double beta1, double beta2, double eps, double weight_decay,
                  int64_t step) {
  size_t n = params.size();
  TORCH_CHECK(grads.size() == n && exp_avgs.size() == n && exp_avg_sqs.size() == n &&
              mirrors.size() == n, "multi_adamw_: list length mismatch");
  float bias_corr1 = 1.0f - powf((float)beta1, (float)step);
  float bias_corr2 = 1.0f - powf((float)beta2, (float)step);
  for (size_t start = 0; start < n; start += MT_CHUNK) {
    MTAdamArgs args{};
    args.n = 0;
    args.cum[0] = 0;
    args.g_bf16_mask = 0;
    for (size_t t = start; t < n && args.n < MT_CHUNK; ++t, ++args.n) {
      auto& p = params[t];
      CHECK_GPU(p); CHECK_CONTIG(p);
      TORCH_CHECK(p.scalar_type() == torch::kFloat32, "multi_adamw_ params must be fp32 masters");
      auto& g = grads[t];
      TORCH_CHECK(g.is_contiguous() && g.numel() == p.numel(), "bad grad ", t);
      int i = args.n;
      args.p[i] = p.data_ptr<float>();
      args.g[i] = g.data_ptr();
      if (g.scalar_type() == torch::kBFloat16) args.g_bf16_mask |= (1ull << i);
      else TORCH_CHECK(g.scalar_type() == torch::kFloat32, "grads must be fp32 or bf16");
      args.m[i] = exp_avgs[t].data_ptr<float>();
      args.v[i] = exp_avg_sqs[t].data_ptr<float>();
      args.mirror[i] = mirrors[t].has_value()
                           ? (mt_ushort*)mirrors[t]->data_ptr() : nullptr;
      args.cum[i + 1] = args.cum[i] + p.numel();
    }
    launch_multi_tensor_adamw(&args, (float)lr, (float)beta1, (float)beta2, (float)eps,
                              (float)weight_decay, bias_corr1, bias_corr2,
                              (void*)current_stream());
  }
}

void multi_accumulate_(std::vector<torch::Tensor> accs, std::vector<torch::Tensor> xs,
                       double alpha) {
  size_t n = accs.size();
  TORCH_CHECK(xs.size() == n, "multi_accumulate_: list length mismatch");
  for (size_t start = 0; start < n; start += MT_CHUNK) {
    MTAccArgs args{};
    args.n = 0;
    args.cum[0] = 0;
    args.x_bf16_mask = 0;
    for (size_t t = start; t < n && args.n < MT_CHUNK; ++t, ++args.n) {
      auto& a = accs[t];
      CHECK_GPU(a); CHECK_CONTIG(a);
      TORCH_CHECK(a.scalar_type() == torch::kFloat32, "accumulators must be fp32");
      auto& x = xs[t];
      TORCH_CHECK(x.is_contiguous() && x.numel() == a.numel(), "bad addend ", t);
      int i = args.n;
      args.acc[i] = a.data_ptr<float>();
      args.x[i] = x.data_ptr();
      if (x.scalar_type() == torch::kBFloat16) args.x_bf16_mask |= (1ull << i);
      else TORCH_CHECK(x.scalar_type() == torch::kFloat32, "addends must be fp32 or bf16");
      args.cum[i + 1] = args.cum[i] + a.numel();
    }
    launch_multi_tensor_accumulate(&args, (float)alpha, (void*)current_stream());
  }
}

// ------------------------------------------------------------- attention

static void check_flash_shapes(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() && t.scalar_type() == torch::kBFloat16,
              name, " must be a contiguous CUDA bf16 tensor");
  TORCH_CHECK(t.dim() == 4, name, " must be [B, H, S, D]");
}

// strided [B,H,S,D] views are fine as long as the last dim is contiguous and
// every row start stays 16B-aligned (the staging loads are 16B vectors); the
// model's qkv-GEMM views qualify, so q/k/v arrive with ZERO copies
static bool flash_strided_ok(const torch::Tensor& t) {
  return t.is_cuda() && t.dim() == 4 && t.scalar_type() == torch::kBFloat16 &&
         t.stride(3) == 1 && t.stride(2) % 8 == 0 && t.stride(1) % 8 == 0 &&
         t.stride(0) % 8 == 0 && (reinterpret_cast<uintptr_t>(t.data_ptr()) % 16 == 0);
}

static torch::Tensor flash_input(const torch::Tensor& t) {
  return flash_strided_ok(t) ? t : t.contiguous();
}

static void flash_strides(const torch::Tensor& t, long long out[3]) {
  out[0] = t.stride(0);
  out[1] = t.stride(1);
  out[2] = t.stride(2);
}

torch::Tensor transpose_bhsd(torch::Tensor t) {
  // [B,H,S,D] -> [B,H,D,S] via the in-register 8x8 transpose kernel
  auto tc = flash_input(t);
  TORCH_CHECK(tc.is_cuda() && tc.dim() == 4 && tc.scalar_type() == torch::kBFloat16,
              "transpose input must be CUDA bf16 [B,H,S,D]");
  int B = tc.size(0), H = tc.size(1), S = tc.size(2), D = tc.size(3);
  TORCH_CHECK(S % 64 == 0 && D % 64 == 0, "transpose_bhsd needs S, D multiples of 64");
  auto out = torch::empty({B, H, D, S}, tc.options());
  launch_transpose_bhsd(tc.data_ptr(), out.data_ptr(), (long long)B * H, H, S, D,
                        tc.stride(0), tc.stride(1), tc.stride(2), (void*)current_stream());
  return out;
}

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q_in, torch::Tensor k_in,
                                          torch::Tensor v_in, bool causal, double scale) {
  auto q = flash_input(q_in), k = flash_input(k_in), v = flash_input(v_in);
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  int Hkv = k.size(1);
  TORCH_CHECK(D == 64 || D == 128, "flash attention requires head_dim 64 or 128, got ", D);
  TORCH_CHECK(S % 64 == 0, "flash attention requires seq_len % 64 == 0, got ", S);
  TORCH_CHECK(k.size(2) == S && v.size(2) == S, "kv seq_len mismatch");
  TORCH_CHECK(H % Hkv == 0, "H must be a multiple of Hkv");
  auto o = torch::empty({B, H, S, D}, q.options());
  auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));
  // one [B,H,S,D]->[B,H,D,S] copy so the PV operand stages with vector
  // LDS writes (in-kernel scalar transposes dominated the tile cost)
  auto vt = transpose_bhsd(v);
  long long qs[3], ks[3];
  flash_strides(q, qs); flash_strides(k, ks);
  launch_flash_fwd(q.data_ptr(), k.data_ptr(), vt.data_ptr(), o.data_ptr(), lse.data_ptr(),
                   B, H, Hkv, S, D, (float)scale, causal ? 1 : 0, qs, ks,
                   (void*)current_stream());
  return {o, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor d_out, torch::Tensor q_in,
                                          torch::Tensor k_in, torch::Tensor v_in,
                                          torch::Tensor o, torch::Tensor lse,
                                          bool causal, double scale) {
  auto q = flash_input(q_in), k = flash_input(k_in), v = flash_input(v_in);
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(k.size(1) == H, "flash attention backward requires H == Hkv (expand kv first)");
  auto d_out_c = d_out.contiguous();
  check_flash_shapes(d_out_c, "d_out");
  auto delta = torch::empty({B, H, S}, q.options().dtype(torch::kFloat32));
  launch_flash_delta(d_out_c.data_ptr(), o.data_ptr(), delta.data_ptr(),
                     (long long)B * H * S, D, (void*)current_stream());
  // pre-transposed operand copies: one pass each instead of per-tile scalar
  // LDS transposes inside every workgroup's kv/q loop
  auto dot = transpose_bhsd(d_out_c);
  auto qt = transpose_bhsd(q);
  auto kt = transpose_bhsd(k);
  auto dq = torch::empty({B, H, S, D}, q.options());
  auto dk = torch::empty({B, H, S, D}, q.options());
  auto dv = torch::empty({B, H, S, D}, q.options());
  long long qs[3], ks[3], vs[3];
  flash_strides(q, qs); flash_strides(k, ks); flash_strides(v, vs);
  launch_flash_bwd_dkdv(d_out_c.data_ptr(), dot.data_ptr(), q.data_ptr(), qt.data_ptr(),
                        k.data_ptr(), v.data_ptr(),
                        lse.data_ptr(), delta.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                        B, H, S, D, (float)scale, causal ? 1 : 0, qs, ks, vs,
                        (void*)current_stream());
  launch_flash_bwd_dq(d_out_c.data_ptr(), q.data_ptr(), k.data_ptr(), kt.data_ptr(), v.data_ptr(),
                      lse.data_ptr(), delta.data_ptr(), dq.data_ptr(),
                      B, H, S, D, (float)scale, causal ? 1 : 0, qs, ks, vs,
                      (void*)current_stream());
  return {dq, dk, dv};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("stage_probe", &stage_probe, "gload_lds staging probe");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 operand layout probe");
  m.def("mfma_linear_bf16", &mfma_linear_bf16, "hand-written MFMA GEMM: x @ W^T (+bias+gelu)");
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (bf16)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward");
  m.def("swiglu_fwd", &swiglu_fwd, "silu(gate) * up");
  m.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward");
  m.def("rope_apply", &rope_apply, "rotary embedding (direction=+1 fwd, -1 bwd)");
  m.def("apply_delta_", &apply_delta_, "tensor += alpha * delta (in place)");
  m.def("weighted_accumulate_", &weighted_accumulate_, "acc += w * x (in place)");
  m.def("compress_fp16", &compress_fp16_gpu, "clamp + cast to fp16");
  m.def("decompress_fp16", &decompress_fp16_gpu, "fp16 -> fp32");
  m.def("quantize_blockwise", &quantize_blockwise_gpu, "blockwise int8 quantize -> (q, absmax)");
  m.def("dequantize_blockwise", &dequantize_blockwise_gpu, "blockwise int8 dequantize");
  m.def("fused_adamw_", &fused_adamw_, "fused AdamW step on fp32 master params");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused MLM cross-entropy fwd -> (loss_sum, valid, lse)");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused cross-entropy bwd -> dlogits");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "out = gelu(x + bias)");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "backward of bias+gelu (from saved pre-activation)");
  m.def("bias_gelu_bwd_xb", &bias_gelu_bwd_xb, "backward of bias+gelu (recomputes x+bias)");
  m.def("layernorm_fwd", &layernorm_fwd, "fused (residual+)layernorm forward");
  m.def("layernorm_bwd", &layernorm_bwd, "layernorm backward");
  m.def("flash_attn_fwd", &flash_attn_fwd, "CDNA4 flash attention forward -> (O, logsumexp)");
  m.def("flash_attn_bwd", &flash_attn_bwd, "CDNA4 flash attention backward -> (dQ, dK, dV)");
  m.def("transpose_bhsd", &transpose_bhsd, "[B,H,S,D] -> [B,H,D,S] in-register 8x8 transpose");
  m.def("multi_adamw_", &multi_adamw_, "one-launch AdamW step over a parameter list");
  m.def("multi_accumulate_", &multi_accumulate_, "one-launch acc += alpha*x over a tensor list");
}
