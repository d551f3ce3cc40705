// Shared split-K reduce + epilogue for the gemv family (bf16 and NF4 paths).
// Header-static so each TU gets its own instantiation (no cross-TU device
// linking needed).
#pragma once

#include "common.h"
#include <torch/extension.h>
// ROCm-native stream API: this header is consumed verbatim (torch's hipify
// only rewrites the .hip TUs, not local headers)
#include <c10/hip/HIPStream.h>

enum GemvEpilogue : int {
  EPI_PLAIN_F32 = 0,     // y_f32[b, out] = sum
  EPI_PLAIN_BF16 = 1,    // y_bf16[b, out] = sum
  EPI_RESIDUAL_BF16 = 2, // y_bf16 = residual_bf16 + sum
  EPI_SWIGLU_F32 = 3,    // out = silu(sum[:half]) * sum[half:], y_f32[b, half]
};

static __global__ void gemv_reduce_kernel_impl(
    const float* __restrict__ partials,  // [n_splits, batch, out]
    const unsigned short* __restrict__ residual,  // [batch, out] or null
    void* __restrict__ y,
    int n_splits,
    int batch,
    int out_dim,
    int epilogue) {
  const int half = out_dim >> 1;
  const int n_out = (epilogue == EPI_SWIGLU_F32) ? half : out_dim;
  const int total = batch * n_out;
  for (int idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total; idx += gridDim.x * blockDim.x) {
    const int b = idx / n_out;
    const int o = idx - b * n_out;
    if (epilogue == EPI_SWIGLU_F32) {
      float g = 0.f, u = 0.f;
      for (int s = 0; s < n_splits; ++s) {
        const float* base = partials + ((size_t)s * batch + b) * out_dim;
        g += base[o];
        u += base[o + half];
      }
      const float act = g / (1.f + __expf(-g)) * u;
      reinterpret_cast<float*>(y)[(size_t)b * half + o] = act;
    } else {
      float sum = 0.f;
      for (int s = 0; s < n_splits; ++s) sum += partials[((size_t)s * batch + b) * out_dim + o];
      if (epilogue == EPI_PLAIN_F32) {
        reinterpret_cast<float*>(y)[(size_t)b * out_dim + o] = sum;
      } else if (epilogue == EPI_PLAIN_BF16) {
        reinterpret_cast<unsigned short*>(y)[(size_t)b * out_dim + o] = f32_to_bf16(sum);
      } else {  // EPI_RESIDUAL_BF16
        const float r = bf16_to_f32(residual[(size_t)b * out_dim + o]);
        reinterpret_cast<unsigned short*>(y)[(size_t)b * out_dim + o] = f32_to_bf16(r + sum);
      }
    }
  }
}

// allocates the output tensor, launches the reduce, returns y
static inline torch::Tensor launch_gemv_reduce(
    torch::Tensor partials,
    c10::optional<torch::Tensor> residual,
    int n_splits,
    int batch,
    int out_dim,
    int epilogue,
    const torch::TensorOptions& f32_opts,
    const torch::TensorOptions& bf16_opts) {
  const int half = out_dim / 2;
  const int n_out = (epilogue == EPI_SWIGLU_F32) ? half : out_dim;
  torch::Tensor y;
  if (epilogue == EPI_PLAIN_F32) y = torch::empty({batch, out_dim}, f32_opts);
  else if (epilogue == EPI_SWIGLU_F32) y = torch::empty({batch, half}, f32_opts);
  else y = torch::empty({batch, out_dim}, bf16_opts);
  const unsigned short* res_p = nullptr;
  if (epilogue == EPI_RESIDUAL_BF16) {
    TORCH_CHECK(residual.has_value(), "residual required for EPI_RESIDUAL_BF16");
    TORCH_CHECK(residual->is_contiguous());
    res_p = reinterpret_cast<const unsigned short*>(residual->data_ptr());
  }
  const int total = batch * n_out;
  int rblocks = std::min((total + 255) / 256, 2048);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  gemv_reduce_kernel_impl<<<rblocks, 256, 0, stream>>>(
      partials.data_ptr<float>(), res_p, y.data_ptr(), n_splits, batch, out_dim, epilogue);
  HIP_CHECK_LAST();
  return y;
}
