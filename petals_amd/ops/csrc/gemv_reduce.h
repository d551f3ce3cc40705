// Shared split-K reduce + epilogue for the gemv family (bf16 and NF4 paths).
// Header-static so each TU gets its own instantiation (no cross-TU device
// linking needed).
#pragma once

#include "common.h"
#include <torch/extension.h>
// ROCm-native stream API: this header is consumed verbatim (torch's hipify
// only rewrites the .hip TUs, not local headers)
#include <c10/hip/HIPStream.h>
#include <cstdlib>

enum GemvEpilogue : int {
  EPI_PLAIN_F32 = 0,     // y_f32[b, out] = sum
  EPI_PLAIN_BF16 = 1,    // y_bf16[b, out] = sum
  EPI_RESIDUAL_BF16 = 2, // y_bf16 = residual_bf16 + sum
  EPI_SWIGLU_F32 = 3,    // out = silu(sum[:half]) * sum[half:], y_f32[b, half]
  EPI_GELU_F32 = 4,      // y_f32[b, out] = gelu_tanh(sum) (BLOOM/Falcon MLPs)
};

// deterministic split sum with 4 independent accumulators so the loads
// pipeline (the naive single-chain version was latency-bound and cost as much
// as the gemv itself on small projections)
static __device__ __forceinline__ float reduce_splits(
    const float* __restrict__ partials, int n_splits, size_t stride, size_t off) {
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int s = 0;
  for (; s + 4 <= n_splits; s += 4) {
    s0 += partials[(size_t)(s + 0) * stride + off];
    s1 += partials[(size_t)(s + 1) * stride + off];
    s2 += partials[(size_t)(s + 2) * stride + off];
    s3 += partials[(size_t)(s + 3) * stride + off];
  }
  for (; s < n_splits; ++s) s0 += partials[(size_t)s * stride + off];
  return (s0 + s1) + (s2 + s3);
}

static __device__ __forceinline__ float gelu_tanh_f32(float x) {
  // matches torch F.gelu(approximate="tanh") (= HF BLOOM/Falcon GELU)
  const float c = 0.79788456080286535588f;  // sqrt(2/pi)
  const float t = tanhf(c * (x + 0.044715f * x * x * x));
  return 0.5f * x * (1.f + t);
}

// 512 threads = 64 outputs x 8 split groups per workgroup: the one-thread-
// per-output version launched only `out/64` single-wave workgroups (0.15
// waves/SIMD for a 10k-output gemv) and was latency-bound at ~7 us — as much
// as the small gemvs themselves. Split groups multiply resident waves 8x and
// cut each thread's serial chain 8x; a tiny LDS tree combines (deterministic
// order preserved: fixed group partition, fixed add order).
#define GEMV_REDUCE_GROUPS 16  // 16 split groups x 64 outputs (1024-thread
// wgs): halves each thread's serial split chain vs the round-2 value of 8 —
// +1.3 tok/s on every alternating A/B pair (profiles/qrr_groups_ab.log);
// PETALS_REDUCE_GROUPS=8 selects the old width

template <int G>
static __global__ __launch_bounds__(64 * G) void gemv_reduce_kernel_impl(
    const float* __restrict__ partials,  // [n_splits, batch, out]
    const unsigned short* __restrict__ residual,  // [batch, out] or null
    const unsigned short* __restrict__ bias,      // [out] bf16 or null
    void* __restrict__ y,
    float* __restrict__ sumsq_out,  // [batch, out/64] per-wg sum(y^2) or null —
    // feeds the folded-RMSNorm decode path (the next gemv derives inv_rms from
    // these instead of a separate norm kernel); computed over the bf16-ROUNDED
    // stored values so it matches what a norm kernel reading y would see
    int n_splits,
    int batch,
    int out_dim,
    int epilogue) {
  __shared__ float acc_g[G][64];
  __shared__ float acc_u[G][64];
  const int half = out_dim >> 1;
  const int n_out = (epilogue == EPI_SWIGLU_F32) ? half : out_dim;
  const int total = batch * n_out;
  const int ol = threadIdx.x & 63;
  const int sg = threadIdx.x >> 6;
  const int idx = blockIdx.x * 64 + ol;
  const bool live = idx < total;
  const int b = live ? idx / n_out : 0;
  const int o = live ? idx - b * n_out : 0;
  const size_t stride = (size_t)batch * out_dim;
  const size_t row = (size_t)b * out_dim;

  float g = 0.f, u = 0.f;
  if (live) {
    for (int s = sg; s < n_splits; s += G) g += partials[(size_t)s * stride + row + o];
    if (epilogue == EPI_SWIGLU_F32)
      for (int s = sg; s < n_splits; s += G) u += partials[(size_t)s * stride + row + o + half];
  }
  acc_g[sg][ol] = g;
  acc_u[sg][ol] = u;
  __syncthreads();
  if (sg != 0 || !live) return;
  float sum = 0.f, usum = 0.f;
#pragma unroll
  for (int k = 0; k < G; ++k) {
    sum += acc_g[k][ol];
    usum += acc_u[k][ol];
  }
  if (epilogue == EPI_SWIGLU_F32) {
    if (bias) {
      sum += bf16_to_f32(bias[o]);
      usum += bf16_to_f32(bias[o + half]);
    }
    const float act = sum / (1.f + __expf(-sum)) * usum;
    reinterpret_cast<float*>(y)[(size_t)b * half + o] = act;
  } else {
    if (bias) sum += bf16_to_f32(bias[o]);
    if (epilogue == EPI_PLAIN_F32) {
      reinterpret_cast<float*>(y)[row + o] = sum;
    } else if (epilogue == EPI_PLAIN_BF16) {
      reinterpret_cast<unsigned short*>(y)[row + o] = f32_to_bf16(sum);
    } else if (epilogue == EPI_GELU_F32) {
      reinterpret_cast<float*>(y)[row + o] = gelu_tanh_f32(sum);
    } else {  // EPI_RESIDUAL_BF16
      const float r = bf16_to_f32(residual[row + o]);
      const unsigned short packed = f32_to_bf16(r + sum);
      reinterpret_cast<unsigned short*>(y)[row + o] = packed;
      if (sumsq_out) {
        const float v = bf16_to_f32(packed);
        float sq = wave_reduce_sum(v * v);  // 64 lanes of sg==0 = one wave
        if (ol == 0) sumsq_out[(size_t)b * (out_dim >> 6) + (o >> 6)] = sq;
      }
    }
  }
}

// allocates the output tensor, launches the reduce, returns y
static inline torch::Tensor launch_gemv_reduce(
    torch::Tensor partials,
    c10::optional<torch::Tensor> residual,
    c10::optional<torch::Tensor> bias,
    int n_splits,
    int batch,
    int out_dim,
    int epilogue,
    const torch::TensorOptions& f32_opts,
    const torch::TensorOptions& bf16_opts,
    c10::optional<torch::Tensor> sumsq_out = c10::nullopt) {
  const int half = out_dim / 2;
  const int n_out = (epilogue == EPI_SWIGLU_F32) ? half : out_dim;
  torch::Tensor y;
  if (epilogue == EPI_PLAIN_F32 || epilogue == EPI_GELU_F32)
    y = torch::empty({batch, out_dim}, f32_opts);
  else if (epilogue == EPI_SWIGLU_F32)
    y = torch::empty({batch, half}, f32_opts);
  else
    y = torch::empty({batch, out_dim}, bf16_opts);
  const unsigned short* res_p = nullptr;
  if (epilogue == EPI_RESIDUAL_BF16) {
    TORCH_CHECK(residual.has_value(), "residual required for EPI_RESIDUAL_BF16");
    TORCH_CHECK(residual->is_contiguous());
    res_p = reinterpret_cast<const unsigned short*>(residual->data_ptr());
  }
  const unsigned short* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    TORCH_CHECK(bias->is_contiguous() && bias->dtype() == torch::kBFloat16);
    TORCH_CHECK(bias->numel() == out_dim, "bias must be [out_dim] (pre-activation)");
    bias_p = reinterpret_cast<const unsigned short*>(bias->data_ptr());
  }
  float* ss_p = nullptr;
  if (sumsq_out.has_value() && sumsq_out->defined() && sumsq_out->numel() > 0) {
    TORCH_CHECK(epilogue == EPI_RESIDUAL_BF16, "sumsq_out only with the residual epilogue");
    TORCH_CHECK(out_dim % 64 == 0, "sumsq_out needs out_dim % 64 == 0");
    TORCH_CHECK(sumsq_out->dtype() == torch::kFloat32 && sumsq_out->is_contiguous());
    TORCH_CHECK(sumsq_out->numel() >= (int64_t)batch * (out_dim / 64));
    ss_p = sumsq_out->data_ptr<float>();
  }
  const int total = batch * n_out;
  static const int red_g = [] {
    const char* e = std::getenv("PETALS_REDUCE_GROUPS");
    return e ? std::atoi(e) : GEMV_REDUCE_GROUPS;
  }();
  int rblocks = (total + 63) / 64;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (red_g == 16)
    gemv_reduce_kernel_impl<16><<<rblocks, 64 * 16, 0, stream>>>(
        partials.data_ptr<float>(), res_p, bias_p, y.data_ptr(), ss_p, n_splits, batch, out_dim, epilogue);
  else
    gemv_reduce_kernel_impl<8><<<rblocks, 64 * 8, 0, stream>>>(
        partials.data_ptr<float>(), res_p, bias_p, y.data_ptr(), ss_p, n_splits, batch, out_dim, epilogue);
  HIP_CHECK_LAST();
  return y;
}
