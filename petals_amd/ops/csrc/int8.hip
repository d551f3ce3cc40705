// Weight-only int8 decode GEMV for CDNA4 (gfx950).
//
// Same split-K structure as gemv.hip: a wave owns 512 output columns (8 per
// lane, one 8-byte load per input row), x[i] is a wave-uniform scalar load
// broadcast into 8 FMAs, the input dimension is split across gridDim.y and
// the shared reduce (gemv_reduce.h) fuses the epilogue. Per-column absmax
// scales (bf16 [out]) are applied once per partial write, so the inner loop
// is pure s8->f32 FMA. Weights are exactly half the bytes of bf16, so the
// HBM-bound decode runs ~2x the bf16 GEMV per weight read.
//
// Parity: the reference's optional LLM.int8 path (bitsandbytes Linear8bitLt,
// reference utils/convert_block.py:87-111); here weights are per-out-column
// symmetric int8 (quantized on load in torch) with bf16/f32 activations, so
// there is no outlier decomposition to fuse.

#include "common.h"
#include "gemv_reduce.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#ifndef GEMV_OUT_PER_WAVE
#define GEMV_OUT_PER_WAVE 512
#endif

using char8 = __attribute__((ext_vector_type(8))) char;

template <int BATCH>
__global__ __launch_bounds__(WAVE) void gemv_int8_kernel(
    const signed char* __restrict__ q,   // [in, out]
    const unsigned short* __restrict__ scale,  // [out] bf16 (absmax/127)
    const float* __restrict__ x,         // [BATCH, in]
    float* __restrict__ partials,        // [n_splits, BATCH, out]
    int in_dim,
    int out_dim,
    int i_per_split) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int out0 = blockIdx.x * GEMV_OUT_PER_WAVE + lane * 8;
  if (out0 >= out_dim) return;
  const int split = blockIdx.y;
  const int i_begin = split * i_per_split;
  const int i_end = min(i_begin + i_per_split, in_dim);

  float acc[BATCH][8];
#pragma unroll
  for (int b = 0; b < BATCH; ++b)
#pragma unroll
    for (int v = 0; v < 8; ++v) acc[b][v] = 0.f;

  const bool full = (out0 + 8) <= out_dim;
  if (full) {
    constexpr int UNROLL = 16;  // 128 B of weight loads in flight per lane batch
    const signed char* wp = q + (size_t)i_begin * out_dim + out0;
    int i = i_begin;
    for (; i + UNROLL <= i_end; i += UNROLL) {
      char8 w8[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        w8[u] = *reinterpret_cast<const char8*>(wp + (size_t)u * out_dim);
      float xs[BATCH][UNROLL];
#pragma unroll
      for (int b = 0; b < BATCH; ++b)
#pragma unroll
        for (int u = 0; u < UNROLL; ++u) xs[b][u] = x[(size_t)b * in_dim + i + u];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        float wf[8];
#pragma unroll
        for (int v = 0; v < 8; ++v) wf[v] = (float)w8[u][v];
#pragma unroll
        for (int b = 0; b < BATCH; ++b)
#pragma unroll
          for (int v = 0; v < 8; ++v) acc[b][v] = fmaf(wf[v], xs[b][u], acc[b][v]);
      }
      wp += (size_t)UNROLL * out_dim;
    }
    for (; i < i_end; ++i) {
      const char8 w8 = *reinterpret_cast<const char8*>(q + (size_t)i * out_dim + out0);
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = x[(size_t)b * in_dim + i];
#pragma unroll
        for (int v = 0; v < 8; ++v) acc[b][v] = fmaf((float)w8[v], xv, acc[b][v]);
      }
    }
  } else {
    for (int i = i_begin; i < i_end; ++i) {
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = x[(size_t)b * in_dim + i];
        // compile-time trip count + guard: runtime-bounded indexing of acc[][]
        // would force it onto the scratch stack (see the NF4 twin)
#pragma unroll
        for (int v = 0; v < 8; ++v)
          if (out0 + v < out_dim)
            acc[b][v] = fmaf((float)q[(size_t)i * out_dim + out0 + v], xv, acc[b][v]);
      }
    }
  }

  // fold the per-column scale into the partial write (once per split)
  float sc[8];
#pragma unroll
  for (int v = 0; v < 8; ++v) sc[v] = (out0 + v < out_dim) ? bf16_to_f32(scale[out0 + v]) : 0.f;
#pragma unroll
  for (int b = 0; b < BATCH; ++b) {
    float* dst = partials + ((size_t)split * BATCH + b) * out_dim + out0;
    if (full) {
      float4v* d4 = reinterpret_cast<float4v*>(dst);
      d4[0] = float4v{acc[b][0] * sc[0], acc[b][1] * sc[1], acc[b][2] * sc[2], acc[b][3] * sc[3]};
      d4[1] = float4v{acc[b][4] * sc[4], acc[b][5] * sc[5], acc[b][6] * sc[6], acc[b][7] * sc[7]};
    } else {
#pragma unroll
      for (int v = 0; v < 8; ++v)
        if (out0 + v < out_dim) dst[v] = acc[b][v] * sc[v];
    }
  }
}

torch::Tensor gemv_int8(
    torch::Tensor q,         // [in, out] int8
    torch::Tensor scale,     // [out] bf16
    torch::Tensor x,         // [batch, in] f32
    torch::Tensor workspace,
    c10::optional<torch::Tensor> residual,
    int64_t epilogue,
    int64_t splits_override,
    c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kInt8 && q.dim() == 2);
  TORCH_CHECK(scale.dtype() == torch::kBFloat16 && scale.numel() == q.size(1));
  TORCH_CHECK(x.dtype() == torch::kFloat32 && x.dim() == 2);
  const int in_dim = q.size(0), out_dim = q.size(1);
  TORCH_CHECK(x.size(1) == in_dim, "x/in mismatch");
  const int batch = x.size(0);
  TORCH_CHECK(batch <= 8, "decode gemv supports batch <= 8");

  const long out_waves = (out_dim + GEMV_OUT_PER_WAVE - 1) / GEMV_OUT_PER_WAVE;
  // int8 matrices are 2x smaller than bf16: allow chunks down to 128 rows
  long splits = splits_override > 0 ? splits_override : (1024 + out_waves - 1) / out_waves;
  long max_splits = (in_dim + 127) / 128;
  if (splits > max_splits) splits = max_splits;
  if (splits < 1) splits = 1;
  const int i_per_split = (in_dim + splits - 1) / splits;

  torch::Tensor partials;
  auto f32opts = x.options();
  if (workspace.numel() >= (int64_t)splits * batch * out_dim) {
    partials = workspace;
  } else {
    partials = torch::empty({(int64_t)splits, batch, out_dim}, f32opts);
  }
  dim3 grid(out_waves, splits);
  auto stream = at::cuda::getCurrentCUDAStream();

#define LAUNCH_I8(B)                                                          \
  gemv_int8_kernel<B><<<grid, WAVE, 0, stream>>>(                             \
      reinterpret_cast<const signed char*>(q.data_ptr()),                     \
      reinterpret_cast<const unsigned short*>(scale.data_ptr()),              \
      x.data_ptr<float>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split)
  switch (batch) {
    case 1: LAUNCH_I8(1); break;
    case 2: LAUNCH_I8(2); break;
    case 3: LAUNCH_I8(3); break;
    case 4: LAUNCH_I8(4); break;
    case 5: LAUNCH_I8(5); break;
    case 6: LAUNCH_I8(6); break;
    case 7: LAUNCH_I8(7); break;
    case 8: LAUNCH_I8(8); break;
    default: TORCH_CHECK(false, "decode gemv supports batch <= 8");
  }
#undef LAUNCH_I8
  HIP_CHECK_LAST();

  if (epilogue < 0) {
    // RAW mode: the caller fuses its own reduce+epilogue (e.g. the qkv
    // rope+cache-write reduce); the view's shape carries the split count
    return partials.view(-1).narrow(0, 0, (int64_t)splits * batch * out_dim)
        .view({(int64_t)splits, (int64_t)batch, (int64_t)out_dim});
  }
  torch::Tensor y = launch_gemv_reduce(
      partials, residual, bias, (int)splits, batch, out_dim, (int)epilogue, f32opts, scale.options());
  return y;
}
