// Decode-path GEMV suite for CDNA4 (gfx950).
//
// Weights are stored TRANSPOSED: Wt[in, out] bf16 row-major (= W^T). A wave
// owns a strip of 512 output columns (8 per lane, one short8 = 16 B/lane per
// input row) and iterates over a chunk of the input dimension; x[i] is a
// wave-uniform scalar load (s_load, L2-resident) broadcast into 8 FMAs.
// The input dimension is split across gridDim.y workgroups so 256 CUs stay
// busy even for a single-token GEMV; each split writes a partial fp32 strip,
// and gemv_reduce sums splits deterministically while fusing the epilogue
// (plain / +residual / SwiGLU-pair) and dtype conversion.
//
// Replaces (capability-wise) the per-token GEMMs the reference runs through
// torch/CUDA inside HF blocks (reference models/llama/block.py:44-127); the
// design follows /opt/skills/guides/cdna_hip_programming.md Appendix B
// ("Element-wise"/"GEMM" vectorization rules; G13).

#include "common.h"
#include "gemv_reduce.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#ifndef GEMV_OUT_PER_WAVE
#define GEMV_OUT_PER_WAVE 512  // 64 lanes x 8 bf16
#endif

// ---------------------------------------------------------------- gemv core

template <int BATCH>
__global__ __launch_bounds__(WAVE) void gemv_bf16_kernel(
    const unsigned short* __restrict__ wt,  // [in, out]
    const float* __restrict__ x,            // [BATCH, in]
    float* __restrict__ partials,           // [n_splits, BATCH, out]
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
    // UNROLL x 16B weight loads in flight per wave: HBM latency (~900 cyc)
    // needs many outstanding loads; the scalar x loads are wave-uniform
    constexpr int UNROLL = 16;  // 256 B of weight loads in flight per wave
    const unsigned short* wp = wt + (size_t)i_begin * out_dim + out0;
    int i = i_begin;
    for (; i + UNROLL <= i_end; i += UNROLL) {
      short8 w8[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        w8[u] = *reinterpret_cast<const short8*>(wp + (size_t)u * out_dim);
      float xs[BATCH][UNROLL];
#pragma unroll
      for (int b = 0; b < BATCH; ++b)
#pragma unroll
        for (int u = 0; u < UNROLL; ++u) xs[b][u] = x[(size_t)b * in_dim + i + u];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        float wf[8];
#pragma unroll
        for (int v = 0; v < 8; ++v) wf[v] = bf16_to_f32((unsigned short)w8[u][v]);
#pragma unroll
        for (int b = 0; b < BATCH; ++b)
#pragma unroll
          for (int v = 0; v < 8; ++v) acc[b][v] = fmaf(wf[v], xs[b][u], acc[b][v]);
      }
      wp += (size_t)UNROLL * out_dim;
    }
    for (; i < i_end; ++i) {
      const short8 w8 = *reinterpret_cast<const short8*>(wt + (size_t)i * out_dim + out0);
      float wf[8];
#pragma unroll
      for (int v = 0; v < 8; ++v) wf[v] = bf16_to_f32((unsigned short)w8[v]);
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = x[(size_t)b * in_dim + i];
#pragma unroll
        for (int v = 0; v < 8; ++v) acc[b][v] = fmaf(wf[v], xv, acc[b][v]);
      }
    }
  } else {
    for (int i = i_begin; i < i_end; ++i) {
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = x[(size_t)b * in_dim + i];
        // compile-time trip count with a guard: a runtime-bounded loop here
        // makes acc[][] dynamically indexed, forcing it onto the SCRATCH
        // stack for the whole kernel (observed on the NF4 twin at BATCH>=5)
#pragma unroll
        for (int v = 0; v < 8; ++v)
          if (out0 + v < out_dim)
            acc[b][v] = fmaf(bf16_to_f32(wt[(size_t)i * out_dim + out0 + v]), xv, acc[b][v]);
      }
    }
  }

#pragma unroll
  for (int b = 0; b < BATCH; ++b) {
    float* dst = partials + ((size_t)split * BATCH + b) * out_dim + out0;
    if (full) {
      float4v* d4 = reinterpret_cast<float4v*>(dst);
      d4[0] = float4v{acc[b][0], acc[b][1], acc[b][2], acc[b][3]};
      d4[1] = float4v{acc[b][4], acc[b][5], acc[b][6], acc[b][7]};
    } else {
#pragma unroll
      for (int v = 0; v < 8; ++v)
        if (out0 + v < out_dim) dst[v] = acc[b][v];
    }
  }
}

// ------------------------------------------------------------------- host

static int pick_splits(int in_dim, int out_dim) {
  const long out_waves = (out_dim + GEMV_OUT_PER_WAVE - 1) / GEMV_OUT_PER_WAVE;
  // measured knee (profiles/gemv_sweep): ~768 single-wave workgroups (3/CU)
  // with input chunks of >=256 rows; more splits shrink chunks below the
  // latency-amortization point, fewer leave CUs idle
  long splits = (768 + out_waves - 1) / out_waves;
  long max_splits = (in_dim + 255) / 256;
  if (splits > max_splits) splits = max_splits;
  if (splits < 1) splits = 1;
  return (int)splits;
}

torch::Tensor gemv_bf16(
    torch::Tensor wt,        // [in, out] bf16
    torch::Tensor x,         // [batch, in] f32
    torch::Tensor workspace, // [n_splits_max, batch, out] f32 (preallocated, may be empty)
    c10::optional<torch::Tensor> residual,  // [batch, out] bf16
    int64_t epilogue,
    int64_t splits_override,
    c10::optional<torch::Tensor> bias) {  // [out] bf16, added pre-activation
  TORCH_CHECK(wt.is_cuda() && wt.dtype() == torch::kBFloat16 && wt.dim() == 2);
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kFloat32 && x.dim() == 2);
  const int in_dim = wt.size(0), out_dim = wt.size(1);
  TORCH_CHECK(x.size(1) == in_dim, "x/in mismatch");
  const int batch = x.size(0);
  TORCH_CHECK(batch <= 8, "decode gemv supports batch <= 8");

  int splits = splits_override > 0 ? (int)splits_override : pick_splits(in_dim, out_dim);
  // UNROLL(16)-aligned chunks: unaligned splits push rows onto the slow
  // per-row tail path (same fix as the NF4 gemv)
  const int i_per_split = (int)(((in_dim + splits - 1) / splits + 15) & ~15);
  splits = (in_dim + i_per_split - 1) / i_per_split;

  torch::Tensor partials;
  if (workspace.numel() >= (int64_t)splits * batch * out_dim) {
    partials = workspace;
  } else {
    partials = torch::empty({(int64_t)splits, batch, out_dim}, x.options());
  }

  const int out_waves = (out_dim + GEMV_OUT_PER_WAVE - 1) / GEMV_OUT_PER_WAVE;
  dim3 grid(out_waves, splits);
  dim3 block(WAVE);
  auto stream = at::cuda::getCurrentCUDAStream();

  const unsigned short* wt_p = reinterpret_cast<const unsigned short*>(wt.data_ptr());
  const float* x_p = x.data_ptr<float>();
  float* part_p = partials.data_ptr<float>();

#define LAUNCH_GEMV(B)                                                        \
  gemv_bf16_kernel<B><<<grid, block, 0, stream>>>(wt_p, x_p, part_p, in_dim, out_dim, i_per_split)
  switch (batch) {
    case 1: LAUNCH_GEMV(1); break;
    case 2: LAUNCH_GEMV(2); break;
    case 3: LAUNCH_GEMV(3); break;
    case 4: LAUNCH_GEMV(4); break;
    case 5: LAUNCH_GEMV(5); break;
    case 6: LAUNCH_GEMV(6); break;
    case 7: LAUNCH_GEMV(7); break;
    case 8: LAUNCH_GEMV(8); break;
    default: TORCH_CHECK(false, "decode gemv supports batch <= 8");
  }
#undef LAUNCH_GEMV
  HIP_CHECK_LAST();

  // epilogue (shared with the NF4 path)
  if (epilogue < 0) {
    // RAW mode: the caller fuses its own reduce+epilogue (e.g. the qkv
    // rope+cache-write reduce); the view's shape carries the split count
    return partials.view(-1).narrow(0, 0, (int64_t)splits * batch * out_dim)
        .view({(int64_t)splits, (int64_t)batch, (int64_t)out_dim});
  }
  torch::Tensor y = launch_gemv_reduce(
      partials, residual, bias, splits, batch, out_dim, (int)epilogue, x.options(), wt.options());
  return y;
}
